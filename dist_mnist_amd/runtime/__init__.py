from .graphs import GraphedStep  # noqa: F401
from .prefetch import Prefetcher  # noqa: F401
