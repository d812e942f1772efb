from .prefetch import Prefetcher  # noqa: F401
