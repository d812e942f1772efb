from .checkpoint import load_checkpoint, save_checkpoint  # noqa: F401
from .timers import StepTimer  # noqa: F401
