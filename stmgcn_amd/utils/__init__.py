from .misc import seed_everything  # noqa: F401
from .profiling import StepTimer  # noqa: F401
