from .misc import seed_everything, StepTimer  # noqa: F401
