from .ddp import GradReducer, init_distributed, cleanup_distributed, dist_env  # noqa: F401
