from .trainer import ModelTrainer  # noqa: F401
from .fused_adam import FusedAdam  # noqa: F401
from .metrics import MSE, RMSE, MAE, MAPE, PCC  # noqa: F401
