from .config import EngineConfig, TrainConfig, DataConfig
from .dataset import EstimationDataset
from .trainer import Trainer, TrainResult
from .checkpoint import save_checkpoint, load_checkpoint

__all__ = [
    "EngineConfig",
    "TrainConfig",
    "DataConfig",
    "EstimationDataset",
    "Trainer",
    "TrainResult",
    "save_checkpoint",
    "load_checkpoint",
]
