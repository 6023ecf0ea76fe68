from .config import TrainerConfig
from .control import (
    TrainTask,
    InferenceTask,
    ModelProvider,
    DatasetProvider,
    OptimizerProvider,
    LRSchedulerProvider,
)
from .train import TrainingConfigurator, Trainer
from .inference import InferenceConfigurator

__all__ = [
    "TrainerConfig",
    "TrainTask",
    "InferenceTask",
    "ModelProvider",
    "DatasetProvider",
    "OptimizerProvider",
    "LRSchedulerProvider",
    "TrainingConfigurator",
    "Trainer",
    "InferenceConfigurator",
]
