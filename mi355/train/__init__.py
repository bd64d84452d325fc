from . import session
from .hf import Trainer, TrainingArguments
from .hf_trainer import HuggingFaceTrainer
from .gbt_trainer import XGBoostTrainer
from .trainer import DataParallelTrainer, TorchTrainer

__all__ = ["session", "Trainer", "TrainingArguments", "HuggingFaceTrainer",
           "DataParallelTrainer", "TorchTrainer", "XGBoostTrainer"]
