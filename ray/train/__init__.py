from mi355.train import (DataParallelTrainer, Trainer, TorchTrainer,
                         TrainingArguments, XGBoostTrainer, session)
from . import huggingface, predictor, batch_predictor, xgboost, torch  # noqa: F401

__all__ = ["DataParallelTrainer", "TorchTrainer", "XGBoostTrainer",
           "Trainer", "TrainingArguments", "session", "huggingface",
           "predictor", "batch_predictor", "xgboost", "torch"]
