from .batch_predictor import BatchPredictor
from .hf_predictor import HuggingFacePredictor, TorchPredictor
from .predictor import Predictor
from .xgboost_predictor import XGBoostPredictor

__all__ = ["BatchPredictor", "Predictor", "HuggingFacePredictor", "TorchPredictor", "XGBoostPredictor"]
