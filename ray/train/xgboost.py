from mi355.predict import XGBoostPredictor
from mi355.train import XGBoostTrainer

__all__ = ["XGBoostTrainer", "XGBoostPredictor"]
