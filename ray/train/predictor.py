from mi355.predict import Predictor

__all__ = ["Predictor"]
