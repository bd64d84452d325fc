from mi355.air import HuggingFaceCheckpoint
from mi355.predict import HuggingFacePredictor
from mi355.train import HuggingFaceTrainer

__all__ = ["HuggingFaceTrainer", "HuggingFaceCheckpoint", "HuggingFacePredictor"]
