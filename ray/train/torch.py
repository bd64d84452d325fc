from mi355.predict import TorchPredictor
from mi355.train import TorchTrainer
from mi355.air import TorchCheckpoint

__all__ = ["TorchTrainer", "TorchPredictor", "TorchCheckpoint"]
