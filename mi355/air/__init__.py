from .checkpoint import Checkpoint, HuggingFaceCheckpoint, TorchCheckpoint
from .config import CheckpointConfig, FailureConfig, RunConfig, ScalingConfig
from .result import Result, ResultGrid

__all__ = ["Checkpoint", "HuggingFaceCheckpoint", "TorchCheckpoint",
           "CheckpointConfig", "FailureConfig", "RunConfig", "ScalingConfig",
           "Result", "ResultGrid"]
