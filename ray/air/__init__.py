from mi355.air import (Checkpoint, CheckpointConfig, FailureConfig, Result,
                       ResultGrid, RunConfig, ScalingConfig)
from . import config  # noqa: F401

__all__ = ["Checkpoint", "CheckpointConfig", "FailureConfig", "Result",
           "ResultGrid", "RunConfig", "ScalingConfig", "config"]
