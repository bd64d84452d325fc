from mi355.air import (CheckpointConfig, FailureConfig, RunConfig,
                       ScalingConfig)

__all__ = ["CheckpointConfig", "FailureConfig", "RunConfig", "ScalingConfig"]
