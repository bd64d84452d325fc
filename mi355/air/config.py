"""AIR config dataclasses — API-compatible with the
reference's call sites (ScalingConfig(num_workers, use_gpu) at
flan-t5-batch-inference.py:98; RunConfig/CheckpointConfig at :103-109)."""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class ScalingConfig:
    num_workers: int = 1
    use_gpu: bool = False
    resources_per_worker: dict | None = None
    trainer_resources: dict | None = None
    placement_strategy: str = "PACK"

    @property
    def num_gpus_per_worker(self) -> float:
        if self.resources_per_worker and "GPU" in self.resources_per_worker:
            return self.resources_per_worker["GPU"]
        return 1.0 if self.use_gpu else 0.0


@dataclass
class CheckpointConfig:
    num_to_keep: int | None = None
    checkpoint_score_attribute: str | None = None
    checkpoint_score_order: str = "max"
    checkpoint_frequency: int = 0
    checkpoint_at_end: bool | None = None


@dataclass
class FailureConfig:
    max_failures: int = 0


@dataclass
class RunConfig:
    name: str | None = None
    local_dir: str | None = None
    storage_path: str | None = None
    checkpoint_config: CheckpointConfig | None = None
    failure_config: FailureConfig | None = None
    verbose: int = 1

    def run_dir(self) -> str:
        import os
        import time

        base = self.storage_path or self.local_dir or os.path.expanduser("~/mi355_results")
        name = self.name or f"run_{int(time.time())}"
        return os.path.join(base, name)
