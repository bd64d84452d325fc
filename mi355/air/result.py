"""Result / ResultGrid (reference
Model_finetuning_and_batch_inference.ipynb:cell-47, Introduction_to_Ray_AI_
Runtime.ipynb:cell-36/39-40, cell-48-52)."""
from __future__ import annotations

from dataclasses import dataclass, field

from .checkpoint import Checkpoint


@dataclass
class Result:
    metrics: dict = field(default_factory=dict)
    checkpoint: Checkpoint | None = None
    error: Exception | None = None
    path: str | None = None
    best_checkpoints: list = field(default_factory=list)  # [(Checkpoint, metrics)]
    metrics_dataframe: object = None
    config: dict = field(default_factory=dict)

    @property
    def metrics_df(self):
        return self.metrics_dataframe


class ResultGrid:
    def __init__(self, results: list[Result]):
        self._results = list(results)

    def __len__(self):
        return len(self._results)

    def __getitem__(self, i) -> Result:
        return self._results[i]

    def __iter__(self):
        return iter(self._results)

    @property
    def errors(self):
        return [r.error for r in self._results if r.error is not None]

    def get_best_result(self, metric: str | None = None, mode: str | None = None) -> Result:
        metric = metric or getattr(self, "_default_metric", None)
        mode = mode or getattr(self, "_default_mode", "min")
        ok = [r for r in self._results if r.error is None and metric in r.metrics]
        if not ok:
            raise RuntimeError("no successful trial reported the metric "
                               f"{metric!r}; errors={self.errors}")
        keyfn = lambda r: r.metrics[metric]
        return min(ok, key=keyfn) if mode == "min" else max(ok, key=keyfn)

    def get_dataframe(self):
        import pandas as pd

        return pd.DataFrame([{**r.config, **r.metrics} for r in self._results])
