"""Trial schedulers: ASHA — asynchronous successive halving
with rung-based promotion (reference ASHAScheduler(max_t=16) at
Model_finetuning_and_batch_inference.ipynb:cell-51/57).

ASHA semantics: rungs at t = grace_period * reduction_factor^k up to max_t.
When a trial reports at a rung boundary, it CONTINUES only if its metric is
in the top 1/reduction_factor of completed results at that rung (async: with
fewer than reduction_factor results so far, it continues optimistically).
"""
from __future__ import annotations

import math
from collections import defaultdict


class FIFOScheduler:
    CONTINUE = "continue"
    STOP = "stop"

    def on_result(self, trial_id, metrics) -> str:
        return self.CONTINUE


class ASHAScheduler(FIFOScheduler):
    def __init__(self, time_attr: str = "training_iteration", metric: str | None = None,
                 mode: str | None = None, max_t: int = 100, grace_period: int = 1,
                 reduction_factor: float = 4, brackets: int = 1):
        self.time_attr = time_attr
        self.metric = metric
        self.mode = mode
        self.max_t = max_t
        self.grace_period = grace_period
        self.rf = reduction_factor
        # rung levels: grace, grace*rf, grace*rf^2, ... < max_t
        self.rungs: list[int] = []
        t = grace_period
        while t < max_t:
            self.rungs.append(int(t))
            t *= reduction_factor
        self._rung_results: dict[int, list[float]] = defaultdict(list)
        self._trial_iter: dict = defaultdict(int)

    def _sign(self):
        return 1.0 if self.mode == "min" else -1.0

    def on_result(self, trial_id, metrics) -> str:
        t = int(metrics.get(self.time_attr, self._trial_iter[trial_id] + 1))
        self._trial_iter[trial_id] = t
        if self.metric not in metrics:
            return self.CONTINUE
        value = self._sign() * float(metrics[self.metric])
        if t >= self.max_t:
            return self.STOP
        for rung in self.rungs:
            if t == rung:
                results = self._rung_results[rung]
                results.append(value)
                k = max(1, int(len(results) / self.rf))
                cutoff = sorted(results)[k - 1]  # top-1/rf smallest signed
                if value > cutoff:
                    return self.STOP
        return self.CONTINUE
