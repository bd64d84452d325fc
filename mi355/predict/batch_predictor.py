"""BatchPredictor (call stack §3.3): actor-pool batch
inference over Dataset blocks — reference
flan-t5-batch-inference.py:119-134, Model_finetuning_and_batch_inference.
ipynb:cell-64/67, Scaling_batch_inference.ipynb:cell-76/78.

One predictor replica per GPU (8 shards on a full node); each replica loads
the model from the checkpoint ONCE (the stateful-actor rationale at
cell-103) and maps over batches. Without the runtime initialised it runs
in-process (single replica) — same results, used by CPU tests."""
from __future__ import annotations

import pandas as pd

from ..air import Checkpoint
from ..data import Dataset
from ..data.dataset import _to_table


class BatchPredictor:
    def __init__(self, checkpoint: Checkpoint, predictor_cls, **predictor_kwargs):
        self._checkpoint = checkpoint
        self._predictor_cls = predictor_cls
        self._predictor_kwargs = predictor_kwargs

    @classmethod
    def from_checkpoint(cls, checkpoint: Checkpoint, predictor_cls,
                        **predictor_kwargs) -> "BatchPredictor":
        return cls(checkpoint, predictor_cls, **predictor_kwargs)

    def predict(self, data: Dataset, *, batch_size: int = 4096,
                min_scoring_workers: int = 1, max_scoring_workers: int | None = None,
                num_gpus_per_worker: float = 0, num_cpus_per_worker: float = 1,
                feature_columns=None, keep_columns=None, **predict_kwargs) -> Dataset:
        from .. import runtime

        n_workers = max_scoring_workers or min_scoring_workers
        if (max_scoring_workers is None and num_gpus_per_worker
                and runtime.is_initialized()):
            # reference default: one replica per GPU (8 shards on a full node)
            from ..runtime.core import _rt

            n_workers = max(n_workers,
                            int(_rt().num_gpus // max(num_gpus_per_worker, 1)))

        if runtime.is_initialized() and n_workers > 1:
            return self._predict_actor_pool(data, batch_size, n_workers,
                                            num_gpus_per_worker, feature_columns,
                                            keep_columns, predict_kwargs)
        predictor = self._predictor_cls.from_checkpoint(self._checkpoint,
                                                        **self._predictor_kwargs)
        out_blocks = []
        for batch in data.iter_batches(batch_size=batch_size, batch_format="pandas"):
            out_blocks.append(_to_table(_predict_one(predictor, batch,
                                                     feature_columns, keep_columns,
                                                     predict_kwargs)))
        return Dataset(out_blocks)

    def _predict_actor_pool(self, data, batch_size, n_workers, num_gpus,
                            feature_columns, keep_columns, predict_kwargs):
        import cloudpickle as cp

        from .. import runtime
        from ..runtime import ActorPool

        ckpt = self._checkpoint
        cls_b = cp.dumps(self._predictor_cls)
        kw_b = cp.dumps(self._predictor_kwargs)

        class _Replica:
            def __init__(self, ckpt, cls_b, kw_b):
                import cloudpickle as cp2

                cls = cp2.loads(cls_b)
                self.predictor = cls.from_checkpoint(ckpt, **cp2.loads(kw_b))

            def predict(self, df, feature_columns, keep_columns, kwargs):
                return _predict_one(self.predictor, df, feature_columns,
                                    keep_columns, kwargs)

        actor_cls = runtime.remote(_Replica).options(num_gpus=num_gpus)
        actors = [actor_cls.remote(ckpt, cls_b, kw_b) for _ in range(n_workers)]
        pool = ActorPool(actors)
        batches = list(data.iter_batches(batch_size=batch_size, batch_format="pandas"))
        outs = pool.map(
            lambda a, df: a.predict.remote(df, feature_columns, keep_columns,
                                           predict_kwargs),
            batches,
        )
        for a in actors:
            a._kill()
        return Dataset([_to_table(o) for o in outs])


def _predict_one(predictor, df: pd.DataFrame, feature_columns, keep_columns,
                 kwargs) -> pd.DataFrame:
    inp = df[feature_columns] if feature_columns else df
    out = predictor.predict(inp, **kwargs)
    if not isinstance(out, pd.DataFrame):
        out = pd.DataFrame({"predictions": list(out)})
    if keep_columns:
        for c in keep_columns:
            out[c] = df[c].values
    return out
