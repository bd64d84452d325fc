"""XGBoostTrainer-compatible distributed GBT trainer (SURVEY.md C18;
reference Introduction_to_Ray_AI_Runtime.ipynb:cell-32:
XGBoostTrainer(label_column, num_boost_round, ScalingConfig(num_workers=5),
params, datasets, preprocessor)). Row shards per worker; per-level histogram
all-reduce over gloo (M8's rabit analog); rank 0 owns checkpointing."""
from __future__ import annotations

from ..air import RunConfig, ScalingConfig
from .trainer import DataParallelTrainer


def _gbt_loop(config):
    import os

    import numpy as np
    import torch
    import torch.distributed as dist

    from ..models.gbt import GBTModel
    from . import session

    ctx = session.get_context()
    train = session.get_dataset_shard("train")
    valid = session.get_dataset_shard("valid") or session.get_dataset_shard("validation")
    label = config["label_column"]
    df = train.to_pandas()
    y = df[label].to_numpy(dtype=float)
    X = df.drop(columns=[label]).to_numpy(dtype=float)
    feature_names = [c for c in df.columns if c != label]
    evals = {}
    if valid is not None and ctx.world_rank == 0:
        vdf = valid.to_pandas()
        evals["valid"] = (vdf.drop(columns=[label]).to_numpy(dtype=float),
                          vdf[label].to_numpy(dtype=float))

    model = GBTModel(config.get("params", {}))
    model.feature_names = feature_names

    world = ctx.world_size

    def allreduce(arr):
        if world <= 1 or not dist.is_initialized():
            return arr
        t = torch.from_numpy(np.ascontiguousarray(arr))
        dist.all_reduce(t)
        return t.numpy()

    # bin edges must agree across shards: rank 0 computes, broadcast
    if world > 1 and dist.is_initialized():
        import cloudpickle

        if ctx.world_rank == 0:
            model.make_bins(X)
            blob = cloudpickle.dumps(model.bin_edges)
            size = torch.tensor([len(blob)])
        else:
            size = torch.tensor([0])
        dist.broadcast(size, src=0)
        buf = torch.zeros(int(size), dtype=torch.uint8)
        if ctx.world_rank == 0:
            buf.copy_(torch.frombuffer(bytearray(blob), dtype=torch.uint8))
        dist.broadcast(buf, src=0)
        model.bin_edges = cloudpickle.loads(bytes(buf.numpy().tobytes()))

    out_dir = config.get("output_dir", "/tmp/mi355_gbt")

    def cb(row):
        ckpt = None
        if ctx.world_rank == 0:
            ckpt = os.path.join(out_dir, f"round_{row['training_iteration']}")
            model.save_pretrained(ckpt)
        return session.report(row, checkpoint=ckpt)

    model.fit(X, y, num_boost_round=config.get("num_boost_round", 10),
              evals=evals, allreduce=allreduce, callback=cb)


class XGBoostTrainer(DataParallelTrainer):
    def __init__(self, *, label_column: str, params: dict | None = None,
                 num_boost_round: int = 10, scaling_config: ScalingConfig | None = None,
                 datasets=None, run_config: RunConfig | None = None,
                 preprocessor=None, **extra):
        self.label_column = label_column
        self.params = params or {}
        self.num_boost_round = num_boost_round
        cfg = {"label_column": label_column, "params": self.params,
               "num_boost_round": num_boost_round,
               "output_dir": (run_config or RunConfig()).run_dir() + "/worker_ckpts"}
        super().__init__(
            train_loop_per_worker=_gbt_loop, train_loop_config=cfg,
            scaling_config=scaling_config, datasets=datasets,
            run_config=run_config, preprocessor=preprocessor, **extra)

    def _with_config(self, config) -> "XGBoostTrainer":
        params = {**self.params, **config.get("params", config)}
        return XGBoostTrainer(
            label_column=self.label_column, params=params,
            num_boost_round=config.get("num_boost_round", self.num_boost_round),
            scaling_config=self.scaling_config, datasets=self.datasets,
            run_config=self.run_config, preprocessor=self.preprocessor)
