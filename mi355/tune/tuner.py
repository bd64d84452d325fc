"""Tuner: hyperparameter search over trainers (call stack
§3.2). One trial = one full DataParallelTrainer.fit() tree; trials run
concurrently (default: one 1-GPU trial per GPU on the node — BASELINE.json
config 4: 8 concurrent 1-GPU trials), ASHA early-stops via the trainer's
report callback killing the trial's workers between epochs."""
from __future__ import annotations

import random
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass

from ..air import Result, ResultGrid, RunConfig
from .schedulers import ASHAScheduler, FIFOScheduler
from .search import apply_overrides, expand_grid, sample_space


@dataclass
class TuneConfig:
    metric: str | None = None
    mode: str = "min"
    num_samples: int = 1
    scheduler: object = None
    max_concurrent_trials: int | None = None
    search_alg: object = None
    seed: int | None = 1234


class Tuner:
    def __init__(self, trainable, *, param_space=None, tune_config: TuneConfig | None = None,
                 run_config: RunConfig | None = None):
        self.trainable = trainable
        self.param_space = param_space or {}
        self.tune_config = tune_config or TuneConfig()
        self.run_config = run_config or RunConfig()

    def _make_trial(self, config: dict, index: int):
        from ..train.trainer import DataParallelTrainer, TorchTrainer

        if isinstance(self.trainable, DataParallelTrainer):
            tic = config.get("trainer_init_config", config)
            if hasattr(self.trainable, "_with_config"):
                trainer = self.trainable._with_config(tic)
            else:
                trainer = self.trainable
                trainer._train_loop_config = {**(trainer._train_loop_config or {}),
                                              **config.get("train_loop_config", config)}
            import copy

            trainer.run_config = RunConfig(
                name=f"{self.run_config.name or 'tune'}/trial_{index:04d}",
                local_dir=self.run_config.local_dir,
                storage_path=self.run_config.storage_path,
                checkpoint_config=self.run_config.checkpoint_config,
                failure_config=self.run_config.failure_config,
            )
            trainer._trial_name = f"trial_{index:04d}"
            return trainer
        # plain function trainable(config)
        fn = self.trainable

        return TorchTrainer(
            fn, train_loop_config=config,
            run_config=RunConfig(
                name=f"{self.run_config.name or 'tune'}/trial_{index:04d}",
                local_dir=self.run_config.local_dir,
                checkpoint_config=self.run_config.checkpoint_config,
            ),
        )

    def fit(self) -> ResultGrid:
        tc = self.tune_config
        rng = random.Random(tc.seed)
        scheduler = tc.scheduler or FIFOScheduler()
        if isinstance(scheduler, ASHAScheduler):
            scheduler.metric = scheduler.metric or tc.metric
            scheduler.mode = scheduler.mode or tc.mode
            if scheduler.time_attr == "training_iteration":
                scheduler.time_attr = "epoch"

        # build trial configs: grid combos x num_samples random draws
        grid_combos = expand_grid(self.param_space)
        configs = []
        for _ in range(tc.num_samples):
            for overrides in grid_combos:
                cfg = sample_space(
                    {k: v for k, v in self.param_space.items()}, rng)
                cfg = apply_overrides(cfg, overrides)
                configs.append(cfg)

        import torch

        max_conc = tc.max_concurrent_trials or max(torch.cuda.device_count(), 1)

        def run_trial(i_cfg):
            i, cfg = i_cfg
            trial_id = f"trial_{i:04d}"
            trainer = self._make_trial(cfg, i)

            def cb(metrics):
                metrics = dict(metrics)
                metrics.setdefault("training_iteration", metrics.get("epoch", 0))
                return scheduler.on_result(trial_id, metrics)

            trainer._report_callback = cb
            try:
                res = trainer.fit()
            except Exception as e:  # trial isolation
                res = Result(error=e)
            res.config = cfg
            return res

        with ThreadPoolExecutor(max_workers=max_conc) as ex:
            results = list(ex.map(run_trial, enumerate(configs)))

        grid = ResultGrid(results)
        grid._default_metric = tc.metric
        grid._default_mode = tc.mode
        return grid
