"""HuggingFaceTrainer (reference call sites
Model_finetuning_and_batch_inference.ipynb:cell-40,
flan-t5-batch-inference.py:96-111): a DataParallelTrainer whose per-worker
loop builds the user's Trainer via ``trainer_init_per_worker(train_shard,
eval_shard, **trainer_init_config)`` and runs it, reporting per-epoch
metrics/checkpoints to the driver (metric flow M6)."""
from __future__ import annotations

from ..air import RunConfig, ScalingConfig
from .trainer import DataParallelTrainer


def _make_hf_loop(init_bytes, init_config):
    import cloudpickle

    def loop():
        import cloudpickle as cp

        from . import session

        init_fn = cp.loads(init_bytes)
        train_shard = session.get_dataset_shard("train")
        eval_shard = (session.get_dataset_shard("evaluation")
                      or session.get_dataset_shard("eval"))
        trainer = init_fn(train_shard, eval_shard, **(init_config or {}))
        resume = session.get_checkpoint()
        trainer.train(resume_from_checkpoint=resume.path if resume else None)

    return loop


class HuggingFaceTrainer(DataParallelTrainer):
    def __init__(self, trainer_init_per_worker=None, *, trainer_init_config=None,
                 scaling_config: ScalingConfig | None = None, datasets=None,
                 run_config: RunConfig | None = None, preprocessor=None,
                 resume_from_checkpoint=None, **extra):
        import cloudpickle

        self.trainer_init_per_worker = trainer_init_per_worker
        self.trainer_init_config = trainer_init_config or {}
        init_bytes = cloudpickle.dumps(trainer_init_per_worker)
        super().__init__(
            train_loop_per_worker=_make_hf_loop(init_bytes, self.trainer_init_config),
            train_loop_config=None,
            scaling_config=scaling_config,
            datasets=datasets,
            run_config=run_config,
            preprocessor=preprocessor,
            resume_from_checkpoint=resume_from_checkpoint,
            **extra,
        )

    def _with_config(self, trainer_init_config) -> "HuggingFaceTrainer":
        """Tuner hook: clone with a trial's hyperparameters."""
        return HuggingFaceTrainer(
            trainer_init_per_worker=self.trainer_init_per_worker,
            trainer_init_config={**self.trainer_init_config, **trainer_init_config},
            scaling_config=self.scaling_config,
            datasets=self.datasets,
            run_config=self.run_config,
            preprocessor=self.preprocessor,
        )
