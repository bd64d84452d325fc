"""Failure detection / retry-from-checkpoint: a worker that
dies mid-training surfaces the error in Result; with FailureConfig the
trainer retries and resumes from the last good checkpoint."""
import os

import pytest

from mi355.air import CheckpointConfig, FailureConfig, RunConfig, ScalingConfig
from mi355.train import TorchTrainer, session


def crashy_loop(config):
    """Reports 4 epochs with a checkpoint each; crashes at epoch 2 unless a
    resume checkpoint exists (fault-injection hook via config)."""
    import json

    ckpt = session.get_checkpoint()
    start = 0
    if ckpt is not None:
        with open(os.path.join(ckpt.path, "state.json")) as f:
            start = json.load(f)["epoch"]
    for epoch in range(start, 4):
        if epoch == 2 and ckpt is None and config.get("inject_crash", True):
            os._exit(17)  # hard kill, as a GPU fault would
        d = os.path.join(config["dir"], f"w{epoch}")
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "state.json"), "w") as f:
            json.dump({"epoch": epoch + 1}, f)
        session.report({"epoch": epoch + 1, "loss": 1.0 / (epoch + 1)},
                       checkpoint=d)


@pytest.mark.timeout(600)
def test_worker_crash_surfaces_error(tmp_path):
    t = TorchTrainer(crashy_loop, train_loop_config={"dir": str(tmp_path / "w")},
                     scaling_config=ScalingConfig(num_workers=1),
                     run_config=RunConfig(name="crash", local_dir=str(tmp_path)))
    res = t.fit()
    assert res.error is not None
    assert res.checkpoint is not None  # epoch-1 checkpoint was kept


@pytest.mark.timeout(600)
def test_failure_config_retries_from_checkpoint(tmp_path):
    t = TorchTrainer(
        crashy_loop, train_loop_config={"dir": str(tmp_path / "w")},
        scaling_config=ScalingConfig(num_workers=1),
        run_config=RunConfig(
            name="retry", local_dir=str(tmp_path),
            checkpoint_config=CheckpointConfig(num_to_keep=2),
            failure_config=FailureConfig(max_failures=1)),
    )
    res = t.fit()
    assert res.error is None, res.error
    assert res.metrics["epoch"] == 4  # resumed past the crash to completion


def test_task_worker_crash_detected_and_respawned():
    """A stateless-task worker that dies mid-task fails its ref with
    WorkerCrashedError and is respawned, so later tasks still run."""
    import pytest

    import mi355
    from mi355.runtime.core import WorkerCrashedError

    mi355.init(num_workers=2)
    try:
        @mi355.remote
        def die_hard():
            import os

            os._exit(23)

        @mi355.remote
        def fine(x):
            return x * 2

        ref = die_hard.remote()
        with pytest.raises(WorkerCrashedError):
            mi355.get(ref, timeout=30)
        # pool recovered: full fan-out still works
        assert mi355.get([fine.remote(i) for i in range(6)]) == \
            [0, 2, 4, 6, 8, 10]
    finally:
        mi355.shutdown()
