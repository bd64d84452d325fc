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


def test_actor_replay_on_restart_rebuilds_state():
    """Opt-in call replay (replay_on_restart=True): after a crash-restart
    the recorded method calls are re-issued so mutable state is rebuilt;
    the default stays fresh-state (reference max_restarts semantics)."""
    import mi355

    mi355.init(num_workers=2)

    @mi355.remote
    class Counter:
        def __init__(self):
            self.n = 0

        def incr(self, by):
            self.n += by
            return self.n

        def get(self):
            return self.n

        def crash(self):
            import os

            os._exit(1)

    a = Counter.options(max_restarts=1, replay_on_restart=True).remote()
    assert mi355.get(a.incr.remote(5)) == 5
    assert mi355.get(a.incr.remote(7)) == 12
    a.crash.remote()
    import time

    time.sleep(0.5)
    # next call detects the death, restarts, replays incr(5), incr(7),
    # crash... crash is also in the log! -> replay must not re-execute it
    # blindly; the log records only COMPLETED queries before the crash is
    # detected. Accept >= the pre-crash value via the rebuild:
    val = mi355.get(a.get.remote())
    assert val == 12, f"replayed state should be 12, got {val}"
