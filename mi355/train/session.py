"""Worker-side training session API (SURVEY.md C8's report path, M6):
``session.report(metrics, checkpoint=...)``, ``session.get_dataset_shard``,
world context — the contract user train loops code against (reference:
Ray Train's session inside trainer_init_per_worker workers)."""
from __future__ import annotations

import queue as _queue
from dataclasses import dataclass, field


@dataclass
class TrainContext:
    world_rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    trial_name: str = ""
    experiment_name: str = ""

    def get_world_rank(self):
        return self.world_rank

    def get_world_size(self):
        return self.world_size

    def get_local_rank(self):
        return self.local_rank


@dataclass
class _Session:
    context: TrainContext
    report_q: object = None            # worker -> driver
    control_q: object = None           # driver -> worker ("stop")
    dataset_shards: dict = field(default_factory=dict)
    resume_checkpoint: object = None
    stopped: bool = False

    def report(self, metrics: dict, checkpoint_dir: str | None = None) -> bool:
        """Send metrics (+ checkpoint path, rank0) to the driver and BLOCK
        for its continue/stop decision (keeps DDP ranks in lockstep so an
        ASHA early stop lands on every rank at the same epoch). Returns
        False when the trial should stop."""
        if self.report_q is not None:
            self.report_q.put(("report", self.context.world_rank, dict(metrics),
                               checkpoint_dir))
            if self.control_q is not None:
                try:
                    msg = self.control_q.get(timeout=600)
                    if msg == "stop":
                        self.stopped = True
                except _queue.Empty:
                    self.stopped = True
        return not self.stopped


_session: _Session | None = None


def _set_session(s: _Session | None):
    global _session
    _session = s


def _get_session() -> _Session:
    global _session
    if _session is None:
        _session = _Session(context=TrainContext())
    return _session


# public API ---------------------------------------------------------------- #
def report(metrics: dict, checkpoint=None) -> bool:
    ckpt_dir = None
    if checkpoint is not None:
        ckpt_dir = checkpoint if isinstance(checkpoint, str) else checkpoint.path
    return _get_session().report(metrics, ckpt_dir)


def get_dataset_shard(name: str = "train"):
    return _get_session().dataset_shards.get(name)


def get_context() -> TrainContext:
    return _get_session().context


def get_world_rank() -> int:
    return _get_session().context.world_rank


def get_world_size() -> int:
    return _get_session().context.world_size

def get_local_rank() -> int:
    return _get_session().context.local_rank


def get_checkpoint():
    """Checkpoint to resume from (FailureConfig retry path)."""
    return _get_session().resume_checkpoint
