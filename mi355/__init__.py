"""mi355 — an MI355X-native distributed ML stack with the capability surface
of the Ray-AIR workshop reference: task/actor runtime with a
shared-memory object store, Arrow-block datasets + preprocessors, a DDP
trainer on RCCL-over-xGMI, ASHA tuning, batch + online inference, and a
model zoo running hand-written HIP/CDNA4 kernels."""
from .runtime import (ObjectRef, init, is_initialized, shutdown, put, get,
                      wait, remote, ActorPool, timeline)
from .runtime.actor import ActorDiedError
from . import metrics

__version__ = "0.1.0"

from . import ops  # noqa: E402  (dispatch layer; harmless on CPU)

__all__ = ["init", "is_initialized", "shutdown", "put", "get", "wait",
           "remote", "ActorPool", "ObjectRef", "ActorDiedError", "ops",
           "metrics", "timeline", "__version__"]
