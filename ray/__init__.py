"""Drop-in `ray` namespace over the mi355 runtime (the reference
notebooks' call sites keep working verbatim — `ray.init()`, `ray.put/get`,
`@ray.remote`, `ray.data.from_huggingface`, `ray.train.huggingface.
HuggingFaceTrainer`, `ray.tune`, `ray.air`, `ray.serve`. This is an alias
layer only; the implementation is mi355's MI355X-native stack."""
from mi355 import (ActorPool, ObjectRef, get, init, is_initialized, put, timeline,
                   remote, shutdown, wait)
from mi355 import serve, tune  # noqa: F401
from . import air, data, exceptions, train, util  # noqa: F401

__version__ = "2.3.1+mi355"

__all__ = ["init", "shutdown", "is_initialized", "put", "get", "wait",
           "remote", "ActorPool", "ObjectRef", "air", "data", "train",
           "tune", "serve", "util"]
