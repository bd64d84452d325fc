from .core import (ObjectRef, init, is_initialized, shutdown, put, get, wait, timeline,
                   remote, Runtime)
from .actor import ActorHandle, ActorClass
from .actor_pool import ActorPool

__all__ = ["ObjectRef", "init", "is_initialized", "shutdown", "put", "get", "timeline",
           "wait", "remote", "Runtime", "ActorHandle", "ActorClass", "ActorPool"]
