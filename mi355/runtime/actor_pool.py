"""ActorPool with ordered/unordered mapping (reference
Scaling_batch_inference.ipynb:cell-123-129 `ActorPool.map_unordered`)."""
from __future__ import annotations

from . import core


class ActorPool:
    def __init__(self, actors):
        self._idle = list(actors)
        self._future_to_actor = {}
        self._pending = []

    def submit(self, fn, value):
        """fn(actor, value) -> ObjectRef (matches ray.util.ActorPool)."""
        if not self._idle:
            ready, _ = core.wait(list(self._future_to_actor), num_returns=1)
            for r in ready:
                self._idle.append(self._future_to_actor.pop(r))
                self._pending.append(r)
        actor = self._idle.pop(0)
        ref = fn(actor, value)
        self._future_to_actor[ref] = actor
        return ref

    def get_next_unordered(self, timeout=None):
        if self._pending:
            return core.get(self._pending.pop(0))
        if not self._future_to_actor:
            raise StopIteration
        ready, _ = core.wait(list(self._future_to_actor), num_returns=1,
                             timeout=timeout)
        if not ready:
            raise TimeoutError
        ref = ready[0]
        self._idle.append(self._future_to_actor.pop(ref))
        return core.get(ref)

    def has_next(self):
        return bool(self._future_to_actor) or bool(self._pending)

    def map_unordered(self, fn, values):
        values = list(values)
        for v in values:
            self.submit(fn, v)
        out = []
        while self.has_next():
            out.append(self.get_next_unordered())
        return out

    def map(self, fn, values):
        refs = []
        values = list(values)
        for v in values:
            # keep order: remember submit order
            if not self._idle:
                ready, _ = core.wait(list(self._future_to_actor), num_returns=1)
                for r in ready:
                    self._idle.append(self._future_to_actor.pop(r))
            actor = self._idle.pop(0)
            ref = fn(actor, v)
            self._future_to_actor[ref] = actor
            refs.append(ref)
        out = []
        for r in refs:
            out.append(core.get(r))
            a = self._future_to_actor.pop(r, None)
            if a is not None:
                self._idle.append(a)
        return out
