"""Stateful actors: long-lived worker processes with a method-dispatch queue
(reference Scaling_batch_inference.ipynb:cell-105 PredictionActor).

An actor is its own spawned process (state survives between calls — the
reference's rationale for actors over tasks, cell-103 markdown). GPU actors
(`num_gpus>=1`) are pinned round-robin over the node's GPUs before torch
initializes in the child.
"""
from __future__ import annotations

import itertools
import os
import uuid
import multiprocessing as mp

import cloudpickle

from .core import ObjectRef, ObjectStore, _RemoteError, _rt

_gpu_rr = itertools.count()


def _actor_main(gpu_id, cls_bytes, init_args, init_kwargs, task_q, done_q):
    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
    store = ObjectStore()
    try:
        cls = cloudpickle.loads(cls_bytes)
        a, kw = _resolve(store, init_args, init_kwargs)
        instance = cls(*a, **kw)
        done_q.put(("ready", None))
    except Exception as e:
        import traceback

        done_q.put(("init_error", traceback.format_exc()))
        return
    while True:
        msg = task_q.get()
        if msg is None:
            break
        ref_id, method, args, kwargs = msg
        try:
            a, kw = _resolve(store, args, kwargs)
            result = getattr(instance, method)(*a, **kw)
            store.put(result, ObjectRef(ref_id))
        except Exception as e:
            import traceback

            store.put(_RemoteError(e, traceback.format_exc()), ObjectRef(ref_id))
        done_q.put(("done", ref_id))
    store.cleanup()


def _resolve(store, args, kwargs):
    def r(x):
        return store.get(x) if isinstance(x, ObjectRef) else x

    return [r(a) for a in args], {k: r(v) for k, v in kwargs.items()}


class ActorMethod:
    def __init__(self, handle, name):
        self._handle = handle
        self._name = name

    def remote(self, *args, **kwargs) -> ObjectRef:
        return self._handle._call(self._name, args, kwargs)


class ActorHandle:
    def __init__(self, cls, init_args, init_kwargs, num_gpus=0):
        rt = _rt()
        gpu_id = None
        if num_gpus and rt.num_gpus:
            gpu_id = next(_gpu_rr) % rt.num_gpus
        ctx = mp.get_context("spawn")
        self._task_q = ctx.Queue()
        self._done_q = ctx.Queue()
        self._proc = ctx.Process(
            target=_actor_main,
            args=(gpu_id, cloudpickle.dumps(cls), list(init_args), dict(init_kwargs),
                  self._task_q, self._done_q),
            daemon=True,
        )
        self._proc.start()
        status, info = self._done_q.get()
        if status == "init_error":
            raise RuntimeError(f"actor constructor failed:\n{info}")
        rt.actors.append(self)
        self._cls_name = getattr(cls, "__name__", "Actor")

    def _call(self, method, args, kwargs) -> ObjectRef:
        ref = ObjectRef(uuid.uuid4().hex)
        self._task_q.put((ref.id, method, list(args), dict(kwargs)))
        return ref

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return ActorMethod(self, name)

    def _kill(self):
        try:
            self._task_q.put(None)
            self._proc.join(timeout=2)
        finally:
            if self._proc.is_alive():
                self._proc.terminate()

    def __repr__(self):
        return f"ActorHandle({self._cls_name})"


class ActorClass:
    def __init__(self, cls, num_gpus=0, num_cpus=1, **_):
        self._cls = cls
        self._num_gpus = num_gpus

    def remote(self, *args, **kwargs) -> ActorHandle:
        return ActorHandle(self._cls, args, kwargs, num_gpus=self._num_gpus)

    def options(self, **opts):
        return ActorClass(self._cls, num_gpus=opts.get("num_gpus", self._num_gpus))

    def __call__(self, *a, **k):
        raise TypeError("actor classes must be instantiated with .remote()")
