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


class ActorDiedError(RuntimeError):
    """The actor process exited (crash/OOM/os._exit) with calls in flight."""


def _actor_main(gpu_id, cls_bytes, init_bytes, task_q, done_q):
    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
    store = ObjectStore()
    try:
        cls = cloudpickle.loads(cls_bytes)
        # init args cloudpickled as one blob: plain pickle (mp spawn) cannot
        # carry local classes / closures that actor constructors often take
        init_args, init_kwargs = cloudpickle.loads(init_bytes)
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
    def __init__(self, cls, init_args, init_kwargs, num_gpus=0, max_restarts=0):
        rt = _rt()
        self._gpu_id = None
        if num_gpus and rt.num_gpus:
            self._gpu_id = rt.gpu_ids[next(_gpu_rr) % rt.num_gpus]
        self._cls_bytes = cloudpickle.dumps(cls)
        self._init_bytes = cloudpickle.dumps((list(init_args), dict(init_kwargs)))
        self._restarts_left = max_restarts
        self._pending: set[str] = set()
        self._spawn()
        rt.actors.append(self)
        self._cls_name = getattr(cls, "__name__", "Actor")

    def _spawn(self):
        ctx = mp.get_context("spawn")
        self._task_q = ctx.Queue()
        self._done_q = ctx.Queue()
        self._proc = ctx.Process(
            target=_actor_main,
            args=(self._gpu_id, self._cls_bytes, self._init_bytes,
                  self._task_q, self._done_q),
            daemon=True,
        )
        self._proc.start()
        status, info = self._done_q.get()
        if status == "init_error":
            raise RuntimeError(f"actor constructor failed:\n{info}")

    def _drain_done(self):
        import queue as _q

        while True:
            try:
                kind, ref_id = self._done_q.get_nowait()
            except (_q.Empty, OSError, EOFError):
                return
            if kind == "done":
                self._pending.discard(ref_id)

    def _check_alive(self):
        """Detect a crashed actor process; fail in-flight calls and restart it
        with fresh state (reference semantics: `max_restarts` re-runs the
        constructor, it does not replay lost calls)."""
        self._drain_done()
        if self._proc.is_alive():
            return
        store = _rt().store
        err = ActorDiedError(
            f"actor {getattr(self, '_cls_name', 'Actor')} died "
            f"(exitcode {self._proc.exitcode})")
        for ref_id in self._pending:
            try:
                store.put(_RemoteError(err, str(err)), ObjectRef(ref_id))
            except Exception:
                pass  # result landed before the crash: keep it
        self._pending.clear()
        if self._restarts_left > 0:
            self._restarts_left -= 1
            from .. import metrics as _prom

            _prom.ACTOR_RESTARTS.labels(actor=self._cls_name).inc()
            self._spawn()
        else:
            raise err

    def _call(self, method, args, kwargs) -> ObjectRef:
        self._check_alive()
        ref = ObjectRef(uuid.uuid4().hex)
        self._pending.add(ref.id)
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
    def __init__(self, cls, num_gpus=0, num_cpus=1, max_restarts=0, **_):
        self._cls = cls
        self._num_gpus = num_gpus
        self._max_restarts = max_restarts

    def remote(self, *args, **kwargs) -> ActorHandle:
        return ActorHandle(self._cls, args, kwargs, num_gpus=self._num_gpus,
                           max_restarts=self._max_restarts)

    def options(self, **opts):
        return ActorClass(self._cls, num_gpus=opts.get("num_gpus", self._num_gpus),
                          max_restarts=opts.get("max_restarts", self._max_restarts))

    def __call__(self, *a, **k):
        raise TypeError("actor classes must be instantiated with .remote()")
