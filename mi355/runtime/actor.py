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
    def __init__(self, cls, init_args, init_kwargs, num_gpus=0, max_restarts=0,
                 replay_on_restart=False, max_replay_log=1000):
        rt = _rt()
        self._gpu_id = None
        if num_gpus and rt.num_gpus:
            self._gpu_id = rt.gpu_ids[next(_gpu_rr) % rt.num_gpus]
        self._cls_bytes = cloudpickle.dumps(cls)
        self._init_bytes = cloudpickle.dumps((list(init_args), dict(init_kwargs)))
        self._restarts_left = max_restarts
        self._pending: set[str] = set()
        # opt-in call replay (r1 ROADMAP): record every completed method
        # call; on a crash-restart re-issue the log (in order, results
        # discarded) so state-mutating call sequences are reconstructed.
        # Off by default — the reference's max_restarts restarts with fresh
        # state — and bounded by max_replay_log (older entries dropped, so
        # replay is exact only while the log covers the actor's lifetime).
        self._replay = bool(replay_on_restart)
        self._max_replay_log = int(max_replay_log)
        self._call_log: list[tuple] = []
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
        if self._replay:
            # replay only calls that COMPLETED before the crash — the call
            # that killed the actor (and anything queued behind it) is
            # still in _pending and must not be re-issued
            self._call_log = [e for e in self._call_log
                              if e[0] not in self._pending]
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
            if self._replay and self._call_log:
                # re-issue the completed calls to rebuild state; replay
                # results go to throwaway refs (log entries keep their new
                # refs so a repeat crash prunes correctly)
                relog = []
                for _old_ref, method, args, kwargs in list(self._call_log):
                    rref = ObjectRef(uuid.uuid4().hex)
                    self._pending.add(rref.id)
                    relog.append((rref.id, method, args, kwargs))
                    self._task_q.put((rref.id, method, list(args),
                                      dict(kwargs)))
                self._call_log = relog
        else:
            raise err

    def _call(self, method, args, kwargs) -> ObjectRef:
        self._check_alive()
        ref = ObjectRef(uuid.uuid4().hex)
        self._pending.add(ref.id)
        if self._replay:
            self._call_log.append((ref.id, method, list(args), dict(kwargs)))
            if len(self._call_log) > self._max_replay_log:
                del self._call_log[0]
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
    def __init__(self, cls, num_gpus=0, num_cpus=1, max_restarts=0,
                 replay_on_restart=False, **_):
        self._cls = cls
        self._num_gpus = num_gpus
        self._max_restarts = max_restarts
        self._replay = replay_on_restart

    def remote(self, *args, **kwargs) -> ActorHandle:
        return ActorHandle(self._cls, args, kwargs, num_gpus=self._num_gpus,
                           max_restarts=self._max_restarts,
                           replay_on_restart=self._replay)

    def options(self, **opts):
        return ActorClass(self._cls, num_gpus=opts.get("num_gpus", self._num_gpus),
                          max_restarts=opts.get("max_restarts", self._max_restarts),
                          replay_on_restart=opts.get("replay_on_restart",
                                                     self._replay))

    def __call__(self, *a, **k):
        raise TypeError("actor classes must be instantiated with .remote()")
