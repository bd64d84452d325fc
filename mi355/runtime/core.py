"""MI355X-native single-node task/actor runtime.

Replaces the Ray core the reference exercises (`ray.init`, `@ray.remote`,
`ray.put/get/wait`, actors — reference call sites Overview_of_Ray.ipynb:
cell-31/34/41/44, Scaling_batch_inference.ipynb:cell-88/91/105/115) with a
from-scratch design sized for ONE 8-GPU MI355X node:

 * worker pool of spawned processes, one task queue per worker;
 * object store = POSIX shared memory segments (`/dev/shm/mi355_<ref>`),
   written once then read-only (sealed-object discipline = race freedom,
   SURVEY.md §5.2); payloads serialized with cloudpickle protocol 5 so
   numpy/arrow buffers go out-of-band into the segment;
 * ObjectRef resolution is decentralized: any process attaches the segment
   by its deterministic name, so workers can `get()` without a round-trip
   through the driver;
 * GPU workers: worker i is pinned to GPU (i mod n_gpus) via
   HIP_VISIBLE_DEVICES before torch initializes.
"""
from __future__ import annotations

import atexit
import io
import os
import pickle
import queue
import struct
import tempfile
import threading
import time
import uuid
import multiprocessing as mp
from dataclasses import dataclass, field
from multiprocessing import shared_memory

import cloudpickle

def _session_prefix() -> str:
    sid = os.environ.setdefault("MI355_SESSION", uuid.uuid4().hex[:8])
    return f"mi355_{sid}"


def _untrack(name: str):
    # Python 3.10's resource_tracker unlinks attached segments when ANY
    # process exits; we manage lifetime ourselves (driver sweep at shutdown).
    try:
        from multiprocessing import resource_tracker

        resource_tracker.unregister("/" + name, "shared_memory")
    except Exception:
        pass


# --------------------------------------------------------------------------- #
# Object store                                                                #
# --------------------------------------------------------------------------- #
@dataclass(frozen=True)
class ObjectRef:
    id: str

    def hex(self):
        return self.id

    def __repr__(self):
        return f"ObjectRef({self.id[:12]})"


def _shm_name(ref: ObjectRef) -> str:
    return f"{_session_prefix()}_{ref.id}"


def _serialize(obj) -> bytes:
    buffers = []
    payload = cloudpickle.dumps(obj, protocol=5, buffer_callback=buffers.append)
    out = io.BytesIO()
    out.write(struct.pack("<I", len(buffers)))
    out.write(struct.pack("<I", len(payload)))
    out.write(payload)
    for b in buffers:
        raw = b.raw()
        out.write(struct.pack("<Q", raw.nbytes))
        out.write(raw)
    return out.getvalue()


def _deserialize(buf: memoryview):
    off = 0
    (nbuf,) = struct.unpack_from("<I", buf, off); off += 4
    (plen,) = struct.unpack_from("<I", buf, off); off += 4
    payload = bytes(buf[off:off + plen]); off += plen
    buffers = []
    for _ in range(nbuf):
        (blen,) = struct.unpack_from("<Q", buf, off); off += 8
        buffers.append(buf[off:off + blen]); off += blen
    return pickle.loads(payload, buffers=buffers)


try:  # native C++ store core (csrc/store/shmstore.cpp) — release/acquire
    from mi355 import _store as _native_store  # type: ignore
except Exception:  # pragma: no cover - fallback when not built
    _native_store = None


class ObjectStore:
    """Sealed shared-memory objects addressable by ref id from any process.

    Backed by the native C++ core (mi355/_store.so: shm_open + mmap +
    release/acquire seal word) when built; multiprocessing.shared_memory
    otherwise. Attached segments are cached for the store's lifetime:
    deserialized numpy/arrow buffers alias the mapped memory (zero-copy)."""

    def __init__(self):
        self._owned: list[str] = []
        self._attached: dict[str, shared_memory.SharedMemory] = {}
        self._native = _native_store
        # spill-to-disk (SURVEY §2.2 C4 / r1 ROADMAP): objects larger than
        # MI355_SPILL_THRESHOLD_MB, or any put that fails with a full
        # /dev/shm, land as regular files under MI355_SPILL_DIR; readers
        # check the spill path after a shm miss.  Write-then-rename keeps
        # the sealed-then-immutable discipline (§5.2) on disk too.
        self._spill_dir = os.environ.get(
            "MI355_SPILL_DIR", os.path.join(tempfile.gettempdir(),
                                            f"{_session_prefix()}_spill"))
        thr = os.environ.get("MI355_SPILL_THRESHOLD_MB")
        self._spill_threshold = int(float(thr) * 1024 * 1024) if thr else None
        self._spilled: list[str] = []
        # LRU eviction of SEALED owned segments (r2 ROADMAP item delivered
        # early): when the owned-shm footprint exceeds MI355_SHM_BUDGET_MB,
        # or a put hits a full /dev/shm, the least-recently-read owned
        # segments are copied to the spill dir and unlinked.  Readers in
        # other processes keep working: live mmaps survive the unlink
        # (POSIX) and NEW readers fall through to the spill file, which
        # get()/ready() already consult after a shm miss.
        bud = os.environ.get("MI355_SHM_BUDGET_MB")
        self._shm_budget = int(float(bud) * 1024 * 1024) if bud else None
        from collections import OrderedDict
        self._owned_sizes: "OrderedDict[str, int]" = OrderedDict()
        self._shm_bytes = 0

    # -- disk spill ------------------------------------------------------ #
    def _spill_path(self, name: str) -> str:
        return os.path.join(self._spill_dir, name)

    def _spill_put(self, data: bytes, name: str):
        os.makedirs(self._spill_dir, exist_ok=True)
        tmp = self._spill_path(name) + ".tmp"
        with open(tmp, "wb") as f:
            f.write(data)
        os.rename(tmp, self._spill_path(name))  # atomic seal
        self._spilled.append(name)

    def _spill_get(self, name: str):
        # mmap readback: deserialized numpy/arrow buffers alias the mapping
        # (the returned memoryview keeps the mmap alive), so a spilled-block
        # get never copies the file through driver heap
        import mmap as _mmap

        try:
            with open(self._spill_path(name), "rb") as f:
                mm = _mmap.mmap(f.fileno(), 0, access=_mmap.ACCESS_READ)
            return memoryview(mm)
        except (FileNotFoundError, ValueError):
            return None

    def _evict_lru(self) -> bool:
        """Move the least-recently-read owned sealed segment to disk and
        unlink its shm. Returns False when nothing evictable remains."""
        while self._owned_sizes:
            name, sz = self._owned_sizes.popitem(last=False)
            self._shm_bytes -= sz
            try:
                buf = self._native.attach(name)
            except Exception:
                buf = None
            if buf is None:
                continue  # unsealed or already gone — not evictable
            try:
                self._spill_put(bytes(buf), name)
            except OSError:
                return False  # disk full too: stop evicting
            try:
                self._native.unlink(name)
                self._native.close(name)
            except Exception:
                pass
            if name in self._owned:
                self._owned.remove(name)
            return True
        return False

    # -- native-core fast paths ----------------------------------------- #
    def _nput(self, data: bytes, name: str):
        mv = self._native.create(name, len(data))
        mv[:len(data)] = data
        self._native.seal(name, len(data))
        self._owned.append(name)
        self._owned_sizes[name] = len(data)
        self._shm_bytes += len(data)
        if self._shm_budget is not None:
            while self._shm_bytes > self._shm_budget and len(self._owned_sizes) > 1:
                if not self._evict_lru():
                    break

    def _nget(self, name):
        return self._native.attach(name)  # None until sealed

    def put(self, obj, ref: ObjectRef | None = None) -> ObjectRef:
        ref = ref or ObjectRef(uuid.uuid4().hex)
        data = _serialize(obj)
        name = _shm_name(ref)
        if self._spill_threshold is not None and len(data) > self._spill_threshold:
            self._spill_put(data, name)
            return ref
        if self._native is not None:
            try:
                self._nput(data, name)
            except (OSError, MemoryError, RuntimeError):
                # /dev/shm full: evict LRU owned segments to disk and retry;
                # only if nothing is evictable does the NEW object spill
                stored = False
                while self._evict_lru():
                    try:
                        self._nput(data, name)
                        stored = True
                        break
                    except (OSError, MemoryError, RuntimeError):
                        continue
                if not stored:
                    self._spill_put(data, name)
            return ref
        seg = shared_memory.SharedMemory(name=name, create=True, size=len(data) + 8)
        _untrack(name)
        # header: 8-byte payload length; the final length write "seals" it
        seg.buf[8:8 + len(data)] = data
        struct.pack_into("<Q", seg.buf, 0, len(data))
        self._owned.append(name)
        self._attached[name] = seg
        return ref

    def _attach(self, name):
        seg = self._attached.get(name)
        if seg is None:
            seg = shared_memory.SharedMemory(name=name)
            _untrack(name)
            self._attached[name] = seg
        return seg

    def get(self, ref: ObjectRef, timeout: float | None = None):
        name = _shm_name(ref)
        if name in self._owned_sizes:  # LRU recency bump for owned objects
            self._owned_sizes.move_to_end(name)
        deadline = None if timeout is None else time.monotonic() + timeout
        delay = 0.0005
        while True:
            if self._native is not None:
                buf = self._nget(name)
                if buf is not None:
                    return _deserialize(memoryview(buf))
                data = self._spill_get(name)
                if data is not None:
                    return _deserialize(data)
            else:
                try:
                    seg = self._attach(name)
                    (n,) = struct.unpack_from("<Q", seg.buf, 0)
                    if n > 0:
                        return _deserialize(memoryview(seg.buf)[8:8 + n])
                except FileNotFoundError:
                    data = self._spill_get(name)
                    if data is not None:
                        return _deserialize(data)
            if deadline is not None and time.monotonic() > deadline:
                raise TimeoutError(f"object {ref} not available after {timeout}s")
            time.sleep(delay)
            delay = min(delay * 1.5, 0.02)

    def ready(self, ref: ObjectRef) -> bool:
        name = _shm_name(ref)
        if self._native is not None:
            return self._native.ready(name) or \
                os.path.exists(self._spill_path(name))
        try:
            seg = self._attach(name)
            (n,) = struct.unpack_from("<Q", seg.buf, 0)
            return n > 0
        except FileNotFoundError:
            return os.path.exists(self._spill_path(name))

    def cleanup(self):
        import glob

        if self._native is not None:
            for name in self._owned:
                try:
                    self._native.unlink(name)
                    self._native.close(name)
                except Exception:
                    pass
            self._owned.clear()
        for name, seg in list(self._attached.items()):
            try:
                seg.close()
            except BufferError:
                pass  # a live array still aliases it; unlink below suffices
            self._attached.pop(name, None)
        for name in self._owned:
            try:
                shared_memory.SharedMemory(name=name).unlink()
            except FileNotFoundError:
                pass
            except Exception:
                pass
        self._owned.clear()

    def sweep_session(self):
        """Driver-side: unlink every segment of this session (incl. worker-
        created results) at shutdown."""
        import glob

        for path in glob.glob(os.path.join(self._spill_dir, "*")):
            try:
                os.unlink(path)
            except OSError:
                pass
        for path in glob.glob(f"/dev/shm/{_session_prefix()}_*"):
            try:
                os.unlink(path)
            except OSError:
                pass


# --------------------------------------------------------------------------- #
# Worker process                                                              #
# --------------------------------------------------------------------------- #
def _resolve_args(store, args, kwargs):
    def r(x):
        return store.get(x) if isinstance(x, ObjectRef) else x

    return [r(a) for a in args], {k: r(v) for k, v in kwargs.items()}


def _worker_main(wid, gpu_id, task_q, done_q):
    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
    store = ObjectStore()
    while True:
        msg = task_q.get()
        if msg is None:
            break
        kind = msg[0]
        if kind == "task":
            _, ref_id, fn_bytes, args, kwargs = msg
            ref = ObjectRef(ref_id)
            t0 = time.monotonic()
            name = "task"
            try:
                fn = cloudpickle.loads(fn_bytes)
                name = getattr(fn, "__name__", "task")
                a, kw = _resolve_args(store, args, kwargs)
                result = fn(*a, **kw)
                store.put(result, ref)
            except Exception as e:  # store the exception; raised at get()
                import traceback

                store.put(_RemoteError(e, traceback.format_exc()), ref)
            done_q.put((wid, ref_id, name, t0, time.monotonic()))
    store.cleanup()


@dataclass
class _RemoteError:
    exc: Exception
    tb: str


class WorkerCrashedError(RuntimeError):
    """A stateless-task worker process died while running a task."""


# --------------------------------------------------------------------------- #
# Runtime (driver side)                                                       #
# --------------------------------------------------------------------------- #
class Runtime:
    def __init__(self, num_workers: int | None = None, num_gpus: int | None = None):
        import torch

        _session_prefix()  # pin the session id BEFORE spawn so workers inherit it

        vis = os.environ.get("MI355_VISIBLE_DEVICES")
        self.gpu_ids = ([int(x) for x in vis.split(",") if x != ""]
                        if vis is not None
                        else list(range(torch.cuda.device_count())))
        if num_gpus is not None:
            self.gpu_ids = self.gpu_ids[:num_gpus]
        self.num_gpus = len(self.gpu_ids)
        if num_workers is None:
            num_workers = max(2, min(os.cpu_count() or 4, 16))
            if self.num_gpus:
                num_workers = max(num_workers, self.num_gpus)
        ctx = mp.get_context("spawn")
        self.store = ObjectStore()
        self.workers = []
        self.done_q = ctx.Queue()
        self._idle: "queue.Queue[int]" = queue.Queue()
        self._inflight: dict[str, int] = {}
        self._lock = threading.Lock()
        for wid in range(num_workers):
            gpu_id = self.gpu_ids[wid % self.num_gpus] if self.num_gpus else None
            tq = ctx.Queue()
            p = ctx.Process(target=_worker_main, args=(wid, gpu_id, tq, self.done_q),
                            daemon=True)
            p.start()
            self.workers.append((p, tq, gpu_id))
            self._idle.put(wid)
        self._pending: "queue.Queue[tuple]" = queue.Queue()
        self._events: list[tuple] = []  # (wid, task_name, t_start, t_end)
        self._alive = True
        self._reaper = threading.Thread(target=self._reap, daemon=True)
        self._reaper.start()
        self._dispatcher = threading.Thread(target=self._dispatch, daemon=True)
        self._dispatcher.start()
        self.actors: list = []
        atexit.register(self.shutdown)

    def _reap(self):
        last_liveness = time.monotonic()
        while self._alive:
            try:
                wid, ref_id, name, t0, t1 = self.done_q.get(timeout=0.2)
            except queue.Empty:
                if time.monotonic() - last_liveness > 1.0:
                    self._check_workers()
                    last_liveness = time.monotonic()
                continue
            except (EOFError, OSError):
                return
            with self._lock:
                self._inflight.pop(ref_id, None)
                self._events.append((wid, name, t0, t1))
            self._idle.put(wid)

    def _check_workers(self):
        """Failure detection: a worker that died mid-task
        (crash / OOM-kill / os._exit) fails its in-flight ref with a
        WorkerCrashedError and is respawned with fresh queues, so the pool
        keeps its capacity and a blocked get() raises instead of hanging."""
        ctx = mp.get_context("spawn")
        for wid, (p, tq, gpu_id) in enumerate(list(self.workers)):
            if p.is_alive():
                continue
            dead_refs = []
            with self._lock:
                for ref_id, w in list(self._inflight.items()):
                    if w == wid:
                        dead_refs.append(ref_id)
                        self._inflight.pop(ref_id)
            for ref_id in dead_refs:
                err = WorkerCrashedError(
                    f"worker {wid} died (exitcode {p.exitcode}) while running "
                    f"task {ref_id}")
                try:
                    self.store.put(_RemoteError(err, str(err)), ObjectRef(ref_id))
                except Exception:
                    pass  # worker finished the write before dying: keep it
            ntq = ctx.Queue()
            np_ = ctx.Process(target=_worker_main,
                              args=(wid, gpu_id, ntq, self.done_q), daemon=True)
            np_.start()
            self.workers[wid] = (np_, ntq, gpu_id)
            # if the worker was idle when it died its token is still in
            # _idle; the extra token only lets the fresh worker be handed two
            # tasks back-to-back once (they queue; accounting self-corrects)
            self._idle.put(wid)

    # -- tasks ---------------------------------------------------------- #
    # Shared-backlog pull scheduling: submit never blocks (Ray semantics) —
    # tasks land in one queue and the dispatcher hands each to the next idle
    # worker (idle workers "steal" from the common backlog, so an unlucky
    # long task never strands work behind it).
    def submit(self, fn, args, kwargs, num_gpus: float = 0) -> ObjectRef:
        ref = ObjectRef(uuid.uuid4().hex)
        fn_bytes = fn if isinstance(fn, bytes) else cloudpickle.dumps(fn)
        with self._lock:
            self._inflight[ref.id] = -1  # queued, not yet on a worker
        self._pending.put((ref.id, fn_bytes, list(args), dict(kwargs)))
        return ref

    def _dispatch(self):
        while self._alive:
            try:
                ref_id, fn_bytes, args, kwargs = self._pending.get(timeout=0.2)
            except queue.Empty:
                continue
            wid = self._idle.get()  # next worker to go idle takes the task
            with self._lock:
                if ref_id in self._inflight:
                    self._inflight[ref_id] = wid
            self.workers[wid][1].put(("task", ref_id, fn_bytes, args, kwargs))

    def timeline(self, filename: str | None = None):
        """Chrome-trace JSON of completed task events per worker (SURVEY.md
        §5.1; reference surface: ray.timeline()). Load in chrome://tracing
        or Perfetto."""
        with self._lock:
            events = list(self._events)
        trace = [{"name": name, "ph": "X", "pid": 0, "tid": wid,
                  "ts": int(t0 * 1e6), "dur": int((t1 - t0) * 1e6),
                  "cat": "task"} for wid, name, t0, t1 in events]
        if filename:
            import json

            with open(filename, "w") as f:
                json.dump(trace, f)
        return trace

    # -- store ---------------------------------------------------------- #
    def put(self, obj) -> ObjectRef:
        return self.store.put(obj)

    def get(self, refs, timeout: float | None = None):
        single = isinstance(refs, ObjectRef)
        lst = [refs] if single else list(refs)
        out = []
        for r in lst:
            v = self.store.get(r, timeout=timeout)
            if isinstance(v, _RemoteError):
                from .actor import ActorDiedError

                if isinstance(v.exc, (ActorDiedError, WorkerCrashedError)):
                    raise v.exc
                raise RuntimeError(f"remote task failed:\n{v.tb}") from v.exc
            out.append(v)
        return out[0] if single else out

    def wait(self, refs, num_returns=1, timeout=None):
        refs = list(refs)
        deadline = None if timeout is None else time.monotonic() + timeout
        ready: list[ObjectRef] = []
        while True:
            ready = [r for r in refs if self.store.ready(r)]
            if len(ready) >= num_returns:
                break
            if deadline is not None and time.monotonic() > deadline:
                break
            time.sleep(0.001)
        # ray.wait semantics: at most num_returns in ready; the rest (ready
        # or not) stay in not_ready so `done, pending = wait(pending)` loops
        # never drop refs (the old max() cap was a no-op).
        ready = ready[:num_returns]
        ready_ids = {r.id for r in ready}
        not_ready = [r for r in refs if r.id not in ready_ids]
        return ready, not_ready

    def shutdown(self):
        if not self._alive:
            return
        self._alive = False
        for a in list(self.actors):
            try:
                a._kill()
            except Exception:
                pass
        for p, tq, _ in self.workers:
            try:
                tq.put(None)
            except Exception:
                pass
        for p, _, _ in self.workers:
            p.join(timeout=2)
            if p.is_alive():
                p.terminate()
        self.store.cleanup()
        self.store.sweep_session()


# --------------------------------------------------------------------------- #
# Module-level API (ray-shaped)                                               #
# --------------------------------------------------------------------------- #
_runtime: Runtime | None = None


def init(num_workers: int | None = None, num_gpus: int | None = None,
         ignore_reinit_error: bool = True, **_ignored) -> Runtime:
    global _runtime
    if _runtime is not None:
        if ignore_reinit_error:
            return _runtime
        raise RuntimeError("mi355 runtime already initialized")
    _runtime = Runtime(num_workers=num_workers, num_gpus=num_gpus)
    dash = ""
    if os.environ.get("MI355_DASHBOARD", "1") != "0":
        try:
            from ..dashboard import start_dashboard

            host, port = start_dashboard()
            dash = f", dashboard at http://{host}:{port}"
        except Exception:
            pass
    # the reference's env smoke test is "ray.init() prints a started line"
    # (Install_locally.md) — mirror that signal, incl. the dashboard URL
    # (Install_locally.md:64-67)
    print(f"mi355 runtime started: {len(_runtime.workers)} workers, "
          f"{_runtime.num_gpus} GPU(s), session {_session_prefix()}{dash}")
    return _runtime


def is_initialized() -> bool:
    return _runtime is not None


def shutdown():
    global _runtime
    if _runtime is not None:
        _runtime.shutdown()
        _runtime = None


def _rt() -> Runtime:
    if _runtime is None:
        init()
    return _runtime


def timeline(filename: str | None = None):
    """Chrome-trace JSON of completed task events (ray.timeline parity)."""
    return _rt().timeline(filename)


def put(obj) -> ObjectRef:
    return _rt().put(obj)


def get(refs, timeout: float | None = None):
    return _rt().get(refs, timeout=timeout)


def wait(refs, num_returns: int = 1, timeout: float | None = None):
    return _rt().wait(refs, num_returns=num_returns, timeout=timeout)


# -- @remote ---------------------------------------------------------------- #
class RemoteFunction:
    def __init__(self, fn, num_gpus=0, num_cpus=1, **_):
        self._fn = fn
        self._fn_bytes = None
        self._num_gpus = num_gpus

    def remote(self, *args, **kwargs) -> ObjectRef:
        if self._fn_bytes is None:
            self._fn_bytes = cloudpickle.dumps(self._fn)
        return _rt().submit(self._fn_bytes, args, kwargs, num_gpus=self._num_gpus)

    def options(self, **opts):
        return RemoteFunction(self._fn, num_gpus=opts.get("num_gpus", self._num_gpus))

    def __call__(self, *a, **k):
        raise TypeError("remote functions must be called with .remote()")


def remote(*args, **kwargs):
    """@remote decorator for functions and classes."""
    from .actor import ActorClass

    if len(args) == 1 and callable(args[0]) and not kwargs:
        target = args[0]
        if isinstance(target, type):
            return ActorClass(target)
        return RemoteFunction(target)

    def deco(target):
        if isinstance(target, type):
            return ActorClass(target, **kwargs)
        return RemoteFunction(target, **kwargs)

    return deco
