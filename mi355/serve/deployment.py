"""serve internals: deployments, replicas, HTTP proxy."""
from __future__ import annotations

import itertools
import json
import threading
import time
from dataclasses import dataclass, field, replace
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pandas as pd

from .. import metrics


# -- HTTP adapters (reference: ray.serve.http_adapters.pandas_read_json) ---- #
def pandas_read_json(body: bytes) -> pd.DataFrame:
    return pd.DataFrame(json.loads(body))


def json_request(body: bytes):
    return json.loads(body)


# -- deployments ------------------------------------------------------------ #
@dataclass
class Deployment:
    func_or_class: object = None
    name: str = "default"
    num_replicas: int = 1
    route_prefix: str = "/"
    ray_actor_options: dict = field(default_factory=dict)
    # ray-style autoscaling_config: {"min_replicas", "max_replicas",
    # "target_num_ongoing_requests_per_replica", "interval_s"} — a controller
    # thread resizes the replica pool toward
    # ceil(ongoing / target) clamped to [min, max]
    autoscaling_config: dict | None = None

    def options(self, name=None, num_replicas=None, route_prefix=None,
                ray_actor_options=None, autoscaling_config=None, **_):
        return replace(
            self,
            name=name or self.name,
            num_replicas=num_replicas or self.num_replicas,
            route_prefix=route_prefix or self.route_prefix,
            ray_actor_options=ray_actor_options or self.ray_actor_options,
            autoscaling_config=autoscaling_config or self.autoscaling_config,
        )

    def bind(self, *args, **kwargs) -> "Application":
        return Application(self, args, kwargs)


def deployment(cls_or_fn=None, **opts):
    if cls_or_fn is not None:
        return Deployment(func_or_class=cls_or_fn)

    def deco(target):
        return Deployment(func_or_class=target, **opts)

    return deco


@dataclass
class Application:
    deployment: Deployment
    args: tuple
    kwargs: dict


def _runtime_up() -> bool:
    import mi355

    return mi355.is_initialized()


class _ActorReplica:
    """Serve replica hosted in its own actor process (loads the model once)."""

    def __init__(self, predictor_cls, checkpoint, kwargs):
        self.predictor = predictor_cls.from_checkpoint(checkpoint, **kwargs)

    def predict(self, data, kw):
        return self.predictor.predict(data, **kw)


def _invoke_actor(rep, data, kw):
    import mi355

    return mi355.get(rep.predict.remote(data, kw))


class _PredictorReplicaGroup:
    """Round-robin pool of predictor replicas for a PredictorDeployment."""

    def __init__(self, app: Application):
        d = app.deployment
        self._dep = d
        self.http_adapter = app.kwargs.get("http_adapter", pandas_read_json)
        rest = {k: v for k, v in app.kwargs.items() if k != "http_adapter"}
        self._is_actor = False
        if app.args and hasattr(app.args[0], "from_checkpoint"):
            # PredictorDeployment.bind(PredictorCls, checkpoint, ...)
            predictor_cls, checkpoint = app.args[0], app.args[1]
            if d.ray_actor_options and _runtime_up():
                # replica = dedicated actor process (GPU-pinned round-robin
                # when num_gpus>0) — the reference's serve replica model
                from ..runtime.actor import ActorClass

                num_gpus = d.ray_actor_options.get("num_gpus", 0)
                cls = ActorClass(_ActorReplica, num_gpus=num_gpus)
                self._make_replica = lambda: cls.remote(predictor_cls,
                                                        checkpoint, rest)
                self._invoke = _invoke_actor
                self._is_actor = True
            else:
                self._make_replica = lambda: predictor_cls.from_checkpoint(
                    checkpoint, **rest)
                self._invoke = lambda rep, data, kw: rep.predict(data, **kw)
        elif isinstance(d.func_or_class, type):
            # generic @serve.deployment class: instance is the handler
            self._make_replica = lambda: d.func_or_class(*app.args, **rest)
            self._invoke = lambda rep, data, kw: rep(data, **kw)
        else:
            # @serve.deployment function
            self._make_replica = lambda: d.func_or_class
            self._invoke = lambda rep, data, kw: rep(data, **kw)
        self.replicas = [self._make_replica()
                         for _ in range(max(d.num_replicas, 1))]
        self._rr = itertools.count()
        self._locks = [threading.Lock() for _ in self.replicas]
        self._pool_lock = threading.Lock()
        self._ongoing = 0
        self._stop = threading.Event()
        if d.autoscaling_config:
            threading.Thread(target=self._autoscale_loop, daemon=True).start()

    # -- autoscaling ----------------------------------------------------- #
    def _autoscale_loop(self):
        cfg = self._dep.autoscaling_config
        lo = int(cfg.get("min_replicas", 1))
        hi = int(cfg.get("max_replicas", max(len(self.replicas), lo)))
        target = float(cfg.get("target_num_ongoing_requests_per_replica", 1))
        interval = float(cfg.get("interval_s", 2.0))
        import math

        while not self._stop.wait(interval):
            desired = max(lo, min(hi, math.ceil(self._ongoing / max(target, 1e-9))))
            with self._pool_lock:
                cur = len(self.replicas)
                if desired > cur:
                    for _ in range(desired - cur):
                        self.replicas.append(self._make_replica())
                        self._locks.append(threading.Lock())
                elif desired < cur:
                    dead = self.replicas[desired:]
                    del self.replicas[desired:]
                    del self._locks[desired:]
                    for rep in dead:
                        self._kill_replica(rep)

    def _kill_replica(self, rep):
        if self._is_actor:
            try:
                rep._kill()
            except Exception:
                pass

    def num_replicas(self) -> int:
        return len(self.replicas)

    def shutdown(self):
        """Drain in-flight requests (acquire every replica lock) and stop."""
        self._stop.set()
        with self._pool_lock:
            for lk, rep in zip(list(self._locks), list(self.replicas)):
                with lk:
                    self._kill_replica(rep)
            self.replicas = []
            self._locks = []

    def handle(self, body: bytes, query_kwargs: dict):
        data = self.http_adapter(body)
        self._ongoing += 1
        try:
            with self._pool_lock:
                n = len(self.replicas)
                if n == 0:
                    raise RuntimeError("deployment shut down")
                i = next(self._rr) % n
                lock, rep = self._locks[i], self.replicas[i]
            with lock:
                out = self._invoke(rep, data, query_kwargs)
        finally:
            self._ongoing -= 1
        if isinstance(out, pd.DataFrame):
            return out.to_dict(orient="records")
        import numpy as np

        if isinstance(out, (list, np.ndarray)):
            return [{"prediction": _py(v)} for v in out]
        return out


def _py(v):
    import numpy as np

    if isinstance(v, np.generic):
        return v.item()
    if isinstance(v, np.ndarray):
        return v.tolist()
    return v


PredictorDeployment = Deployment(func_or_class=_PredictorReplicaGroup,
                                 name="PredictorDeployment")


# -- proxy / controller ------------------------------------------------------ #
class _ServeState:
    def __init__(self):
        self.routes: dict[str, _PredictorReplicaGroup] = {}
        self.apps: dict[str, Application] = {}
        self.server: ThreadingHTTPServer | None = None
        self.thread: threading.Thread | None = None


_state = _ServeState()


class _Handler(BaseHTTPRequestHandler):
    def log_message(self, *a):  # quiet
        pass

    def do_POST(self):
        length = int(self.headers.get("Content-Length", 0))
        body = self.rfile.read(length)
        route = self.path.split("?")[0].rstrip("/") or "/"
        group = _state.routes.get(route)
        if group is None:
            self.send_response(404)
            self.end_headers()
            self.wfile.write(b'{"error": "no such route"}')
            return
        t0 = time.perf_counter()
        try:
            result = group.handle(body, {})
            payload = json.dumps(result).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.end_headers()
            self.wfile.write(payload)
        except Exception as e:
            self.send_response(500)
            self.end_headers()
            self.wfile.write(json.dumps({"error": str(e)}).encode())
        finally:
            metrics.SERVE_REQUESTS.labels(route=route).inc()
            metrics.SERVE_REQUEST_SECONDS.labels(route=route).inc(
                time.perf_counter() - t0)

    def do_GET(self):
        if self.path in ("/", "/-/routes"):
            self.send_response(200)
            self.end_headers()
            self.wfile.write(json.dumps(
                {r: d.deployment.name for r, d in _state.apps.items()}).encode())
        elif self.path.split("?")[0] in ("/-/metrics", "/metrics"):
            body = metrics.REGISTRY.render().encode()
            self.send_response(200)
            self.send_header("Content-Type", metrics.CONTENT_TYPE)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        else:
            self.do_POST()


def run(app: Application, host: str = "127.0.0.1", port: int = 8000,
        _blocking: bool = False, **_):
    """Deploy the application and (idempotently) start the HTTP proxy.

    Graceful update: re-running a route builds the NEW replica group first,
    atomically swaps it into the router, then drains and kills the old
    group's replicas — in-flight requests finish on the old replicas."""
    d = app.deployment
    group = _PredictorReplicaGroup(app)
    route = d.route_prefix.rstrip("/") or "/"
    old = _state.routes.get(route)
    _state.routes[route] = group
    _state.apps[route] = app
    if old is not None:
        threading.Thread(target=old.shutdown, daemon=True).start()
    if _state.server is None:
        _state.server = ThreadingHTTPServer((host, port), _Handler)
        _state.thread = threading.Thread(target=_state.server.serve_forever,
                                         daemon=True)
        _state.thread.start()
    return RayServeHandle(group)


@dataclass
class RayServeHandle:
    group: _PredictorReplicaGroup

    def predict(self, df):
        # Route through the group's _invoke under the replica lock — the
        # direct replicas[i].predict(df) call broke for actor-backed replicas
        # (ActorMethod has no __call__) and bypassed the HTTP path's
        # serialization.
        g = self.group
        with g._pool_lock:
            i = next(g._rr) % len(g.replicas)
            lock, rep = g._locks[i], g.replicas[i]
        with lock:
            return g._invoke(rep, df, {})


def get_deployment(name: str):
    for route, app in _state.apps.items():
        if app.deployment.name == name:
            return app.deployment
    raise KeyError(name)


def shutdown():
    if _state.server is not None:
        _state.server.shutdown()
        _state.server = None
        _state.thread = None
    _state.routes.clear()
    _state.apps.clear()
