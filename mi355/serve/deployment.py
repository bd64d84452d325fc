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

    def options(self, name=None, num_replicas=None, route_prefix=None,
                ray_actor_options=None, **_):
        return replace(
            self,
            name=name or self.name,
            num_replicas=num_replicas or self.num_replicas,
            route_prefix=route_prefix or self.route_prefix,
            ray_actor_options=ray_actor_options or self.ray_actor_options,
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
        self.http_adapter = app.kwargs.get("http_adapter", pandas_read_json)
        rest = {k: v for k, v in app.kwargs.items() if k != "http_adapter"}
        if app.args and hasattr(app.args[0], "from_checkpoint"):
            # PredictorDeployment.bind(PredictorCls, checkpoint, ...)
            predictor_cls, checkpoint = app.args[0], app.args[1]
            if d.ray_actor_options and _runtime_up():
                # replica = dedicated actor process (GPU-pinned round-robin
                # when num_gpus>0) — the reference's serve replica model
                from ..runtime.actor import ActorClass

                num_gpus = d.ray_actor_options.get("num_gpus", 0)
                cls = ActorClass(_ActorReplica, num_gpus=num_gpus)
                self.replicas = [cls.remote(predictor_cls, checkpoint, rest)
                                 for _ in range(d.num_replicas)]
                self._invoke = _invoke_actor
            else:
                self.replicas = [predictor_cls.from_checkpoint(checkpoint, **rest)
                                 for _ in range(d.num_replicas)]
                self._invoke = lambda rep, data, kw: rep.predict(data, **kw)
        elif isinstance(d.func_or_class, type):
            # generic @serve.deployment class: instance is the handler
            self.replicas = [d.func_or_class(*app.args, **rest)
                             for _ in range(d.num_replicas)]
            self._invoke = lambda rep, data, kw: rep(data, **kw)
        else:
            # @serve.deployment function
            self.replicas = [d.func_or_class] * max(d.num_replicas, 1)
            self._invoke = lambda rep, data, kw: rep(data, **kw)
        self._rr = itertools.cycle(range(len(self.replicas)))
        self._locks = [threading.Lock() for _ in self.replicas]

    def handle(self, body: bytes, query_kwargs: dict):
        data = self.http_adapter(body)
        i = next(self._rr)
        with self._locks[i]:
            out = self._invoke(self.replicas[i], data, query_kwargs)
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
    """Deploy the application and (idempotently) start the HTTP proxy."""
    d = app.deployment
    group = _PredictorReplicaGroup(app)
    route = d.route_prefix.rstrip("/") or "/"
    _state.routes[route] = group
    _state.apps[route] = app
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
        i = next(self.group._rr)
        with self.group._locks[i]:
            return self.group._invoke(self.group.replicas[i], df, {})


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
