"""Prometheus-style metrics (SURVEY.md C25 observability; reference exposes
Ray's dashboard/Prometheus metrics — ref README.md "Ray Dashboard" links).

A tiny in-process registry with Counter/Gauge metrics rendered in the
Prometheus text exposition format (0.0.4), served from:
  * `serve.run`'s HTTP proxy at GET /-/metrics (scrapable alongside the
    model routes), and
  * an optional standalone server via `start_metrics_server(port)`.

Trainers set `mi355_train_*` gauges on every worker report; serve increments
`mi355_serve_requests_total` / `..._request_seconds_total` per request. No
external client library — the text format is simple enough to emit directly.
"""
from __future__ import annotations

import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


def _fmt_value(v: float) -> str:
    if v == int(v) and abs(v) < 1e15:
        return str(int(v))
    return repr(float(v))


def _fmt_labels(labels: dict) -> str:
    if not labels:
        return ""
    inner = ",".join(
        '%s="%s"' % (k, str(v).replace("\\", "\\\\").replace('"', '\\"').replace("\n", "\\n"))
        for k, v in sorted(labels.items()))
    return "{" + inner + "}"


class _Metric:
    kind = "untyped"

    def __init__(self, name: str, help_text: str = ""):
        self.name = name
        self.help = help_text
        self._lock = threading.Lock()
        self._series: dict[tuple, float] = {}

    def labels(self, **labels) -> "_Child":
        return _Child(self, tuple(sorted(labels.items())))

    def _set(self, key: tuple, value: float):
        with self._lock:
            self._series[key] = float(value)

    def _add(self, key: tuple, delta: float):
        with self._lock:
            self._series[key] = self._series.get(key, 0.0) + float(delta)

    def render(self) -> str:
        lines = []
        if self.help:
            lines.append(f"# HELP {self.name} {self.help}")
        lines.append(f"# TYPE {self.name} {self.kind}")
        with self._lock:
            series = dict(self._series) or {(): 0.0}
        for key, value in sorted(series.items()):
            lines.append(f"{self.name}{_fmt_labels(dict(key))} {_fmt_value(value)}")
        return "\n".join(lines)


class _Child:
    def __init__(self, parent: _Metric, key: tuple):
        self._parent = parent
        self._key = key

    def inc(self, amount: float = 1.0):
        self._parent._add(self._key, amount)

    def set(self, value: float):
        self._parent._set(self._key, value)

    def get(self) -> float:
        with self._parent._lock:
            return self._parent._series.get(self._key, 0.0)


class Counter(_Metric):
    kind = "counter"

    def inc(self, amount: float = 1.0):
        self._add((), amount)


class Gauge(_Metric):
    kind = "gauge"

    def set(self, value: float):
        self._set((), value)

    def inc(self, amount: float = 1.0):
        self._add((), amount)


class Registry:
    def __init__(self):
        self._lock = threading.Lock()
        self._metrics: dict[str, _Metric] = {}

    def counter(self, name: str, help_text: str = "") -> Counter:
        return self._get_or_make(name, help_text, Counter)

    def gauge(self, name: str, help_text: str = "") -> Gauge:
        return self._get_or_make(name, help_text, Gauge)

    def _get_or_make(self, name, help_text, cls):
        with self._lock:
            m = self._metrics.get(name)
            if m is None:
                m = self._metrics[name] = cls(name, help_text)
            elif not isinstance(m, cls):
                raise TypeError(f"metric {name!r} already registered as {m.kind}")
            return m

    def render(self) -> str:
        with self._lock:
            metrics = list(self._metrics.values())
        return "\n".join(m.render() for m in metrics) + "\n"


REGISTRY = Registry()

# Framework-level metrics. Series appear lazily (a metric with no series
# renders one unlabeled 0 sample so scrapes always see the families).
TRAIN_REPORTS = REGISTRY.counter(
    "mi355_train_reports_total", "Worker reports received by the trainer driver")
TRAIN_METRIC = REGISTRY.gauge(
    "mi355_train_metric", "Latest value of each numeric metric reported by rank 0")
TRAIN_WORKERS = REGISTRY.gauge(
    "mi355_train_workers", "Live training workers per run")
SERVE_REQUESTS = REGISTRY.counter(
    "mi355_serve_requests_total", "HTTP requests handled per route")
SERVE_REQUEST_SECONDS = REGISTRY.counter(
    "mi355_serve_request_seconds_total", "Cumulative request handling time per route")
ACTOR_RESTARTS = REGISTRY.counter(
    "mi355_actor_restarts_total", "Actor processes restarted after a crash")


def record_train_report(run_name: str, metrics: dict):
    TRAIN_REPORTS.labels(run=run_name).inc()
    for k, v in metrics.items():
        if isinstance(v, (int, float)) and not isinstance(v, bool):
            TRAIN_METRIC.labels(run=run_name, name=k).set(v)


CONTENT_TYPE = "text/plain; version=0.0.4; charset=utf-8"


class _MetricsHandler(BaseHTTPRequestHandler):
    registry: Registry = REGISTRY

    def log_message(self, *a):
        pass

    def do_GET(self):
        if self.path.split("?")[0] not in ("/metrics", "/"):
            self.send_response(404)
            self.end_headers()
            return
        body = self.registry.render().encode()
        self.send_response(200)
        self.send_header("Content-Type", CONTENT_TYPE)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


def _parse_rocm_smi(payload: dict) -> dict:
    """{'card0': {'GPU use (%)': '37', 'GPU Memory Allocated (VRAM%)': '12'}}
    -> {'0': (37.0, 12.0)} (tolerant of key variants across rocm-smi versions)."""
    out = {}
    for card, fields in payload.items():
        if not card.startswith("card") or not isinstance(fields, dict):
            continue
        use = mem = None
        for k, v in fields.items():
            lk = k.lower()
            try:
                if "gpu use" in lk:
                    use = float(v)
                elif "vram%" in lk or "memory allocated" in lk:
                    mem = float(v)
            except (TypeError, ValueError):
                continue
        out[card[4:]] = (use, mem)
    return out


def start_gpu_sampler(interval: float = 15.0, registry: Registry = REGISTRY):
    """Sample per-GPU utilization + VRAM via rocm-smi into gauges
    (mi355_gpu_utilization_percent / mi355_gpu_vram_percent). Daemon thread;
    exits quietly if rocm-smi is unavailable (CPU-only hosts)."""
    import json
    import subprocess
    import time

    use_g = registry.gauge("mi355_gpu_utilization_percent", "GPU busy percent")
    mem_g = registry.gauge("mi355_gpu_vram_percent", "VRAM allocated percent")

    def loop():
        while True:
            try:
                raw = subprocess.run(
                    ["rocm-smi", "--showuse", "--showmemuse", "--json"],
                    capture_output=True, text=True, timeout=10).stdout
                for gpu, (use, mem) in _parse_rocm_smi(json.loads(raw)).items():
                    if use is not None:
                        use_g.labels(gpu=gpu).set(use)
                    if mem is not None:
                        mem_g.labels(gpu=gpu).set(mem)
            except Exception:
                return  # no rocm-smi / bad output: stop sampling quietly
            time.sleep(interval)

    t = threading.Thread(target=loop, daemon=True)
    t.start()
    return t


def start_metrics_server(port: int = 0, host: str = "127.0.0.1",
                         registry: Registry = REGISTRY):
    """Serve `registry` at http://host:port/metrics; returns (server, port).

    port=0 picks a free port. The server runs on a daemon thread; call
    `server.shutdown()` to stop it.
    """
    handler = type("_Handler", (_MetricsHandler,), {"registry": registry})
    server = ThreadingHTTPServer((host, port), handler)
    threading.Thread(target=server.serve_forever, daemon=True).start()
    return server, server.server_address[1]
