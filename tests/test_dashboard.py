"""Web dashboard (reference prescribes the Ray dashboard at
127.0.0.1:8265, Install_locally.md:64-67): HTML status page + JSON state +
Prometheus + timeline from one stdlib HTTP server."""
import json
import urllib.request

import pytest


@pytest.fixture()
def dash_url():
    from mi355 import dashboard

    host, port = dashboard.start_dashboard(port=0)
    yield f"http://{host}:{port}"
    dashboard.stop_dashboard()


def _get(url):
    with urllib.request.urlopen(url, timeout=10) as r:
        return r.status, r.headers.get("Content-Type", ""), r.read()


def test_html_page_renders(dash_url, tmp_path, monkeypatch):
    # seed a fake results tree so the Runs table has content
    run = tmp_path / "demo-run"
    (run / "checkpoint_000001").mkdir(parents=True)
    (run / "metrics.jsonl").write_text(json.dumps({"epoch": 1, "loss": 0.5}) + "\n")
    monkeypatch.setenv("MI355_RESULTS_DIR", str(tmp_path))
    code, ctype, body = _get(dash_url + "/")
    assert code == 200 and "text/html" in ctype
    text = body.decode()
    assert "mi355 dashboard" in text
    assert "demo-run" in text and "loss=0.5" in text


def test_api_state_json(dash_url, tmp_path, monkeypatch):
    monkeypatch.setenv("MI355_RESULTS_DIR", str(tmp_path))
    code, ctype, body = _get(dash_url + "/api/state")
    assert code == 200 and "json" in ctype
    st = json.loads(body)
    assert set(st) >= {"gpus", "runtime", "runs"}


def test_metrics_endpoint_prometheus(dash_url):
    from mi355 import metrics

    metrics.REGISTRY.gauge("mi355_dash_test_gauge", "t").labels(x="1").set(7)
    code, ctype, body = _get(dash_url + "/metrics")
    assert code == 200 and "text/plain" in ctype
    assert b"mi355_dash_test_gauge" in body


def test_timeline_endpoint(dash_url):
    import mi355

    mi355.init(num_workers=2)

    @mi355.remote
    def f(x):
        return x * 2

    assert mi355.get(f.remote(21)) == 42
    code, ctype, body = _get(dash_url + "/timeline")
    assert code == 200
    events = json.loads(body)
    assert isinstance(events, list)
    # the completed task shows up as a chrome-trace "X" slice
    assert any(e.get("ph") == "X" for e in events)


def test_init_prints_dashboard_url(capsys):
    import importlib

    import mi355
    from mi355 import dashboard

    dashboard.stop_dashboard()
    mi355.shutdown()
    mi355.init(num_workers=2)
    out = capsys.readouterr().out
    assert "mi355 runtime started" in out
    assert "dashboard at http://" in out
    dashboard.stop_dashboard()
