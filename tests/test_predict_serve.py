"""BatchPredictor / Predictor / serve tests (C14-C17) — mirror the reference
inference architectures."""
import json
import urllib.request

import numpy as np
import pandas as pd
import pytest
import torch

from mi355.air import Checkpoint
from mi355.data import Dataset, from_pandas
from mi355.predict import BatchPredictor, HuggingFacePredictor, Predictor


class DoublingPredictor(Predictor):
    """Custom Predictor subclass via _predict_pandas (reference
    Scaling_batch_inference.ipynb:cell-73 pattern)."""

    def __init__(self, factor, preprocessor=None):
        super().__init__(preprocessor)
        self.factor = factor

    @classmethod
    def from_checkpoint(cls, ckpt, **kw):
        d = ckpt.to_dict()
        return cls(d["factor"], preprocessor=ckpt.get_preprocessor())

    def _predict_pandas(self, df, **kw):
        return pd.DataFrame({"out": df["x"] * self.factor})


def test_checkpoint_dict_predictor_roundtrip():
    ckpt = Checkpoint.from_dict({"factor": 3})
    p = DoublingPredictor.from_checkpoint(ckpt)
    out = p.predict(pd.DataFrame({"x": [1, 2, 3]}))
    assert out["out"].tolist() == [3, 6, 9]


def test_batch_predictor_over_dataset():
    ds = from_pandas(pd.DataFrame({"x": list(range(100))})).repartition(4)
    bp = BatchPredictor.from_checkpoint(Checkpoint.from_dict({"factor": 2}),
                                        DoublingPredictor)
    out = bp.predict(ds, batch_size=16)
    assert out.count() == 100
    assert out.to_pandas()["out"].tolist() == [2 * i for i in range(100)]


def test_batch_predictor_keep_columns():
    ds = from_pandas(pd.DataFrame({"x": [1, 2], "id": ["a", "b"]}))
    bp = BatchPredictor.from_checkpoint(Checkpoint.from_dict({"factor": 10}),
                                        DoublingPredictor)
    out = bp.predict(ds, feature_columns=["x"], keep_columns=["id"]).to_pandas()
    assert out["id"].tolist() == ["a", "b"] and out["out"].tolist() == [10, 20]


def test_hf_predictor_generate_from_checkpoint(tmp_path):
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration

    torch.manual_seed(0)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1,
                    d_model=32, d_ff=64, num_heads=2, d_kv=16, vocab_size=128,
                    dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    model.save_pretrained(str(tmp_path))
    ckpt = Checkpoint.from_directory(str(tmp_path))
    pred = HuggingFacePredictor.from_checkpoint(ckpt, task="text2text-generation")
    df = pd.DataFrame({"input_ids": [np.arange(8) + 2] * 3,
                       "attention_mask": [np.ones(8, dtype=np.int64)] * 3})
    out = pred.predict(df, max_new_tokens=4, min_new_tokens=4)
    assert len(out) == 3
    assert all(len(t) == 4 for t in out["generated_tokens"])


@pytest.mark.timeout(120)
def test_serve_http_roundtrip():
    from mi355 import serve

    app = serve.PredictorDeployment.options(
        name="DoubleService", num_replicas=2, route_prefix="/double",
    ).bind(DoublingPredictor, Checkpoint.from_dict({"factor": 5}),
           http_adapter=serve.pandas_read_json)
    serve.run(app, port=18755)
    try:
        req = urllib.request.Request(
            "http://127.0.0.1:18755/double",
            data=json.dumps([{"x": 1}, {"x": 4}]).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=30) as resp:
            body = json.loads(resp.read())
        assert body == [{"out": 5}, {"out": 20}]
        # unknown route -> 404
        req2 = urllib.request.Request("http://127.0.0.1:18755/nope", data=b"[]")
        try:
            urllib.request.urlopen(req2, timeout=30)
            assert False, "expected 404"
        except urllib.error.HTTPError as e:
            assert e.code == 404
    finally:
        serve.shutdown()


def test_serve_actor_backed_replicas():
    """ray_actor_options + initialized runtime -> each replica is its own
    actor process (the reference's serve replica model)."""
    import json
    import socket
    import urllib.request

    import mi355
    from mi355 import serve
    from mi355.air.checkpoint import Checkpoint
    from mi355.predict.predictor import Predictor

    class EchoPredictor(Predictor):
        def __init__(self, scale, preprocessor=None):
            super().__init__(preprocessor)
            self.scale = scale

        @classmethod
        def from_checkpoint(cls, checkpoint, **kw):
            return cls(checkpoint.to_dict()["scale"])

        def predict(self, df, **kw):
            return df.assign(y=df["x"] * self.scale)

    mi355.init(num_workers=1)
    try:
        ckpt = Checkpoint.from_dict({"scale": 3})
        s = socket.socket(); s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]; s.close()
        app = serve.PredictorDeployment.options(
            name="actor-echo", route_prefix="/actor-echo", num_replicas=2,
            ray_actor_options={"num_gpus": 0},
        ).bind(EchoPredictor, ckpt, http_adapter=serve.pandas_read_json)
        serve.run(app, port=port)
        import sys

        dmod = sys.modules["mi355.serve.deployment"]
        group = dmod._state.routes["/actor-echo"]
        from mi355.runtime.actor import ActorHandle

        assert all(isinstance(r, ActorHandle) for r in group.replicas)
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/actor-echo",
            data=json.dumps({"x": [1, 2]}).encode(), method="POST")
        with urllib.request.urlopen(req) as r:
            body = json.loads(r.read())
        assert [row["y"] for row in body] == [3, 6]
    finally:
        serve.shutdown()
        mi355.shutdown()


def test_serve_routes_listing_and_metrics_route():
    """GET / lists deployed routes (the proxy's route map); /-/metrics serves
    the Prometheus registry."""
    import json
    import socket
    import urllib.request

    import pandas as pd

    from mi355 import serve

    @serve.deployment(name="lister", route_prefix="/lister")
    class Echo:
        def __call__(self, df: pd.DataFrame):
            return df

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    serve.run(Echo.bind(), host="127.0.0.1", port=port)
    try:
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/-/routes") as r:
            routes = json.loads(r.read())
        assert routes == {"/lister": "lister"}
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/-/metrics") as r:
            assert "mi355_serve_requests_total" in r.read().decode()
        assert serve.get_deployment("lister").name == "lister"
    finally:
        serve.shutdown()


def test_serve_autoscaling_grows_and_shrinks():
    """autoscaling_config resizes the replica pool toward
    ceil(ongoing/target) clamped to [min, max]."""
    import time

    from mi355 import serve
    from mi355.serve.deployment import _state

    calls = []

    @serve.deployment
    class Slowish:
        def __call__(self, data, **kw):
            calls.append(1)
            time.sleep(0.3)
            return {"ok": True}

    app = Slowish.options(
        name="auto", route_prefix="/auto", num_replicas=1,
        autoscaling_config={"min_replicas": 1, "max_replicas": 3,
                            "target_num_ongoing_requests_per_replica": 1,
                            "interval_s": 0.1}).bind()
    serve.run(app, port=0) if False else serve.run(app)
    group = _state.routes["/auto"]
    assert group.num_replicas() == 1
    # simulate load: bump the ongoing counter directly and let the
    # controller react
    group._ongoing = 3
    time.sleep(0.5)
    assert group.num_replicas() == 3
    group._ongoing = 0
    time.sleep(0.5)
    assert group.num_replicas() == 1
    group.shutdown()


def test_serve_graceful_update_swaps_replicas():
    """Re-running a route swaps in the new deployment; requests keep
    succeeding across the update (old group drains in the background)."""
    from mi355 import serve
    from mi355.serve.deployment import _state

    @serve.deployment
    class V1:
        def __call__(self, data, **kw):
            return {"version": 1}

    @serve.deployment
    class V2:
        def __call__(self, data, **kw):
            return {"version": 2}

    serve.run(V1.options(name="v", route_prefix="/ver").bind())
    g1 = _state.routes["/ver"]
    assert g1.handle(b"{}", {})["version"] == 1
    serve.run(V2.options(name="v", route_prefix="/ver").bind())
    g2 = _state.routes["/ver"]
    assert g2 is not g1
    assert g2.handle(b"{}", {})["version"] == 2
    import time

    time.sleep(0.3)  # old group drained
    assert g1.num_replicas() == 0
