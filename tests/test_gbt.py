"""GBT / XGBoostTrainer tests (C18, M8) — the Intro-notebook pipeline on a
synthetic binary-classification task, incl. 2-worker histogram allreduce."""
import numpy as np
import pandas as pd
import pytest

import mi355.data as mdata
from mi355.air import Checkpoint, RunConfig, ScalingConfig
from mi355.models.gbt import GBTModel
from mi355.predict import BatchPredictor, XGBoostPredictor
from mi355.train import XGBoostTrainer


def make_data(n=400, seed=0):
    rng = np.random.default_rng(seed)
    x1 = rng.normal(size=n)
    x2 = rng.normal(size=n)
    y = ((x1 + 0.5 * x2 + rng.normal(scale=0.3, size=n)) > 0).astype(float)
    return pd.DataFrame({"x1": x1, "x2": x2, "is_big_tip": y})


def test_gbt_model_learns():
    df = make_data()
    X = df[["x1", "x2"]].to_numpy()
    y = df["is_big_tip"].to_numpy()
    m = GBTModel({"objective": "binary:logistic", "max_depth": 3, "eta": 0.3})
    hist = m.fit(X, y, num_boost_round=20)
    assert hist[-1]["train-logloss"] < 0.35
    assert hist[-1]["train-logloss"] < hist[0]["train-logloss"]
    acc = ((m.predict(X) > 0.5) == (y > 0.5)).mean()
    assert acc > 0.9


def test_gbt_regression():
    rng = np.random.default_rng(1)
    X = rng.normal(size=(300, 3))
    y = 2 * X[:, 0] - X[:, 1] + 0.1 * rng.normal(size=300)
    m = GBTModel({"objective": "reg:squarederror", "max_depth": 4})
    hist = m.fit(X, y, num_boost_round=30)
    assert hist[-1]["train-rmse"] < 0.6


def test_gbt_save_load_roundtrip(tmp_path):
    df = make_data(100)
    m = GBTModel({"objective": "binary:logistic", "max_depth": 3})
    m.feature_names = ["x1", "x2"]
    m.fit(df[["x1", "x2"]].to_numpy(), df["is_big_tip"].to_numpy(), num_boost_round=5)
    m.save_pretrained(str(tmp_path))
    m2 = GBTModel.from_pretrained(str(tmp_path))
    X = df[["x1", "x2"]].to_numpy()
    np.testing.assert_allclose(m.predict(X), m2.predict(X))


@pytest.mark.timeout(600)
def test_xgboost_trainer_two_workers(tmp_path):
    df = make_data(400)
    ds = mdata.from_pandas(df)
    train, valid = ds.train_test_split(test_size=0.25)
    trainer = XGBoostTrainer(
        label_column="is_big_tip",
        params={"objective": "binary:logistic", "max_depth": 3, "eta": 0.3,
                "eval_metric": ["logloss", "error"]},
        num_boost_round=15,
        scaling_config=ScalingConfig(num_workers=2, use_gpu=False),
        datasets={"train": train, "valid": valid},
        run_config=RunConfig(name="gbt2", local_dir=str(tmp_path)),
    )
    res = trainer.fit()
    assert res.error is None, res.error
    assert res.metrics["train-logloss"] < 0.45
    assert "valid-logloss" in res.metrics
    # distributed result must match single-process training closely
    m1 = GBTModel({"objective": "binary:logistic", "max_depth": 3, "eta": 0.3})
    tdf = train.to_pandas()
    m1.fit(tdf[["x1", "x2"]].to_numpy(), tdf["is_big_tip"].to_numpy(),
           num_boost_round=15)
    dist_model = res.checkpoint.get_model()
    X = tdf[["x1", "x2"]].to_numpy()
    corr = np.corrcoef(m1.predict(X), dist_model.predict(X))[0, 1]
    # bin edges come from rank-0's shard (approximate sketch, like xgboost's
    # distributed hist) so trees differ slightly from exact single-process
    assert corr > 0.9, corr
    # predictor path (C16)
    bp = BatchPredictor.from_checkpoint(res.checkpoint, XGBoostPredictor)
    out = bp.predict(valid.drop_columns(["is_big_tip"]), batch_size=64).to_pandas()
    assert len(out) == valid.count()
    assert out["predictions"].between(0, 1).all()


def test_serve_xgboost_service(tmp_path):
    """The Intro notebook's exact serve cell: PredictorDeployment.options(
    name='XGBoostService', num_replicas=2, route_prefix='/rayair').bind(
    XGBoostPredictor, checkpoint, http_adapter=pandas_read_json) + POST."""
    import json
    import socket
    import urllib.request

    from ray import serve
    from ray.serve import PredictorDeployment
    from ray.serve.http_adapters import pandas_read_json

    df = make_data(200)
    m = GBTModel({"objective": "binary:logistic", "max_depth": 3, "eta": 0.3})
    m.fit(df[["x1", "x2"]].to_numpy(), df["is_big_tip"].to_numpy(),
          num_boost_round=8)
    ckpt_dir = str(tmp_path / "gbt")
    m.save_pretrained(ckpt_dir)
    from mi355.air import Checkpoint

    ckpt = Checkpoint.from_directory(ckpt_dir)

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    serve.run(PredictorDeployment.options(
        name="XGBoostService", num_replicas=2, route_prefix="/rayair",
    ).bind(XGBoostPredictor, ckpt, http_adapter=pandas_read_json), port=port)
    try:
        body = json.dumps({"x1": [0.1, 0.9], "x2": [0.2, 0.8]}).encode()
        req = urllib.request.Request(f"http://127.0.0.1:{port}/rayair",
                                     data=body, method="POST")
        with urllib.request.urlopen(req) as r:
            rows = json.loads(r.read())
        assert len(rows) == 2
        assert all(0.0 <= row["predictions"] <= 1.0 for row in rows)
    finally:
        serve.shutdown()
