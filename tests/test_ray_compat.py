"""The reference notebooks' `ray.*` call surface resolves through the compat
namespace (SURVEY.md §2.2 call sites)."""
import pytest


def test_ray_core_surface():
    import ray

    ray.init(num_workers=2, num_gpus=0)
    try:
        assert ray.is_initialized()

        @ray.remote
        def f(x):
            return x + 1

        ref = ray.put(41)
        assert ray.get(f.remote(ray.get(ref))) == 42
        ready, _ = ray.wait([f.remote(1)], num_returns=1)
        assert ready
    finally:
        ray.shutdown()
    assert not ray.is_initialized()


def test_ray_air_train_tune_imports():
    from ray import tune
    from ray.air import Checkpoint, Result, ResultGrid
    from ray.air.config import CheckpointConfig, RunConfig, ScalingConfig
    from ray.data import from_items
    from ray.data.preprocessors import BatchMapper, MinMaxScaler
    from ray.train.batch_predictor import BatchPredictor
    from ray.train.huggingface import HuggingFaceCheckpoint, HuggingFaceTrainer
    from ray.train.predictor import Predictor
    from ray.train.xgboost import XGBoostPredictor, XGBoostTrainer
    from ray.tune import ASHAScheduler, TuneConfig, Tuner
    from ray.util import ActorPool

    sc = ScalingConfig(num_workers=2, use_gpu=False)
    assert sc.num_workers == 2
    assert tune.choice([1, 2]).categories == [1, 2]
    ds = from_items([{"a": 1}, {"a": 2}])
    assert ds.count() == 2


def test_ray_serve_surface():
    from ray import serve

    d = serve.PredictorDeployment.options(name="X", num_replicas=2,
                                          route_prefix="/x")
    assert d.num_replicas == 2 and d.route_prefix == "/x"


def test_ray_timeline_and_exceptions():
    import ray
    import ray.exceptions

    assert callable(ray.timeline)
    assert issubclass(ray.exceptions.RayActorError, RuntimeError)
    from mi355 import ActorDiedError

    assert ray.exceptions.RayActorError is ActorDiedError


def test_reference_import_forms():
    """Every `from ray...` import statement that appears in the reference
    notebooks must resolve (SURVEY.md §2.2 C-table citations)."""
    from ray import serve, tune  # noqa: F401
    from ray.air import Checkpoint  # noqa: F401
    from ray.air.config import (CheckpointConfig, RunConfig,  # noqa: F401
                                ScalingConfig)
    from ray.data.preprocessors import BatchMapper, MinMaxScaler  # noqa: F401
    from ray.serve import PredictorDeployment  # noqa: F401
    from ray.serve.http_adapters import pandas_read_json  # noqa: F401
    from ray.train.batch_predictor import BatchPredictor  # noqa: F401
    from ray.train.huggingface import (HuggingFaceCheckpoint,  # noqa: F401
                                       HuggingFacePredictor,
                                       HuggingFaceTrainer)
    from ray.train.predictor import Predictor  # noqa: F401
    from ray.train.xgboost import (XGBoostPredictor,  # noqa: F401
                                   XGBoostTrainer)
    from ray.tune import Tuner  # noqa: F401
    from ray.tune.schedulers.async_hyperband import ASHAScheduler  # noqa: F401
    from ray.tune.tuner import TuneConfig  # noqa: F401
    from ray.util.actor_pool import ActorPool  # noqa: F401


def test_raylet_objectref_and_serve_deployment_resolve():
    """Overview_of_Ray.ipynb cell-34 shows futures as ray._raylet.ObjectRef;
    Introduction_to_Ray_AI_Runtime.ipynb uses @serve.deployment."""
    import ray
    import ray._raylet
    from ray import serve

    ref = None
    assert ray._raylet.ObjectRef is not None
    assert callable(serve.deployment)
    # the runtime's actual future type IS the advertised one
    from mi355.runtime.core import ObjectRef
    assert ray._raylet.ObjectRef is ObjectRef
