"""Online inference serving (reference
Introduction_to_Ray_AI_Runtime.ipynb:cell-70-78):

    serve.run(PredictorDeployment.options(name=..., num_replicas=2,
              route_prefix="/rayair").bind(PredictorCls, checkpoint,
              http_adapter=pandas_read_json))
    requests.post("http://localhost:8000/rayair", json=[{...}])

HTTP proxy on :8000 routes by prefix to a round-robin pool of predictor
replicas (GET /-/routes lists them, /-/metrics serves Prometheus text).
Replicas live in-process by default (model loaded once each); pass
``ray_actor_options={"num_gpus": 1}`` with the mi355 runtime initialised and
each replica becomes its own GPU-pinned actor process.
"""
from .deployment import (Deployment, PredictorDeployment, deployment,
                         pandas_read_json, json_request, run, shutdown,
                         get_deployment)

__all__ = ["Deployment", "PredictorDeployment", "deployment", "pandas_read_json",
           "json_request", "run", "shutdown", "get_deployment"]
