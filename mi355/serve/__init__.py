"""Online inference serving (reference
Introduction_to_Ray_AI_Runtime.ipynb:cell-70-78):

    serve.run(PredictorDeployment.options(name=..., num_replicas=2,
              route_prefix="/rayair").bind(PredictorCls, checkpoint,
              http_adapter=pandas_read_json))
    requests.post("http://localhost:8000/rayair", json=[{...}])

HTTP proxy on :8000 routes by prefix to a round-robin pool of predictor
replicas. Replicas live in-process (model loaded once each); with the mi355
runtime initialised and num_gpus>0 they are GPU-pinned actor processes.
"""
from .deployment import (Deployment, PredictorDeployment, deployment,
                         pandas_read_json, json_request, run, shutdown,
                         get_deployment)

__all__ = ["Deployment", "PredictorDeployment", "deployment", "pandas_read_json",
           "json_request", "run", "shutdown", "get_deployment"]
