from mi355.serve import (Deployment, PredictorDeployment, deployment,
                         get_deployment, json_request, pandas_read_json,
                         run, shutdown)

__all__ = ["run", "shutdown", "Deployment", "PredictorDeployment",
           "deployment", "pandas_read_json", "json_request",
           "get_deployment"]
