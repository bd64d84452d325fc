from mi355.tune import (ASHAScheduler, TuneConfig, Tuner, choice, grid_search,
                        loguniform, qrandint, quniform, randint, randn,
                        sample_from, uniform)
from mi355.tune import schedulers  # noqa: F401

__all__ = ["Tuner", "TuneConfig", "ASHAScheduler", "choice", "uniform",
           "quniform", "randint", "qrandint", "randn", "loguniform",
           "grid_search", "sample_from", "schedulers"]
