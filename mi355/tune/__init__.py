from .search import (choice, grid_search, loguniform, qrandint, quniform,
                     randint, randn, sample_from, uniform)
from .schedulers import ASHAScheduler, FIFOScheduler
from .tuner import TuneConfig, Tuner

__all__ = ["choice", "uniform", "quniform", "randint", "qrandint", "randn",
           "loguniform", "grid_search", "sample_from", "ASHAScheduler",
           "FIFOScheduler", "Tuner", "TuneConfig"]
