from mi355.tune import ASHAScheduler
from mi355.tune.schedulers import FIFOScheduler

__all__ = ["ASHAScheduler", "FIFOScheduler"]
