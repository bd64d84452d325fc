from mi355.tune import Tuner, TuneConfig

__all__ = ["Tuner", "TuneConfig"]
