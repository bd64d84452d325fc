from mi355.tune import ASHAScheduler

AsyncHyperBandScheduler = ASHAScheduler

__all__ = ["ASHAScheduler", "AsyncHyperBandScheduler"]
