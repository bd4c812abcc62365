"""gluon.contrib.estimator (reference parity: MXNet 1.6 estimator API)."""
from .estimator import Estimator
from .event_handler import (TrainBegin, TrainEnd, EpochBegin, EpochEnd,
                            BatchBegin, BatchEnd, StoppingHandler,
                            MetricHandler, LoggingHandler,
                            CheckpointHandler, EarlyStoppingHandler)

__all__ = ["Estimator", "TrainBegin", "TrainEnd", "EpochBegin", "EpochEnd",
           "BatchBegin", "BatchEnd", "StoppingHandler", "MetricHandler",
           "LoggingHandler", "CheckpointHandler", "EarlyStoppingHandler"]
