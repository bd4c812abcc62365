"""Event handlers for the Estimator fit loop (reference parity:
python/mxnet/gluon/contrib/estimator/event_handler.py)."""
from __future__ import annotations

import logging
import time


class TrainBegin:
    def train_begin(self, estimator, *args, **kwargs):
        pass


class TrainEnd:
    def train_end(self, estimator, *args, **kwargs):
        pass


class EpochBegin:
    def epoch_begin(self, estimator, *args, **kwargs):
        pass


class EpochEnd:
    def epoch_end(self, estimator, *args, **kwargs):
        pass


class BatchBegin:
    def batch_begin(self, estimator, *args, **kwargs):
        pass


class BatchEnd:
    def batch_end(self, estimator, *args, **kwargs):
        pass


class StoppingHandler(TrainBegin, BatchEnd, EpochEnd):
    """Stop after max_epoch epochs or max_batch batches."""

    def __init__(self, max_epoch=None, max_batch=None):
        self.max_epoch = max_epoch
        self.max_batch = max_batch
        self.current_batch = 0
        self.current_epoch = 0
        self.stop_training = False

    def train_begin(self, estimator, *args, **kwargs):
        self.current_batch = 0
        self.current_epoch = 0

    def batch_end(self, estimator, *args, **kwargs):
        self.current_batch += 1
        if self.max_batch and self.current_batch >= self.max_batch:
            self.stop_training = True

    def epoch_end(self, estimator, *args, **kwargs):
        self.current_epoch += 1
        if self.max_epoch and self.current_epoch >= self.max_epoch:
            self.stop_training = True


class MetricHandler(EpochBegin, BatchEnd):
    """Reset metrics at epoch begin, update on every batch end."""

    def __init__(self, metrics):
        self.metrics = metrics

    def epoch_begin(self, estimator, *args, **kwargs):
        for m in self.metrics:
            m.reset()

    def batch_end(self, estimator, *args, **kwargs):
        pred = kwargs.get("pred")
        label = kwargs.get("label")
        loss = kwargs.get("loss")
        for m in self.metrics:
            if m.name.startswith("loss") and loss is not None:
                m.update(None, loss)
            elif pred is not None:
                m.update(label, pred)


class LoggingHandler(TrainBegin, TrainEnd, EpochEnd, BatchEnd):
    def __init__(self, log_interval="epoch", metrics=None):
        self.log_interval = log_interval
        self.metrics = metrics or []
        self.batch_index = 0
        self.logger = logging.getLogger("estimator")

    def train_begin(self, estimator, *args, **kwargs):
        self.train_start = time.time()
        self.logger.info("Training begin")

    def train_end(self, estimator, *args, **kwargs):
        self.logger.info("Training done in %.1fs",
                         time.time() - self.train_start)

    def batch_end(self, estimator, *args, **kwargs):
        self.batch_index += 1
        if self.log_interval != "epoch" and \
                self.batch_index % int(self.log_interval) == 0:
            msg = " ".join(f"{m.name}={m.get()[1]:.4f}"
                           for m in self.metrics)
            self.logger.info("[batch %d] %s", self.batch_index, msg)

    def epoch_end(self, estimator, *args, **kwargs):
        msg = " ".join(f"{m.name}={m.get()[1]:.4f}" for m in self.metrics)
        self.logger.info("[epoch end] %s", msg)


class CheckpointHandler(TrainBegin, EpochEnd):
    """Save net parameters (and trainer states) every epoch."""

    def __init__(self, model_dir, model_prefix="model",
                 save_best=False, monitor=None, max_checkpoints=5):
        self.model_dir = model_dir
        self.model_prefix = model_prefix
        self.save_best = save_best
        self.monitor = monitor
        self.epoch = 0
        self.best = None

    def train_begin(self, estimator, *args, **kwargs):
        import os
        os.makedirs(self.model_dir, exist_ok=True)

    def epoch_end(self, estimator, *args, **kwargs):
        import os
        path = os.path.join(self.model_dir,
                            f"{self.model_prefix}-epoch{self.epoch}.params")
        estimator.net.save_parameters(path)
        if estimator.trainer is not None:
            estimator.trainer.save_states(path.replace(".params", ".states"))
        if self.save_best and self.monitor is not None:
            val = self.monitor.get()[1]
            if self.best is None or val > self.best:
                self.best = val
                estimator.net.save_parameters(os.path.join(
                    self.model_dir, f"{self.model_prefix}-best.params"))
        self.epoch += 1


class EarlyStoppingHandler(TrainBegin, EpochEnd):
    def __init__(self, monitor, min_delta=0, patience=0, mode="max"):
        self.monitor = monitor
        self.min_delta = min_delta
        self.patience = patience
        self.mode = mode
        self.best = None
        self.wait = 0
        self.stop_training = False

    def train_begin(self, estimator, *args, **kwargs):
        self.best = None
        self.wait = 0
        self.stop_training = False

    def epoch_end(self, estimator, *args, **kwargs):
        val = self.monitor.get()[1]
        improved = self.best is None or (
            val > self.best + self.min_delta if self.mode == "max"
            else val < self.best - self.min_delta)
        if improved:
            self.best = val
            self.wait = 0
        else:
            self.wait += 1
            if self.wait > self.patience:
                self.stop_training = True
