"""Estimator: a batteries-included Gluon fit loop (reference parity:
python/mxnet/gluon/contrib/estimator/estimator.py, MXNet 1.6)."""
from __future__ import annotations

from .... import autograd
from ... import Trainer
from ....metric import EvalMetric, Loss as LossMetric
from .event_handler import (TrainBegin, TrainEnd, EpochBegin, EpochEnd,
                            BatchBegin, BatchEnd, StoppingHandler,
                            MetricHandler, LoggingHandler)


class Estimator:
    def __init__(self, net, loss, train_metrics=None, trainer=None,
                 context=None):
        self.net = net
        self.loss = loss
        self.train_metrics = train_metrics or []
        if isinstance(self.train_metrics, EvalMetric):
            self.train_metrics = [self.train_metrics]
        self.train_loss_metric = LossMetric(name="loss")
        self.context = context
        self.trainer = trainer or Trainer(
            net.collect_params(), "sgd", {"learning_rate": 0.001})
        self.stop_training = False

    def _arrange_handlers(self, event_handlers):
        handlers = list(event_handlers or [])
        if not any(isinstance(h, StoppingHandler) for h in handlers):
            handlers.append(self._stop_handler)
        if not any(isinstance(h, MetricHandler) for h in handlers):
            handlers.append(MetricHandler(
                [self.train_loss_metric] + self.train_metrics))
        groups = {}
        for base in (TrainBegin, EpochBegin, BatchBegin, BatchEnd,
                     EpochEnd, TrainEnd):
            groups[base] = [h for h in handlers if isinstance(h, base)]
        return groups

    def evaluate(self, val_data, val_metrics):
        for m in val_metrics:
            m.reset()
        for batch in val_data:
            data, label = batch.data[0], batch.label[0]
            pred = self.net(data)
            for m in val_metrics:
                m.update(label, pred)
        return val_metrics

    def fit(self, train_data, val_data=None, epochs=None, batches=None,
            event_handlers=None):
        self._stop_handler = StoppingHandler(max_epoch=epochs,
                                             max_batch=batches)
        groups = self._arrange_handlers(event_handlers)
        self.stop_training = False
        for h in groups[TrainBegin]:
            h.train_begin(self)
        while not self.stop_training:
            for h in groups[EpochBegin]:
                h.epoch_begin(self)
            for batch in train_data:
                data, label = batch.data[0], batch.label[0]
                for h in groups[BatchBegin]:
                    h.batch_begin(self, batch=batch)
                with autograd.record():
                    pred = self.net(data)
                    loss = self.loss(pred, label)
                loss.backward()
                bs = data.shape[0]
                self.trainer.step(bs)
                for h in groups[BatchEnd]:
                    h.batch_end(self, batch=batch, pred=pred, label=label,
                                loss=loss)
                self.stop_training = any(
                    getattr(h, "stop_training", False)
                    for hs in groups.values() for h in hs)
                if self.stop_training:
                    break
            for h in groups[EpochEnd]:
                h.epoch_end(self)
            if hasattr(train_data, "reset"):
                train_data.reset()
            self.stop_training = self.stop_training or any(
                getattr(h, "stop_training", False)
                for hs in groups.values() for h in hs)
        for h in groups[TrainEnd]:
            h.train_end(self)
