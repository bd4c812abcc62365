"""Evaluation metrics (parity with python/mxnet/metric.py)."""
from __future__ import annotations

import numpy as np

from .ndarray.ndarray import NDArray

__all__ = ["EvalMetric", "Accuracy", "TopKAccuracy", "F1", "MAE", "MSE",
           "RMSE", "CrossEntropy", "NegativeLogLikelihood", "Perplexity",
           "Loss", "PearsonCorrelation", "CompositeEvalMetric",
           "CustomMetric", "create", "np_fn"]

_REGISTRY = {}


def register(cls):
    _REGISTRY[cls.__name__.lower()] = cls
    return cls


def create(metric, *args, **kwargs):
    if isinstance(metric, EvalMetric):
        return metric
    if callable(metric):
        return CustomMetric(metric, *args, **kwargs)
    if isinstance(metric, (list, tuple)):
        composite = CompositeEvalMetric()
        for m in metric:
            composite.add(create(m, *args, **kwargs))
        return composite
    name = metric.lower()
    aliases = {"acc": "accuracy", "ce": "crossentropy", "nll_loss":
               "negativeloglikelihood", "top_k_accuracy": "topkaccuracy"}
    name = aliases.get(name, name)
    return _REGISTRY[name](*args, **kwargs)


def _np(x):
    if isinstance(x, NDArray):
        return x.asnumpy()
    return np.asarray(x)


class EvalMetric:
    def __init__(self, name, output_names=None, label_names=None):
        self.name = name
        self.output_names = output_names
        self.label_names = label_names
        self.reset()

    def update(self, labels, preds):
        raise NotImplementedError

    def reset(self):
        self.num_inst = 0
        self.sum_metric = 0.0

    def get(self):
        if self.num_inst == 0:
            return (self.name, float("nan"))
        return (self.name, self.sum_metric / self.num_inst)

    def get_name_value(self):
        name, value = self.get()
        if not isinstance(name, list):
            name = [name]
        if not isinstance(value, list):
            value = [value]
        return list(zip(name, value))

    def __str__(self):
        return f"EvalMetric: {dict(self.get_name_value())}"


def _as_list(x):
    return x if isinstance(x, (list, tuple)) else [x]


@register
class Accuracy(EvalMetric):
    def __init__(self, axis=1, name="accuracy", **kw):
        super().__init__(name, **kw)
        self.axis = axis

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            p = _np(pred)
            l = _np(label).astype(np.int64)
            if p.ndim > l.ndim:
                p = np.argmax(p, axis=self.axis)
            p = p.astype(np.int64).flatten()
            l = l.flatten()
            self.sum_metric += (p == l).sum()
            self.num_inst += len(l)


@register
class TopKAccuracy(EvalMetric):
    def __init__(self, top_k=1, name="top_k_accuracy", **kw):
        super().__init__(f"{name}_{top_k}", **kw)
        self.top_k = top_k

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            p = _np(pred)
            l = _np(label).astype(np.int64).flatten()
            topk = np.argsort(p, axis=-1)[:, ::-1][:, :self.top_k]
            for j in range(self.top_k):
                self.sum_metric += (topk[:, j] == l).sum()
            self.num_inst += len(l)


@register
class F1(EvalMetric):
    def __init__(self, name="f1", average="macro", **kw):
        super().__init__(name, **kw)
        self.average = average

    def reset(self):
        super().reset()
        self.tp = self.fp = self.fn = 0

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            p = _np(pred)
            l = _np(label).astype(np.int64).flatten()
            if p.ndim > 1:
                p = np.argmax(p, axis=-1)
            p = p.astype(np.int64).flatten()
            self.tp += ((p == 1) & (l == 1)).sum()
            self.fp += ((p == 1) & (l == 0)).sum()
            self.fn += ((p == 0) & (l == 1)).sum()
            self.num_inst = 1

    def get(self):
        prec = self.tp / max(self.tp + self.fp, 1)
        rec = self.tp / max(self.tp + self.fn, 1)
        f1 = 2 * prec * rec / max(prec + rec, 1e-12)
        return (self.name, f1)


@register
class MAE(EvalMetric):
    def __init__(self, name="mae", **kw):
        super().__init__(name, **kw)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l, p = _np(label), _np(pred)
            self.sum_metric += np.abs(l - p.reshape(l.shape)).mean()
            self.num_inst += 1


@register
class MSE(EvalMetric):
    def __init__(self, name="mse", **kw):
        super().__init__(name, **kw)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l, p = _np(label), _np(pred)
            self.sum_metric += ((l - p.reshape(l.shape)) ** 2).mean()
            self.num_inst += 1


@register
class RMSE(EvalMetric):
    def __init__(self, name="rmse", **kw):
        super().__init__(name, **kw)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l, p = _np(label), _np(pred)
            self.sum_metric += np.sqrt(((l - p.reshape(l.shape)) ** 2).mean())
            self.num_inst += 1


@register
class CrossEntropy(EvalMetric):
    def __init__(self, eps=1e-12, name="cross-entropy", **kw):
        super().__init__(name, **kw)
        self.eps = eps

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l = _np(label).astype(np.int64).flatten()
            p = _np(pred)
            prob = p[np.arange(l.shape[0]), l]
            self.sum_metric += (-np.log(prob + self.eps)).sum()
            self.num_inst += l.shape[0]


@register
class NegativeLogLikelihood(CrossEntropy):
    def __init__(self, eps=1e-12, name="nll-loss", **kw):
        super().__init__(eps=eps, name=name, **kw)


@register
class Perplexity(EvalMetric):
    def __init__(self, ignore_label=None, axis=-1, name="perplexity", **kw):
        super().__init__(name, **kw)
        self.ignore_label = ignore_label
        self.axis = axis

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l = _np(label).astype(np.int64).flatten()
            p = _np(pred).reshape(-1, _np(pred).shape[-1])
            prob = p[np.arange(l.shape[0]), l]
            if self.ignore_label is not None:
                ignore = (l == self.ignore_label)
                prob = prob[~ignore]
            self.sum_metric += (-np.log(np.maximum(prob, 1e-10))).sum()
            self.num_inst += prob.shape[0]

    def get(self):
        if self.num_inst == 0:
            return (self.name, float("nan"))
        return (self.name, float(np.exp(self.sum_metric / self.num_inst)))


@register
class Loss(EvalMetric):
    def __init__(self, name="loss", **kw):
        super().__init__(name, **kw)

    def update(self, _, preds):
        for pred in _as_list(preds):
            loss = _np(pred).sum()
            self.sum_metric += loss
            self.num_inst += _np(pred).size


@register
class PearsonCorrelation(EvalMetric):
    def __init__(self, name="pearsonr", **kw):
        super().__init__(name, **kw)

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            l, p = _np(label).flatten(), _np(pred).flatten()
            self.sum_metric += np.corrcoef(l, p)[0, 1]
            self.num_inst += 1


class CompositeEvalMetric(EvalMetric):
    def __init__(self, metrics=None, name="composite", **kw):
        super().__init__(name, **kw)
        self.metrics = [create(m) for m in (metrics or [])]

    def add(self, metric):
        self.metrics.append(create(metric))

    def update(self, labels, preds):
        for m in self.metrics:
            m.update(labels, preds)

    def reset(self):
        for m in getattr(self, "metrics", []):
            m.reset()

    def get(self):
        names, values = [], []
        for m in self.metrics:
            n, v = m.get()
            names.append(n)
            values.append(v)
        return (names, values)


class CustomMetric(EvalMetric):
    def __init__(self, feval, name="custom", allow_extra_outputs=False, **kw):
        super().__init__(f"custom({name})", **kw)
        self._feval = feval

    def update(self, labels, preds):
        for label, pred in zip(_as_list(labels), _as_list(preds)):
            v = self._feval(_np(label), _np(pred))
            if isinstance(v, tuple):
                s, n = v
                self.sum_metric += s
                self.num_inst += n
            else:
                self.sum_metric += v
                self.num_inst += 1


def np_fn(numpy_feval, name=None, allow_extra_outputs=False):
    name = name or numpy_feval.__name__
    return CustomMetric(numpy_feval, name, allow_extra_outputs)


np_metric = np_fn


@register
class MCC(EvalMetric):
    """Matthews correlation coefficient for binary classification
    (reference: python/mxnet/metric.py — MCC)."""

    def __init__(self, name="mcc", output_names=None, label_names=None):
        super().__init__(name, output_names=output_names,
                         label_names=label_names)
        self._tp = self._tn = self._fp = self._fn = 0

    def reset(self):
        super().reset()
        self._tp = self._tn = self._fp = self._fn = 0

    def update(self, labels, preds):
        labels, preds = _as_list(labels), _as_list(preds)
        for label, pred in zip(labels, preds):
            p = _np(pred)
            pred_cls = p.argmax(axis=-1) if p.ndim > 1 else (p > 0.5)
            lab = _np(label).astype("int64").ravel()
            pred_cls = pred_cls.astype("int64").ravel()
            self._tp += int(((pred_cls == 1) & (lab == 1)).sum())
            self._tn += int(((pred_cls == 0) & (lab == 0)).sum())
            self._fp += int(((pred_cls == 1) & (lab == 0)).sum())
            self._fn += int(((pred_cls == 0) & (lab == 1)).sum())
            self.num_inst += lab.size
        tp, tn, fp, fn = self._tp, self._tn, self._fp, self._fn
        denom = ((tp + fp) * (tp + fn) * (tn + fp) * (tn + fn)) ** 0.5
        self.sum_metric = self.num_inst * (
            (tp * tn - fp * fn) / denom if denom else 0.0)
