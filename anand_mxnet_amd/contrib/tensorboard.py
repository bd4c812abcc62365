"""TensorBoard logging hook (reference parity:
python/mxnet/contrib/tensorboard.py — LogMetricsCallback).

The reference depends on the external `tensorboard` SummaryWriter; this
image has no tensorboard package, so the callback degrades to a
structured JSONL event log that tensorboard-compatible tooling (or the
estimator's LoggingHandler) can consume.  If a `tensorboardX`/`torch`
SummaryWriter is importable it is used directly.
"""
from __future__ import annotations

import json
import os
import time


class LogMetricsCallback:
    """Batch/epoch callback: logs eval metrics to an event file.

    Reference usage pattern::

        mod.fit(..., batch_end_callback=LogMetricsCallback('logs/train'))
    """

    def __init__(self, logging_dir, prefix=None):
        self.prefix = prefix
        self._writer = None
        os.makedirs(logging_dir, exist_ok=True)
        try:
            from torch.utils.tensorboard import SummaryWriter
            self._writer = SummaryWriter(logging_dir)
        except Exception:
            self._path = os.path.join(logging_dir, "events.jsonl")
        self._step = 0

    def __call__(self, param):
        if param.eval_metric is None:
            return
        self._step += 1
        for name, value in param.eval_metric.get_name_value():
            if self.prefix is not None:
                name = f"{self.prefix}-{name}"
            if self._writer is not None:
                self._writer.add_scalar(name, value, self._step)
            else:
                with open(self._path, "a") as f:
                    f.write(json.dumps({"time": time.time(), "step":
                                        self._step, "tag": name,
                                        "value": float(value)}) + "\n")
