"""Monitor — per-op output statistics hooks (parity with
python/mxnet/monitor.py; installed as Gluon forward hooks instead of the
reference's executor monitor callback)."""
from __future__ import annotations

import re

from .ndarray.ndarray import NDArray

__all__ = ["Monitor"]


class Monitor:
    def __init__(self, interval, stat_func=None, pattern=".*", sort=False):
        self.interval = interval
        self.stat_func = stat_func or (lambda x: x.norm() / (x.size ** 0.5))
        self.re_prog = re.compile(pattern)
        self.sort = sort
        self.queue = []
        self.step = 0
        self.activated = False
        self._installed = []

    def install(self, block):
        """Attach to every child block's forward output."""
        def hook(blk, inputs, output):
            if not self.activated:
                return
            outs = output if isinstance(output, (list, tuple)) else [output]
            for i, o in enumerate(outs):
                if isinstance(o, NDArray) and \
                        self.re_prog.match(blk.name or ""):
                    self.queue.append((self.step, f"{blk.name}_output{i}",
                                       self.stat_func(o)))
        block.apply(lambda b: b.register_forward_hook(hook))

    install_to = install

    def tic(self):
        if self.step % self.interval == 0:
            self.queue = []
            self.activated = True
        self.step += 1

    def toc(self):
        if not self.activated:
            return []
        self.activated = False
        res = []
        for n, k, v in self.queue:
            res.append((n, k, float(v.asscalar()) if isinstance(v, NDArray)
                        else v))
        if self.sort:
            res.sort(key=lambda x: x[1])
        self.queue = []
        return res

    def toc_print(self):
        for n, k, v in self.toc():
            print(f"Batch: {n:7d} {k:30s} {v:.5f}")
