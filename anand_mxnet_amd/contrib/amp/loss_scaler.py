"""Dynamic loss scaling (parity with contrib/amp/loss_scaler.py).

bf16 shares fp32's exponent range, so scaling is rarely REQUIRED on this
stack (unlike fp16); it is kept for API parity and for fp16/fp8 paths."""
from __future__ import annotations

import numpy as np

from ... import ndarray as nd


class LossScaler:
    def __init__(self, init_scale=2.0 ** 16, scale_factor=2.0,
                 scale_window=2000, tolerance=0.0):
        self.loss_scale = init_scale
        self._scale_factor = scale_factor
        self._scale_window = scale_window
        self._unskipped = 0

    def has_overflow(self, params):
        """Check grads for inf/nan; called before optimizer step."""
        for p in params:
            if p.grad_req == "null":
                continue
            for g in p.list_grad():
                t = g._t
                import torch
                if not bool(torch.isfinite(t.detach().float().sum())):
                    return True
        return False

    def update_scale(self, overflow: bool):
        if overflow:
            self.loss_scale = max(self.loss_scale / self._scale_factor, 1.0)
            self._unskipped = 0
        else:
            self._unskipped += 1
            if self._unskipped >= self._scale_window:
                self.loss_scale *= self._scale_factor
                self._unskipped = 0
        return not overflow
