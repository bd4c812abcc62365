"""2-bit threshold gradient compression with error-feedback residual
(reference parity: src/kvstore/gradient_compression.{cc,cu} — the
`kv.set_gradient_compression({'type': '2bit', 'threshold': t})` path).

Semantics per worker push:
  residual += grad
  q = +t where residual >  t ; -t where residual < -t ; else 0
  residual -= q ; send q (2 bits/value on the wire)

The quantize/dequantize math runs as vectorized torch ops (CPU or GPU);
`pack`/`unpack` produce the reference's 16-values-per-u32 wire words for
byte-level parity tests.
"""
from __future__ import annotations

import torch


class GradientCompression:
    def __init__(self, threshold=0.5):
        if threshold <= 0:
            raise ValueError("2bit compression threshold must be > 0")
        self.threshold = float(threshold)
        self._residuals = {}

    # ---------------- worker-side quantize with error feedback ----------
    def compress(self, key, grad: torch.Tensor) -> torch.Tensor:
        """Returns the dequantized form of the 2-bit message (values in
        {-t, 0, +t}) and updates the per-key residual."""
        res = self._residuals.get(key)
        if res is None or res.shape != grad.shape:
            res = torch.zeros_like(grad, dtype=torch.float32)
        res = res + grad.float()
        t = self.threshold
        q = torch.where(res > t, torch.full_like(res, t),
                        torch.where(res < -t, torch.full_like(res, -t),
                                    torch.zeros_like(res)))
        self._residuals[key] = res - q
        return q.to(grad.dtype)

    # ---------------- wire format (16 x 2-bit codes per u32) ------------
    @staticmethod
    def pack(q: torch.Tensor, threshold: float) -> torch.Tensor:
        """codes: 0 = zero, 1 = +t, 2 = -t; little-end first."""
        flat = q.reshape(-1).float()
        codes = torch.zeros(flat.shape, dtype=torch.int64,
                            device=flat.device)
        codes[flat > 0] = 1
        codes[flat < 0] = 2
        n = flat.numel()
        pad = (-n) % 16
        if pad:
            codes = torch.cat([codes, codes.new_zeros(pad)])
        codes = codes.reshape(-1, 16)
        shifts = torch.arange(0, 32, 2, device=flat.device)
        words = (codes << shifts).sum(dim=1)
        return words.to(torch.int64)

    @staticmethod
    def unpack(words: torch.Tensor, numel: int, threshold: float,
               dtype=torch.float32) -> torch.Tensor:
        shifts = torch.arange(0, 32, 2, device=words.device)
        codes = (words.reshape(-1, 1) >> shifts) & 0x3
        vals = torch.zeros(codes.shape, dtype=torch.float32,
                           device=words.device)
        vals[codes == 1] = threshold
        vals[codes == 2] = -threshold
        return vals.reshape(-1)[:numel].to(dtype)
