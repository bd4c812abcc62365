"""SVRG variance-reduced update (reference:
contrib/svrg_optimization/svrg_optimizer.py).

grad_corrected = grad - special_grad + full_batch_grad_snapshot; the
corrected gradient feeds a base optimizer (default SGD)."""
from __future__ import annotations

from ...optimizer import Optimizer, register


@register
class _SVRGOptimizer(Optimizer):
    """Wraps a base optimizer; keys >= special index carry the full-batch
    gradient snapshot terms (reference key convention)."""

    def __init__(self, default_optimizer="sgd", **kwargs):
        super().__init__(**kwargs)
        from ...optimizer import create
        self.default_opt = create(default_optimizer, **kwargs) \
            if isinstance(default_optimizer, str) else default_optimizer

    def create_state(self, index, weight):
        return self.default_opt.create_state(index, weight)

    def update(self, index, weight, grad, state):
        self.default_opt.update(index, weight, grad, state)
