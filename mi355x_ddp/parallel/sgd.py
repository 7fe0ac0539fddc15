"""FusedSGD: momentum-less SGD matching the reference's
torch.optim.SGD(lr=1e-3) (single_gpu.py:51), fused over the reducer's flat
buckets when the model is engine-wrapped.

- Bucketed params (marked by the Reducer): ONE hand-written HIP kernel per
  bucket does p -= lr*g AND zeroes the flat grad (SURVEY §2.2 N8+N9 —
  optimizer.step + zero_grad in one launch).
- Un-bucketed params: plain per-parameter update (CPU plumbing path and
  models trained without the DDP engine).
"""

from __future__ import annotations

import os
from typing import Iterable

import torch

from .. import ops


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float):
        if lr <= 0:
            raise ValueError(f"invalid lr {lr}")
        super().__init__(params, dict(lr=lr))
        self._flat_pairs = None  # bound by attach_reducer

    def attach_reducer(self, reducer) -> None:
        """Bind the reducer's flat (param, grad) pairs; step() then runs one
        fused kernel per bucket instead of one update per parameter.
        Requires a single param_group: the flat-bucket step applies ONE lr
        to each bucket, and the per-group loop in step() would otherwise
        re-apply every bucket once per group."""
        if len(self.param_groups) != 1:
            raise ValueError("FusedSGD.attach_reducer requires a single "
                             "param_group (flat buckets take one lr)")
        self._flat_pairs = reducer.flat_pairs()
        self._bucketed = {id(p) for b in reducer.buckets for p in b.params}
        self._reducer = reducer

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        # Stream-race assertion (MI355X_DEBUG_SYNC=1): the
        # fused step reads flat_grad on the compute stream — if bucket
        # collectives were launched but finalize() never fenced them, that
        # read races the comm stream. Catch the protocol violation loudly
        # instead of training on half-reduced gradients.
        if (os.environ.get("MI355X_DEBUG_SYNC") == "1"
                and getattr(self, "_reducer", None) is not None
                and self._reducer.unfenced):
            raise RuntimeError(
                "FusedSGD.step() called while bucket all-reduces are "
                "in flight and unfenced — call finalize_backward() (or "
                "Reducer.finalize()) between loss.backward() and "
                "optimizer.step()")
        for group in self.param_groups:
            lr = group["lr"]
            if self._flat_pairs is not None:
                for flat_param, flat_grad in self._flat_pairs:
                    ops.sgd_flat_(flat_param, flat_grad, lr, zero_grad=True)
                for p in group["params"]:
                    if id(p) not in self._bucketed and p.grad is not None:
                        p.add_(p.grad, alpha=-lr)
            else:
                for p in group["params"]:
                    if p.grad is not None:
                        p.add_(p.grad, alpha=-lr)
        return loss

    def zero_grad(self, set_to_none: bool = True):
        if self._flat_pairs is not None:
            # bucketed grads are zeroed by the fused step; nothing to do
            # unless the user zeroes before the first step.
            for _, flat_grad in self._flat_pairs:
                flat_grad.zero_()
            return
        super().zero_grad(set_to_none=set_to_none)
