"""Toy model: Linear(20, 1) with hand-written MFMA kernels on device.

Parity target: `torch.nn.Linear(20, 1)` (reference single_gpu.py:50,
multigpu.py:67 etc.) — same parameter shapes/names (weight [out,in],
bias [out]) and the same default initialization, so checkpoints are
interchangeable with the reference's.
"""

from __future__ import annotations

import math

import torch
from torch import nn

from .. import ops


class HipLinear(nn.Module):
    """nn.Linear drop-in whose forward/backward run the CDNA4 MFMA kernels
    on device; CPU tensors use torch (plumbing path)."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features)) if bias else None
        self.reset_parameters()

    def reset_parameters(self) -> None:
        # identical to torch.nn.Linear.reset_parameters
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            fan_in = self.in_features
            bound = 1 / math.sqrt(fan_in) if fan_in > 0 else 0
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype != self.weight.dtype:
            x = x.to(self.weight.dtype)  # bf16 models take f32 loader batches
        return ops.linear(x, self.weight, self.bias)

    def extra_repr(self) -> str:
        return (f"in_features={self.in_features}, "
                f"out_features={self.out_features}, bias={self.bias is not None}")


def toy_model(in_features: int = 20, out_features: int = 1) -> HipLinear:
    return HipLinear(in_features, out_features)
