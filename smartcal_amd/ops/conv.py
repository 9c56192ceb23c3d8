"""Direct-conv op (k5, s2, no padding) — the CNN agents' conv layers.

Hand-written CDNA4 kernels (`ops/csrc/conv2d.hip`) on the
GPU path; `F.conv2d` on CPU (also the numerics oracle). `FusedConv2d`
is a drop-in for ``nn.Conv2d(cin, cout, kernel_size=5, stride=2)`` with
the same parameter names, so state_dicts interchange.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from . import ext, use_hip

__all__ = ["conv2d_k5s2", "FusedConv2d"]


class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W, b):
        y = ext().conv2d_k5s2_fwd(x, W, b)
        ctx.save_for_backward(x, W)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, W = ctx.saved_tensors
        dx, dW, db = ext().conv2d_k5s2_bwd(dy.contiguous(), x, W)
        return dx, dW, (db if ctx.has_bias else None)


def conv2d_k5s2(x: torch.Tensor, W: torch.Tensor,
                b: Optional[torch.Tensor] = None) -> torch.Tensor:
    if use_hip(x):
        return _Conv2dFn.apply(x.contiguous(), W.contiguous(),
                               b.contiguous() if b is not None else None)
    return F.conv2d(x, W, b, stride=2)


class FusedConv2d(torch.nn.Module):
    """Conv2d(k=5, s=2) over the direct HIP kernel; nn.Conv2d-compatible
    parameters (weight, bias) and default init."""

    def __init__(self, in_channels: int, out_channels: int):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.weight = torch.nn.Parameter(
            torch.empty(out_channels, in_channels, 5, 5))
        self.bias = torch.nn.Parameter(torch.empty(out_channels))
        # nn.Conv2d default init (kaiming_uniform a=sqrt(5) + fan-in bias)
        torch.nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        fan_in = in_channels * 25
        bound = 1.0 / math.sqrt(fan_in)
        torch.nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return conv2d_k5s2(x, self.weight, self.bias)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size=5, stride=2")
