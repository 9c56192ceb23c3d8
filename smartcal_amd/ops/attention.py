"""Fused small-sequence attention (N: transformer fwd/bwd kernels).

The reference's attention (`calibration/transformer_models.py:76-118`)
is `scaled_dot_product` = two batched GEMMs + a softmax, applied either
over the HEADS of one sample (supervised classifier: T = num_heads <= 8)
or, in the RL token encoder, over T = M+2 <= 32 sky tokens. Both are
tiny-sequence shapes, so the MI355X design fuses the whole attention —
S = Q K^T / sqrt(dh), row softmax, O = A V — into ONE kernel launch per
call (``csrc/attention.hip``: one workgroup per (batch x head), operands
staged in LDS, GEMM pieces on v_mfma_f32_16x16x4_f32 tiles), and the
full backward (dV/dA/softmax-bwd/dQ/dK) into a second single launch.

CPU tensors (and shapes beyond T=32 / dh=96) run the torch composition,
which doubles as the numerics oracle in ``tests/``.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from . import ext, use_hip

_TMAX = 32
_DMAX = 96


def _torch_sdp(q, k, v):
    d_k = q.size(-1)
    logits = torch.matmul(q, k.transpose(-2, -1)) / math.sqrt(d_k)
    attn = F.softmax(logits, dim=-1)
    return torch.matmul(attn, v), attn


class _AttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v):
        o, a = ext().attn_fwd(q, k, v)
        ctx.save_for_backward(a, q, k, v)
        ctx.set_materialize_grads(False)
        return o, a

    @staticmethod
    def backward(ctx, do, da):
        # da (grad wrt the returned attention map) is unused by every
        # in-repo consumer (the map is returned for inspection only);
        # supporting it would mean a second softmax-backward path.
        assert da is None, \
            "backprop through the returned attention map is unsupported"
        a, q, k, v = ctx.saved_tensors
        if do is None:
            return None, None, None
        dq, dk, dv = ext().attn_bwd(do.contiguous(), a, q, k, v)
        return dq, dk, dv


def scaled_dot_product(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor):
    """Drop-in for the reference's ``scaled_dot_product`` (values, attn).

    Accepts (..., T, dh); the HIP path flattens leading dims to one
    group axis. Falls back to torch off-GPU or beyond the T/dh caps.
    """
    T, dh = q.shape[-2], q.shape[-1]
    if use_hip(q) and T <= _TMAX and dh <= _DMAX and q.dtype == torch.float32:
        lead = q.shape[:-2]
        qf = q.contiguous().reshape(-1, T, dh)
        kf = k.contiguous().reshape(-1, T, dh)
        vf = v.contiguous().reshape(-1, T, dh)
        o, a = _AttnFn.apply(qf, kf, vf)
        return o.reshape(*lead, T, dh), a.reshape(*lead, T, T)
    return _torch_sdp(q, k, v)
