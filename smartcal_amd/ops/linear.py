"""Fused Linear(+LayerNorm)(+ELU) — the actor/critic MLP hot op (N1).

The reference's MLP layers are ``F.elu(LayerNorm(Linear(x)))`` chains
(reference ``elasticnet/enet_sac.py:436-444`` actor, ``:382-394`` critic).
Here the whole layer is ONE CDNA4 HIP kernel on the forward path: an
MFMA(f32)-tiled GEMM with LDS-staged X/W tiles whose epilogue computes the
row mean/variance with wave ``shfl_xor`` reductions and applies the affine
LayerNorm + ELU before a single store — no separate normalisation kernel, no
extra HBM round-trip for the pre-activations (gfx950 has exact f32-input MFMA
at the f32 vector rate; there is no TF32 on CDNA4).

Backward is three kernels: (1) ELU'+LayerNorm backward with fused
dgamma/dbeta column reductions, (2) dX = dZ @ W, (3) dW = dZ^T @ X + db —
both MFMA f32 GEMMs.

CPU tensors run the equivalent torch composition (also the test oracle).

CONTRACT RESTRICTION (deliberate): on the HIP path the backward
accumulates parameter gradients straight into ``.grad`` (the agents' flat
gradient pools) and returns ``None`` for the W/b/gamma/beta grad outputs.
This removes every per-parameter AccumulateGrad/zero kernel from the
learn graph, but it means ``torch.autograd.grad(loss, params)`` and
double-backward (``create_graph=True`` — e.g.
``autograd_tools.hessian_vec_prod``) are NOT supported through the HIP
fused path; influence/HVP tooling uses plain ``nn.Linear`` models, which
is what every in-repo consumer of ``autograd_tools`` does. The CPU path
has no such restriction.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from . import ext, use_hip

ACT_NONE = 0
ACT_ELU = 1
ACT_RELU = 2
ACT_TANH = 3

_ACT_CODES = {"none": ACT_NONE, "elu": ACT_ELU, "relu": ACT_RELU,
              "tanh": ACT_TANH}

# Process-wide compute dtype for the fused-linear family ("fp32" | "bf16").
# bf16 keeps fp32 master weights, fp32 tensors and fp32 LayerNorm/epilogue;
# only the GEMM multiplies run on the bf16 MFMA pipe
# (v_mfma_f32_16x16x32_bf16, fp32 accumulate) — BASELINE config 2.
_COMPUTE_DTYPE = "fp32"


def set_compute_dtype(dtype: str) -> None:
    global _COMPUTE_DTYPE
    assert dtype in ("fp32", "bf16"), dtype
    _COMPUTE_DTYPE = dtype


def get_compute_dtype() -> str:
    return _COMPUTE_DTYPE


_scratch: dict = {}


def _scratch_buf(n: int, device) -> torch.Tensor:
    """Reusable throwaway accumulation buffer for dgamma/dbeta when the
    parameter grads are not wanted (frozen-critic actor phase): avoids
    two at::zeros fill kernels per layer per backward."""
    key = (device.type, device.index)
    buf = _scratch.get(key)
    if buf is None or buf.numel() < n:
        buf = torch.empty(max(n, 1024), device=device)
        _scratch[key] = buf
    return buf[:n]


def _grad_view(p: torch.Tensor) -> torch.Tensor:
    """The tensor the kernels accumulate this parameter's gradient into.

    Agents bind each parameter's .grad to a slice of one flat pool
    (``utils.flatten.FlatParams``), so the backward kernels can write
    with += directly — no per-parameter autograd zero/add kernels in the
    learn graph. Standalone tensors (tests) get a zeroed .grad created
    on first use, which matches autograd's accumulate semantics.
    """
    if p.grad is None:
        p.grad = torch.zeros_like(p)
    return p.grad


class _FusedLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W, b, gamma, beta, act_code: int, with_ln: bool):
        bf16 = _COMPUTE_DTYPE == "bf16"
        fwd = ext().fused_linear_bf16_fwd if bf16 \
            else ext().fused_linear_fwd
        y, zhat, rstd = fwd(x, W, b, gamma, beta, act_code, with_ln)
        ctx.save_for_backward(x, W, gamma, zhat, rstd, y)
        # python refs to the parameter objects for direct grad
        # accumulation (not part of the autograd graph)
        ctx.param_refs = (W, b, gamma, beta)
        ctx.act_code = act_code
        ctx.with_ln = with_ln
        ctx.bf16 = bf16
        return y

    @staticmethod
    def backward(ctx, dy):
        x, W, gamma, zhat, rstd, y = ctx.saved_tensors
        Wp, bp, gp, bep = ctx.param_refs
        dy = dy.contiguous()
        want_w = ctx.needs_input_grad[1]
        if ctx.with_ln and want_w:
            dz = ext().fused_linear_bwd_dz_into(
                dy, y, zhat, rstd, gamma, ctx.act_code, True,
                _grad_view(gp), _grad_view(bep))
        elif ctx.with_ln:
            # grads not wanted (frozen params): accumulate into a
            # reusable scratch buffer instead of fresh zeros
            n = dy.shape[1]
            sb = _scratch_buf(2 * n, dy.device)
            dz = ext().fused_linear_bwd_dz_into(
                dy, y, zhat, rstd, gamma, ctx.act_code, True,
                sb[:n], sb[n:2 * n])
        else:
            dz, _, _ = ext().fused_linear_bwd_dz(
                dy, y, zhat, rstd, gamma, ctx.act_code, False)
        if getattr(ctx, "bf16", False):
            dx = ext().mfma_gemm_nn_bf16(dz, W)
            if want_w:
                ext().mfma_gemm_tn_bias_into_bf16(dz, x, _grad_view(Wp),
                                                  _grad_view(bp))
        else:
            dx = ext().mfma_gemm_nn(dz, W)      # (B,N) @ (N,K) -> (B,K)
            if want_w:
                # dW = dz^T @ x and db = col-sum(dz), accumulated straight
                # into the flat gradient pool by the kernels
                ext().mfma_gemm_tn_bias_into(dz, x, _grad_view(Wp),
                                             _grad_view(bp))
        return dx, None, None, None, None, None, None


def fused_linear(x: torch.Tensor, W: torch.Tensor,
                 b: Optional[torch.Tensor] = None,
                 gamma: Optional[torch.Tensor] = None,
                 beta: Optional[torch.Tensor] = None,
                 act: str = "none") -> torch.Tensor:
    """y = act(LayerNorm(x @ W.T + b; gamma, beta)).

    LayerNorm is applied iff gamma is not None. ``x``: (B, K) or (K,);
    ``W``: (N, K). fp32.
    """
    squeeze = x.dim() == 1
    if squeeze:
        x = x.unsqueeze(0)
    with_ln = gamma is not None
    act_code = _ACT_CODES[act]
    if use_hip(x):
        y = _FusedLinearFn.apply(x.contiguous(), W, b, gamma, beta,
                                 act_code, with_ln)
    else:
        z = F.linear(x, W, b)
        if with_ln:
            z = F.layer_norm(z, (W.shape[0],), gamma, beta)
        if act == "elu":
            z = F.elu(z)
        elif act == "relu":
            z = F.relu(z)
        elif act == "tanh":
            z = torch.tanh(z)
        y = z
    return y.squeeze(0) if squeeze else y


class FusedLinear(torch.nn.Module):
    """Linear(+LayerNorm)(+activation) module over the fused kernel.

    Weight/bias init follows the reference's fan-based uniform init
    (``enet_sac.py:18-22``: sc = 1/sqrt(weight.size(0)) unless given).
    """

    def __init__(self, in_features: int, out_features: int,
                 ln: bool = True, act: str = "elu",
                 init_scale: Optional[float] = None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.act = act
        sc = init_scale if init_scale is not None else 1.0 / (out_features ** 0.5)
        self.weight = torch.nn.Parameter(
            torch.empty(out_features, in_features).uniform_(-sc, sc))
        self.bias = torch.nn.Parameter(
            torch.empty(out_features).uniform_(-sc, sc))
        if ln:
            self.ln_weight = torch.nn.Parameter(torch.ones(out_features))
            self.ln_bias = torch.nn.Parameter(torch.zeros(out_features))
        else:
            self.register_parameter("ln_weight", None)
            self.register_parameter("ln_bias", None)

    def forward(self, x):
        return fused_linear(x, self.weight, self.bias,
                            self.ln_weight, self.ln_bias, self.act)

    def extra_repr(self):
        return (f"in={self.in_features}, out={self.out_features}, "
                f"ln={self.ln_weight is not None}, act={self.act}")


class _FusedChainFn(torch.autograd.Function):
    """Whole-chain forward (one kernel) with per-layer direct-accumulate
    backward. flat params layout: [W, b, gamma, beta] per layer."""

    @staticmethod
    def forward(ctx, x, acts, *params):
        L = len(acts)
        Ws = [params[4 * l + 0] for l in range(L)]
        bs = [params[4 * l + 1] for l in range(L)]
        gs = [params[4 * l + 2] for l in range(L)]
        bes = [params[4 * l + 3] for l in range(L)]
        bf16 = _COMPUTE_DTYPE == "bf16"
        ys, zhats, rstds = ext().mlp_chain_fwd(
            x, Ws, bs, gs, bes, [int(a) for a in acts], bf16)
        ctx.save_for_backward(x, *ys, *zhats, *rstds)
        ctx.param_refs = params
        ctx.acts = acts
        ctx.L = L
        ctx.bf16 = bf16
        return ys[-1]

    @staticmethod
    def backward(ctx, dy):
        L = ctx.L
        saved = ctx.saved_tensors
        x = saved[0]
        ys = saved[1:1 + L]
        zhats = saved[1 + L:1 + 2 * L]
        rstds = saved[1 + 2 * L:1 + 3 * L]
        params = ctx.param_refs
        dy = dy.contiguous()
        for l in range(L - 1, -1, -1):
            W, b, g, be = params[4 * l:4 * l + 4]
            want_w = ctx.needs_input_grad[2 + 4 * l]
            if want_w:
                dz = ext().fused_linear_bwd_dz_into(
                    dy, ys[l], zhats[l], rstds[l], g, ctx.acts[l], True,
                    _grad_view(g), _grad_view(be))
            else:
                n = dy.shape[1]
                sb = _scratch_buf(2 * n, dy.device)
                dz = ext().fused_linear_bwd_dz_into(
                    dy, ys[l], zhats[l], rstds[l], g, ctx.acts[l], True,
                    sb[:n], sb[n:2 * n])
            x_l = x if l == 0 else ys[l - 1]
            if getattr(ctx, "bf16", False):
                if want_w:
                    ext().mfma_gemm_tn_bias_into_bf16(dz, x_l,
                                                      _grad_view(W),
                                                      _grad_view(b))
                dy = ext().mfma_gemm_nn_bf16(dz, W)
            else:
                if want_w:
                    ext().mfma_gemm_tn_bias_into(dz, x_l, _grad_view(W),
                                                 _grad_view(b))
                dy = ext().mfma_gemm_nn(dz, W)
        return (dy, None) + (None,) * (4 * L)


def fused_chain(x: torch.Tensor, layers) -> torch.Tensor:
    """Run a stack of LN+act FusedLinear layers as ONE kernel on GPU
    (activations stay in LDS between layers); CPU falls back to the
    per-layer path."""
    squeeze = x.dim() == 1
    if squeeze:
        x = x.unsqueeze(0)
    if use_hip(x) and len(layers) >= 2 \
            and all(m.ln_weight is not None for m in layers):
        acts = tuple(_ACT_CODES[m.act] for m in layers)
        params = []
        for m in layers:
            params += [m.weight, m.bias, m.ln_weight, m.ln_bias]
        y = _FusedChainFn.apply(x.contiguous(), acts, *params)
    else:
        y = x
        for m in layers:
            y = m(y)
    return y.squeeze(0) if squeeze else y
