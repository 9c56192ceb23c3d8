"""Transformer actor/critic networks for dict-observation SAC.

BASELINE.json names "calibenv SAC with transformer actor/critic
(transformer_models.py)" as a target config; the reference's
`transformer_models.py` only ships a supervised classifier whose MHA
attends over the heads of ONE sample (seq len = 1,
`transformer_models.py:105-118`), so the actor/critic variant is built
here as a proper token-sequence encoder: the (M+1, 7) per-direction sky
metadata rows become tokens, the influence-map ConvEncoder embedding is
one more token (and the action, for the critic), standard multi-head
attention mixes them, and the pooled embedding feeds the policy/value
heads. All projections run on the fused-linear HIP kernels
(`ops/csrc/fused_linear.hip`), like `models/transformer.py`.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.distributions import Normal

from ..ops.linear import FusedLinear
from .conv_networks import ConvEncoder, EPS


def _tokens_of(meta: torch.Tensor, token_dim: int) -> torch.Tensor:
    """(B, meta_dim) → (B, T, token_dim); whole-meta single token when
    meta_dim is not a multiple of token_dim."""
    B, D = meta.shape
    if D % token_dim == 0:
        return meta.reshape(B, D // token_dim, token_dim)
    return meta.reshape(B, 1, D)


class TokenMHA(nn.Module):
    """Standard multi-head attention over a token sequence (B, T, d),
    fused-qkv projections on the HIP linear kernels."""

    def __init__(self, d_model: int, num_heads: int):
        super().__init__()
        assert d_model % num_heads == 0
        self.h = num_heads
        self.dh = d_model // num_heads
        self.d = d_model
        self.qkv = FusedLinear(d_model, 3 * d_model, ln=False, act="none")
        self.o = FusedLinear(d_model, d_model, ln=False, act="none")
        nn.init.xavier_uniform_(self.qkv.weight)
        self.qkv.bias.data.fill_(0)
        nn.init.xavier_uniform_(self.o.weight)
        self.o.bias.data.fill_(0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops.attention import scaled_dot_product as _sdp
        B, T, d = x.shape
        qkv = self.qkv(x.reshape(B * T, d)).reshape(B, T, 3, self.h,
                                                    self.dh)
        q, k, v = qkv.permute(2, 0, 3, 1, 4)          # (B, h, T, dh) each
        # fused attention kernel on GPU (one launch per call)
        vals, _ = _sdp(q.contiguous(), k.contiguous(), v.contiguous())
        out = vals.transpose(1, 2).reshape(B * T, d)
        return self.o(out).reshape(B, T, d)


class TokenBlock(nn.Module):
    """Post-LN residual encoder block over tokens (mirrors the layout of
    `models/transformer.py::EncoderBlock`)."""

    def __init__(self, d_model: int, num_heads: int, d_ff: int):
        super().__init__()
        self.attn = TokenMHA(d_model, num_heads)
        self.ff1 = FusedLinear(d_model, d_ff, ln=False, act="none",
                               init_scale=1.0 / d_model ** 0.5)
        self.ff2 = FusedLinear(d_ff, d_model, ln=False, act="none",
                               init_scale=1.0 / d_ff ** 0.5)
        self.norm1 = nn.LayerNorm(d_model)
        self.norm2 = nn.LayerNorm(d_model)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.norm1(x + self.attn(x))
        B, T, d = x.shape
        h = self.ff2(F.relu(self.ff1(x.reshape(B * T, d))))
        return self.norm2(x + h.reshape(B, T, d))


class _ObsEncoder(nn.Module):
    """{img, meta[, action]} → pooled (B, d_model) embedding."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int = 0,
                 token_dim: int = 7, d_model: int = 64, num_heads: int = 4,
                 num_layers: int = 2):
        super().__init__()
        h, w = img_hw
        self.token_dim = token_dim
        tdim = token_dim if meta_dim % token_dim == 0 else meta_dim
        self.enc = ConvEncoder(h, w, act="elu")
        self.img_proj = FusedLinear(self.enc.out_dim, d_model, ln=False,
                                    act="none")
        self.tok_proj = FusedLinear(tdim, d_model, ln=False, act="none")
        self.act_proj = (FusedLinear(n_actions, d_model, ln=False,
                                     act="none") if n_actions else None)
        self.blocks = nn.ModuleList(
            TokenBlock(d_model, num_heads, 2 * d_model)
            for _ in range(num_layers))
        self.d_model = d_model

    def forward(self, img, meta, action=None) -> torch.Tensor:
        B = meta.shape[0]
        toks = _tokens_of(torch.flatten(meta, 1), self.token_dim)
        T = toks.shape[1]
        x = self.tok_proj(toks.reshape(B * T, -1)).reshape(B, T, -1)
        seq = [self.img_proj(self.enc(img)).unsqueeze(1), x]
        if self.act_proj is not None:
            seq.append(self.act_proj(action).unsqueeze(1))
        x = torch.cat(seq, dim=1)
        for blk in self.blocks:
            x = blk(x)
        return x.mean(dim=1)


class TransformerCritic(nn.Module):
    """Q(img, meta, action) with a token-transformer trunk."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int,
                 d_model: int = 64):
        super().__init__()
        self.trunk = _ObsEncoder(img_hw, meta_dim, n_actions,
                                 d_model=d_model)
        self.head = nn.Linear(d_model, 1)
        nn.init.uniform_(self.head.weight, -3e-3, 3e-3)
        nn.init.zeros_(self.head.bias)

    def forward(self, img, meta, action):
        return self.head(self.trunk(img, meta, action))


class SACActorTransformer(nn.Module):
    """Tanh-squashed Gaussian policy with a token-transformer trunk
    (same sampling contract as `conv_networks.SACActorCNN`)."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int,
                 max_action: float = 1.0, d_model: int = 64):
        super().__init__()
        self.max_action = max_action
        self.reparam_noise = EPS
        self.trunk = _ObsEncoder(img_hw, meta_dim, 0, d_model=d_model)
        self.mu = nn.Linear(d_model, n_actions)
        self.sigma = nn.Linear(d_model, n_actions)
        for head in (self.mu, self.sigma):
            nn.init.uniform_(head.weight, -3e-3, 3e-3)
            nn.init.zeros_(head.bias)

    def forward(self, img, meta):
        x = self.trunk(img, meta)
        mu = self.mu(x)
        sigma = torch.clamp(self.sigma(x), min=self.reparam_noise, max=1.0)
        return mu, sigma

    def sample_normal(self, img, meta, reparameterize: bool = True):
        mu, sigma = self.forward(img, meta)
        # validate_args syncs the host (.all() on a GPU bool) —
        # illegal inside hipGraph capture
        # Normal kept for log_prob only (elementwise, capture-safe);
        # the DRAW is explicit reparameterization — the two-tensor
        # torch.normal(loc, scale) that rsample()/sample() lower to is
        # hipErrorStreamCaptureUnsupported on ROCm, randn_like is fine
        dist = Normal(mu, sigma, validate_args=False)
        raw = mu + sigma * torch.randn_like(mu)
        if not reparameterize:
            raw = raw.detach()
        action = torch.tanh(raw) * self.max_action
        log_probs = dist.log_prob(raw) \
            - torch.log(1 - action.pow(2) + self.reparam_noise)
        return action, log_probs.sum(1, keepdim=True)


class DeterministicActorTransformer(nn.Module):
    """tanh deterministic policy with the token-transformer trunk
    (TD3/DDPG counterpart of :class:`SACActorTransformer`)."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int,
                 d_model: int = 64):
        super().__init__()
        self.trunk = _ObsEncoder(img_hw, meta_dim, 0, d_model=d_model)
        self.out = nn.Linear(d_model, n_actions)
        nn.init.uniform_(self.out.weight, -3e-3, 3e-3)
        nn.init.zeros_(self.out.bias)

    def forward(self, img, meta):
        return torch.tanh(self.out(self.trunk(img, meta)))
