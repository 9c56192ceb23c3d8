"""Fused tanh-squashed Gaussian sampling + log-prob (N3).

One elementwise HIP kernel computes, from (mu, logsigma, eps):
action = max_action * tanh(mu + sigma*eps) and the squashed log-probability
log N(z; mu, sigma) - log(max_action*(1 - tanh(z)^2) + 1e-6) row-summed —
the reference composes ~10 torch ops for this per actor call
(reference ``elasticnet/enet_sac.py:446-466``). Backward is a second fused
elementwise kernel (reparameterized path).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from . import ext, use_hip

_LOG_SQRT_2PI = 0.5 * math.log(2.0 * math.pi)
REPARAM_NOISE = 1e-6


class _TanhGaussFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, mu, logsigma, eps, max_action: float):
        action, logprob, a_t = ext().tanh_gauss_fwd(mu, logsigma, eps,
                                                    max_action)
        ctx.save_for_backward(logsigma, eps, a_t)
        ctx.max_action = max_action
        return action, logprob

    @staticmethod
    def backward(ctx, daction, dlogprob):
        logsigma, eps, a_t = ctx.saved_tensors
        dmu, dlogsigma = ext().tanh_gauss_bwd(
            daction.contiguous(), dlogprob.contiguous(), logsigma, eps, a_t,
            ctx.max_action)
        return dmu, dlogsigma, None, None


def tanh_gauss_sample(mu: torch.Tensor, logsigma: torch.Tensor,
                      max_action: float = 1.0,
                      reparameterize: bool = True,
                      eps: Optional[torch.Tensor] = None
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sample a = M*tanh(mu + sigma*eps) and its squashed log-prob.

    Matches the reference actor's ``sample_normal``: the non-reparameterized
    path detaches the sample from the graph (``.sample()`` semantics).
    Returns (action (B,A), logprob (B,1)).
    """
    squeeze = mu.dim() == 1
    if squeeze:
        mu = mu.unsqueeze(0)
        logsigma = logsigma.unsqueeze(0)
    if eps is None:
        eps = torch.randn_like(mu)
    if use_hip(mu):
        mu_c = mu.contiguous()
        logsigma_c = logsigma.contiguous()
        if reparameterize:
            action, logprob = _TanhGaussFn.apply(mu_c, logsigma_c, eps,
                                                 float(max_action))
        else:
            with torch.no_grad():
                action, logprob, _ = ext().tanh_gauss_fwd(
                    mu_c, logsigma_c, eps, float(max_action))
    else:
        sigma = logsigma.exp()
        if reparameterize:
            z = mu + sigma * eps
        else:
            z = (mu + sigma * eps).detach()
        a_t = torch.tanh(z)
        action = a_t * max_action
        log_probs = (-0.5 * ((z - mu) / sigma) ** 2 - logsigma
                     - _LOG_SQRT_2PI)
        log_probs = log_probs - torch.log(
            max_action * (1.0 - a_t.pow(2)) + REPARAM_NOISE)
        logprob = log_probs.sum(1, keepdim=True)
    return action, logprob
