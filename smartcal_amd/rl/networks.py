"""Actor/critic MLPs over the fused HIP linear kernels (N1).

Architectures mirror the reference agents' nets:

* SAC/TD3/DDPG actor (MLP): in -> 512 -> 256 -> 128 -> heads, each hidden
  layer Linear+LayerNorm+ELU fused into one kernel
  (reference ``elasticnet/enet_sac.py:407-466``). SAC heads (mu, logsigma)
  are a single fused (128 -> 2A) projection split on output; TD3/DDPG head
  is a tanh projection.
* Critic: state branch in->512->256, action branch A->128->64, concat,
  linear head to 1 (reference ``enet_sac.py:352-394``).

Weight init follows the reference's ``init_layer`` (uniform +-1/sqrt(fan),
0.003 for heads, ``enet_sac.py:18-22``).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops.linear import FusedLinear, fused_linear, fused_chain
from ..ops.sampling import tanh_gauss_sample

LOGSIG_MIN = -20.0
LOGSIG_MAX = 2.0


class SACActorMLP(nn.Module):
    def __init__(self, input_dim: int, n_actions: int, max_action: float = 1.0):
        super().__init__()
        self.n_actions = n_actions
        self.max_action = max_action
        self.l1 = FusedLinear(input_dim, 512)
        self.l2 = FusedLinear(512, 256)
        self.l3 = FusedLinear(256, 128)
        self.head = FusedLinear(128, 2 * n_actions, ln=False, act="none",
                                init_scale=0.003)

    def forward(self, x: torch.Tensor):
        # whole hidden stack as ONE chain kernel (activations in LDS)
        h = fused_chain(x, (self.l1, self.l2, self.l3))
        out = self.head(h)
        mu = out[..., :self.n_actions]
        logsigma = out[..., self.n_actions:].clamp(LOGSIG_MIN, LOGSIG_MAX)
        return mu, logsigma

    def sample_normal(self, state: torch.Tensor, reparameterize: bool = True):
        mu, logsigma = self.forward(state)
        if mu.dim() == 1:
            mu = mu.unsqueeze(0)
            logsigma = logsigma.unsqueeze(0)
        return tanh_gauss_sample(mu, logsigma, self.max_action,
                                 reparameterize)


class DeterministicActorMLP(nn.Module):
    """TD3/DDPG actor: tanh-squashed deterministic policy."""

    def __init__(self, input_dim: int, n_actions: int):
        super().__init__()
        self.l1 = FusedLinear(input_dim, 512)
        self.l2 = FusedLinear(512, 256)
        self.l3 = FusedLinear(256, 128)
        self.head = FusedLinear(128, n_actions, ln=False, act="tanh",
                                init_scale=0.003)

    def forward(self, x: torch.Tensor):
        return self.head(fused_chain(x, (self.l1, self.l2, self.l3)))


class CriticMLP(nn.Module):
    def __init__(self, input_dim: int, n_actions: int):
        super().__init__()
        self.s1 = FusedLinear(input_dim, 512)
        self.s2 = FusedLinear(512, 256)
        self.a1 = FusedLinear(n_actions, 128)
        self.a2 = FusedLinear(128, 64)
        self.head = FusedLinear(256 + 64, 1, ln=False, act="none",
                                init_scale=0.003)

    def forward(self, state: torch.Tensor, action: torch.Tensor):
        x = fused_chain(state, (self.s1, self.s2))
        y = fused_chain(action, (self.a1, self.a2))
        z = torch.cat((x, y), dim=-1)
        return self.head(z)
