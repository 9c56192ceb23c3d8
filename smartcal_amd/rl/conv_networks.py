"""Convolutional actor/critic networks for image+metadata observations.

Same architecture family as the reference CNN agents
(`calibration/calib_sac.py:90-250`, `demixing_rl/demix_sac.py:372-518`):
3 convs (1→16→32→32, k5 s2, BatchNorm) on the influence map, a 2-layer
FC on metadata (+action for critics), concatenated into a head.

The convolutions are hand-written direct (im2col-free) CDNA4 kernels,
forward and backward (`ops/csrc/conv2d.hip` via `ops.conv.FusedConv2d`)
— these tiny-channel k5/s2 shapes favor direct per-output accumulation
over MFMA implicit GEMM. BatchNorm stays a torch op (cheap reductions,
exact running-stat semantics). The surrounding agents keep parameters
in flat pools so the optimizer / polyak / DP all-reduce are single
fused kernels, and the twin critic/target forwards overlap on side HIP
streams (`utils.streams.StreamFork`).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.distributions import Normal

from ..ops.conv import FusedConv2d

EPS = 1e-6


def _init_layer(layer, w: float | None = None):
    """fan-in uniform init, matching the reference's `init_layer`
    (`calib_sac.py:76-87`)."""
    if w is None:
        fan_in = layer.weight.data.size()[0]
        w = 1.0 / math.sqrt(fan_in)
    nn.init.uniform_(layer.weight.data, -w, w)
    if layer.bias is not None:
        nn.init.uniform_(layer.bias.data, -w, w)


def _conv_out(size, k=5, s=2):
    return (size - (k - 1) - 1) // s + 1


class ConvEncoder(nn.Module):
    """1→16→32→32 k5 s2 conv stack with BatchNorm; ELU or ReLU."""

    def __init__(self, h: int, w: int, act: str = "elu"):
        super().__init__()
        # hand-written direct-conv HIP kernels (ops/csrc/conv2d.hip) on
        # GPU; F.conv2d on CPU. BatchNorm stays a torch op (cheap
        # reductions; running-stat semantics preserved exactly).
        self.conv1 = FusedConv2d(1, 16)
        self.bn1 = nn.BatchNorm2d(16)
        self.conv2 = FusedConv2d(16, 32)
        self.bn2 = nn.BatchNorm2d(32)
        self.conv3 = FusedConv2d(32, 32)
        self.bn3 = nn.BatchNorm2d(32)
        for c in (self.conv1, self.conv2, self.conv3):
            _init_layer(c)
        cw = _conv_out(_conv_out(_conv_out(w)))
        ch = _conv_out(_conv_out(_conv_out(h)))
        self.out_dim = cw * ch * 32
        self.act = F.elu if act == "elu" else F.relu

    def forward(self, x):
        x = self.act(self.bn1(self.conv1(x)))
        x = self.act(self.bn2(self.conv2(x)))
        x = self.act(self.bn3(self.conv3(x)))
        return torch.flatten(x, start_dim=1)


class CriticCNN(nn.Module):
    """Q(s={img, meta}, a) — conv on img + FC(action‖meta) → head
    (`calib_sac.py:90-147`)."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int):
        super().__init__()
        h, w = img_hw
        self.enc = ConvEncoder(h, w, act="relu")
        self.fc1 = nn.Linear(n_actions + meta_dim, 128)
        self.fc2 = nn.Linear(128, 16)
        self.head = nn.Linear(self.enc.out_dim + 16, 1)
        _init_layer(self.fc1)
        _init_layer(self.fc2)
        _init_layer(self.head, 0.003)

    def forward(self, img, meta, action):
        x = self.enc(img)
        y = torch.cat((torch.flatten(action, 1), torch.flatten(meta, 1)), 1)
        y = F.relu(self.fc1(y))
        y = F.relu(self.fc2(y))
        return self.head(torch.cat((x, y), 1))


class SACActorCNN(nn.Module):
    """Tanh-squashed Gaussian policy over {img, meta}
    (`calib_sac.py:156-250`)."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int,
                 max_action: float = 1.0):
        super().__init__()
        h, w = img_hw
        self.max_action = max_action
        self.reparam_noise = EPS
        self.enc = ConvEncoder(h, w, act="elu")
        self.fc11 = nn.Linear(meta_dim, 128)
        self.fc12 = nn.Linear(128, 16)
        self.fc21 = nn.Linear(self.enc.out_dim + 16, 128)
        self.mu = nn.Linear(128, n_actions)
        self.sigma = nn.Linear(128, n_actions)
        for l in (self.fc11, self.fc12, self.fc21):
            _init_layer(l)
        _init_layer(self.mu, 0.003)
        _init_layer(self.sigma, 0.003)

    def forward(self, img, meta):
        x = self.enc(img)
        z = F.relu(self.fc11(torch.flatten(meta, 1)))
        z = F.relu(self.fc12(z))
        x = F.elu(self.fc21(torch.cat((x, z), 1)))
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


class DeterministicActorCNN(nn.Module):
    """tanh deterministic policy for TD3/DDPG CNN agents
    (`calib_td3.py`, `calib_ddpg.py`)."""

    def __init__(self, img_hw, meta_dim: int, n_actions: int):
        super().__init__()
        h, w = img_hw
        self.enc = ConvEncoder(h, w, act="elu")
        self.fc11 = nn.Linear(meta_dim, 128)
        self.fc12 = nn.Linear(128, 16)
        self.fc21 = nn.Linear(self.enc.out_dim + 16, 128)
        self.out = nn.Linear(128, n_actions)
        for l in (self.fc11, self.fc12, self.fc21):
            _init_layer(l)
        _init_layer(self.out, 0.003)

    def forward(self, img, meta):
        x = self.enc(img)
        z = F.relu(self.fc11(torch.flatten(meta, 1)))
        z = F.relu(self.fc12(z))
        x = F.elu(self.fc21(torch.cat((x, z), 1)))
        return torch.tanh(self.out(x))
