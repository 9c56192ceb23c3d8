"""Vectorized elastic-net environment: E independent ENetEnv instances
stepped with ONE kernel launch.

The reference env is strictly single-instance (its reward-clamp comment
at ``elasticnet/enetenv.py:150`` notes clamping is "only useful for
multiple environments" — which it never builds). The MI355X solver
kernels are batched over environments from the start (one workgroup per
environment, ``ops/csrc/enet_solver.hip``), so a vectorized rollout of
E envs costs one ``enet_lbfgs_solve`` + one ``enet_influence`` launch
instead of E python-dispatched steps — the natural way to feed a replay
buffer at device speed. Semantics per instance are identical to
:class:`~smartcal_amd.envs.enet.ENetEnv`.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..ops import enet as enet_ops
from ..utils.device import default_device
from .enet import HIGH, LOW


class VecENetEnv:
    """E independent elastic-net tuning problems, batch-stepped."""

    def __init__(self, E: int, M: int = 5, N: int = 15,
                 reward_clamp: Optional[float] = None,
                 device: Optional[torch.device] = None):
        """``reward_clamp``: optional symmetric clamp on rewards — the
        reference leaves exactly this commented out with the note "only
        useful for multiple environments" (``enetenv.py:150``); this IS
        the multiple-environments case, so it is exposed here (off by
        default for reward parity with the single env)."""
        self.E = int(E)
        self.M = int(M)
        self.N = int(N)
        self.K = 2
        self.reward_clamp = reward_clamp
        self.device = device if device is not None else default_device()
        self.SNR = 0.1
        self.y: Optional[torch.Tensor] = None
        self._sample_problems()

    def _sample_problems(self):
        dev = self.device
        E, N, M = self.E, self.N, self.M
        A = torch.randn(E, N, M, dtype=torch.float32, device=dev)
        self.A = A / A.flatten(1).norm(dim=1).reshape(E, 1, 1)
        self.x0 = torch.zeros(E, M, dtype=torch.float32, device=dev)
        for e in range(E):  # per-env sparse ground truth (enetenv.py:55-60)
            Mo = int(torch.randint(3, M, (1,)).item())
            idx = np.random.randint(0, M, Mo)
            self.x0[e, idx] = torch.randn(Mo, device=dev)
        self.y0 = torch.einsum("enm,em->en", self.A, self.x0)
        self.x = torch.zeros(E, M, dtype=torch.float32, device=dev)

    def _observe_y(self):
        n = torch.randn(self.E, self.N, dtype=torch.float32,
                        device=self.device)
        scale = (self.SNR * self.y0.norm(dim=1)
                 / n.norm(dim=1).clamp(min=1e-12))
        self.y = self.y0 + scale.unsqueeze(1) * n

    def reset(self):
        self._sample_problems()
        self.y = None
        return {"A": self.A.reshape(self.E, -1),
                "eig": torch.zeros(self.E, self.N, dtype=torch.float32,
                                   device=self.device)}

    def step(self, actions, keepnoise: bool = False):
        """actions (E, 2) in [-1, 1] → (obs dict, rewards (E,), done
        (E,) bool, info). Same per-instance semantics as ENetEnv.step."""
        if torch.is_tensor(actions):
            a = actions.detach().to(dtype=torch.float32) \
                .reshape(self.E, self.K)
        else:
            a = torch.as_tensor(np.asarray(actions), dtype=torch.float32) \
                .reshape(self.E, self.K)
        if a.device != self.device:
            a = a.to(self.device)
        a = torch.nan_to_num(a.detach(), nan=0.0, posinf=1.0, neginf=-1.0)
        scaled = a * (HIGH - LOW) / 2 + (HIGH + LOW) / 2
        penalty = -0.1 * ((scaled < LOW).sum(1)
                          + (scaled > HIGH).sum(1)).to(torch.float32)
        rho = scaled.clamp(LOW, HIGH)
        self.rho = rho
        if not keepnoise or self.y is None:
            self._observe_y()
        x, EE, reward = enet_ops.solve_and_influence_batch(
            self.A, self.y, rho, penalty)
        self.x = x
        EE = torch.nan_to_num(EE, nan=0.0, posinf=1e6, neginf=-1e6)
        reward = torch.nan_to_num(reward, nan=-100.0, posinf=1e6,
                                  neginf=-1e6)
        if self.reward_clamp is not None:
            reward = reward.clamp(-self.reward_clamp, self.reward_clamp)
        obs = {"A": self.A.reshape(self.E, -1), "eig": EE}
        done = torch.zeros(self.E, dtype=torch.bool, device=self.device)
        return obs, reward, done, {}

    def solution_error(self) -> torch.Tensor:
        """Per-env relative error vs the ground truth (E,)."""
        return ((self.x0 - self.x).norm(dim=1)
                / self.x0.norm(dim=1).clamp(min=1e-12))
