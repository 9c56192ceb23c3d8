"""TD3 agent over {img, meta} dict observations (CNN actor/critic).

Covers `calibration/calib_td3.py:216-405` and
`demixing_rl/demix_td3.py:364-647`: target actor + twin target critics,
target-policy smoothing, delayed actor updates, warmup random actions,
PER with TD-error priority updates, and the adaptive-ADMM hint loop
(Barzilai-Borwein ρ adaptation, `enet_td3.py:310-361` /
`demix_td3.py:546-603`).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..utils.device import default_device
from ..utils.flatten import FlatParams, FusedAdam
from ..utils.streams import StreamFork
from .buffers_dict import DictPERBuffer, DictReplayBuffer
from .conv_networks import CriticCNN, DeterministicActorCNN


class Agent:
    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.005, M=3, meta_dim=None,
                 update_actor_interval=2, warmup=100, noise=0.1,
                 name_prefix="", use_hint=False, prioritized=True,
                 normalize_reward=False, arch="cnn",
                 admm_rho=0.1, device: Optional[torch.device] = None,
                 checkpoint_dir="./", grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.max_action = 1.0
        self.min_action = -1.0
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.name_prefix = name_prefix
        self.grad_hook = grad_hook
        self.update_actor_interval = update_actor_interval
        self.warmup = warmup
        self.noise = noise
        self.learn_step_cntr = 0
        self.time_step = 0

        img_shape = tuple(input_dims)
        hw = img_shape[-2:]
        if meta_dim is None:
            meta_dim = 7 * (M + 1)
        self.meta_dim = meta_dim
        self.prioritized = prioritized
        if prioritized:
            # reference `demix_td3.py:162-166,380`: PER with optional
            # reward standardization over the stored buffer
            self.replaymem = DictPERBuffer(max_mem_size, img_shape,
                                           meta_dim, n_actions,
                                           device=self.device,
                                           normalize_reward=normalize_reward)
        else:
            self.replaymem = DictReplayBuffer(max_mem_size, img_shape,
                                              meta_dim, n_actions,
                                              device=self.device)

        if arch == "transformer":
            from .transformer_networks import (DeterministicActorTransformer,
                                               TransformerCritic)
            mk_actor = lambda: DeterministicActorTransformer(hw, meta_dim,
                                                             n_actions)
            mk_critic = lambda: TransformerCritic(hw, meta_dim, n_actions)
        else:
            mk_actor = lambda: DeterministicActorCNN(hw, meta_dim, n_actions)
            mk_critic = lambda: CriticCNN(hw, meta_dim, n_actions)
        self.actor = mk_actor().to(self.device)
        self.target_actor = mk_actor().to(self.device)
        self.critic_1 = mk_critic().to(self.device)
        self.critic_2 = mk_critic().to(self.device)
        self.target_critic_1 = mk_critic().to(self.device)
        self.target_critic_2 = mk_critic().to(self.device)

        self.actor_fp = FlatParams(self.actor)
        self.target_actor_fp = FlatParams(self.target_actor)
        self.critic_1_fp = FlatParams(self.critic_1)
        self.critic_2_fp = FlatParams(self.critic_2)
        self.target_critic_1_fp = FlatParams(self.target_critic_1)
        self.target_critic_2_fp = FlatParams(self.target_critic_2)
        self.actor_opt = FusedAdam(self.actor_fp, lr=lr_a)
        self.critic_1_opt = FusedAdam(self.critic_1_fp, lr=lr_c)
        self.critic_2_opt = FusedAdam(self.critic_2_fp, lr=lr_c)

        self.use_hint = use_hint
        self._fork = StreamFork(self.device)
        self.admm_rho = admm_rho
        self.Nadmm = 5
        self.adaptive_admm = True
        self.update_network_parameters(tau=1.0)

    # ------------------------------------------------------------------
    def update_network_parameters(self, tau=None):
        if tau is None:
            tau = self.tau
        self.target_actor_fp.polyak_from(self.actor_fp, tau)
        self.target_critic_1_fp.polyak_from(self.critic_1_fp, tau)
        self.target_critic_2_fp.polyak_from(self.critic_2_fp, tau)

    def store_transition(self, state, action, reward, state_, terminal,
                         hint=None):
        self.replaymem.store_transition(state, action, reward, state_,
                                        terminal, hint)

    def _obs_tensors(self, observation):
        img = torch.as_tensor(np.asarray(observation.get(
            "img", observation.get("infmap"))), dtype=torch.float32,
            device=self.device)[None]
        meta_key = "sky" if "sky" in observation else "metadata"
        meta = torch.as_tensor(np.asarray(observation[meta_key]).reshape(-1),
                               dtype=torch.float32, device=self.device)[None]
        return img, meta

    def choose_action(self, observation):
        """Warmup: random normal actions (`enet_td3.py:207-209`)."""
        if self.time_step < self.warmup:
            self.time_step += 1
            mu = np.clip(np.random.normal(scale=0.5, size=self.n_actions),
                         self.min_action, self.max_action)
            return mu.astype(np.float32)
        self.actor.eval()
        img, meta = self._obs_tensors(observation)
        with torch.no_grad():
            mu = self.actor(img, meta)
            mu = mu + torch.clamp(
                torch.randn_like(mu) * self.noise, -0.5, 0.5)
            mu = torch.clamp(mu, self.min_action, self.max_action)
        self.actor.train()
        self.time_step += 1
        return mu.cpu().numpy()[0]

    # ------------------------------------------------------------------
    def learn(self):
        if len(self.replaymem) < self.batch_size:
            return
        if self.prioritized:
            batch, idx, is_w = self.replaymem.sample_buffer(self.batch_size)
            is_w = is_w.unsqueeze(1)
        else:
            batch = self.replaymem.sample_buffer(self.batch_size)
            idx = is_w = None
        (img, meta, action, reward, img_, meta_, done, hint) = batch
        reward = reward.unsqueeze(1)
        done = done.unsqueeze(1)

        with torch.no_grad():
            ta = self.target_actor(img_, meta_)
            # target-policy smoothing (`enet_td3.py:247-251`)
            ta = ta + torch.clamp(torch.randn_like(ta) * 0.2, -0.5, 0.5)
            ta = torch.clamp(ta, self.min_action, self.max_action)
        def _t1():
            with torch.no_grad():
                return self.target_critic_1(img_, meta_, ta)

        def _t2():
            with torch.no_grad():
                return self.target_critic_2(img_, meta_, ta)

        q1, q2, q1_t, q2_t = self._fork(
            lambda: self.critic_1(img, meta, action),
            lambda: self.critic_2(img, meta, action), _t1, _t2)
        with torch.no_grad():
            q_t = torch.min(q1_t, q2_t).masked_fill(done, 0.0)
            target = reward + self.gamma * q_t
        if is_w is not None:
            c_loss = (is_w * (q1 - target).pow(2)).mean() \
                + (is_w * (q2 - target).pow(2)).mean()
        else:
            c_loss = F.mse_loss(q1, target) + F.mse_loss(q2, target)
        self.critic_1_opt.zero_grad()
        self.critic_2_opt.zero_grad()
        c_loss.backward()
        self._fork.join()   # side-stream backwards must land first
        if self.grad_hook is not None:
            self.grad_hook([self.critic_1_fp, self.critic_2_fp])
        self.critic_1_opt.step()
        self.critic_2_opt.step()
        if idx is not None:
            self.replaymem.update_priorities(idx, (q1 - target).detach())

        self.learn_step_cntr += 1
        if self.learn_step_cntr % self.update_actor_interval != 0:
            return

        if not self.use_hint:
            actions = self.actor(img, meta)
            actor_loss = -self.critic_1(img, meta, actions).mean()
            self.actor_opt.zero_grad()
            actor_loss.backward()
            self._fork.join()   # side-stream backwards must land first
            if self.grad_hook is not None:
                self.grad_hook([self.actor_fp])
            self.actor_opt.step()
        else:
            # adaptive-ADMM hint-constrained actor update
            # (`enet_td3.py:310-361`)
            admm_rho = self.admm_rho
            lagrange_y = torch.zeros(self.batch_size * self.n_actions,
                                     device=self.device)
            y0 = None
            a0 = None
            for admm in range(self.Nadmm):
                actions = self.actor(img, meta)
                aloss = -self.critic_1(img, meta, actions).mean()
                penalty = (lagrange_y.view_as(actions) * (actions - hint)
                           ).mean() + admm_rho / 2 * F.mse_loss(actions,
                                                                hint)
                actor_loss = aloss + penalty
                self.actor_opt.zero_grad()
                actor_loss.backward()
                self._fork.join()   # side-stream backwards must land first
                if self.grad_hook is not None:
                    self.grad_hook([self.actor_fp])
                self.actor_opt.step()
                with torch.no_grad():
                    actions = self.actor(img, meta)
                    resid = (actions - hint).view(-1)
                    lagrange_new = lagrange_y + admm_rho * resid
                    if self.adaptive_admm:
                        if admm == 0:
                            y0 = lagrange_new.clone()
                            a0 = resid.clone()
                        elif admm % 3 == 0 and admm < self.Nadmm - 1:
                            dy = lagrange_new - y0
                            da = resid - a0
                            denom = torch.dot(da, da)
                            if float(denom) > 0:
                                a_hat = float(torch.dot(da, dy) / denom)
                                corr = float(
                                    torch.dot(da, dy)
                                    / (torch.linalg.vector_norm(da)
                                       * torch.linalg.vector_norm(dy)
                                       + 1e-12))
                                if (corr > 0.2
                                        and 0.1 * self.admm_rho < a_hat
                                        < 10 * self.admm_rho):
                                    admm_rho = a_hat
                            y0 = lagrange_new.clone()
                            a0 = resid.clone()
                    lagrange_y = lagrange_new
        self.update_network_parameters()

    # -- checkpointing -----------------------------------------------------
    _NAMES = {"actor": "a_eval_td3_actor.model",
              "target_actor": "a_target_td3_actor.model",
              "critic_1": "q_eval_1_td3_critic.model",
              "critic_2": "q_eval_2_td3_critic.model",
              "target_critic_1": "q_target_1_td3_critic.model",
              "target_critic_2": "q_target_2_td3_critic.model"}

    def _path(self, name):
        return os.path.join(self.checkpoint_dir, f"{self.name_prefix}{name}")

    def save_models(self):
        for attr, fname in self._NAMES.items():
            torch.save(getattr(self, attr).state_dict(), self._path(fname))
        self.replaymem.save_checkpoint(self._path(self.replaymem.filename))

    def load_models(self):
        for attr, fname in self._NAMES.items():
            getattr(self, attr).load_state_dict(
                torch.load(self._path(fname), map_location=self.device,
                           weights_only=True))
        self.update_network_parameters(tau=1.0)
