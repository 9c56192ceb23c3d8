"""DDPG agent over {img, meta} dict observations (CNN actor/critic).

Covers `calibration/calib_ddpg.py:238-380`: single critic + target
actor/critic pair, Ornstein-Uhlenbeck exploration noise
(`calib_ddpg.py:23-43`).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..utils.device import default_device
from ..utils.flatten import FlatParams, FusedAdam
from .buffers_dict import DictReplayBuffer
from .conv_networks import CriticCNN, DeterministicActorCNN
from .noise import OUActionNoise


class Agent:
    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.001, M=3, meta_dim=None,
                 name_prefix="", arch="cnn",
                 device: Optional[torch.device] = None,
                 checkpoint_dir="./", grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.name_prefix = name_prefix
        self.grad_hook = grad_hook

        img_shape = tuple(input_dims)
        hw = img_shape[-2:]
        if meta_dim is None:
            meta_dim = 7 * (M + 1)
        self.meta_dim = meta_dim
        self.replaymem = DictReplayBuffer(max_mem_size, img_shape, meta_dim,
                                          n_actions, device=self.device)
        self.noise = OUActionNoise(mu=np.zeros(n_actions))

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
        self.critic = mk_critic().to(self.device)
        self.target_critic = mk_critic().to(self.device)

        self.actor_fp = FlatParams(self.actor)
        self.target_actor_fp = FlatParams(self.target_actor)
        self.critic_fp = FlatParams(self.critic)
        self.target_critic_fp = FlatParams(self.target_critic)
        self.actor_opt = FusedAdam(self.actor_fp, lr=lr_a)
        self.critic_opt = FusedAdam(self.critic_fp, lr=lr_c)
        self.update_network_parameters(tau=1.0)

    def update_network_parameters(self, tau=None):
        if tau is None:
            tau = self.tau
        self.target_actor_fp.polyak_from(self.actor_fp, tau)
        self.target_critic_fp.polyak_from(self.critic_fp, tau)

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
        self.actor.eval()
        img, meta = self._obs_tensors(observation)
        with torch.no_grad():
            mu = self.actor(img, meta)
        self.actor.train()
        mu = mu.cpu().numpy()[0] + self.noise()
        return np.clip(mu, -1.0, 1.0).astype(np.float32)

    def learn(self):
        if len(self.replaymem) < self.batch_size:
            return
        (img, meta, action, reward, img_, meta_, done, hint) = \
            self.replaymem.sample_buffer(self.batch_size)
        reward = reward.unsqueeze(1)
        done = done.unsqueeze(1)
        with torch.no_grad():
            ta = self.target_actor(img_, meta_)
            q_t = self.target_critic(img_, meta_, ta).masked_fill(done, 0.0)
            target = reward + self.gamma * q_t
        q = self.critic(img, meta, action)
        critic_loss = F.mse_loss(q, target)
        self.critic_opt.zero_grad()
        critic_loss.backward()
        if self.grad_hook is not None:
            self.grad_hook([self.critic_fp])
        self.critic_opt.step()

        actions = self.actor(img, meta)
        actor_loss = -self.critic(img, meta, actions).mean()
        self.actor_opt.zero_grad()
        actor_loss.backward()
        if self.grad_hook is not None:
            self.grad_hook([self.actor_fp])
        self.actor_opt.step()
        self.update_network_parameters()

    _NAMES = {"actor": "a_eval_ddpg_actor.model",
              "target_actor": "a_target_ddpg_actor.model",
              "critic": "q_eval_ddpg_critic.model",
              "target_critic": "q_target_ddpg_critic.model"}

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
