"""DDPG agent (MLP) — single critic + target actor/critic pair, OU noise.

Parity with reference ``elasticnet/enet_ddpg.py:192-334``: OU exploration
noise added to the deterministic policy, target networks soft-updated each
learn step, critic regression on the masked bootstrap target, actor ascent
on Q(s, pi(s)).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..envs.enet import obs_to_state
from ..utils.device import default_device
from ..utils.flatten import FlatParams, FusedAdam
from .buffers import ReplayBuffer
from .networks import CriticMLP, DeterministicActorMLP
from .noise import OUActionNoise


class Agent:
    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.001, use_hint=False,
                 device: Optional[torch.device] = None,
                 checkpoint_dir="./", grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.max_action = 1.0
        self.min_action = -1.0
        self.use_hint = use_hint
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.grad_hook = grad_hook

        input_dim = input_dims[0] if isinstance(input_dims, (list, tuple)) \
            else int(input_dims)
        self.replaymem = ReplayBuffer(max_mem_size, [input_dim], n_actions,
                                      device=self.device)
        self.noise = OUActionNoise(mu=np.zeros(n_actions))

        self.actor = DeterministicActorMLP(input_dim,
                                           n_actions).to(self.device)
        self.critic = CriticMLP(input_dim, n_actions).to(self.device)
        self.target_actor = DeterministicActorMLP(input_dim,
                                                  n_actions).to(self.device)
        self.target_critic = CriticMLP(input_dim, n_actions).to(self.device)

        self.actor_fp = FlatParams(self.actor)
        self.critic_fp = FlatParams(self.critic)
        self.target_actor_fp = FlatParams(self.target_actor)
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
        if hint is None:
            hint = np.zeros(self.n_actions, dtype=np.float32)
        self.replaymem.store_transition(obs_to_state(state), action, reward,
                                        obs_to_state(state_), terminal, hint)

    def choose_action(self, observation):
        state = obs_to_state(observation).to(self.device)
        with torch.no_grad():
            mu = self.actor(state).reshape(-1)
        mu_prime = mu + torch.tensor(self.noise(), dtype=torch.float32,
                                     device=self.device)
        return mu_prime.clamp(self.min_action,
                              self.max_action).cpu().numpy()

    def _grad_sync(self, fps):
        if self.grad_hook is not None:
            self.grad_hook(fps)

    def learn(self):
        if self.replaymem.mem_cntr < self.batch_size:
            return
        state, action, reward, new_state, done, _hint = \
            self.replaymem.sample_buffer(self.batch_size)
        state_batch = state.to(self.device)
        new_state_batch = new_state.to(self.device)
        action_batch = action.to(self.device)
        reward_batch = reward.to(self.device)
        terminal_batch = done.to(self.device)

        with torch.no_grad():
            target_actions = self.target_actor(new_state_batch)
            q_ = self.target_critic(new_state_batch, target_actions)
            q_[terminal_batch] = 0.0
            target = (reward_batch + self.gamma * q_.view(-1)) \
                .view(self.batch_size, 1)

        q = self.critic(state_batch, action_batch)
        critic_loss = F.mse_loss(target, q)
        self.critic_opt.zero_grad()
        critic_loss.backward()
        self._grad_sync([self.critic_fp])
        self.critic_opt.step()

        self.actor_opt.zero_grad()
        actor_loss = -torch.mean(
            self.critic(state_batch, self.actor(state_batch)))
        actor_loss.backward()
        self._grad_sync([self.actor_fp])
        self.actor_opt.step()

        self.update_network_parameters()

    def _path(self, name):
        return os.path.join(self.checkpoint_dir, name)

    def _save_net(self, net, fname):
        sd = {k: v.detach().cpu().clone() for k, v in net.state_dict().items()}
        torch.save(sd, self._path(fname))

    def save_models(self):
        self._save_net(self.actor, "a_eval_ddpg_actor.model")
        self._save_net(self.target_actor, "a_target_ddpg_actor.model")
        self._save_net(self.critic, "q_eval_ddpg_critic.model")
        self._save_net(self.target_critic, "q_target_ddpg_critic.model")
        self.replaymem.save_checkpoint(self._path("replaymem_ddpg.model"))

    def load_models(self):
        ml = self.device
        for net, fn in [(self.actor, "a_eval_ddpg_actor.model"),
                        (self.target_actor, "a_target_ddpg_actor.model"),
                        (self.critic, "q_eval_ddpg_critic.model"),
                        (self.target_critic, "q_target_ddpg_critic.model")]:
            net.load_state_dict(torch.load(self._path(fn), map_location=ml))
        try:
            self.replaymem.load_checkpoint(self._path("replaymem_ddpg.model"))
        except FileNotFoundError:
            pass
        self.update_network_parameters(tau=1.0)
