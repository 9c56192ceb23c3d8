"""TD3 agent (MLP) — target actor + twin target critics, policy smoothing,
delayed actor updates, optional PER and adaptive-ADMM hint constraint.

Algorithm parity with reference ``elasticnet/enet_td3.py:124-405``:
warmup-phase random actions, exploration noise + clamp in
``choose_action`` (``:207-218``), target-policy smoothing noise
clamp(N(0,0.2), -0.5, 0.5) (``:247-251``), terminal masking, PER priorities
from TD error (``:263-269``), delayed actor update every
``update_actor_interval`` steps, and the hint path's adaptive-ADMM inner
loop (Nadmm=5) with Barzilai-Borwein-style rho adaptation and correlation
gate (``:310-361``).
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
from .buffers import PERBuffer, ReplayBuffer
from .networks import CriticMLP, DeterministicActorMLP


class Agent:
    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.001, update_actor_interval=2,
                 warmup=1000, noise=0.1, prioritized=False, use_hint=False,
                 device: Optional[torch.device] = None,
                 checkpoint_dir="./", grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.max_action = 1.0
        self.min_action = -1.0
        self.learn_step_cntr = 0
        self.time_step = 0
        self.warmup = warmup
        self.update_actor_interval = update_actor_interval
        self.prioritized = prioritized
        self.use_hint = use_hint
        self.admm_rho = 0.1
        self.Nadmm = 5
        self.adaptive_admm = True
        self.corr_min = 0.5
        self.noise = noise
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.grad_hook = grad_hook

        input_dim = input_dims[0] if isinstance(input_dims, (list, tuple)) \
            else int(input_dims)
        buf_cls = PERBuffer if prioritized else ReplayBuffer
        self.replaymem = buf_cls(max_mem_size, [input_dim], n_actions,
                                 device=self.device)

        self.actor = DeterministicActorMLP(input_dim,
                                           n_actions).to(self.device)
        self.critic_1 = CriticMLP(input_dim, n_actions).to(self.device)
        self.critic_2 = CriticMLP(input_dim, n_actions).to(self.device)
        self.target_actor = DeterministicActorMLP(input_dim,
                                                  n_actions).to(self.device)
        self.target_critic_1 = CriticMLP(input_dim, n_actions).to(self.device)
        self.target_critic_2 = CriticMLP(input_dim, n_actions).to(self.device)

        self.actor_fp = FlatParams(self.actor)
        self.critic_1_fp = FlatParams(self.critic_1)
        self.critic_2_fp = FlatParams(self.critic_2)
        self.target_actor_fp = FlatParams(self.target_actor)
        self.target_critic_1_fp = FlatParams(self.target_critic_1)
        self.target_critic_2_fp = FlatParams(self.target_critic_2)
        self.actor_opt = FusedAdam(self.actor_fp, lr=lr_a)
        self.critic_1_opt = FusedAdam(self.critic_1_fp, lr=lr_c)
        self.critic_2_opt = FusedAdam(self.critic_2_fp, lr=lr_c)

        self.update_network_parameters(tau=1.0)

    def update_network_parameters(self, tau=None):
        if tau is None:
            tau = self.tau
        self.target_actor_fp.polyak_from(self.actor_fp, tau)
        self.target_critic_1_fp.polyak_from(self.critic_1_fp, tau)
        self.target_critic_2_fp.polyak_from(self.critic_2_fp, tau)

    def store_transition(self, state, action, reward, state_, terminal,
                         hint):
        self.replaymem.store_transition(obs_to_state(state), action, reward,
                                        obs_to_state(state_), terminal, hint)

    def choose_action(self, observation):
        if self.time_step < self.warmup:
            mu = torch.tensor(np.random.normal(scale=self.noise,
                                               size=(self.n_actions,)),
                              dtype=torch.float32, device=self.device)
        else:
            state = obs_to_state(observation).to(self.device)
            with torch.no_grad():
                mu = self.actor(state).reshape(-1)
        noise = torch.tensor(np.random.normal(scale=self.noise,
                                              size=(self.n_actions,)),
                             dtype=torch.float32, device=self.device)
        mu_prime = (mu + noise).clamp(self.min_action, self.max_action)
        self.time_step += 1
        return mu_prime.cpu().numpy()

    def choose_action_tensor(self, observation):
        """Device-resident action (no host sync) — the bench/vectorized
        rollout path; ``choose_action`` keeps the reference's numpy API."""
        if self.time_step < self.warmup:
            mu = torch.randn(self.n_actions, device=self.device) * self.noise
        else:
            state = obs_to_state(observation).to(self.device)
            with torch.no_grad():
                mu = self.actor(state).reshape(-1)
        noise = torch.randn(self.n_actions, device=self.device) * self.noise
        self.time_step += 1
        return (mu + noise).clamp(self.min_action, self.max_action)

    def _grad_sync(self, fps):
        if self.grad_hook is not None:
            self.grad_hook(fps)

    def learn(self):
        if self.replaymem.mem_cntr < self.batch_size:
            return

        if self.prioritized:
            (state, action, reward, new_state, done, hint), idxs, is_w = \
                self.replaymem.sample_buffer(self.batch_size)
            is_weight = is_w.to(self.device).unsqueeze(1)
        else:
            state, action, reward, new_state, done, hint = \
                self.replaymem.sample_buffer(self.batch_size)

        state_batch = state.to(self.device)
        new_state_batch = new_state.to(self.device)
        action_batch = action.to(self.device)
        reward_batch = reward.to(self.device)
        terminal_batch = done.to(self.device)
        hint_batch = hint.to(self.device)

        with torch.no_grad():
            target_actions = self.target_actor(new_state_batch)
            smooth = torch.clamp(
                torch.randn((), device=self.device) * 0.2, -0.5, 0.5)
            target_actions = (target_actions + smooth).clamp(
                self.min_action, self.max_action)
            q1_ = self.target_critic_1(new_state_batch, target_actions)
            q2_ = self.target_critic_2(new_state_batch, target_actions)
            q1_[terminal_batch] = 0.0
            q2_[terminal_batch] = 0.0
            critic_value_ = torch.min(q1_.view(-1), q2_.view(-1))
            target = (reward_batch + self.gamma * critic_value_) \
                .view(self.batch_size, 1)

        if self.prioritized:
            with torch.no_grad():
                e1 = (self.critic_1(state_batch, action_batch)
                      - target).abs()
                e2 = (self.critic_2(state_batch, action_batch)
                      - target).abs()
            self.replaymem.update_priorities(idxs, 0.5 * (e1 + e2))

        q1 = self.critic_1(state_batch, action_batch)
        q2 = self.critic_2(state_batch, action_batch)
        if self.prioritized:
            critic_loss = (is_weight * (q1 - target).pow(2)).mean() \
                + (is_weight * (q2 - target).pow(2)).mean()
        else:
            critic_loss = F.mse_loss(target, q1) + F.mse_loss(target, q2)
        self.critic_1_opt.zero_grad()
        self.critic_2_opt.zero_grad()
        critic_loss.backward()
        self._grad_sync([self.critic_1_fp, self.critic_2_fp])
        self.critic_1_opt.step()
        self.critic_2_opt.step()

        self.learn_step_cntr += 1
        if self.learn_step_cntr % self.update_actor_interval != 0:
            return

        if not self.use_hint:
            self.actor_opt.zero_grad()
            q1_pi = self.critic_1(state_batch, self.actor(state_batch))
            if self.prioritized:
                actor_loss = -torch.mean(q1_pi * is_weight)
            else:
                actor_loss = -torch.mean(q1_pi)
            actor_loss.backward()
            self._grad_sync([self.actor_fp])
            self.actor_opt.step()
        else:
            # augmented-Lagrangian / adaptive-ADMM hint constraint
            lagrange_y = torch.zeros(hint_batch.numel(), device=self.device)
            lagrange_y0 = None
            actions0 = None
            admm_rho = self.admm_rho
            for admm in range(self.Nadmm):
                self.actor_opt.zero_grad()
                actions = self.actor(state_batch)
                q1_pi = self.critic_1(state_batch, actions)
                if self.prioritized:
                    actor_loss = -torch.mean(q1_pi * is_weight)
                else:
                    actor_loss = -torch.mean(q1_pi)
                diff1 = (actions - hint_batch).view(-1)
                pen = (torch.dot(lagrange_y, diff1)
                       + admm_rho / 2 * F.mse_loss(actions, hint_batch))
                if self.prioritized:
                    loss1 = (pen * is_weight).mean() / actions.numel()
                else:
                    loss1 = pen.mean() / actions.numel()
                (actor_loss + loss1).backward()
                self._grad_sync([self.actor_fp])
                self.actor_opt.step()
                with torch.no_grad():
                    lagrange_y = lagrange_y \
                        + admm_rho * (actions - hint_batch).view(-1)
                    if self.adaptive_admm:
                        if admm == 0:
                            lagrange_y0 = actions.view(-1).detach().clone()
                            actions0 = actions.view(-1).detach().clone()
                        elif admm % 3 == 0 and admm < self.Nadmm - 1:
                            ly1 = lagrange_y + admm_rho \
                                * (actions - hint_batch).view(-1)
                            dy = ly1 - lagrange_y0
                            du = actions.view(-1).detach() - actions0
                            d11 = torch.dot(dy, dy)
                            d12 = torch.dot(dy, du)
                            d22 = torch.dot(du, du)
                            lagrange_y0 = ly1
                            actions0 = actions.view(-1).detach().clone()
                            if d11 > 0 and d12 > 0 and d22 > 0:
                                corr = d12 / torch.sqrt(d11 * d22)
                                a_sd = d11 / d12
                                a_mg = d12 / d22
                                a_hat = a_mg if 2 * a_mg > a_sd \
                                    else a_sd - 0.5 * a_mg
                                if (corr > self.corr_min
                                        and a_hat < 10 * self.admm_rho
                                        and a_hat > 0.1 * self.admm_rho):
                                    admm_rho = float(a_hat)

        self.update_network_parameters()

    # -- checkpointing ----------------------------------------------------
    def _path(self, name):
        return os.path.join(self.checkpoint_dir, name)

    def _save_net(self, net, fname):
        sd = {k: v.detach().cpu().clone() for k, v in net.state_dict().items()}
        torch.save(sd, self._path(fname))

    def save_models(self):
        self._save_net(self.actor, "a_eval_td3_actor.model")
        self._save_net(self.target_actor, "a_target_td3_actor.model")
        self._save_net(self.critic_1, "q_eval_1_td3_critic.model")
        self._save_net(self.critic_2, "q_eval_2_td3_critic.model")
        self._save_net(self.target_critic_1, "q_target_1_td3_critic.model")
        self._save_net(self.target_critic_2, "q_target_2_td3_critic.model")
        self.replaymem.save_checkpoint(self._path(
            ("prioritized_" if self.prioritized else "")
            + "replaymem_td3.model"))

    def load_models(self):
        ml = self.device
        for net, fn in [(self.actor, "a_eval_td3_actor.model"),
                        (self.target_actor, "a_target_td3_actor.model"),
                        (self.critic_1, "q_eval_1_td3_critic.model"),
                        (self.critic_2, "q_eval_2_td3_critic.model"),
                        (self.target_critic_1, "q_target_1_td3_critic.model"),
                        (self.target_critic_2, "q_target_2_td3_critic.model")]:
            net.load_state_dict(torch.load(self._path(fn), map_location=ml))
        try:
            self.replaymem.load_checkpoint(self._path(
                ("prioritized_" if self.prioritized else "")
                + "replaymem_td3.model"))
        except FileNotFoundError:
            pass
        self.update_network_parameters(tau=1.0)

    def load_models_for_eval(self):
        self.load_models()
        self.actor.eval()
