"""SAC agent over {img, meta} dict observations (CNN actor/critic).

Covers the reference's `calibration/calib_sac.py:253-427` and
`demixing_rl/demix_sac.py:529-722` / `demixing_fuzzy/demix_sac.py`
(`use_influence=False` drops the conv branch): twin critics + target
critics, fixed or learnable temperature, KLD-based hint penalty with
augmented Lagrangian ratcheted every 10 learn steps, optional PER.
Parameters live in flat pools (fused Adam / one-kernel polyak / one
RCCL all-reduce in DP mode).
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.distributions import Normal

from ..utils.device import default_device
from ..utils.flatten import FlatParams, FusedAdam
from ..utils.streams import StreamFork
from .buffers_dict import DictPERBuffer, DictReplayBuffer
from .conv_networks import CriticCNN, SACActorCNN, EPS, _init_layer


class _MetaOnlySACActor(nn.Module):
    """Metadata-only actor (demixing_fuzzy `use_influence=False` path:
    conv branch dropped, INNER_DIM=32)."""

    def __init__(self, meta_dim, n_actions, max_action=1.0, inner=32):
        super().__init__()
        self.max_action = max_action
        self.reparam_noise = EPS
        self.fc1 = nn.Linear(meta_dim, 128)
        self.fc2 = nn.Linear(128, inner)
        self.fc3 = nn.Linear(inner, 128)
        self.mu = nn.Linear(128, n_actions)
        self.sigma = nn.Linear(128, n_actions)
        for l in (self.fc1, self.fc2, self.fc3):
            _init_layer(l)
        _init_layer(self.mu, 0.003)
        _init_layer(self.sigma, 0.003)

    def forward(self, img, meta):
        z = F.relu(self.fc1(torch.flatten(meta, 1)))
        z = F.relu(self.fc2(z))
        z = F.elu(self.fc3(z))
        return self.mu(z), torch.clamp(self.sigma(z),
                                       min=self.reparam_noise, max=1.0)

    def sample_normal(self, img, meta, reparameterize=True):
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


class _MetaOnlyCritic(nn.Module):
    def __init__(self, meta_dim, n_actions, inner=32):
        super().__init__()
        self.fc1 = nn.Linear(n_actions + meta_dim, 128)
        self.fc2 = nn.Linear(128, inner)
        self.head = nn.Linear(inner, 1)
        _init_layer(self.fc1)
        _init_layer(self.fc2)
        _init_layer(self.head, 0.003)

    def forward(self, img, meta, action):
        y = torch.cat((torch.flatten(action, 1), torch.flatten(meta, 1)), 1)
        y = F.relu(self.fc1(y))
        y = F.relu(self.fc2(y))
        return self.head(y)


EPS_KLD = 0.01


class Agent:
    """SAC over dict observations. ``use_influence=False`` builds
    metadata-only MLP networks (demixing_fuzzy variant)."""

    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.001, M=3, reward_scale=2,
                 alpha=0.1, hint_threshold=0.1, admm_rho=1.0,
                 name_prefix="", use_hint=False, prioritized=False,
                 use_influence=True, meta_dim=None, learn_alpha=False,
                 arch="cnn",
                 device: Optional[torch.device] = None,
                 checkpoint_dir="./", grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.max_action = 1.0
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.name_prefix = name_prefix
        self.grad_hook = grad_hook
        self.use_influence = use_influence

        img_shape = tuple(input_dims)          # e.g. (1, 128, 128)
        hw = img_shape[-2:]
        if meta_dim is None:
            meta_dim = 7 * (M + 1)             # calib metadata (M+1,7)
        self.meta_dim = meta_dim

        buf_cls = DictPERBuffer if prioritized else DictReplayBuffer
        self.prioritized = prioritized
        self.replaymem = buf_cls(max_mem_size, img_shape, meta_dim,
                                 n_actions, device=self.device)

        if use_influence and arch == "transformer":
            # BASELINE.json config: "calibenv SAC with transformer
            # actor/critic" — token-sequence encoder over the sky
            # metadata rows + influence-map embedding
            from .transformer_networks import (SACActorTransformer,
                                               TransformerCritic)
            mk_actor = lambda: SACActorTransformer(hw, meta_dim, n_actions,
                                                   self.max_action)
            mk_critic = lambda: TransformerCritic(hw, meta_dim, n_actions)
        elif use_influence:
            mk_actor = lambda: SACActorCNN(hw, meta_dim, n_actions,
                                           self.max_action)
            mk_critic = lambda: CriticCNN(hw, meta_dim, n_actions)
        else:
            mk_actor = lambda: _MetaOnlySACActor(meta_dim, n_actions,
                                                 self.max_action)
            mk_critic = lambda: _MetaOnlyCritic(meta_dim, n_actions)
        self.actor = mk_actor().to(self.device)
        self.critic_1 = mk_critic().to(self.device)
        self.critic_2 = mk_critic().to(self.device)
        self.target_critic_1 = mk_critic().to(self.device)
        self.target_critic_2 = mk_critic().to(self.device)

        self.actor_fp = FlatParams(self.actor)
        self.critic_1_fp = FlatParams(self.critic_1)
        self.critic_2_fp = FlatParams(self.critic_2)
        self.target_critic_1_fp = FlatParams(self.target_critic_1)
        self.target_critic_2_fp = FlatParams(self.target_critic_2)
        self.actor_opt = FusedAdam(self.actor_fp, lr=lr_a)
        self.critic_1_opt = FusedAdam(self.critic_1_fp, lr=lr_c)
        self.critic_2_opt = FusedAdam(self.critic_2_fp, lr=lr_c)

        self.alpha = torch.tensor(float(alpha), device=self.device)
        self.scale = reward_scale
        self.learn_alpha = learn_alpha
        if learn_alpha:
            self.log_alpha = torch.tensor(float(np.log(alpha)),
                                          device=self.device,
                                          requires_grad=True)
            self.alpha_opt = torch.optim.Adam([self.log_alpha], lr=1e-4)
            self.target_entropy = -float(n_actions)

        self.use_hint = use_hint
        self.zero_tensor = torch.tensor(0.0, device=self.device)
        self.hint_threshold = hint_threshold
        self.rho = torch.tensor(0.0, device=self.device)
        self.admm_rho = admm_rho
        self.learn_counter = 0
        self._fork = StreamFork(self.device)
        self._hard_sync()

    # ------------------------------------------------------------------
    def _hard_sync(self):
        self.target_critic_1_fp.polyak_from(self.critic_1_fp, 1.0)
        self.target_critic_2_fp.polyak_from(self.critic_2_fp, 1.0)

    def update_network_parameters(self, tau=None):
        if tau is None:
            tau = self.tau
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
                               dtype=torch.float32,
                               device=self.device)[None]
        return img, meta

    def choose_action(self, observation):
        self.actor.eval()
        img, meta = self._obs_tensors(observation)
        with torch.no_grad():
            actions, _ = self.actor.sample_normal(img, meta,
                                                  reparameterize=False)
        self.actor.train()
        return actions.cpu().numpy()[0]

    # ------------------------------------------------------------------
    def _kld_loss(self, action, hint):
        """KLD hint penalty (`calib_sac.py:361-366`)."""
        noise = getattr(self.actor, "reparam_noise", EPS)
        a = torch.clamp(0.5 * action + 0.5 + noise, min=EPS_KLD, max=1.0)
        h = torch.clamp(0.5 * hint + 0.5 + noise, min=EPS_KLD, max=1.0)
        return h * (torch.log(h) - torch.log(a))

    def learn(self):
        if len(self.replaymem) < self.batch_size:
            return
        if getattr(self, "_graph", None) is not None:
            return self._learn_graphed()
        if self.prioritized:
            batch, idx, is_w = self.replaymem.sample_buffer(self.batch_size)
            is_w = is_w.unsqueeze(1)
        else:
            batch = self.replaymem.sample_buffer(self.batch_size)
            idx = is_w = None
        (img, meta, action, reward, img_, meta_, done, hint) = batch
        self._learn_body(img, meta, action, reward.unsqueeze(1),
                         img_, meta_, done.unsqueeze(1), hint, is_w, idx)

    def _learn_body(self, img, meta, action, reward, img_, meta_, done,
                    hint, is_w=None, idx=None):
        with torch.no_grad():
            na, nlp = self.actor.sample_normal(img_, meta_,
                                               reparameterize=False)

        def _t1():
            with torch.no_grad():
                return self.target_critic_1(img_, meta_, na)

        def _t2():
            with torch.no_grad():
                return self.target_critic_2(img_, meta_, na)

        # all four critic-family forwards overlap on 4 HIP streams
        q1, q2, q1_t, q2_t = self._fork(
            lambda: self.critic_1(img, meta, action),
            lambda: self.critic_2(img, meta, action), _t1, _t2)
        with torch.no_grad():
            tgt = torch.min(q1_t, q2_t) - self.alpha * nlp
            tgt = tgt.masked_fill(done, 0.0)
            # note: the CNN reference does NOT apply reward_scale in the
            # target (`calib_sac.py:345`); self.scale kept for parity only
            new_q = reward + self.gamma * tgt
        if is_w is not None:
            c_loss = (is_w * (q1 - new_q).pow(2)).mean() \
                + (is_w * (q2 - new_q).pow(2)).mean()
        else:
            c_loss = F.mse_loss(q1, new_q) + F.mse_loss(q2, new_q)
        self.critic_1_opt.zero_grad()
        self.critic_2_opt.zero_grad()
        c_loss.backward()
        self._fork.join()   # side-stream backwards must land first
        if self.grad_hook is not None:
            self.grad_hook([self.critic_1_fp, self.critic_2_fp])
        self.critic_1_opt.step()
        self.critic_2_opt.step()
        if idx is not None:
            self.replaymem.update_priorities(idx, (q1 - new_q).detach())

        actions, log_probs = self.actor.sample_normal(img, meta,
                                                      reparameterize=True)
        q1_pi, q2_pi = self._fork(
            lambda: self.critic_1(img, meta, actions),
            lambda: self.critic_2(img, meta, actions))
        critic_value = torch.min(q1_pi, q2_pi)
        actor_loss = (self.alpha * log_probs - critic_value).mean()
        if self.use_hint:
            gfun = torch.max(self.zero_tensor,
                             (self._kld_loss(actions, hint)
                              - self.hint_threshold).mean()).pow(2)
            actor_loss = actor_loss + 0.5 * self.admm_rho * gfun * gfun \
                + self.rho * gfun
        self.actor_opt.zero_grad()
        actor_loss.backward()
        self._fork.join()   # side-stream backwards must land first
        if self.grad_hook is not None:
            self.grad_hook([self.actor_fp])
        self.actor_opt.step()

        if self.learn_alpha:
            a_loss = -(self.log_alpha.exp()
                       * (log_probs + self.target_entropy).detach()).mean()
            self.alpha_opt.zero_grad()
            a_loss.backward()
            self.alpha_opt.step()
            self.alpha = self.log_alpha.exp().detach()

        if self.learn_counter % 10 == 0 and self.use_hint:
            with torch.no_grad():
                gfun = torch.max(self.zero_tensor,
                                 (self._kld_loss(actions, hint)
                                  - self.hint_threshold).mean()).pow(2)
                self.rho = self.rho + self.admm_rho * gfun
        self.learn_counter += 1
        self.update_network_parameters()

    # -- hipGraph capture of the CNN learn step -----------------------------
    def enable_cuda_graph(self):
        """Capture the whole CNN learn step (conv forwards/backwards on
        the direct-conv kernels, BatchNorm batch stats + running-stat
        updates — all static-shape tensor ops — fused Adam, polyak) into
        ONE hipGraph. Plain-replay, no-hint, fixed-alpha path only."""
        assert not self.use_hint and not self.prioritized \
            and not self.learn_alpha
        assert self.device.type == "cuda"
        B = self.batch_size
        mem = self.replaymem
        dev = self.device
        self._g_idx = torch.zeros(B, dtype=torch.long, device=dev)
        self._g_img = torch.zeros((B, *mem.img_memory.shape[1:]),
                                  device=dev)
        self._g_img_ = torch.zeros_like(self._g_img)
        self._g_meta = torch.zeros((B, *mem.meta_memory.shape[1:]),
                                   device=dev)
        self._g_meta_ = torch.zeros_like(self._g_meta)
        self._g_action = torch.zeros(B, self.n_actions, device=dev)
        self._g_reward = torch.zeros(B, 1, device=dev)
        self._g_done = torch.zeros(B, 1, dtype=torch.bool, device=dev)
        self._g_hint = torch.zeros(B, self.n_actions, device=dev)
        self._fill_static()
        torch.cuda.synchronize()
        counter0 = self.learn_counter   # warmup/capture must not count
        # MIOpen batch-norm does find-db lookups/allocations at call time
        # that break stream capture — force the native implementation for
        # the captured body (the graph then always replays native kernels)
        with torch.backends.cudnn.flags(enabled=False):
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    self._learn_body(self._g_img, self._g_meta,
                                     self._g_action, self._g_reward,
                                     self._g_img_, self._g_meta_,
                                     self._g_done, self._g_hint)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._learn_body(self._g_img, self._g_meta, self._g_action,
                                 self._g_reward, self._g_img_,
                                 self._g_meta_, self._g_done, self._g_hint)
        self.learn_counter = counter0
        self._graph = g

    def disable_cuda_graph(self):
        self._graph = None

    def _fill_static(self):
        mem = self.replaymem
        n = len(mem)
        torch.randint(0, n, (self.batch_size,), device=self.device,
                      out=self._g_idx)
        torch.index_select(mem.img_memory, 0, self._g_idx,
                           out=self._g_img)
        torch.index_select(mem.new_img_memory, 0, self._g_idx,
                           out=self._g_img_)
        torch.index_select(mem.meta_memory, 0, self._g_idx,
                           out=self._g_meta)
        torch.index_select(mem.new_meta_memory, 0, self._g_idx,
                           out=self._g_meta_)
        torch.index_select(mem.action_memory, 0, self._g_idx,
                           out=self._g_action)
        torch.index_select(mem.reward_memory, 0, self._g_idx,
                           out=self._g_reward.view(-1))
        torch.index_select(mem.terminal_memory, 0, self._g_idx,
                           out=self._g_done.view(-1))

    def _learn_graphed(self):
        self._fill_static()
        self._graph.replay()
        # the captured body's python-side counter bump does not replay
        self.learn_counter += 1

    # -- checkpointing (reference naming: <prefix>_sac_{actor,critic}) ----
    def _path(self, name):
        return os.path.join(self.checkpoint_dir,
                            f"{self.name_prefix}{name}")

    def save_models(self):
        torch.save(self.actor.state_dict(), self._path("a_eval_sac_actor.model"))
        torch.save(self.critic_1.state_dict(),
                   self._path("q_eval_1_sac_critic.model"))
        torch.save(self.critic_2.state_dict(),
                   self._path("q_eval_2_sac_critic.model"))
        self.replaymem.save_checkpoint(self._path(self.replaymem.filename))

    def load_models(self):
        map_loc = self.device
        self.actor.load_state_dict(torch.load(
            self._path("a_eval_sac_actor.model"), map_location=map_loc,
            weights_only=True))
        self.critic_1.load_state_dict(torch.load(
            self._path("q_eval_1_sac_critic.model"), map_location=map_loc,
            weights_only=True))
        self.critic_2.load_state_dict(torch.load(
            self._path("q_eval_2_sac_critic.model"), map_location=map_loc,
            weights_only=True))
        # load_state_dict copies in place → the flat pools see the new
        # values through the parameter views; only targets need syncing
        self._hard_sync()

    def load_models_for_eval(self):
        self.load_models()
        self.actor.eval()
        self.critic_1.eval()
        self.critic_2.eval()
