"""SAC agent (MLP) — twin critics, target critics, fixed/learnable alpha,
optional hint-constrained actor loss (augmented Lagrangian).

Same algorithm and hyperparameter surface as the reference agent
(reference ``elasticnet/enet_sac.py:478-658``): reward scaling, terminal
masking of the soft target, soft (polyak) target update per learn step,
hint penalty ``0.5*rho_admm*g^2*g^2 + rho*g`` with ``rho`` ratcheted every 10
learn steps. Default ``prioritized=False`` builds a plain buffer (the
reference accepts the flag but always builds a plain buffer,
``enet_sac.py:488-490``; we honor it when True).

MI355X-native runtime: each network's parameters live in one flat fp32 pool
(``utils.flatten.FlatParams``) so the Adam step is one fused kernel, the
polyak update is one axpby, and (in data-parallel mode) the gradient
all-reduce is one RCCL call over xGMI.
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
from .networks import CriticMLP, SACActorMLP


class Agent:
    def __init__(self, gamma, lr_a, lr_c, input_dims, batch_size, n_actions,
                 max_mem_size=100, tau=0.001, reward_scale=2, alpha=0.1,
                 name_prefix="", prioritized=False, use_hint=False,
                 device: Optional[torch.device] = None,
                 checkpoint_dir="./", learn_alpha=False,
                 grad_hook=None):
        self.gamma = gamma
        self.tau = tau
        self.batch_size = batch_size
        self.n_actions = n_actions
        self.max_action = 1.0
        self.min_action = -1.0
        self.prioritized = prioritized
        self.device = device if device is not None else default_device()
        self.checkpoint_dir = checkpoint_dir
        self.name_prefix = name_prefix
        # called with list[FlatParams] after backward, before optimizer step
        # (the DP learner installs an RCCL all-reduce here)
        self.grad_hook = grad_hook

        input_dim = input_dims[0] if isinstance(input_dims, (list, tuple)) \
            else int(input_dims)
        buf_cls = PERBuffer if prioritized else ReplayBuffer
        self.replaymem = buf_cls(max_mem_size, [input_dim], n_actions,
                                 device=self.device)

        self.actor = SACActorMLP(input_dim, n_actions,
                                 self.max_action).to(self.device)
        self.critic_1 = CriticMLP(input_dim, n_actions).to(self.device)
        self.critic_2 = CriticMLP(input_dim, n_actions).to(self.device)
        self.target_critic_1 = CriticMLP(input_dim, n_actions).to(self.device)
        self.target_critic_2 = CriticMLP(input_dim, n_actions).to(self.device)

        # flat parameter pools + fused optimizers
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
        self.zero_tensor = torch.tensor(0.0, device=self.device)
        self.learn_alpha = learn_alpha
        if self.learn_alpha:
            self.target_entropy = -float(np.sum(n_actions))
            self.alpha_lr = 1e-4

        self.use_hint = use_hint
        if self.use_hint:
            self.hint_threshold = 0.1
            self.rho = torch.tensor(0.0, device=self.device)
            self.admm_rho = 0.01

        self.update_network_parameters(tau=1.0)
        self.learn_counter = 0

    # ------------------------------------------------------------------
    def update_network_parameters(self, tau=None):
        if tau is None:
            tau = self.tau
        self.target_critic_1_fp.polyak_from(self.critic_1_fp, tau)
        self.target_critic_2_fp.polyak_from(self.critic_2_fp, tau)

    def store_transition(self, state, action, reward, state_, terminal,
                         hint):
        self.replaymem.store_transition(obs_to_state(state), action, reward,
                                        obs_to_state(state_), terminal, hint)

    def choose_action(self, observation):
        state = obs_to_state(observation).to(self.device)
        with torch.no_grad():
            actions, _ = self.actor.sample_normal(state,
                                                  reparameterize=False)
        return actions.cpu().numpy()[0]

    def choose_action_tensor(self, observation):
        """Device-resident action (no host sync) — the bench/vectorized
        rollout path; ``choose_action`` keeps the reference's numpy API."""
        state = obs_to_state(observation).to(self.device)
        with torch.no_grad():
            actions, _ = self.actor.sample_normal(state,
                                                  reparameterize=False)
        return actions.reshape(-1)

    # ------------------------------------------------------------------
    def _grad_sync(self, fps):
        if self.grad_hook is not None:
            self.grad_hook(fps)

    def _fork(self, *fns):
        """Run independent small-GEMM chains concurrently: fns[0] stays
        on the current stream, each other fn forks onto its own side HIP
        stream (event fork/join). Stream-aware autograd then also runs
        their backwards concurrently, and the fork/join pattern is
        captured into the hipGraph as parallel branches. The twin
        critics/targets are ~tiny GEMMs at ~3% occupancy each — overlap
        hides most of their wall time."""
        if self.device.type != "cuda":
            return tuple(f() for f in fns)
        nside = len(fns) - 1
        if not hasattr(self, "_side_streams"):
            self._side_streams = []
            self._fork_evs = []
            self._join_evs = []
        while len(self._side_streams) < nside:
            self._side_streams.append(torch.cuda.Stream())
            self._fork_evs.append(torch.cuda.Event())
            self._join_evs.append(torch.cuda.Event())
        results = [None] * len(fns)
        for i in range(nside):
            self._fork_evs[i].record()
            with torch.cuda.stream(self._side_streams[i]):
                self._fork_evs[i].wait()
                results[i + 1] = fns[i + 1]()
                self._join_evs[i].record()
        results[0] = fns[0]()
        for i in range(nside):
            self._join_evs[i].wait()
        if not torch.cuda.is_current_stream_capturing():
            # eager mode: guard side-stream allocations against premature
            # reuse once consumed on the main stream
            cs = torch.cuda.current_stream()
            for r in results[1:]:
                for t in (r if isinstance(r, (tuple, list)) else (r,)):
                    if torch.is_tensor(t):
                        t.record_stream(cs)
        return tuple(results)

    def _pair(self, f1, f2):
        return self._fork(f1, f2)

    def _sync_side_streams(self):
        """Join every _fork side stream into the current stream. Backward
        of a forward that ran on a side stream also runs there
        (stream-aware autograd), and the fused-linear backward writes
        ``flat_grad`` directly — bypassing AccumulateGrad, so the
        engine's leaf-stream sync does not cover it. Without this join a
        main-stream grad all-reduce (data-parallel) could read
        partially-written gradients. Event record/wait is graph-capture
        safe (captured as dependency edges)."""
        if self.device.type != "cuda" or not hasattr(self, "_side_streams"):
            return
        if not hasattr(self, "_sync_evs"):
            self._sync_evs = []
        while len(self._sync_evs) < len(self._side_streams):
            self._sync_evs.append(torch.cuda.Event())
        for ev, s in zip(self._sync_evs, self._side_streams):
            ev.record(s)
            ev.wait()

    def _learn_body(self, state_batch, new_state_batch, action_batch,
                    reward_batch, terminal_batch, hint_batch, is_w=None):
        """Tensor-only learn step (hipGraph-capturable on GPU): soft target,
        twin-critic regression, actor update, fused Adam, polyak."""
        with torch.no_grad():
            new_actions, new_log_probs = self.actor.sample_normal(
                new_state_batch, reparameterize=False)

        def _t1():
            with torch.no_grad():
                return self.target_critic_1(new_state_batch, new_actions)

        def _t2():
            with torch.no_grad():
                return self.target_critic_2(new_state_batch, new_actions)

        # all four critic-family forwards run concurrently on 4 streams
        q1, q2, q1_t, q2_t = self._fork(
            lambda: self.critic_1(state_batch, action_batch),
            lambda: self.critic_2(state_batch, action_batch),
            _t1, _t2)
        with torch.no_grad():
            min_next_target = torch.min(q1_t, q2_t) \
                - self.alpha * new_log_probs
            # masked_fill (not boolean indexing): same semantics as the
            # reference's t[mask]=0 but static-shaped => graph-safe
            min_next_target = min_next_target.masked_fill(terminal_batch,
                                                          0.0)
            new_q_value = reward_batch + self.gamma * min_next_target
        if is_w is not None:
            critic_1_loss = (is_w * (q1 - new_q_value).pow(2)).mean()
            critic_2_loss = (is_w * (q2 - new_q_value).pow(2)).mean()
        else:
            critic_1_loss = F.mse_loss(q1, new_q_value)
            critic_2_loss = F.mse_loss(q2, new_q_value)
        critic_loss = critic_1_loss + critic_2_loss
        self.critic_1_opt.zero_grad()
        self.critic_2_opt.zero_grad()
        critic_loss.backward()
        self._sync_side_streams()
        self._grad_sync([self.critic_1_fp, self.critic_2_fp])
        self._fork(self.critic_1_opt.step, self.critic_2_opt.step)

        actions, log_probs = self.actor.sample_normal(state_batch,
                                                      reparameterize=True)
        # actor update: critic-weight grads from this loss are discarded
        # (zeroed before the next critic regression), so freeze critic
        # params here — backward then only propagates through `actions`,
        # skipping all dW GEMMs of both critics
        for p in self.critic_1.parameters():
            p.requires_grad_(False)
        for p in self.critic_2.parameters():
            p.requires_grad_(False)
        q1_pi, q2_pi = self._pair(
            lambda: self.critic_1(state_batch, actions),
            lambda: self.critic_2(state_batch, actions))
        critic_value = torch.min(q1_pi, q2_pi)

        actor_loss = (self.alpha * log_probs - critic_value).mean()
        if self.use_hint:
            gfun = torch.max(
                self.zero_tensor,
                (F.mse_loss(actions, hint_batch)
                 - self.hint_threshold).mean()).pow(2)
            actor_loss = actor_loss + 0.5 * self.admm_rho * gfun * gfun \
                + self.rho * gfun
        self.actor_opt.zero_grad()
        actor_loss.backward()
        for p in self.critic_1.parameters():
            p.requires_grad_(True)
        for p in self.critic_2.parameters():
            p.requires_grad_(True)
        self._sync_side_streams()
        self._grad_sync([self.actor_fp])
        self.actor_opt.step()
        self.update_network_parameters()
        return q1, new_q_value

    # -- hipGraph capture of the learn step -------------------------------
    def enable_cuda_graph(self):
        """Capture the whole learn step (forwards, backwards, fused Adam,
        polyak) into ONE hipGraph. Per learn step afterwards: one randint,
        four index_select gathers into static buffers, one graph replay —
        instead of ~100 python-dispatched launches. Only for the plain
        (non-PER, non-hint) path and after the replay has batch_size
        entries."""
        assert not self.use_hint and not self.prioritized
        assert self.device.type == "cuda"
        B = self.batch_size
        mem = self.replaymem
        self._g_idx = torch.zeros(B, dtype=torch.long, device=self.device)
        self._g_state = torch.zeros(B, mem.state_memory.shape[1],
                                    device=self.device)
        self._g_new_state = torch.zeros_like(self._g_state)
        self._g_action = torch.zeros(B, self.n_actions, device=self.device)
        self._g_reward = torch.zeros(B, 1, device=self.device)
        self._g_done = torch.zeros(B, 1, dtype=torch.bool,
                                   device=self.device)
        self._g_hint = torch.zeros(B, self.n_actions, device=self.device)

        # populate statics with a real sample so warmup trains on data
        n = len(mem)
        if n > 0:
            torch.randint(0, n, (B,), device=self.device, out=self._g_idx)
            torch.index_select(mem.state_memory, 0, self._g_idx,
                               out=self._g_state)
            torch.index_select(mem.new_state_memory, 0, self._g_idx,
                               out=self._g_new_state)
            torch.index_select(mem.action_memory, 0, self._g_idx,
                               out=self._g_action)
            torch.index_select(mem.reward_memory, 0, self._g_idx,
                               out=self._g_reward.view(-1))
            self._g_reward.mul_(self.scale)
            torch.index_select(mem.terminal_memory, 0, self._g_idx,
                               out=self._g_done.view(-1))

        # warmup on a side stream (allocator + autograd settle)
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self._learn_body(self._g_state, self._g_new_state,
                                 self._g_action, self._g_reward,
                                 self._g_done, self._g_hint)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._learn_body(self._g_state, self._g_new_state,
                             self._g_action, self._g_reward,
                             self._g_done, self._g_hint)
        self._graph = g

    def disable_cuda_graph(self):
        """Drop a captured graph and return to the eager learn path
        (used e.g. when ranks must agree on graphed vs eager mode)."""
        self._graph = None

    def _learn_graphed(self):
        mem = self.replaymem
        n = len(mem)
        torch.randint(0, n, (self.batch_size,), device=self.device,
                      out=self._g_idx)
        torch.index_select(mem.state_memory, 0, self._g_idx,
                           out=self._g_state)
        torch.index_select(mem.new_state_memory, 0, self._g_idx,
                           out=self._g_new_state)
        torch.index_select(mem.action_memory, 0, self._g_idx,
                           out=self._g_action)
        torch.index_select(mem.reward_memory, 0, self._g_idx,
                           out=self._g_reward.view(-1))
        self._g_reward.mul_(self.scale)  # body expects scaled rewards
        torch.index_select(mem.terminal_memory, 0, self._g_idx,
                           out=self._g_done.view(-1))
        self._graph.replay()
        self.learn_counter += 1

    def learn(self):
        if self.replaymem.mem_cntr < self.batch_size:
            return

        if getattr(self, "_graph", None) is not None:
            return self._learn_graphed()

        if self.prioritized:
            (state, action, reward, new_state, done, hint), idxs, is_w = \
                self.replaymem.sample_buffer(self.batch_size)
            is_w = is_w.to(self.device).unsqueeze(1)
        else:
            state, action, reward, new_state, done, hint = \
                self.replaymem.sample_buffer(self.batch_size)
            is_w = None

        state_batch = state.to(self.device)
        new_state_batch = new_state.to(self.device)
        action_batch = action.to(self.device)
        reward_batch = self.scale * reward.to(self.device).unsqueeze(1)
        terminal_batch = done.to(self.device).unsqueeze(1)
        hint_batch = hint.to(self.device)

        if self.prioritized:
            q1, new_q_value = self._learn_body(
                state_batch, new_state_batch, action_batch, reward_batch,
                terminal_batch, hint_batch, is_w)
            with torch.no_grad():
                q2 = self.critic_2(state_batch, action_batch)
                td = 0.5 * ((q1 - new_q_value).abs()
                            + (q2 - new_q_value).abs())
            self.replaymem.update_priorities(idxs, td)
        else:
            self._learn_body(state_batch, new_state_batch, action_batch,
                             reward_batch, terminal_batch, hint_batch)

        if self.learn_counter % 10 == 0 and (self.learn_alpha
                                             or self.use_hint):
            with torch.no_grad():
                actions, log_probs = self.actor.sample_normal(
                    state_batch, reparameterize=False)
                if self.learn_alpha:
                    self.alpha = torch.max(
                        self.zero_tensor,
                        self.alpha + self.alpha_lr
                        * (self.target_entropy - (-log_probs)).mean())
                if self.use_hint:
                    gfun = torch.max(
                        self.zero_tensor,
                        (F.mse_loss(actions, hint_batch)
                         - self.hint_threshold).mean()).pow(2)
                    self.rho = self.rho + self.admm_rho * gfun

        self.learn_counter += 1

    # -- checkpointing (reference file layout: <name>_sac_<role>.model) ---
    def _path(self, name):
        return os.path.join(self.checkpoint_dir,
                            self.name_prefix + name)

    def _save_net(self, net, fname):
        sd = {k: v.detach().cpu().clone() for k, v in net.state_dict().items()}
        torch.save(sd, self._path(fname))

    def save_models(self):
        self._save_net(self.actor, "a_eval_sac_actor.model")
        self._save_net(self.critic_1, "q_eval_1_sac_critic.model")
        self._save_net(self.critic_2, "q_eval_2_sac_critic.model")
        self.replaymem.save_checkpoint(self._path(
            ("prioritized_" if self.prioritized else "")
            + "replaymem_sac.model"))

    def load_models(self):
        map_loc = self.device
        self.actor.load_state_dict(torch.load(
            self._path("a_eval_sac_actor.model"), map_location=map_loc))
        self.critic_1.load_state_dict(torch.load(
            self._path("q_eval_1_sac_critic.model"), map_location=map_loc))
        self.critic_2.load_state_dict(torch.load(
            self._path("q_eval_2_sac_critic.model"), map_location=map_loc))
        try:
            self.replaymem.load_checkpoint(self._path(
                ("prioritized_" if self.prioritized else "")
                + "replaymem_sac.model"))
        except FileNotFoundError:
            pass
        self.update_network_parameters(tau=1.0)

    def load_models_for_eval(self):
        self.load_models()
        self.actor.eval()
        self.critic_1.eval()
        self.critic_2.eval()

    def print(self):
        print(self.actor)
        print(self.critic_1)
