"""Replay buffers: plain ring buffer, sum-tree, prioritized replay.

Device-resident by design: all transition storage lives in torch tensors on
the agent's device (288 GB HBM3E per MI355X makes host ring buffers + H2D
copies per learn step — the reference's layout, numpy buffers at
``enet_sac.py:23-80`` — pointless). Sampling indices, priorities and
importance weights are computed on-device (``smartcal_amd.ops.per``:
prefix-sum inverse-CDF sampling, the batched equivalent of the reference's
serial sum-tree descent ``enet_sac.py:82-200,270-323``).

``SumTree`` is kept as a host-side exact-parity structure (used in tests to
show the prefix-sum sampler draws from the same distribution, and for
host-only runs).
"""

from __future__ import annotations

import pickle
from typing import Optional, Tuple

import numpy as np
import torch

from ..ops import per as per_ops


class ReplayBuffer:
    """Uniform ring buffer over device tensors (state/action/reward/next/
    done/hint), same record layout as reference ``enet_sac.py:23-80``."""

    def __init__(self, max_size: int, input_dims, n_actions: int,
                 device: Optional[torch.device] = None):
        self.mem_size = int(max_size)
        self.mem_cntr = 0
        self.device = device if device is not None else torch.device("cpu")
        dims = list(input_dims) if not isinstance(input_dims, int) \
            else [input_dims]
        self.state_memory = torch.zeros((self.mem_size, *dims),
                                        dtype=torch.float32,
                                        device=self.device)
        self.new_state_memory = torch.zeros_like(self.state_memory)
        self.action_memory = torch.zeros((self.mem_size, n_actions),
                                         dtype=torch.float32,
                                         device=self.device)
        self.reward_memory = torch.zeros(self.mem_size, dtype=torch.float32,
                                         device=self.device)
        self.terminal_memory = torch.zeros(self.mem_size, dtype=torch.bool,
                                           device=self.device)
        self.hint_memory = torch.zeros((self.mem_size, n_actions),
                                       dtype=torch.float32,
                                       device=self.device)
        self.filename = "replaymem_sac.model"

    def __len__(self):
        return min(self.mem_cntr, self.mem_size)

    def _as(self, x, like: torch.Tensor):
        if torch.is_tensor(x):
            return x.detach().to(like.device, like.dtype).reshape(like.shape)
        return torch.as_tensor(np.asarray(x), dtype=like.dtype,
                               device=like.device).reshape(like.shape)

    def store_transition(self, state, action, reward, state_, done,
                         hint=None):
        i = self.mem_cntr % self.mem_size
        self.state_memory[i] = self._as(state, self.state_memory[i])
        self.new_state_memory[i] = self._as(state_, self.new_state_memory[i])
        self.action_memory[i] = self._as(action, self.action_memory[i])
        if torch.is_tensor(reward):
            # keep device-resident — no host sync
            self.reward_memory[i] = reward.detach().to(
                self.reward_memory.device, torch.float32).reshape(())
        else:
            self.reward_memory[i] = float(reward)
        self.terminal_memory[i] = bool(done)
        if hint is not None:
            self.hint_memory[i] = self._as(hint, self.hint_memory[i])
        self.mem_cntr += 1

    def store_batch(self, states, actions, rewards, states_, dones,
                    hints=None):
        """Vectorized store of E transitions in one shot (device-resident,
        no per-transition host sync) — the VecENetEnv rollout path. Ring
        semantics identical to E sequential store_transition calls."""
        E = states.shape[0]
        if E > self.mem_size:
            raise ValueError(
                f"store_batch of {E} > capacity {self.mem_size}: duplicate "
                "ring slots in one write are order-undefined on GPU")
        idx = (torch.arange(E, device=self.device)
               + self.mem_cntr) % self.mem_size
        self.state_memory[idx] = states.detach().to(
            self.device, torch.float32).reshape(E, -1)
        self.new_state_memory[idx] = states_.detach().to(
            self.device, torch.float32).reshape(E, -1)
        self.action_memory[idx] = actions.detach().to(
            self.device, torch.float32).reshape(E, -1)
        self.reward_memory[idx] = rewards.detach().to(
            self.device, torch.float32).reshape(-1)
        if torch.is_tensor(dones):
            self.terminal_memory[idx] = dones.to(self.device,
                                                 torch.bool).reshape(-1)
        else:
            self.terminal_memory[idx] = bool(dones)
        if hints is not None:
            self.hint_memory[idx] = hints.detach().to(
                self.device, torch.float32).reshape(E, -1)
        self.mem_cntr += E
        return idx

    def sample_buffer(self, batch_size: int):
        max_mem = len(self)
        idx = torch.randint(0, max_mem, (batch_size,), device=self.device)
        return (self.state_memory[idx], self.action_memory[idx],
                self.reward_memory[idx], self.new_state_memory[idx],
                self.terminal_memory[idx], self.hint_memory[idx])

    # -- checkpointing (reference pickles the whole object,
    #    ``enet_sac.py:59-73``; we save a state dict of host tensors) ------
    def save_checkpoint(self, filename: Optional[str] = None):
        # only the filled prefix is stored (empty slots are zeros)
        fn = filename or self.filename
        n = len(self)
        sd = {}
        for k, v in self.__dict__.items():
            if k == "device":
                continue
            if torch.is_tensor(v) and v.shape[:1] == (self.mem_size,):
                sd[k] = v[:n].cpu().clone()
            elif torch.is_tensor(v):
                sd[k] = v.cpu()
            else:
                sd[k] = v
        with open(fn, "wb") as f:
            pickle.dump(sd, f)

    def load_checkpoint(self, filename: Optional[str] = None):
        fn = filename or self.filename
        with open(fn, "rb") as f:
            sd = pickle.load(f)
        for k, v in sd.items():
            if k == "mem_size":
                # capacity is a property of THIS buffer's tensors, not of
                # the checkpoint: restoring a larger saved mem_size would
                # desync ring indexing from the allocated rows
                continue
            cur = getattr(self, k, None)
            if torch.is_tensor(v) and torch.is_tensor(cur):
                if v.shape == cur.shape:
                    cur.copy_(v.to(self.device))
                else:       # filled-prefix checkpoint; a prefix longer
                    # than the current capacity (checkpoint from a larger
                    # buffer) keeps only what fits
                    n = min(v.shape[0], cur.shape[0])
                    cur[:n].copy_(v[:n].to(self.device))
            else:
                setattr(self, k, v)
        # a counter from a larger buffer over-claims filled entries the
        # copy above could not keep — clamp so len() stays honest
        if sd.get("mem_size", self.mem_size) > self.mem_size \
                and self.mem_cntr > self.mem_size:
            self.mem_cntr = self.mem_size


class SumTree:
    """Host-side binary sum tree (exact parity with reference
    ``enet_sac.py:82-200``); used as the distribution oracle in tests."""

    def __init__(self, capacity: int):
        self.capacity = capacity
        self.tree = np.zeros(2 * capacity - 1, dtype=np.float64)
        self.data_idx = np.zeros(capacity, dtype=np.int64)
        self.write = 0
        self.n_entries = 0

    def total(self) -> float:
        return float(self.tree[0])

    def add(self, priority: float, data_index: int):
        leaf = self.write + self.capacity - 1
        self.data_idx[self.write] = data_index
        self.update(leaf, priority)
        self.write = (self.write + 1) % self.capacity
        self.n_entries = min(self.n_entries + 1, self.capacity)

    def update(self, leaf: int, priority: float):
        change = priority - self.tree[leaf]
        self.tree[leaf] = priority
        while leaf != 0:
            leaf = (leaf - 1) // 2
            self.tree[leaf] += change

    def get_leaf(self, v: float) -> Tuple[int, float, int]:
        idx = 0
        while True:
            left = 2 * idx + 1
            if left >= len(self.tree):
                break
            if v <= self.tree[left]:
                idx = left
            else:
                v -= self.tree[left]
                idx = left + 1
        data_slot = idx - (self.capacity - 1)
        return idx, float(self.tree[idx]), int(self.data_idx[data_slot])


class PERBuffer(ReplayBuffer):
    """Proportional prioritized replay (stratified), device-resident.

    Matches the reference PER semantics (``enet_sac.py:203-330``): new
    transitions get the current max priority (min 1.0); sampling is
    stratified proportional; importance weights (N·P)^-beta normalized by
    the max; priorities updated as (|delta| + eps)^alpha, clipped at 1.
    """

    EPS = 0.01
    ALPHA = 0.6
    BETA0 = 0.4
    BETA_INC = 0.001
    MAX_PRIORITY = 1.0

    def __init__(self, max_size, input_dims, n_actions, device=None):
        super().__init__(max_size, input_dims, n_actions, device)
        self.priorities = torch.zeros(self.mem_size, dtype=torch.float32,
                                      device=self.device)
        self.beta = self.BETA0
        self.filename = "prioritized_replaymem_sac.model"

    def store_transition(self, state, action, reward, state_, done,
                         hint=None):
        i = self.mem_cntr % self.mem_size
        super().store_transition(state, action, reward, state_, done, hint)
        mx = float(self.priorities.max()) if len(self) > 1 else 0.0
        self.priorities[i] = mx if mx > 0 else self.MAX_PRIORITY

    def store_batch(self, states, actions, rewards, states_, dones,
                    hints=None):
        had = len(self) > 1
        idx = super().store_batch(states, actions, rewards, states_, dones,
                                  hints)
        mx = self.priorities.max().clamp(min=0.0) if had \
            else torch.tensor(0.0, device=self.device)
        self.priorities[idx] = torch.where(
            mx > 0, mx, torch.tensor(self.MAX_PRIORITY, device=self.device))
        return idx

    def sample_buffer(self, batch_size: int):
        n = len(self)
        pri = self.priorities[:n]
        self.beta = min(1.0, self.beta + self.BETA_INC)
        idx, probs, weights = per_ops.sample_with_weights(pri, batch_size,
                                                          self.beta)
        batch = (self.state_memory[idx], self.action_memory[idx],
                 self.reward_memory[idx], self.new_state_memory[idx],
                 self.terminal_memory[idx], self.hint_memory[idx])
        return batch, idx, weights

    def update_priorities(self, idx: torch.Tensor, td_errors: torch.Tensor):
        per_ops.update_priorities(self.priorities, idx, td_errors, self.EPS,
                                  self.ALPHA, self.MAX_PRIORITY)
