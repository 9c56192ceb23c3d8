"""Replay buffers for dict observations {img, meta} — device resident.

Mirrors the record layout of the reference CNN-agent buffers
(`calibration/calib_sac.py:26-73`, `demixing_rl/demix_sac.py:310-360`):
image and metadata stored separately, plus action/reward/next/done/hint.
Tensors live on the training device (288 GB HBM3E holds even the
demixing 16000×128² buffer ≈ 4 GB without spilling); sampling is pure
device indexing, no H2D copies per learn step.
"""

from __future__ import annotations

import pickle
from typing import Optional

import numpy as np
import torch

from ..ops import per as per_ops

__all__ = ["DictReplayBuffer", "DictPERBuffer"]


class DictReplayBuffer:
    def __init__(self, max_size: int, img_shape, meta_shape, n_actions: int,
                 device: Optional[torch.device] = None):
        self.mem_size = int(max_size)
        self.mem_cntr = 0
        self.device = device if device is not None else torch.device("cpu")
        z = lambda *s: torch.zeros(s, dtype=torch.float32,
                                   device=self.device)
        self.img_memory = z(self.mem_size, *img_shape)
        self.new_img_memory = z(self.mem_size, *img_shape)
        self.meta_memory = z(self.mem_size, *np.atleast_1d(meta_shape))
        self.new_meta_memory = torch.zeros_like(self.meta_memory)
        self.action_memory = z(self.mem_size, n_actions)
        self.reward_memory = z(self.mem_size)
        self.terminal_memory = torch.zeros(self.mem_size, dtype=torch.bool,
                                           device=self.device)
        self.hint_memory = z(self.mem_size, n_actions)
        self.filename = "replaymem_cnn.model"
        self._img_key = None   # detected from first stored obs

    def __len__(self):
        return min(self.mem_cntr, self.mem_size)

    def _split(self, obs):
        if self._img_key is None:
            self._img_key = "img" if "img" in obs else "infmap"
        meta_key = "sky" if "sky" in obs else "metadata"
        return obs[self._img_key], obs[meta_key]

    def _as(self, x, like):
        if torch.is_tensor(x):
            return x.detach().to(like.device, like.dtype).reshape(like.shape)
        return torch.as_tensor(np.asarray(x), dtype=like.dtype,
                               device=like.device).reshape(like.shape)

    def store_transition(self, state, action, reward, state_, done,
                         hint=None):
        i = self.mem_cntr % self.mem_size
        img, meta = self._split(state)
        img_, meta_ = self._split(state_)
        self.img_memory[i] = self._as(img, self.img_memory[i])
        self.meta_memory[i] = self._as(meta, self.meta_memory[i])
        self.new_img_memory[i] = self._as(img_, self.new_img_memory[i])
        self.new_meta_memory[i] = self._as(meta_, self.new_meta_memory[i])
        self.action_memory[i] = self._as(action, self.action_memory[i])
        self.reward_memory[i] = float(reward) if not torch.is_tensor(reward) \
            else reward.detach().to(self.device, torch.float32).reshape(())
        self.terminal_memory[i] = bool(done)
        if hint is not None:
            self.hint_memory[i] = self._as(hint, self.hint_memory[i])
        self.mem_cntr += 1

    def sample_buffer(self, batch_size: int):
        idx = torch.randint(0, len(self), (batch_size,), device=self.device)
        return self._gather(idx)

    def _gather(self, idx):
        return (self.img_memory[idx], self.meta_memory[idx],
                self.action_memory[idx], self.reward_memory[idx],
                self.new_img_memory[idx], self.new_meta_memory[idx],
                self.terminal_memory[idx], self.hint_memory[idx])

    def save_checkpoint(self, filename=None):
        # store only the filled prefix: a fresh 16000-slot image buffer
        # is ~1.3 GB of zeros otherwise
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

    def load_checkpoint(self, filename=None):
        fn = filename or self.filename
        with open(fn, "rb") as f:
            sd = pickle.load(f)
        for k, v in sd.items():
            cur = getattr(self, k, None)
            if torch.is_tensor(v) and torch.is_tensor(cur):
                if v.shape == cur.shape:
                    cur.copy_(v.to(self.device))
                else:       # filled-prefix checkpoint
                    cur[:v.shape[0]].copy_(v.to(self.device))
            else:
                setattr(self, k, v)


class DictPERBuffer(DictReplayBuffer):
    """Prioritized variant; same priority semantics as the reference's
    per-module SumTree/PER (`demix_sac.py:30-300`) via the batched
    stratified sampler in ``ops.per``."""

    EPS = 0.01
    ALPHA = 0.6
    BETA0 = 0.4
    BETA_INC = 0.001
    MAX_PRIORITY = 1.0

    def __init__(self, max_size, img_shape, meta_shape, n_actions,
                 device=None, normalize_reward: bool = False):
        super().__init__(max_size, img_shape, meta_shape, n_actions, device)
        self.priorities = torch.zeros(self.mem_size, dtype=torch.float32,
                                      device=self.device)
        self.beta = self.BETA0
        self.normalize_reward = normalize_reward
        self.filename = "prioritized_replaymem_cnn.model"

    def store_transition(self, state, action, reward, state_, done,
                         hint=None):
        i = self.mem_cntr % self.mem_size
        super().store_transition(state, action, reward, state_, done, hint)
        mx = float(self.priorities.max()) if len(self) > 1 else 0.0
        self.priorities[i] = mx if mx > 0 else self.MAX_PRIORITY

    def sample_buffer(self, batch_size: int):
        n = len(self)
        self.beta = min(1.0, self.beta + self.BETA_INC)
        idx, probs, weights = per_ops.sample_with_weights(
            self.priorities[:n], batch_size, self.beta)
        batch = self._gather(idx)
        if self.normalize_reward:
            # reference `demix_td3.py:162-166`: standardize sampled rewards
            # by the filled buffer's running statistics
            r = self.reward_memory[:n]
            mu, sd = r.mean(), r.std().clamp(min=1e-6)
            batch = (*batch[:3], (batch[3] - mu) / sd, *batch[4:])
        return batch, idx, weights

    def update_priorities(self, idx, td_errors):
        per_ops.update_priorities(self.priorities, idx, td_errors, self.EPS,
                                  self.ALPHA, self.MAX_PRIORITY)
