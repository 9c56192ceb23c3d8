"""Minimal gym-compatible environment API.

The reference builds its environments on ``gymnasium`` (``gym.Env`` +
``spaces.Box`` / ``spaces.Dict``, see reference ``elasticnet/enetenv.py:1-45``).
gymnasium is not available in this image, so we provide the small subset the
framework needs with the same semantics: ``Env`` with ``reset/step/render/
close``, ``Box.sample()``, ``Dict`` observation spaces. Environments written
against this module also run unmodified against gymnasium (same API surface).
"""

from __future__ import annotations

import numpy as np


class Space:
    def sample(self):
        raise NotImplementedError

    def contains(self, x) -> bool:  # pragma: no cover - convenience
        raise NotImplementedError


class Box(Space):
    """Continuous box space ``low <= x <= high`` with a given shape."""

    def __init__(self, low, high, shape=None, dtype=np.float32):
        low = np.asarray(low, dtype=dtype)
        high = np.asarray(high, dtype=dtype)
        if shape is not None:
            low = np.broadcast_to(low, shape).astype(dtype)
            high = np.broadcast_to(high, shape).astype(dtype)
        self.low = low
        self.high = high
        self.shape = low.shape
        self.dtype = dtype

    def sample(self):
        # uniform in [low, high]; unbounded dims sample standard normal
        finite = np.isfinite(self.low) & np.isfinite(self.high)
        out = np.empty(self.shape, dtype=self.dtype)
        u = np.random.uniform(size=self.shape)
        out[finite] = (self.low + u * (self.high - self.low))[finite]
        if not finite.all():
            out[~finite] = np.random.randn((~finite).sum())
        return out.astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(
            (x >= self.low - 1e-6).all() and (x <= self.high + 1e-6).all()
        )


class Dict(Space):
    """Dict of named sub-spaces (observation spaces)."""

    def __init__(self, spaces_dict):
        self.spaces = dict(spaces_dict)

    def sample(self):
        return {k: s.sample() for k, s in self.spaces.items()}

    def __getitem__(self, k):
        return self.spaces[k]


class Env:
    """Base environment: subclasses implement reset() and step(action)."""

    metadata: dict = {"render.modes": ["human"]}
    action_space: Space
    observation_space: Space

    def reset(self):
        raise NotImplementedError

    def step(self, action):
        raise NotImplementedError

    def render(self, mode="human"):  # pragma: no cover - optional
        pass

    def close(self):
        pass


class spaces:  # namespace mirror of gymnasium.spaces
    Box = Box
    Dict = Dict
