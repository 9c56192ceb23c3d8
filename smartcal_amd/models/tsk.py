"""Takagi-Sugeno-Kang (TSK) fuzzy network — from scratch.

The reference trains a pytsk model (`demixing_rl/train_tsk.py:56-70`):
Gaussian-membership antecedents in high-dim mode (mean of
log-memberships over input dims → softmax over rules), wrapped with
LayerNorm(n_rule)+ReLU, and an order-1 consequent, with tanh output and
two regularizers (inverse center-distance, sigma magnitude,
`train_tsk.py:81-110`). pytsk is not in this image, so the same model is
implemented here directly; parameter roles (`center`, `sigma`,
consequent weights) match so the training script's regularizers apply
unchanged.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

__all__ = ["AntecedentGMF", "TSKModel", "antecedent_init_center",
           "center_difference_loss", "sigma_loss"]


def antecedent_init_center(X: np.ndarray, n_rule: int,
                           seed: int = 0) -> np.ndarray:
    """KMeans-style rule-center init (pytsk's `antecedent_init_center`):
    (in_dim, n_rule) centers from clustering the training inputs."""
    try:
        from sklearn.cluster import KMeans
        km = KMeans(n_clusters=n_rule, n_init=4, random_state=seed)
        km.fit(X)
        return km.cluster_centers_.T.astype(np.float32)
    except Exception:  # pragma: no cover - sklearn edge cases
        rng = np.random.default_rng(seed)
        idx = rng.choice(X.shape[0], n_rule, replace=X.shape[0] < n_rule)
        return X[idx].T.astype(np.float32)


class AntecedentGMF(nn.Module):
    """Gaussian-MF antecedent, high-dim mode: firing levels are the
    softmax over rules of the mean log-membership."""

    def __init__(self, in_dim: int, n_rule: int,
                 init_center: np.ndarray | None = None,
                 init_sigma: float = 1.0):
        super().__init__()
        if init_center is None:
            init_center = np.random.randn(in_dim, n_rule).astype(np.float32)
        self.center = nn.Parameter(
            torch.as_tensor(init_center, dtype=torch.float32))
        self.sigma = nn.Parameter(
            torch.full((in_dim, n_rule), float(init_sigma)))

    def forward(self, x):
        # x (B, in_dim) → z (B, in_dim, n_rule)
        d = x.unsqueeze(2) - self.center.unsqueeze(0)
        z = -d.pow(2) / (2 * self.sigma.pow(2).unsqueeze(0) + 1e-12)
        return torch.softmax(z.mean(dim=1), dim=-1)       # (B, n_rule)


class TSKModel(nn.Module):
    """Full TSK: antecedent (optionally wrapped with LN+ReLU as the
    reference does) + order-0/1 consequent + tanh output."""

    def __init__(self, in_dim: int, out_dim: int, n_rule: int,
                 order: int = 1, init_center: np.ndarray | None = None,
                 layernorm: bool = True):
        super().__init__()
        self.in_dim = in_dim
        self.n_rule = n_rule
        self.order = order
        self.antecedent = AntecedentGMF(in_dim, n_rule, init_center)
        self.post = nn.Sequential(nn.LayerNorm(n_rule), nn.ReLU()) \
            if layernorm else nn.Identity()
        cons_in = (in_dim + 1) if order == 1 else 1
        self.cons = nn.Parameter(
            0.01 * torch.randn(n_rule, cons_in, out_dim))

    def firing(self, x):
        return self.post(self.antecedent(x))

    def forward(self, x):
        frs = self.firing(x)                               # (B, R)
        if self.order == 1:
            x1 = torch.cat([x, torch.ones_like(x[:, :1])], dim=1)
            rule_out = torch.einsum('bi,rio->bro', x1, self.cons)
        else:
            rule_out = self.cons[:, 0, :].unsqueeze(0).expand(
                x.shape[0], -1, -1)
        return torch.tanh(torch.einsum('br,bro->bo', frs, rule_out))


def center_difference_loss(model: TSKModel) -> torch.Tensor:
    """Inverse pairwise center-distance regularizer
    (`train_tsk.py:81-96`)."""
    centers = model.antecedent.center            # (in_dim, R)
    R = centers.shape[1]
    total = centers.new_zeros(())
    for r in range(R):
        for r1 in range(r + 1, R):
            d = centers[:, r] - centers[:, r1]
            total = total + (1.0 / (d.pow(2) + 1e-5)).sum()
    M = centers.shape[0]
    return total / (M * R * (R - 1) / 2)


def sigma_loss(model: TSKModel) -> torch.Tensor:
    """Mean squared sigma regularizer (`train_tsk.py:98-107`)."""
    s = model.antecedent.sigma
    return s.pow(2).sum() / s.numel()
