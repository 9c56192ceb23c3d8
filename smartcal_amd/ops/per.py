"""Prioritized-replay sampling primitives (N5).

The reference implements proportional PER with a serial numpy sum-tree
(``enet_sac.py:82-200``: O(log n) pointer chasing per sample). The
MI355X-native design replaces the tree with a device-resident priority
array + prefix-sum inverse-CDF sampling: one ``cumsum`` (rocPRIM scan under
torch) + one batched ``searchsorted`` — identical sampling distribution
(stratified proportional), fully batched, no host round trip.
"""

from __future__ import annotations

from typing import Tuple

import torch


def stratified_sample(priorities: torch.Tensor, batch_size: int,
                      generator=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stratified proportional sampling from a priority array.

    The total mass is split into ``batch_size`` equal segments and one
    uniform draw is taken per segment (the sum-tree's ``get_leaf`` descent,
    reference ``enet_sac.py:270-312``, computes exactly this inverse CDF).

    Returns (indices (B,), probabilities P(i) (B,)).
    """
    n = priorities.shape[0]
    total = priorities.sum()
    cdf = torch.cumsum(priorities, dim=0)
    seg = total / batch_size
    u = torch.rand(batch_size, device=priorities.device, generator=generator)
    targets = (torch.arange(batch_size, device=priorities.device,
                            dtype=priorities.dtype) + u) * seg
    idx = torch.searchsorted(cdf, targets).clamp_(max=n - 1)
    probs = priorities[idx] / total
    return idx, probs


def importance_weights(probs: torch.Tensor, n_entries: int,
                       beta: float) -> torch.Tensor:
    """(N * P(i))^-beta, normalized by the max weight."""
    w = (n_entries * probs).clamp_min(1e-12).pow(-beta)
    return w / w.max()
