"""Prioritized-replay sampling primitives (N5) — hand-written CDNA4 kernel.

The reference implements proportional PER with a serial numpy sum-tree
(``enet_sac.py:82-200``: O(log n) pointer chasing per sample, stratified
draws at ``:270-312``, batched priority update at ``:313-323``). The
MI355X-native design replaces the tree with a device-resident priority
array and ONE hand-written kernel launch (``csrc/per.hip``): a 256-thread
workgroup stages the whole array's prefix sum in LDS (160 KB/CU covers
every workload's mem_size), resolves all stratified inverse-CDF targets by
binary search on the LDS prefix, and computes max-normalized importance
weights in the same launch — identical sampling distribution, no host
round trip, no cumsum/searchsorted kernel chain.

CPU tensors (and priority arrays beyond the 32k-LDS capacity) run the
equivalent torch composition, which doubles as the numerics oracle in
``tests/``.
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import ext, use_hip

# must match PER_LDS_MAX_N in csrc/bindings.hip
_LDS_MAX_N = 32768


def _torch_stratified(priorities: torch.Tensor, batch_size: int,
                      u: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    n = priorities.shape[0]
    total = priorities.sum()
    cdf = torch.cumsum(priorities, dim=0)
    seg = total / batch_size
    targets = (torch.arange(batch_size, device=priorities.device,
                            dtype=priorities.dtype) + u) * seg
    idx = torch.searchsorted(cdf, targets).clamp_(max=n - 1)
    probs = priorities[idx] / total
    return idx, probs


def stratified_sample(priorities: torch.Tensor, batch_size: int,
                      generator=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stratified proportional sampling from a priority array.

    The total mass is split into ``batch_size`` equal segments and one
    uniform draw is taken per segment (the sum-tree's ``get_leaf`` descent,
    reference ``enet_sac.py:270-312``, computes exactly this inverse CDF).

    Returns (indices (B,), probabilities P(i) (B,)).
    """
    u = torch.rand(batch_size, device=priorities.device, generator=generator)
    if use_hip(priorities) and priorities.shape[0] <= _LDS_MAX_N:
        idx, probs, _ = ext().per_sample(priorities.contiguous(), u, 0.0)
        return idx, probs
    return _torch_stratified(priorities, batch_size, u)


def importance_weights(probs: torch.Tensor, n_entries: int,
                       beta: float) -> torch.Tensor:
    """(N * P(i))^-beta, normalized by the max weight."""
    w = (n_entries * probs).clamp_min(1e-12).pow(-beta)
    return w / w.max()


def sample_with_weights(priorities: torch.Tensor, batch_size: int,
                        beta: float, generator=None
                        ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Fused sample + importance weights: ONE kernel launch on the GPU.

    Returns (indices (B,), probabilities (B,), normalized IS weights (B,)).
    """
    u = torch.rand(batch_size, device=priorities.device, generator=generator)
    if use_hip(priorities) and priorities.shape[0] <= _LDS_MAX_N:
        return ext().per_sample(priorities.contiguous(), u, float(beta))
    idx, probs = _torch_stratified(priorities, batch_size, u)
    return idx, probs, importance_weights(probs, priorities.shape[0], beta)


def update_priorities(priorities: torch.Tensor, idx: torch.Tensor,
                      td_errors: torch.Tensor, eps: float, alpha: float,
                      max_priority: float) -> None:
    """priorities[idx] = min(|td| + eps, max_priority)^alpha, in place
    (reference ``enet_sac.py:313-323``)."""
    td = td_errors.detach().reshape(-1).to(torch.float32)
    if use_hip(priorities):
        ext().per_update(priorities, idx.to(torch.long), td.contiguous(),
                         float(eps), float(alpha), float(max_priority))
        return
    pri = (td.abs() + eps).clamp(max=max_priority).pow(alpha)
    priorities[idx] = pri.to(priorities.dtype)
