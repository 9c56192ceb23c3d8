"""Replay buffers: ring semantics, PER distribution vs sum-tree oracle."""

import numpy as np
import torch

from smartcal_amd.ops import per as per_ops
from smartcal_amd.rl.buffers import PERBuffer, ReplayBuffer, SumTree


def test_ring_buffer_roundtrip():
    buf = ReplayBuffer(4, [3], 2)
    for i in range(6):
        buf.store_transition(torch.full((3,), float(i)),
                             np.array([i, -i], dtype=np.float32),
                             float(i), torch.full((3,), float(i + 1)),
                             False, np.zeros(2, dtype=np.float32))
    assert len(buf) == 4
    # slots hold the last 4 transitions (ring overwrite)
    vals = sorted(buf.state_memory[:, 0].tolist())
    assert vals == [2.0, 3.0, 4.0, 5.0]
    s, a, r, s2, d, h = buf.sample_buffer(8)
    assert s.shape == (8, 3) and a.shape == (8, 2)


def test_stratified_sampler_matches_sumtree_distribution():
    torch.manual_seed(0)
    n = 64
    pri = torch.rand(n) + 0.05
    # empirical distribution from the prefix-sum sampler
    counts = torch.zeros(n)
    for _ in range(300):
        idx, probs = per_ops.stratified_sample(pri, 16)
        for i in idx:
            counts[i] += 1
    emp = counts / counts.sum()
    expected = pri / pri.sum()
    # sum-tree oracle draws from the same proportional distribution
    tree = SumTree(n)
    for i in range(n):
        tree.add(float(pri[i]), i)
    tcounts = torch.zeros(n)
    rng = np.random.default_rng(0)
    for _ in range(300):
        seg = tree.total() / 16
        for j in range(16):
            v = rng.uniform(j * seg, (j + 1) * seg)
            _, _, di = tree.get_leaf(v)
            tcounts[di] += 1
    temp = tcounts / tcounts.sum()
    assert (emp - expected).abs().max() < 0.02
    assert (temp - expected).abs().max() < 0.02
    assert (emp - temp).abs().max() < 0.02


def test_per_buffer_update_and_weights():
    buf = PERBuffer(32, [2], 1)
    for i in range(32):
        buf.store_transition(torch.randn(2), np.array([0.1], np.float32),
                             0.5, torch.randn(2), False,
                             np.zeros(1, np.float32))
    batch, idx, w = buf.sample_buffer(8)
    assert w.shape == (8,)
    assert float(w.max()) <= 1.0 + 1e-6
    buf.update_priorities(idx, torch.full((8,), 2.0))
    # priorities are clipped at 1 then raised to alpha
    assert torch.allclose(buf.priorities[idx],
                          torch.ones(8), atol=1e-5)
