"""Tests: transformer/regressor/TSK models, fuzzy controller, fuzzy env."""

import numpy as np
import pytest
import torch

from smartcal_amd.models import (SupervisedBuffer, TrainingBuffer,
                                 TransformerEncoder, RegressorNet,
                                 TSKModel, center_difference_loss,
                                 sigma_loss)
from smartcal_amd.fuzzy import DemixController, trapmf


def test_supervised_buffer_roundtrip(tmp_path):
    buf = SupervisedBuffer(8, (4,), (2,))
    for i in range(5):
        buf.store_data(np.full(4, i, np.float32), np.full(2, -i, np.float32))
    x, y = buf.sample_minibatch(3)
    assert x.shape == (3, 4) and y.shape == (3, 2)
    fn = str(tmp_path / "b.buffer")
    buf.save_checkpoint(fn)
    buf2 = SupervisedBuffer(8, (4,), (2,))
    buf2.load_checkpoint(fn)
    np.testing.assert_array_equal(buf2.x, buf.x)
    buf2.resize(16)
    assert buf2.x.shape == (16, 4)
    tb = TrainingBuffer(8, 4, 2)
    tb.store(np.ones(4), np.zeros(2))
    x, y = tb.sample(2)
    assert x.shape == (2, 4)


def test_transformer_forward_backward():
    torch.manual_seed(0)
    # the demixing classifier config scaled down: K=6 heads
    net = TransformerEncoder(num_layers=1, input_dim=6 * 24, model_dim=6 * 11,
                             num_classes=5, num_heads=6, dropout=0.0)
    x = torch.randn(4, 6 * 24)
    y = net(x)
    assert y.shape == (4, 5)
    assert (y >= 0).all() and (y <= 1).all()
    loss = torch.nn.functional.binary_cross_entropy(y,
                                                    torch.rand(4, 5))
    loss.backward()
    maps = net.get_attention_maps(x)
    assert len(maps) == 1 and maps[0].shape[-1] == maps[0].shape[-2]


def test_regressor_shapes():
    net = RegressorNet(20, 5)
    y = net(torch.randn(7, 20))
    assert y.shape == (7, 5)
    assert (y.abs() <= 1).all()


def test_tsk_fit_simple():
    """TSK must fit a small piecewise target better than init."""
    torch.manual_seed(0)
    rng = np.random.default_rng(0)
    X = rng.standard_normal((256, 3)).astype(np.float32)
    Y = np.tanh(X[:, :2] * 0.5).astype(np.float32)
    from smartcal_amd.models.tsk import antecedent_init_center
    model = TSKModel(3, 2, n_rule=3,
                     init_center=antecedent_init_center(X, 3))
    xt = torch.from_numpy(X)
    yt = torch.from_numpy(Y)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    def lossfn():
        out = model(xt)
        return (out - yt).norm() ** 2 / X.shape[0] \
            + 1e-4 * center_difference_loss(model) \
            + 1e-4 * sigma_loss(model)
    l0 = float(lossfn().detach())
    for _ in range(150):
        opt.zero_grad()
        l = lossfn()
        l.backward()
        opt.step()
    assert float(lossfn().detach()) < 0.5 * l0


def test_trapmf():
    x = np.array([-1.0, 0.0, 0.5, 1.0, 1.5, 2.0, 3.0])
    y = trapmf(x, [0, 1, 2, 3])
    np.testing.assert_allclose(y, [0, 0, 0.5, 1, 1, 1, 0])


def test_fuzzy_controller_roundtrip():
    ctrl = DemixController(n_action=32)
    # default config → action → limits is a fixed point
    a0 = ctrl.update_action()
    cfg_before = list(ctrl.config["inputs"]["_azimuth"]["medium"])
    ctrl.update_limits(a0)
    assert np.allclose(ctrl.update_action(), a0, atol=1e-9)
    np.testing.assert_allclose(
        ctrl.config["inputs"]["_azimuth"]["medium"], cfg_before, atol=1e-9)
    # evaluation: low elevation should produce a low priority,
    # close separation + high elevation a higher one
    ctrl.create_controller()
    p_low = ctrl.evaluate(0, 0, -60, 40, 100, 0.5, 0.3)
    p_high = ctrl.evaluate(0, 0, 80, 10, 5, 20, 60)
    assert p_high > p_low
    assert ctrl.get_high_priority() == 70


def test_fuzzy_env():
    from smartcal_amd.envs.demix_fuzzy import FuzzyDemixingEnv
    env = FuzzyDemixingEnv(K=3, Nf=2, Ninf=16, Tdelta=4, Ts=1,
                           provide_hint=True, provide_influence=False,
                           N_stations=6, device="cpu", seed=0)
    obs = env.reset()
    assert obs["metadata"].shape == (5 * 3 + 2,)
    assert env.action_space.shape == (24 * 2 + 8,)
    a = env.action_space.sample()
    obs, r, done, hint, info = env.step(a)
    assert np.isfinite(r)
    assert hint.shape == (24 * 2 + 8,)
    assert env.maxiter == 15
    # selection flags set for the chosen clusters
    md = obs["metadata"] / 1e-3
    assert md[5 * 3 - 1] == 1
