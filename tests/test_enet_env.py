"""Elastic-net env: analytic fast path vs fully-generic autograd oracle.

The env computes the influence eigenvalues with the analytic identities
jac(Ax, x) = A and d^2 loss/dx dy^T = -2 A^T (see ops/enet.py docstring);
these tests check both identities against the reference's fully-generic
autograd formulation (reference ``elasticnet/enetenv.py:117-137``).
"""

import numpy as np
import torch

from smartcal_amd import autograd_tools as at
from smartcal_amd.envs.enet import ENetEnv, obs_to_state
from smartcal_amd.ops import enet as enet_ops


def test_analytic_ll_matches_autograd():
    torch.manual_seed(0)
    N, M = 10, 8
    A = torch.randn(N, M)
    A /= A.norm()
    y = torch.randn(N)
    rho1, rho2 = 0.05, 0.01
    x, opt = enet_ops.lbfgs_solve_reference(A, y, rho1, rho2, epochs=10)

    xr = x.clone().requires_grad_(True)

    def lossfn(Ain, yin, xin):
        err = yin - Ain @ xin
        return (err ** 2).sum() + rho1 * (xin ** 2).sum() \
            + rho2 * xin.abs().sum()

    # reference-style: jacobian of d loss/dx wrt y evaluated at e = ones
    def df_dx(yi):
        return at.gradient(lossfn(A, yi, xr), xr)

    e = torch.ones(N)
    ll = torch.autograd.functional.jacobian(df_dx, e)
    assert torch.allclose(ll, -2 * A.t(), atol=1e-5)

    # reference-style: jacobian of the model wrt x is A
    jac = at.jacobian(A @ xr, xr)
    assert torch.allclose(jac, A, atol=1e-5)


def test_influence_eigs_vs_reference_loop():
    torch.manual_seed(0)
    N, M = 10, 8
    A = torch.randn(N, M)
    A /= A.norm()
    y = torch.randn(N)
    x, opt = enet_ops.lbfgs_solve_reference(A, y, 0.05, 0.01, epochs=10)
    Y, S = enet_ops.curvature_stacks(opt)
    EE_fast = enet_ops.influence_eigs_reference(A, Y, S, rho1=0.05)

    # reference formulation: per-column inv_hessian_mult + torch.linalg.eig
    ll = -2 * A.t()
    mm = torch.zeros_like(ll)
    for i in range(N):
        mm[:, i] = at.inv_hessian_mult(opt, ll[:, i].clone())
    B = A @ mm
    E, _ = torch.linalg.eig(B)
    EE_ref = E.real + 1
    assert torch.allclose(EE_fast.sort().values, EE_ref.sort().values,
                          rtol=1e-3, atol=1e-4)


def test_env_step_reset_contract():
    np.random.seed(0)
    torch.manual_seed(0)
    env = ENetEnv(M=6, N=8, device=torch.device("cpu"))
    obs = env.reset()
    assert obs["A"].numel() == 48
    assert obs["eig"].numel() == 8
    a = np.random.uniform(-1, 1, size=2).astype(np.float32)
    obs2, reward, done, info = env.step(a)
    assert torch.isfinite(reward)
    assert not done
    assert obs2["eig"].shape[0] == 8
    state = obs_to_state(obs2)
    assert state.numel() == 8 + 48

    # out-of-range action picks up a penalty and clamps rho
    obs3, r3, _, _ = env.step(np.array([5.0, -5.0], dtype=np.float32))
    assert torch.isfinite(r3)
    assert float(env.rho.min()) >= 1e-3 - 1e-6
    assert float(env.rho.max()) <= 1e-1 + 1e-6


def test_env_hint_in_action_space():
    np.random.seed(0)
    torch.manual_seed(0)
    env = ENetEnv(M=6, N=8, provide_hint=True, device=torch.device("cpu"))
    env.reset()
    obs, reward, done, hint, info = env.step(
        np.zeros(2, dtype=np.float32))
    assert hint.shape == (2,)
    assert (hint >= -1.01).all() and (hint <= 1.01).all()


def test_skenet_gridsearchcv():
    """SKEnet plugs into real sklearn GridSearchCV (`enetenv.py:249-295`)."""
    import numpy as np
    from sklearn.model_selection import GridSearchCV
    from smartcal_amd.envs.enet import SKEnet
    rng = np.random.default_rng(0)
    A = rng.normal(size=(20, 20)).astype("f")
    x = np.zeros(20, "f"); x[3] = 1.0
    y = A @ x
    gs = GridSearchCV(SKEnet(), {"lambda1": [0.001, 0.01],
                                 "lambda2": [0.001, 0.01]}, cv=2)
    gs.fit(A, y)
    assert gs.best_score_ > -0.01
    est = SKEnet(**gs.best_params_).fit(A, y)
    assert np.mean((est.predict(A) - y) ** 2) < 1e-3


def test_env_sanitizes_nonfinite(monkeypatch):
    """NaN from a degenerate solve must not escape the env boundary."""
    import torch
    import numpy as np
    from smartcal_amd.envs import enet as enet_env_mod
    env = enet_env_mod.ENetEnv(8, 8)
    env.reset()

    def bad_solve(A, y, r1, r2, pen):
        x = torch.full((8,), float("nan"))
        EE = torch.full((8,), float("nan"))
        return x, EE, float("nan")

    monkeypatch.setattr(enet_env_mod.enet_ops, "solve_and_influence",
                        bad_solve)
    obs, r, done, info = env.step(np.array([0.5, float("nan")]))
    assert np.isfinite(float(r))
    assert torch.isfinite(obs["eig"]).all()


def test_vec_enet_env_contract():
    """VecENetEnv: batched shapes, finite rewards, per-env independence."""
    import torch
    from smartcal_amd.envs.vec_enet import VecENetEnv
    torch.manual_seed(0)
    np.random.seed(0)
    env = VecENetEnv(3, 8, 8)
    obs = env.reset()
    assert obs["A"].shape == (3, 64) and obs["eig"].shape == (3, 8)
    a = np.random.uniform(-1, 1, size=(3, 2)).astype(np.float32)
    obs2, r, done, info = env.step(a)
    assert r.shape == (3,) and torch.isfinite(r).all()
    assert obs2["eig"].shape == (3, 8)
    assert not done.any()
    err = env.solution_error()
    assert err.shape == (3,) and torch.isfinite(err).all()
    # out-of-range actions get the boundary penalty per env
    obs3, r3, *_ = env.step(np.array([[2.0, 2.0], [0.0, 0.0],
                                      [-2.0, -2.0]], dtype=np.float32))
    assert torch.isfinite(r3).all()
    # optional clamp (the reference's commented-out multi-env clamp)
    envc = VecENetEnv(2, 8, 8, reward_clamp=1.0)
    envc.reset()
    _, rc, *_ = envc.step(np.zeros((2, 2), np.float32))
    assert (rc.abs() <= 1.0).all()
