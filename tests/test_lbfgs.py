"""LBFGSNew unit tests: convergence, state layout, inverse-Hessian replay."""

import numpy as np
import pytest
import torch

from smartcal_amd.autograd_tools import inv_hessian_mult, inv_hessian_mult_mat
from smartcal_amd.optim import LBFGSNew


def _quadratic_problem(n=12, seed=3):
    g = torch.Generator().manual_seed(seed)
    Q = torch.randn(n, n, generator=g)
    H = Q @ Q.t() + n * torch.eye(n)  # SPD
    b = torch.randn(n, generator=g)
    xstar = torch.linalg.solve(H, b)
    return H, b, xstar


def test_converges_on_quadratic():
    H, b, xstar = _quadratic_problem()
    x = torch.zeros(H.shape[0], requires_grad=True)
    opt = LBFGSNew([x], history_size=7, max_iter=10, line_search_fn=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = 0.5 * x @ H @ x - b @ x
        if loss.requires_grad:
            loss.backward()
        return loss

    for _ in range(20):
        opt.step(closure)
    assert torch.allclose(x.detach(), xstar, atol=1e-4)


def test_converges_on_elastic_net():
    torch.manual_seed(1)
    N, M = 20, 20
    A = torch.randn(N, M)
    A /= A.norm()
    y = torch.randn(N)
    rho1, rho2 = 0.05, 0.01
    x = torch.zeros(M, requires_grad=True)
    opt = LBFGSNew([x], history_size=7, max_iter=10, line_search_fn=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        err = y - A @ x
        loss = err.dot(err) + rho1 * x.dot(x) + rho2 * x.abs().sum()
        if loss.requires_grad:
            loss.backward()
        return loss

    losses = []
    for _ in range(20):
        losses.append(float(closure().detach()))
        opt.step(closure)
    final = float(closure().detach())
    assert final < losses[0]
    # compare against scipy L-BFGS-B on the same objective
    from scipy.optimize import minimize

    An, yn = A.numpy(), y.numpy()

    def f(v):
        e = yn - An @ v
        return float(e @ e + rho1 * v @ v + rho2 * np.abs(v).sum())

    res = minimize(f, np.zeros(M), method="L-BFGS-B")
    assert final <= res.fun * 1.05 + 1e-6


def test_state_layout_and_inv_hessian_mult():
    H, b, xstar = _quadratic_problem(n=10)
    x = torch.zeros(10, requires_grad=True)
    opt = LBFGSNew([x], history_size=7, max_iter=10, line_search_fn=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = 0.5 * x @ H @ x - b @ x
        if loss.requires_grad:
            loss.backward()
        return loss

    for _ in range(15):
        opt.step(closure)
    st = opt.state[opt._params[0]]
    assert "old_dirs" in st and "old_stps" in st
    assert len(st["old_dirs"]) > 0
    # y_i should equal H s_i exactly for a quadratic
    for yv, sv in zip(st["old_dirs"], st["old_stps"]):
        assert torch.allclose(yv, H @ sv, rtol=1e-3, atol=1e-4)
    # inverse-Hessian application approximates H^{-1} q on a quadratic
    q = torch.randn(10)
    r = inv_hessian_mult(opt, q.clone())
    r_exact = torch.linalg.solve(H, q)
    cos = torch.dot(r, r_exact) / (r.norm() * r_exact.norm())
    assert cos > 0.9


def test_inv_hessian_mult_mat_matches_columns():
    H, b, _ = _quadratic_problem(n=8)
    x = torch.zeros(8, requires_grad=True)
    opt = LBFGSNew([x], history_size=7, max_iter=10, line_search_fn=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        loss = 0.5 * x @ H @ x - b @ x
        if loss.requires_grad:
            loss.backward()
        return loss

    for _ in range(10):
        opt.step(closure)
    st = opt.state[opt._params[0]]
    Y = torch.stack(list(st["old_dirs"]))
    S = torch.stack(list(st["old_stps"]))
    Q = torch.randn(8, 5)
    R_mat = inv_hessian_mult_mat(Y, S, Q.clone())
    for j in range(5):
        rj = inv_hessian_mult(opt, Q[:, j].clone())
        assert torch.allclose(R_mat[:, j], rj, rtol=1e-4, atol=1e-5)


def test_batch_mode_descends():
    torch.manual_seed(0)
    W = torch.randn(30, 5)
    xt = torch.randn(5)
    yt = W @ xt + 0.01 * torch.randn(30)
    x = torch.zeros(5, requires_grad=True)
    opt = LBFGSNew([x], history_size=7, max_iter=4, line_search_fn=False,
                   batch_mode=True)

    def closure():
        if torch.is_grad_enabled():
            opt.zero_grad()
        idx = torch.randint(0, 30, (16,))
        e = W[idx] @ x - yt[idx]
        loss = e.dot(e)
        if loss.requires_grad:
            loss.backward()
        return loss

    for _ in range(30):
        opt.step(closure)
    assert torch.norm(x.detach() - xt) < 0.5


def test_enet_solve_matches_scipy_lbfgsb():
    """Our L-BFGS elastic-net solve reaches the same objective as scipy
    L-BFGS-B (the reference's SKEnet inner solver, `enetenv.py:249-295`)
    on random instances."""
    import numpy as np
    import torch
    from scipy.optimize import minimize
    from smartcal_amd.ops import enet as enet_ops

    rng = np.random.default_rng(0)
    for _ in range(3):
        N = M = 12
        A = rng.normal(size=(N, M)).astype(np.float32)
        A /= np.linalg.norm(A)
        y = (A @ (rng.normal(size=M) * (rng.random(M) > 0.6))) \
            .astype(np.float32)
        rho1, rho2 = 0.05, 0.02

        def f(x):
            r = y - A @ x
            return float(r @ r + rho1 * x @ x + rho2 * np.abs(x).sum())

        sp = minimize(f, np.zeros(M), method="L-BFGS-B")
        x_t, _ = enet_ops.lbfgs_solve_reference(
            torch.from_numpy(A), torch.from_numpy(y), rho1, rho2)
        ours = f(x_t.numpy().astype(np.float64))
        assert ours <= sp.fun * 1.02 + 1e-6, (ours, sp.fun)


def test_enet_solve_kkt_conditions():
    """Approximate KKT optimality of the elastic-net solve: at the
    solution, 2A^T(Ax-y) + 2 rho1 x + rho2 sign(x) ~ 0 on the support
    and |2A^T(Ax-y)_i| <~ rho2 off it (subgradient condition)."""
    import numpy as np
    import torch
    from smartcal_amd.ops import enet as enet_ops

    rng = np.random.default_rng(1)
    for trial in range(3):
        N = M = 16
        A = torch.from_numpy(rng.normal(size=(N, M)).astype(np.float32))
        A = A / A.norm()
        y = A @ torch.from_numpy(
            (rng.normal(size=M) * (rng.random(M) > 0.5)).astype(np.float32))
        rho1, rho2 = 0.03, 0.02
        x, _ = enet_ops.lbfgs_solve_reference(A, y, rho1, rho2)
        g_smooth = 2 * A.t() @ (A @ x - y) + 2 * rho1 * x
        scale = float(g_smooth.abs().max().clamp(min=1.0))
        on = x.abs() > 1e-3
        if on.any():
            kkt_on = (g_smooth[on] + rho2 * torch.sign(x[on])).abs().max()
            assert float(kkt_on) < 0.05 * scale + 0.02, float(kkt_on)
        if (~on).any():
            kkt_off = g_smooth[~on].abs().max()
            assert float(kkt_off) <= rho2 * 1.5 + 0.02, float(kkt_off)
