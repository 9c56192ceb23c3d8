"""Autograd utilities: Jacobian, HVP, inverse-HVP, influence matrices.

Capability-parity with the reference's ``autograd_tools.py`` (reference
``elasticnet/autograd_tools.py`` == ``demixing_rl/autograd_tools.py``):

* ``gradient`` / ``jacobian`` — VJP-based Jacobians
  (reference ``autograd_tools.py:13-29``),
* ``inv_hessian_mult`` — inverse-Hessian × vector via two-loop recursion
  replayed from a converged L-BFGS optimizer's curvature pairs
  (``autograd_tools.py:35-66``),
* ``influence_matrix`` — d(model output)/d(input) through the inverse
  Hessian of the training loss (``autograd_tools.py:94-149``),
* ``hessian_vec_prod`` — Pearlmutter R-op HVP (``autograd_tools.py:159-176``),
* ``inverse_hessian_vec_prod`` — normalized Neumann-series iHVP
  (``autograd_tools.py:183-194``).

Additions for the MI355X-first hot path: ``inv_hessian_mult_mat`` applies the
two-loop recursion to a whole matrix of right-hand sides at once (rank-1
matrix updates instead of a python loop over columns) — this is what the
elastic-net environment's influence computation uses so that a full N-column
solve is a handful of GPU ops instead of N × history python-loop round trips.
"""

from __future__ import annotations

from typing import Sequence, Tuple

import torch


def gradient(y: torch.Tensor, x: torch.Tensor, grad_outputs=None,
             create_graph: bool = True) -> torch.Tensor:
    """VJP: (dy/dx)^T @ grad_outputs (defaults to all-ones)."""
    if grad_outputs is None:
        grad_outputs = torch.ones_like(y)
    return torch.autograd.grad(y, [x], grad_outputs=grad_outputs,
                               create_graph=create_graph)[0]


def jacobian(y: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """Dense Jacobian dy/dx via one-hot VJPs (rows of dy/dx)."""
    jac = y.new_zeros(y.shape[0], x.shape[0])
    for i in range(y.shape[0]):
        e = torch.zeros_like(y)
        e[i] = 1.0
        jac[i] = gradient(y, x, grad_outputs=e)
    return jac


def _curvature_pairs(opt) -> Tuple[Sequence[torch.Tensor], Sequence[torch.Tensor]]:
    """Extract (y-list, s-list) curvature pairs from an L-BFGS optimizer."""
    sd = opt.state_dict()
    st = sd.get("state", {})
    if 0 not in st and len(st) > 0:
        st = {0: next(iter(st.values()))}
    entry = st.get(0, {})
    dirs = entry.get("old_dirs")
    stps = entry.get("old_stps")
    return dirs, stps


def inv_hessian_mult(opt, q: torch.Tensor) -> torch.Tensor:
    """approx inv(Hessian) @ q using the optimizer's L-BFGS curvature pairs.

    ``q`` is modified in place (same contract as the reference,
    ``autograd_tools.py:35-66``). Returns r = H^{-1} q.
    """
    dirs, stps = _curvature_pairs(opt)
    if not dirs or not stps:
        return q
    if q.is_cuda:
        # one two_loop_kernel launch instead of a host sync per history
        # entry per dot (the CPU composition below)
        Y = torch.stack(list(dirs))
        S = torch.stack(list(stps))
        r = inv_hessian_mult_mat(Y, S, q.reshape(-1, 1))[:, 0]
        q.copy_(r)     # reference contract: q is scratch, r returned
        return r
    n = len(dirs)
    ys = dirs[-1].dot(stps[-1])
    yy = dirs[-1].dot(dirs[-1])
    ro = [1.0 / float(dirs[i].dot(stps[i])) for i in range(n)]
    al = [0.0] * n
    for i in range(n - 1, -1, -1):
        al[i] = float(stps[i].dot(q)) * ro[i]
        q.add_(dirs[i], alpha=-al[i])
    r = q * (ys / yy)
    for i in range(n):
        be_i = float(dirs[i].dot(r)) * ro[i]
        r.add_(stps[i], alpha=al[i] - be_i)
    return r


def inv_hessian_mult_mat(Y: torch.Tensor, S: torch.Tensor,
                         Q: torch.Tensor) -> torch.Tensor:
    """Two-loop recursion applied to a matrix of RHS columns at once.

    Args:
        Y: (h, n) stacked gradient-difference vectors (newest last).
        S: (h, n) stacked step vectors (newest last).
        Q: (n, k) right-hand sides.
    Returns:
        (n, k) matrix  H^{-1} Q  — column-for-column identical to calling
        ``inv_hessian_mult`` on each column.
    """
    if Y.numel() == 0:
        return Q
    h = Y.shape[0]
    ro = 1.0 / (Y * S).sum(dim=1)          # (h,)
    ys = float(Y[-1].dot(S[-1]))
    yy = float(Y[-1].dot(Y[-1]))
    from .ops import use_hip
    if use_hip(Q) and h <= 16:
        # whole recursion for all RHS columns in ONE launch
        # (ops/csrc/two_loop.hip — N6/N7); torch composition below is
        # the CPU oracle
        from .ops import ext
        R = ext().two_loop_apply(Y.contiguous(), S.contiguous(),
                                 Q.t().contiguous(), ro.contiguous(),
                                 ys / yy)
        return R.t().contiguous()
    Q = Q.clone()
    al = Q.new_zeros(h, Q.shape[1])
    for i in range(h - 1, -1, -1):
        al[i] = (S[i] @ Q) * ro[i]          # (k,)
        Q -= torch.outer(Y[i], al[i])
    R = Q * (ys / yy)
    for i in range(h):
        be = (Y[i] @ R) * ro[i]
        R += torch.outer(S[i], al[i] - be)
    return R


def gather_flat_grad(model: torch.nn.Module) -> torch.Tensor:
    """Flatten and clear parameter gradients (reference
    ``autograd_tools.py:72-79`` clears grads to avoid graph leaks)."""
    views = []
    for p in model.parameters():
        if p.grad is not None:
            views.append(p.grad.detach().reshape(-1).clone())
            p.grad = None
        else:
            views.append(p.new_zeros(p.numel()))
    return torch.cat(views, 0)


def gather_flat_parameters(parameters) -> torch.Tensor:
    return torch.cat([p.reshape(-1) for p in parameters], 0)


def hessian_vec_prod(model, criterion, inputs, outputs,
                     v: torch.Tensor) -> torch.Tensor:
    """Pearlmutter-trick HVP of the loss wrt model parameters."""
    model.zero_grad()
    pred = model(inputs)
    L = criterion(pred, outputs)
    grads = torch.autograd.grad(L, list(model.parameters()),
                                create_graph=True, retain_graph=True)
    flat_g = gather_flat_parameters(grads)
    w = torch.zeros_like(flat_g, requires_grad=True)
    g2 = torch.autograd.grad(flat_g, list(model.parameters()),
                             grad_outputs=w, create_graph=True,
                             allow_unused=True)
    g2 = [torch.zeros_like(p) if g is None else g
          for g, p in zip(g2, model.parameters())]
    r = torch.autograd.grad(gather_flat_parameters(g2), w, grad_outputs=v,
                            allow_unused=True)
    return r[0].detach()


def inverse_hessian_vec_prod(model, criterion, inputs, outputs,
                             v: torch.Tensor, maxiter: int = 10) -> torch.Tensor:
    """Neumann/Taylor-series iHVP with per-iteration normalization
    (Koh & Liang 2017 §3; reference ``autograd_tools.py:183-194``)."""
    x = v / torch.norm(v)
    for _ in range(maxiter):
        q = hessian_vec_prod(model, criterion, inputs, outputs, x)
        x = v + x - q
        x = x / torch.norm(x)
    return x


def influence_matrix(model, xinput, youtput, opt=None,
                     override_input: bool = False) -> torch.Tensor:
    """Influence of each input element on each output element through the
    inverse training Hessian: If[j, i] = J_theta(y_j) . H^{-1} d^2L/dx_i dtheta.

    Mirrors reference ``autograd_tools.py:94-149`` (M x N result, outer loop
    over inputs, inner over outputs).
    """
    device = xinput.device
    if override_input:
        x = xinput.detach().clone().requires_grad_(True)
    else:
        x = torch.ones(xinput.shape, requires_grad=True, device=device)
    N = x.reshape(-1).shape[0]
    M = youtput.reshape(-1).shape[0]
    labels = youtput if override_input else torch.ones_like(youtput)

    criterion = torch.nn.MSELoss()

    def l2loss(ytrue, xin):
        return criterion(model(xin).reshape(-1), ytrue.reshape(-1))

    If = xinput.new_zeros(M, N)
    for ci in range(N):
        g = gradient(l2loss(labels, x), x).reshape(-1)
        g[ci].backward()
        ddf_dxdtheta = gather_flat_grad(model)
        if opt is not None:
            iddf = inv_hessian_mult(opt, ddf_dxdtheta)
        else:
            iddf = inverse_hessian_vec_prod(model, criterion, x, labels,
                                            ddf_dxdtheta, maxiter=10)
        for cj in range(M):
            y = model(x).reshape(-1)
            y[cj].backward()
            jvec = gather_flat_grad(model)
            If[cj, ci] = torch.dot(iddf, jvec)
    return If
