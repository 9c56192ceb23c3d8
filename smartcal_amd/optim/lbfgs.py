"""L-BFGS with strong-Wolfe (cubic interpolation) line search.

Capability-parity implementation of the reference's ``LBFGSNew`` optimizer
(reference ``elasticnet/lbfgsnew.py``, 759 LoC): limited-memory BFGS with

* two-loop recursion for the search direction (``lbfgsnew.py:637-651``),
* curvature-pair acceptance guard ``y·s > 1e-10 ||s||^2``
  (``lbfgsnew.py:610-622``),
* strong-Wolfe line search with cubic interpolation for full-batch mode
  (``lbfgsnew.py:192-316,412``),
* a bounded backtracking line search for stochastic (``batch_mode``) use
  (``lbfgsnew.py:115-190,592-607``), where a running mean/variance of the
  gradient norm bounds the maximum step ``alphabar``.

This is a fresh implementation (not a copy): standard Nocedal & Wright
algorithms 3.5/3.6, written against torch tensors so it runs on CPU and on the
MI355X GPU alike. The optimizer state intentionally keeps the same layout the
reference exposes — ``state['old_dirs']`` holds the gradient-difference
vectors *y* and ``state['old_stps']`` the step vectors *s* — because
``autograd_tools.inv_hessian_mult`` replays the two-loop recursion directly
from that state (reference ``elasticnet/autograd_tools.py:35-66``).

The hot-path dual of this optimizer is the single-kernel HIP solver in
``smartcal_amd.ops`` (one workgroup runs the whole solve on-device); this
class is the general, composable form.
"""

from __future__ import annotations

from typing import List, Optional

import torch
from torch.optim import Optimizer


def _cubic_interpolate(x1, f1, g1, x2, f2, g2, bounds=None):
    """Minimizer of the cubic through (x1,f1,g1), (x2,f2,g2).

    Falls back to bisection when the cubic has no interior minimum.
    """
    if bounds is not None:
        xmin_bound, xmax_bound = bounds
    else:
        xmin_bound, xmax_bound = (x1, x2) if x1 <= x2 else (x2, x1)
    d1 = g1 + g2 - 3 * (f1 - f2) / (x1 - x2)
    d2_square = d1 * d1 - g1 * g2
    if d2_square >= 0:
        d2 = d2_square ** 0.5
        if x1 <= x2:
            min_pos = x2 - (x2 - x1) * ((g2 + d2 - d1) / (g2 - g1 + 2 * d2))
        else:
            min_pos = x1 - (x1 - x2) * ((g1 + d2 - d1) / (g1 - g2 + 2 * d2))
        return min(max(min_pos, xmin_bound), xmax_bound)
    return (xmin_bound + xmax_bound) / 2.0


class LBFGSNew(Optimizer):
    """L-BFGS optimizer (closure-based), drop-in for the reference LBFGSNew.

    Args:
        params: iterable of parameters (single group).
        lr: step length used when no line search is active.
        max_iter: maximal L-BFGS iterations per ``step()`` call.
        history_size: number of curvature pairs kept (reference default 7).
        line_search_fn: True => strong-Wolfe line search (full-batch mode),
            matching the reference's ``line_search_fn=True``.
        batch_mode: stochastic mode — backtracking line search with a
            running-statistics bound on the step size.
        tolerance_grad / tolerance_change: convergence thresholds.
    """

    def __init__(self, params, lr=1.0, max_iter=10, history_size=7,
                 tolerance_grad=1e-7, tolerance_change=1e-9,
                 line_search_fn=True, batch_mode=False,
                 cost_use_gradient=False):
        defaults = dict(lr=lr, max_iter=max_iter, history_size=history_size,
                        tolerance_grad=tolerance_grad,
                        tolerance_change=tolerance_change,
                        line_search_fn=line_search_fn, batch_mode=batch_mode,
                        cost_use_gradient=cost_use_gradient)
        super().__init__(params, defaults)
        if len(self.param_groups) != 1:
            raise ValueError("LBFGSNew supports a single parameter group")
        self._params = self.param_groups[0]["params"]

    # -- flat-vector helpers (reference lbfgsnew.py:84-113) ---------------
    def _gather_flat_grad(self) -> torch.Tensor:
        views = []
        for p in self._params:
            if p.grad is None:
                views.append(p.new_zeros(p.numel()))
            else:
                views.append(p.grad.reshape(-1))
        return torch.cat(views, 0)

    def _add_grad(self, step_size: float, update: torch.Tensor) -> None:
        offset = 0
        for p in self._params:
            n = p.numel()
            with torch.no_grad():
                p.add_(update[offset:offset + n].view_as(p), alpha=step_size)
            offset += n

    def _clone_param(self) -> List[torch.Tensor]:
        return [p.detach().clone(memory_format=torch.contiguous_format)
                for p in self._params]

    def _set_param(self, params_data: List[torch.Tensor]) -> None:
        with torch.no_grad():
            for p, pdata in zip(self._params, params_data):
                p.copy_(pdata)

    def _directional_evaluate(self, closure, x0, t, d):
        """phi(t), phi'(t) for the line search along d from x0."""
        self._set_param(x0)
        self._add_grad(t, d)
        loss = float(closure().detach())
        flat_grad = self._gather_flat_grad()
        return loss, flat_grad

    # -- strong-Wolfe line search (Nocedal & Wright alg. 3.5/3.6) --------
    def _strong_wolfe(self, closure, x0, t, d, f0, g0, gtd0,
                      c1=1e-4, c2=0.9, max_ls=25):
        d_norm = d.abs().max()
        g0 = g0.clone()
        f_prev, g_prev, t_prev = f0, g0, 0.0
        gtd_prev = gtd0
        done = False
        ls_iter = 0
        # bracketing phase
        while ls_iter < max_ls:
            f_new, g_new = self._directional_evaluate(closure, x0, t, d)
            gtd_new = float(g_new.dot(d))
            if f_new > (f0 + c1 * t * gtd0) or (ls_iter > 0 and f_new >= f_prev):
                bracket = [t_prev, t]
                bracket_f = [f_prev, f_new]
                bracket_g = [g_prev, g_new.clone()]
                bracket_gtd = [gtd_prev, gtd_new]
                break
            if abs(gtd_new) <= -c2 * gtd0:
                bracket = [t, t]
                bracket_f = [f_new, f_new]
                bracket_g = [g_new, g_new]
                done = True
                break
            if gtd_new >= 0:
                bracket = [t_prev, t]
                bracket_f = [f_prev, f_new]
                bracket_g = [g_prev, g_new.clone()]
                bracket_gtd = [gtd_prev, gtd_new]
                break
            # extrapolate
            min_step = t + 0.01 * (t - t_prev)
            max_step = t * 10
            t_next = _cubic_interpolate(t_prev, f_prev, gtd_prev, t, f_new,
                                        gtd_new, bounds=(min_step, max_step))
            t_prev, f_prev, g_prev, gtd_prev = t, f_new, g_new.clone(), gtd_new
            t = t_next
            ls_iter += 1
        else:
            bracket = [0.0, t]
            bracket_f = [f0, f_new]
            bracket_g = [g0, g_new]
            bracket_gtd = [gtd0, gtd_new]

        # zoom phase
        insuf_progress = False
        low_pos, high_pos = (0, 1) if bracket_f[0] <= bracket_f[-1] else (1, 0)
        while not done and ls_iter < max_ls:
            if abs(bracket[1] - bracket[0]) * d_norm < 1e-10:
                break
            t = _cubic_interpolate(bracket[0], bracket_f[0], bracket_gtd[0],
                                   bracket[1], bracket_f[1], bracket_gtd[1])
            # guard against stagnation at the bracket edge
            eps = 0.1 * (max(bracket) - min(bracket))
            if min(max(bracket) - t, t - min(bracket)) < eps:
                if insuf_progress or t >= max(bracket) or t <= min(bracket):
                    if abs(t - max(bracket)) < abs(t - min(bracket)):
                        t = max(bracket) - eps
                    else:
                        t = min(bracket) + eps
                    insuf_progress = False
                else:
                    insuf_progress = True
            else:
                insuf_progress = False

            f_new, g_new = self._directional_evaluate(closure, x0, t, d)
            gtd_new = float(g_new.dot(d))
            if f_new > (f0 + c1 * t * gtd0) or f_new >= bracket_f[low_pos]:
                bracket[high_pos] = t
                bracket_f[high_pos] = f_new
                bracket_g[high_pos] = g_new.clone()
                bracket_gtd[high_pos] = gtd_new
                low_pos, high_pos = ((0, 1) if bracket_f[0] <= bracket_f[1]
                                     else (1, 0))
            else:
                if abs(gtd_new) <= -c2 * gtd0:
                    done = True
                elif gtd_new * (bracket[high_pos] - bracket[low_pos]) >= 0:
                    bracket[high_pos] = bracket[low_pos]
                    bracket_f[high_pos] = bracket_f[low_pos]
                    bracket_g[high_pos] = bracket_g[low_pos]
                    bracket_gtd[high_pos] = bracket_gtd[low_pos]
                bracket[low_pos] = t
                bracket_f[low_pos] = f_new
                bracket_g[low_pos] = g_new.clone()
                bracket_gtd[low_pos] = gtd_new
            ls_iter += 1

        t = bracket[low_pos] if not done else t
        f_new = bracket_f[low_pos] if not done else f_new
        g_new = bracket_g[low_pos] if not done else g_new
        return f_new, g_new, t

    # -- backtracking line search for stochastic batch mode --------------
    def _backtrack(self, closure, x0, t, d, f0, gtd0, alphabar,
                   c1=1e-4, max_ls=20):
        t = min(t, alphabar)
        for _ in range(max_ls):
            self._set_param(x0)
            self._add_grad(t, d)
            with torch.no_grad():
                f_new = float(closure())
            if f_new <= f0 + c1 * t * gtd0 and f_new == f_new:  # not NaN
                return t
            t *= 0.5
        return t

    @torch.enable_grad()
    def step(self, closure):
        group = self.param_groups[0]
        lr = group["lr"]
        max_iter = group["max_iter"]
        history_size = group["history_size"]
        tol_grad = group["tolerance_grad"]
        tol_change = group["tolerance_change"]
        line_search = group["line_search_fn"]
        batch_mode = group["batch_mode"]

        state = self.state[self._params[0]]
        state.setdefault("func_evals", 0)
        state.setdefault("n_iter", 0)

        orig_loss = closure()
        loss = float(orig_loss.detach())
        state["func_evals"] += 1
        flat_grad = self._gather_flat_grad()
        if flat_grad.abs().max() <= tol_grad:
            return orig_loss

        old_dirs: List[torch.Tensor] = state.get("old_dirs", [])
        old_stps: List[torch.Tensor] = state.get("old_stps", [])
        ro: List[torch.Tensor] = state.get("ro", [])
        prev_flat_grad: Optional[torch.Tensor] = state.get("prev_flat_grad")
        prev_loss = state.get("prev_loss")
        d = state.get("d")
        t = state.get("t", lr)
        H_diag = state.get("H_diag", 1.0)

        # running statistics for batch-mode step bound (reference
        # lbfgsnew.py:592-607 keeps running mean/var of the gradient to set
        # the max step alphabar)
        g_running_mean = state.get("g_running_mean", 0.0)
        g_running_var = state.get("g_running_var", 0.0)
        g_count = state.get("g_count", 0)

        n_local_iter = 0
        while n_local_iter < max_iter:
            n_local_iter += 1
            state["n_iter"] += 1

            # ---- curvature update (n_iter >= 2) ----
            if state["n_iter"] > 1 and prev_flat_grad is not None \
                    and d is not None:
                y = flat_grad.sub(prev_flat_grad)
                s = d.mul(t)
                ys = float(y.dot(s))
                # curvature guard (reference lbfgsnew.py:610-622)
                if ys > 1e-10 * float(s.dot(s)):
                    if len(old_dirs) == history_size:
                        old_dirs.pop(0)
                        old_stps.pop(0)
                        ro.pop(0)
                    old_dirs.append(y)
                    old_stps.append(s)
                    ro.append(torch.tensor(1.0 / ys, device=y.device))
                    H_diag = ys / float(y.dot(y))

            # ---- direction ----
            if not old_dirs:
                d = flat_grad.neg()
                H_diag = 1.0
            else:
                # two-loop recursion (reference lbfgsnew.py:637-651)
                num_old = len(old_dirs)
                from ..ops import use_hip
                if use_hip(flat_grad) and num_old <= 16:
                    # whole recursion in ONE kernel launch
                    # (ops/csrc/two_loop.hip) — the host composition
                    # below syncs once per history entry per dot
                    from ..ops import ext
                    Y = torch.stack(list(old_dirs))
                    S = torch.stack(list(old_stps))
                    rov = torch.stack([torch.as_tensor(
                        r, device=flat_grad.device).reshape(())
                        for r in ro]).to(torch.float32)
                    d = ext().two_loop_apply(
                        Y.contiguous(), S.contiguous(),
                        flat_grad.neg().reshape(1, -1).contiguous(),
                        rov.contiguous(), float(H_diag)).reshape(-1)
                else:
                    al = [None] * num_old
                    q = flat_grad.neg()
                    for i in range(num_old - 1, -1, -1):
                        al[i] = float(old_stps[i].dot(q)) * float(ro[i])
                        q.add_(old_dirs[i], alpha=-al[i])
                    d = q.mul(H_diag)
                    for i in range(num_old):
                        be_i = float(old_dirs[i].dot(d)) * float(ro[i])
                        d.add_(old_stps[i], alpha=al[i] - be_i)

            if prev_flat_grad is None:
                prev_flat_grad = flat_grad.clone(memory_format=torch.contiguous_format)
            else:
                prev_flat_grad.copy_(flat_grad)
            prev_loss = loss

            # ---- step length ----
            gtd = float(flat_grad.dot(d))
            if gtd > -tol_change:
                break  # not a descent direction / converged

            if state["n_iter"] == 1:
                t = min(1.0, 1.0 / float(flat_grad.abs().sum())) * lr
            else:
                t = lr

            if batch_mode:
                # bound the step by gradient-noise statistics
                gnorm = float(flat_grad.norm())
                g_count += 1
                delta = gnorm - g_running_mean
                g_running_mean += delta / g_count
                g_running_var += delta * (gnorm - g_running_mean)
                var = g_running_var / max(g_count - 1, 1)
                alphabar = 1.0 / (1.0 + var / max(g_running_mean ** 2, 1e-12))
                x0 = self._clone_param()
                t = self._backtrack(closure, x0, t, d, loss, gtd, alphabar)
                self._set_param(x0)
                self._add_grad(t, d)
                loss = float(closure().detach())
                flat_grad = self._gather_flat_grad()
                state["func_evals"] += 2
            elif line_search:
                x0 = self._clone_param()
                loss, flat_grad, t = self._strong_wolfe(
                    closure, x0, t, d, loss, flat_grad, gtd)
                self._set_param(x0)
                self._add_grad(t, d)
                state["func_evals"] += 1
            else:
                self._add_grad(t, d)
                if n_local_iter != max_iter:
                    with torch.enable_grad():
                        loss = float(closure().detach())
                    flat_grad = self._gather_flat_grad()
                    state["func_evals"] += 1

            # ---- convergence ----
            if flat_grad.abs().max() <= tol_grad:
                break
            if d.mul(t).abs().max() <= tol_change:
                break
            if abs(loss - prev_loss) < tol_change:
                break

        state["old_dirs"] = old_dirs
        state["old_stps"] = old_stps
        state["ro"] = ro
        state["prev_flat_grad"] = prev_flat_grad
        state["prev_loss"] = prev_loss
        state["d"] = d
        state["t"] = t
        state["H_diag"] = H_diag
        state["g_running_mean"] = g_running_mean
        state["g_running_var"] = g_running_var
        state["g_count"] = g_count
        return orig_loss
