"""Direction-dependent calibration solver with consensus ADMM.

The in-repo replacement for the external ``sagecal_gpu`` /
``mpirun sagecal-mpi_gpu`` binaries the reference shells out to
(`calibration/docal.sh:12`, `demixing_rl/demixingenv.py:129`,
SURVEY.md §2.2 N13/N14). MI355X-first design:

* **J-step** — joint full-Jones alternating least squares: fixing all
  other stations, each station's stacked per-direction Jones
  G_p = [J¹_p … J^K_p] (2×2K) has a closed-form solution from 2K×2K
  normal equations. All stations, solution intervals and frequencies are
  solved as ONE batched ``torch.linalg.solve`` per sweep (GEMM-shaped
  work), with 0.5 averaging damping à la StefCal. The ADMM proximal term
  ρ_k‖J^k − B_f Z^k + Y^k/ρ_k‖² enters the normal equations exactly.
* **Z-step** — consensus polynomial over frequency (`-P` order, ordinary
  or Bernstein basis as `consensus_poly`, `calibration_tools.py:551-585`):
  closed-form per direction via the (Ne×Ne) pinv, with federated-averaging
  spatial regularization α. When ``torch.distributed`` is initialized the
  per-frequency partial sums are combined with ONE flat all_reduce (RCCL
  over xGMI); single-process mode sums locally over its frequencies.
* **Y-step** — dual ascent.

Interfaces are in-memory tensors (`radio.sim.VisData`); solutions are
returned in the reference's J layout (K, 2N·Ts, 2) so the influence
pipeline (`radio.influence`) and text writers consume them directly.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch
import torch.distributed as dist

from .consensus import bpoly
from .hessian import baseline_pq
from .coherency import predict_coherencies_uvw
from .sim import VisData, apply_jones

__all__ = ["CalSolution", "calibrate"]


@dataclass
class CalSolution:
    J: torch.Tensor          # (Nf_local, K, Ts, N, 2, 2) complex64
    Z: torch.Tensor          # (K, Ne, Ts, N, 2, 2) complex64 (global)
    residual: torch.Tensor   # (Nf_local, S, 4) complex64 = data − model
    freqs: np.ndarray        # local frequencies
    rho: np.ndarray          # per-direction spectral rho used

    def J_ref_layout(self, fi: int) -> torch.Tensor:
        """J for local freq fi in the reference layout (K, 2N·Ts, 2)
        (`calibration_tools.py:105-119`)."""
        K = self.J.shape[1]
        return self.J[fi].reshape(K, -1, 2)


def _poly_basis(freqs: np.ndarray, f_all: np.ndarray, f0: float, Ne: int,
                polytype: int) -> np.ndarray:
    """Rows of the frequency polynomial basis for ``freqs``, defined over
    the global frequency set ``f_all`` (Bernstein normalization needs the
    global min/max, as in `consensus_poly`)."""
    if polytype == 0:
        B = np.zeros((len(freqs), Ne))
        B[:, 0] = 1.0
        ff = (freqs - f0) / f0
        for cj in range(1, Ne):
            B[:, cj] = ff ** cj
    else:
        lo, hi = f_all.min(), f_all.max()
        ff = (freqs - lo) / (hi - lo) if hi > lo else np.zeros_like(freqs)
        B = bpoly(ff, Ne - 1).astype(np.float64)
    return B


def _solve_sweeps(data22, C22, J, rho_t, prox_target, p_idx, q_idx, N,
                  Tdelta, n_sweeps, lambda_reg=1e-6):
    """In-place alternating-LS sweeps for J (F?,Ts,K,N,2,2).

    data22: (F,T,B,2,2); C22: (F,K,T,B,2,2); rho_t: (K,) float;
    prox_target: (F,Ts,K,N,2,2) or None — the ADMM prox point
    (B_f Z − Y/ρ).
    """
    F, T, Bn = data22.shape[0], data22.shape[1], data22.shape[2]
    K = C22.shape[1]
    Ts = J.shape[1]
    dev = data22.device
    t_int = (torch.arange(T, device=dev) // Tdelta).clamp_(max=Ts - 1)
    eyeK = torch.eye(2 * K, dtype=data22.dtype, device=dev)
    # gather plan for the per-(interval, station) reduction: station n of
    # interval ti receives exactly Tdelta·(N−1) contributions (as the p
    # side or the q side of a baseline). Precomputing the gather indices
    # into cat([p-side, q-side]) turns four slow complex index_add_
    # scatters per sweep into two coalesced gather+sum passes.
    flat_p = (t_int.view(T, 1) * N + p_idx.view(1, Bn)).reshape(-1)
    flat_q = (t_int.view(T, 1) * N + q_idx.view(1, Bn)).reshape(-1)
    both = torch.cat([flat_p, flat_q])              # (2·T·B,)
    order = torch.argsort(both, stable=True)
    Cnt = T * Bn * 2 // (Ts * N)                    # = Tdelta·(N−1)
    gidx = order.reshape(Ts * N, Cnt)               # (Ts·N, Cnt)
    use_kernel = dev.type == "cuda" and K <= 8
    if use_kernel:
        from ..ops import ext
        p32 = p_idx.to(torch.int32).contiguous()
        q32 = q_idx.to(torch.int32).contiguous()
        t32 = t_int.to(torch.int32).contiguous()
        C22c = C22.contiguous()
    for _ in range(n_sweeps):
        if use_kernel:
            # fused HIP kernel: all per-sample Jones products +
            # normal-equation contributions in ONE launch (the env step
            # is dispatch-bound; see ops/csrc/als_sweep.hip). Output is
            # entry-major (F, X, 2TB) for coalesced writes.
            rhs_cat, nm_cat = ext().als_sweep(
                C22c, data22.contiguous(), J.contiguous(), p32, q32, t32)
        else:
            # A^k for p-side rows: A = C_pq (J^k_q)^H ; for q-side rows:
            # A' = C_pq^H (J^k_p)^H  (from V_pq^H = Σ J^k_q C^H J_p^H)
            Jt = J[:, t_int]                               # (F,T,K,N,2,2)
            Jq = Jt[:, :, :, q_idx]                        # (F,T,K,B,2,2)
            Jp = Jt[:, :, :, p_idx]
            Cp = C22.permute(0, 2, 3, 1, 4, 5)             # (F,T,B,K,2,2)
            # elementwise small-complex products (radio.small_complex):
            # batched 2×2 / k=2 shapes are pathological for rocBLAS
            from .small_complex import mm2H, Hmm2, abH_k2
            A_p = mm2H(Cp, Jq.permute(0, 1, 3, 2, 4, 5))   # (F,T,B,K,2,2)
            A_q = Hmm2(Cp, Jp.permute(0, 1, 3, 2, 4, 5).conj().mT)
            V = data22                                      # (F,T,B,2,2)
            # per (station, interval): normal matrix (2K,2K), rhs (2,2K)
            # W = A stacked over k → (2K,2); contribs V·W^H and W·W^H
            Wp = A_p.reshape(F, T, Bn, 2 * K, 2)
            Wq = A_q.reshape(F, T, Bn, 2 * K, 2)
            rhs_p = abH_k2(V, Wp)                           # (F,T,B,2,2K)
            rhs_q = abH_k2(V.mH, Wq)
            nm_p = abH_k2(Wp, Wp)                           # (F,T,B,2K,2K)
            nm_q = abH_k2(Wq, Wq)
            rhs_cat = torch.cat(
                [rhs_p.reshape(F, T * Bn, 2 * 2 * K),
                 rhs_q.reshape(F, T * Bn, 2 * 2 * K)], dim=1) \
                .permute(0, 2, 1)
            nm_cat = torch.cat(
                [nm_p.reshape(F, T * Bn, 2 * K * 2 * K),
                 nm_q.reshape(F, T * Bn, 2 * K * 2 * K)], dim=1) \
                .permute(0, 2, 1)
        # reduce into (F,Ts,N,…) by interval and station via the
        # precomputed gather plan (see above); cat layout is (F, X, 2TB)
        if use_kernel:
            # fused gather+segment-sum (ops/csrc/gather_sum.hip): the
            # torch composition below materializes the full gathered
            # copy (~260 MB/sweep at LOFAR scale) before reducing
            from ..ops import ext as _ext
            gc = gidx.contiguous()
            rhs = _ext().gather_sum(rhs_cat.contiguous(), gc) \
                .permute(0, 2, 1).reshape(F, Ts, N, 2, 2 * K)
            nm = _ext().gather_sum(nm_cat.contiguous(), gc) \
                .permute(0, 2, 1).reshape(F, Ts, N, 2 * K, 2 * K)
        else:
            rhs = rhs_cat[:, :, gidx.reshape(-1)] \
                .reshape(F, 2 * 2 * K, Ts * N, Cnt).sum(dim=3) \
                .permute(0, 2, 1).reshape(F, Ts, N, 2, 2 * K)
            nm = nm_cat[:, :, gidx.reshape(-1)] \
                .reshape(F, 2 * K * 2 * K, Ts * N, Cnt).sum(dim=3) \
                .permute(0, 2, 1).reshape(F, Ts, N, 2 * K, 2 * K)
        # ADMM prox: + diag(ρ_k I2) and + ρ_k F^k on the rhs
        if prox_target is not None:
            rho_blocks = torch.kron(
                torch.diag(rho_t.to(nm.real.dtype)),
                torch.eye(2, device=dev)).to(nm.dtype)
            nm = nm + rho_blocks
            # prox rhs: (F,Ts,K,N,2,2) → (F,Ts,N,2,2K) with ρ_k weights
            pt = prox_target * rho_t.view(1, 1, K, 1, 1, 1)
            rhs = rhs + pt.permute(0, 1, 3, 4, 2, 5).reshape(
                F, Ts, N, 2, 2 * K)
        nm = nm + lambda_reg * eyeK
        # solve G (2,2K): G nm = rhs → nm^T G^T = rhs^T
        G = torch.linalg.solve(nm.mT, rhs.mT).mT            # (F,Ts,N,2,2K)
        Jnew = G.reshape(F, Ts, N, 2, K, 2).permute(0, 1, 4, 2, 3, 5)
        J.copy_(0.5 * J + 0.5 * Jnew)
    return J


def calibrate(vis: VisData, sky, clusters, rho_spectral,
              admm_iter: int = 10, poly_order: int = 2, polytype: int = 1,
              alpha: float = 0.0, n_sweeps: int = 3, init_sweeps: int = 6,
              smear_bw: float | None = 180e3,
              C_cache: torch.Tensor | None = None,
              freq_group: "dist.ProcessGroup | None" = None) -> CalSolution:
    """Consensus-ADMM direction-dependent calibration of ``vis``.

    rho_spectral: (K,) ADMM regularization per direction (the env action);
    admm_iter: ADMM iterations (the reference's ``-A``);
    poly_order: consensus polynomial terms (``-P``);
    alpha: federated-averaging / spatial regularization (``-X`` style).

    When torch.distributed is initialized, each rank is expected to hold
    a disjoint frequency shard in ``vis``; the Z-step all_reduces the
    basis-projected partial sums (one flat RCCL all_reduce per ADMM
    iteration over xGMI).
    """
    dev = vis.data.device
    N, B, Ts, Tdelta = vis.N, vis.B, vis.Ts, vis.Tdelta
    T = vis.n_time
    K = len(clusters)
    Nf = len(vis.freqs)
    Ne = poly_order
    p_idx, q_idx = baseline_pq(N, dev)
    rho_t = torch.as_tensor(np.asarray(rho_spectral, np.float32), device=dev)

    # model coherencies per local frequency (cacheable across env steps)
    if C_cache is None:
        C = torch.stack([
            predict_coherencies_uvw(sky, clusters, vis.uvw, float(f),
                                    vis.ra0, vis.dec0, smear_bw=smear_bw)
            for f in vis.freqs])                            # (Nf,K,S,4)
    else:
        C = C_cache
    # row-major 2×2 physical convention throughout the solver (matches
    # apply_jones; see sim.apply_jones note)
    C22 = C.reshape(Nf, K, T, B, 2, 2)
    data22 = vis.data.reshape(Nf, T, B, 2, 2)

    # distributed frequency-shard info
    distributed = dist.is_available() and dist.is_initialized() \
        and dist.get_world_size(freq_group) > 1
    if distributed:
        all_freqs = [None] * dist.get_world_size(freq_group)
        dist.all_gather_object(all_freqs, list(vis.freqs), group=freq_group)
        f_all = np.sort(np.concatenate([np.asarray(f) for f in all_freqs]))
    else:
        f_all = np.asarray(vis.freqs)
    f0 = float(np.mean(f_all))
    Bf = _poly_basis(np.asarray(vis.freqs), f_all, f0, Ne, polytype)
    Bf_t = torch.as_tensor(Bf, dtype=torch.float32, device=dev)
    # global Gram matrix of the basis
    Ball = _poly_basis(f_all, f_all, f0, Ne, polytype)
    Gram = Ball.T @ Ball                                     # (Ne,Ne)

    # init J with identity and a few unconstrained sweeps
    J = torch.zeros((Nf, Ts, K, N, 2, 2), dtype=torch.complex64, device=dev)
    J[..., 0, 0] = 1.0
    J[..., 1, 1] = 1.0
    _solve_sweeps(data22, C22, J, rho_t, None, p_idx, q_idx, N, Tdelta,
                  init_sweeps)

    Y = torch.zeros_like(J)                                  # scaled dual
    Z = torch.zeros((K, Ne, Ts, N, 2, 2), dtype=torch.complex64, device=dev)
    rho_k = rho_t.view(1, 1, K, 1, 1, 1)

    for it in range(admm_iter):
        # ---- Z-step: per direction closed form over all freqs ----------
        # partial sum: Σ_f b_f ⊗ ρ (J_f + Y_f/ρ)  → (Ne,K,Ts,N,2,2)
        JY = J + Y / rho_k.clamp(min=1e-12)
        part = torch.einsum('fe,ftknab->ektnab',
                            Bf_t.to(torch.complex64), JY)
        if distributed:
            dist.all_reduce(part, group=freq_group)
        # weight by rho (same for all freqs) and solve (ρ Gram + αI) Z = part
        M = torch.as_tensor(Gram, dtype=torch.complex64, device=dev)
        rhoZ = rho_t.view(K, 1, 1, 1, 1, 1).to(torch.complex64)
        A = (M.unsqueeze(0) * rhoZ.view(K, 1, 1)
             + alpha * torch.eye(Ne, dtype=torch.complex64, device=dev))
        rhs = part.permute(1, 0, 2, 3, 4, 5).reshape(K, Ne, -1) \
            * rhoZ.view(K, 1, 1)
        # inv+GEMM: complex batched trsm is broken for wide RHS on this
        # ROCm build (see radio.hessian.dsolutions_r note); Ne is tiny
        Z = (torch.linalg.inv(A) @ rhs).reshape(K, Ne, Ts, N, 2, 2)
        # ---- J-step with prox toward B_f Z − Y/ρ -----------------------
        BZ = torch.einsum('fe,ektnab->ftknab', Bf_t.to(torch.complex64),
                          Z.permute(1, 0, 2, 3, 4, 5))      # (F,Ts,K,N,2,2)
        prox = BZ - Y / rho_k.clamp(min=1e-12)
        _solve_sweeps(data22, C22, J, rho_t, prox, p_idx, q_idx, N, Tdelta,
                      n_sweeps)
        # ---- dual ascent ----------------------------------------------
        Y = Y + rho_k.to(torch.complex64) * (J - BZ)

    # residual = data − model(J)
    res = torch.empty_like(vis.data)
    for fi in range(Nf):
        model = apply_jones(C[fi], J[fi].permute(1, 0, 2, 3, 4), N, Tdelta)
        res[fi] = vis.data[fi] - model
    return CalSolution(J=J.permute(0, 2, 1, 3, 4, 5).contiguous(), Z=Z,
                       residual=res, freqs=np.asarray(vis.freqs),
                       rho=np.asarray(rho_spectral, np.float32))
