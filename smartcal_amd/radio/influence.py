"""Influence-map computation (the RL state for calibration/demixing).

The in-memory, batched equivalent of `calibration/analysis_torch.py`
(`process_chunk` :16-77, `analysis_uvwdir_loop` :79-186) and the
per-direction variant `calibration/influence_tools.py:219-357`. Where the
reference forks a torch.multiprocessing Pool over timeslot chunks (P3/P4
of SURVEY.md §2.3), here each chunk is a single batched pass on device
(the batched `radio.hessian` ops already saturate the GPU) and chunks run
back-to-back on one stream — no process pool, no shared memory, no MS
round-trip.
"""

from __future__ import annotations

import numpy as np
import torch

from . import hessian as hs
from .consensus import hessian_addition_scalar
from .sim import to_R

__all__ = ["influence_values", "influence_per_direction", "hadd_for"]


def hadd_for(K: int, N: int, Ne: int, freqs: np.ndarray, f0: float,
             fidx: int, rho_spectral, rho_spatial, device,
             polytype: int = 1) -> torch.Tensor:
    """(K,4N,4N) consensus Hessian additions (`analysis_torch.py:141-156`).

    Every per-direction addition is a scalar multiple of the identity
    (see `consensus.hessian_addition_scalar`), so this is K scalar
    evaluations + one batched diag-embed — no 2N×2N pinvs per step."""
    scal = torch.tensor(
        [hessian_addition_scalar(
            Ne, N, freqs, f0, fidx, float(rho_spectral[k]),
            float(rho_spatial[k]) if rho_spatial is not None else 0.0,
            polytype=polytype) for k in range(K)],
        dtype=torch.float32)
    eye = torch.eye(4 * N, dtype=torch.complex64, device=device)
    return scal.to(device).view(K, 1, 1).to(torch.complex64) * eye


def influence_values(residual4: torch.Tensor, C: torch.Tensor,
                     J: torch.Tensor, N: int, Tdelta: int,
                     Hadd: torch.Tensor | None = None,
                     fullpol: bool = False) -> torch.Tensor:
    """Per-sample influence values (S, 4) complex for one frequency.

    residual4: (S, 4) residual visibilities; C: (K, S, 4) coherencies;
    J: (K, 2N·Ts, 2) solutions (reference layout); Tdelta: timeslots per
    solution interval. Mirrors `process_chunk` exactly: per interval,
    H = Hessianres + Hadd → dJ (8 dirs) → dR → column means of every
    4th row accumulated into XX/YY (XY/YX when fullpol), finally scaled
    by 8·B·T.
    """
    S = residual4.shape[0]
    B = N * (N - 1) // 2
    T = S // B
    Ts = T // Tdelta
    out = torch.zeros_like(residual4)
    for ncal in range(Ts):
        s0 = ncal * Tdelta * B
        s1 = s0 + Tdelta * B
        Rchunk = to_R(residual4[s0:s1])
        Cchunk = C[:, s0:s1]
        Jchunk = J[:, ncal * 2 * N:(ncal + 1) * 2 * N]
        H = hs.hessianres(Rchunk, Cchunk, Jchunk, N)
        if Hadd is not None:
            H = H + Hadd
        # analytic row-block means (hs.dres_colmeans): never builds the
        # (8,4B,B) dR nor the 8·K·4N·B dJ — a 4-RHS solve + contractions
        m = hs.dres_colmeans(Cchunk, Jchunk, N, H)           # (8,4,B)
        xx = m[:, 0].sum(dim=0)
        yy = m[:, 3].sum(dim=0)
        out[s0:s1, 0] = xx.repeat(Tdelta)
        out[s0:s1, 3] = yy.repeat(Tdelta)
        if fullpol:
            out[s0:s1, 1] = m[:, 1].sum(dim=0).repeat(Tdelta)
            out[s0:s1, 2] = m[:, 2].sum(dim=0).repeat(Tdelta)
    return out * (8 * B * Tdelta)


def influence_per_direction(residual4: torch.Tensor, C: torch.Tensor,
                            J: torch.Tensor, N: int, Tdelta: int,
                            Hadd: torch.Tensor | None = None):
    """Per-direction influence values (K, S, 4) + per-direction summary
    stats, following `influence_tools.analysis_uvw_perdir:219-357`:
    returns (values, ‖J‖_k, ‖C‖_k, |mean Inf|_k, LLR_k)."""
    K = C.shape[0]
    S = residual4.shape[0]
    B = N * (N - 1) // 2
    T = S // B
    Ts = T // Tdelta
    out = torch.zeros((K,) + residual4.shape, dtype=residual4.dtype,
                      device=residual4.device)
    llr = torch.zeros(K, device=residual4.device)
    for ncal in range(Ts):
        s0 = ncal * Tdelta * B
        s1 = s0 + Tdelta * B
        Rchunk = to_R(residual4[s0:s1])
        Cchunk = C[:, s0:s1]
        Jchunk = J[:, ncal * 2 * N:(ncal + 1) * 2 * N]
        H = hs.hessianres(Rchunk, Cchunk, Jchunk, N)
        if Hadd is not None:
            H = H + Hadd
        m = hs.dres_colmeans(Cchunk, Jchunk, N, H,
                             per_k=True)                       # (8,K,4,B)
        ms = m.sum(dim=0)                                      # (K,4,B)
        for pol in (0, 3):
            out[:, s0:s1, pol] = ms[:, pol, :].repeat(1, Tdelta)
        llr += hs.log_likelihood_ratio(Rchunk, Cchunk, Jchunk, N) / Ts
    out = out * (8 * B * Tdelta)
    Jn = torch.linalg.vector_norm(J.reshape(K, -1), dim=1)
    Cn = torch.linalg.vector_norm(C.reshape(K, -1), dim=1)
    inf_mean = out.mean(dim=(1, 2)).abs()
    return out, Jn, Cn, inf_mean, llr
