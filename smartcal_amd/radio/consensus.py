"""Consensus-polynomial constraint matrices (host-side setup math).

Equivalent of `calibration_tools.py:524-585` (`Bpoly`, `consensus_poly`)
and the Hessian-addition assembly of `analysis_torch.py:141-156`. These
run once per direction per episode (K ≤ ~10, 2N×2N pinv) — setup cost,
not hot — so they stay in numpy on the host.
"""

from __future__ import annotations

import numpy as np

__all__ = ["bpoly", "consensus_poly", "hessian_addition"]


def bpoly(x: np.ndarray, N: int) -> np.ndarray:
    """Bernstein basis: for each x in [0,1], the N+1 values
    C(N,r) x^r (1-x)^(N-r); `calibration_tools.py:524-545`."""
    x = np.asarray(x, dtype=np.float64)
    M = len(x)
    fact = np.cumprod(np.concatenate(([1.0], np.arange(1, N + 1))))
    px = np.ones((N + 1, M))
    p1x = np.ones((N + 1, M))
    for ci in range(1, N + 1):
        px[ci] = px[ci - 1] * x
        p1x[ci] = p1x[ci - 1] * (1.0 - x)
    y = np.zeros((N + 1, M))
    for r in range(N + 1):
        y[r] = fact[N] / (fact[N - r] * fact[r]) * px[r] * p1x[N - r]
    return y.T.astype(np.float32)


def consensus_poly(Ne: int, N: int, freqs: np.ndarray, f0: float, fidx: int,
                   polytype: int = 0, rho: float = 0.0, alpha: float = 0.0):
    """(F, P) constraint matrices; `calibration_tools.py:551-585`.

    Ne: polynomial terms; N: stations; freqs: (Nf,) Hz; f0 reference;
    fidx: working frequency index; polytype 0=ordinary 1=Bernstein;
    rho: spectral ADMM weight; alpha: spatial/federated regularization.
    """
    freqs = np.asarray(freqs, dtype=np.float64)
    Nf = len(freqs)
    if polytype == 0:
        Bfull = np.zeros((Nf, Ne), dtype=np.float64)
        Bfull[:, 0] = 1.0
        ff = (freqs - f0) / f0
        for cj in range(1, Ne):
            Bfull[:, cj] = ff ** cj
    else:
        ff = (freqs - freqs.min()) / (freqs.max() - freqs.min())
        Bfull = bpoly(ff, Ne - 1).astype(np.float64)

    Bi = Bfull.T @ Bfull
    Bi = np.linalg.pinv(rho * Bi + alpha * np.eye(Ne))
    Bf = np.kron(Bfull[fidx], np.eye(2 * N))
    P = np.kron(Bi, np.eye(2 * N)) @ Bf.T
    F = np.eye(2 * N) - rho * (Bf @ P)
    return F.astype(np.float32), P.astype(np.float32)


def hessian_addition(Ne: int, N: int, freqs: np.ndarray, f0: float,
                     fidx: int, rho_spectral: float, rho_spatial: float,
                     polytype: int = 1) -> np.ndarray:
    """Per-direction consensus contribution Hadd (4N×4N real, as float32)
    added to the calibration Hessian; `analysis_torch.py:141-156`."""
    alpha = rho_spatial
    F, P = consensus_poly(Ne, N, freqs, f0, fidx, polytype=polytype,
                          rho=rho_spectral, alpha=alpha)
    F = F.astype(np.float64)
    FF = F.T @ F
    I2N = np.eye(2 * N)
    if alpha > 0.0:
        P = P.astype(np.float64)
        PP = P.T @ P
        H11 = 0.5 * rho_spectral * FF \
            + 0.5 * alpha * rho_spectral ** 2 * PP
        H12 = 0.5 * FF + 0.5 * alpha * rho_spectral * PP
        H22 = -0.5 / rho_spectral * (I2N - FF) + 0.5 * alpha * PP
        Htilde = H11 - H12 @ np.linalg.pinv(H22) @ H12
        Hadd = np.kron(np.eye(2), Htilde)
    else:
        Hadd = 0.5 * rho_spectral * np.kron(
            np.eye(2), FF @ (I2N + np.linalg.pinv(I2N - FF) @ FF))
    return Hadd.astype(np.float32)


def hessian_addition_scalar(Ne: int, N: int, freqs: np.ndarray, f0: float,
                            fidx: int, rho_spectral: float,
                            rho_spatial: float, polytype: int = 1) -> float:
    """Scalar c with Hadd = c·I₄N.

    Every matrix in `hessian_addition` is a scalar multiple of I₂N
    (F = I − ρ·kron(b_f Bi b_fᵀ, I) — the reference notes "F … is
    diagonal scalar", `calibration_tools.py:583`), so the whole Schur
    complement collapses to scalar arithmetic on the Ne×Ne Gram matrix —
    O(Ne³) instead of a 2N×2N pinv per direction per step.
    """
    freqs = np.asarray(freqs, dtype=np.float64)
    Nf = len(freqs)
    alpha = rho_spatial
    rho = rho_spectral
    if polytype == 0:
        Bfull = np.zeros((Nf, Ne))
        Bfull[:, 0] = 1.0
        ff = (freqs - f0) / f0
        for cj in range(1, Ne):
            Bfull[:, cj] = ff ** cj
    else:
        ff = (freqs - freqs.min()) / (freqs.max() - freqs.min())
        Bfull = bpoly(ff, Ne - 1).astype(np.float64)
    Bi = np.linalg.pinv(rho * (Bfull.T @ Bfull) + alpha * np.eye(Ne))
    bf = Bfull[fidx]
    s = float(bf @ Bi @ bf)
    f1 = 1.0 - rho * s                  # F = f1·I
    FF = f1 * f1
    def _pinv_s(x):
        return 1.0 / x if abs(x) > 1e-14 else 0.0
    if alpha > 0.0:
        p = float(bf @ Bi.T @ Bi @ bf)  # PᵀP = p·I
        h11 = 0.5 * rho * FF + 0.5 * alpha * rho * rho * p
        h12 = 0.5 * FF + 0.5 * alpha * rho * p
        h22 = -0.5 / rho * (1.0 - FF) + 0.5 * alpha * p
        return float(h11 - h12 * _pinv_s(h22) * h12)
    return float(0.5 * rho * FF * (1.0 + _pinv_s(1.0 - FF) * FF))
