"""Calibration Hessian, solution/residual derivatives, log-likelihood ratio.

The numerical core of influence-map computation. The reference implements
these as quadruple Python loops over (direction, timeslot, baseline)
(`calibration/calibration_tools.py`: `Hessianres` :589, `Hessianres_torch`
:634, `Dsolutions[_r][_torch]` :680-875, `Dresiduals[_r|_k|_rk][_torch]`
:879-1178, `log_likelihood_ratio` :1181) — ~minutes per 128² map chunk.
Here each is a handful of batched einsum / index_add_ launches over the
whole (K, T·B) sample block at once, numerically identical (complex64) and
device-agnostic; on MI355X the batched forms run as dense GEMM-shaped work.

Conventions (identical to the reference):
 * stations N, baselines B=N(N-1)/2 enumerated (p,q), p<q lexicographic;
 * samples ordered timeslot-major: sample ck = t·B + b;
 * residual R: (2·B·T, 2) complex — consecutive 2×2 blocks per sample;
 * coherencies C: (K, B·T, 4) complex, rows [XX,XY,YX,YY], 2×2 via
   column-major reshape ([[XX,YX],[XY,YY]] per the reference's
   `reshape((2,2),order='F')`);
 * solutions J: (K, 2N, 2) complex for the interval.
"""

from __future__ import annotations

import torch

__all__ = ["baseline_pq", "hessianres", "dsolutions_r", "dresiduals_r",
           "dresiduals_rk", "dres_colmeans", "log_likelihood_ratio",
           "R_VECTORS"]

_EPS = 1e-12


def baseline_pq(N: int, device=None):
    """(p_idx, q_idx) arrays of length B for p<q lexicographic order."""
    p, q = torch.triu_indices(N, N, offset=1, device=device)
    return p.contiguous(), q.contiguous()


def _c22(C: torch.Tensor) -> torch.Tensor:
    """(K,S,4) rows [XX,XY,YX,YY] → (K,S,2,2) column-major 2×2.

    Implemented as reshape+transpose VIEW (zero-copy): the advanced-index
    form `C[..., (0,2,1,3)]` launches a gather kernel and ~8 ms of python
    per call at LOFAR scale."""
    return C.reshape(*C.shape[:-1], 2, 2).mT


def _kron_T_I2(D: torch.Tensor) -> torch.Tensor:
    """kron(D^T, I2) for batched 2×2 D: (...,2,2) → (...,4,4)."""
    E = torch.eye(2, dtype=D.dtype, device=D.device)
    out = torch.einsum('...ji,kl->...ikjl', D, E)
    return out.reshape(*D.shape[:-2], 4, 4)


# the 8 elementary perturbation vectors dVpq (re/im of each 2×2 entry),
# as row-major 2×2 matrices: r//2 selects the entry, odd r is imaginary
# (`calibration_tools.py:700-704`)
def _r_vectors(device, dtype=torch.complex64):
    V = torch.zeros(8, 2, 2, dtype=dtype, device=device)
    for r in range(8):
        m = r // 2
        V[r, m // 2, m % 2] = 1.0 if r % 2 == 0 else 1j
    return V


R_VECTORS = _r_vectors  # exposed for tests


def hessianres(R: torch.Tensor, C: torch.Tensor, J: torch.Tensor,
               N: int) -> torch.Tensor:
    """H (K, 4N, 4N): the calibration Hessian around the residual.

    Batched equivalent of `Hessianres_torch` (`calibration_tools.py:634-676`):
    per (k, sample at baseline p,q) accumulate
      off-diag (p,q): kron(-conj(C), Res)   [+ hermitian mirror]
      diag (p,p): kron((C J_q^H J_q C^H)^T, I)
      diag (q,q): kron((C^H J_p^H J_p C)^T, I)
    averaged over B·T.
    """
    K, S = C.shape[0], C.shape[1]
    B = N * (N - 1) // 2
    T = S // B
    p_idx, q_idx = baseline_pq(N, C.device)
    from ..ops import use_hip
    if use_hip(R.real):
        # single-launch assembly kernel (ops/csrc/hessianres.hip); the
        # torch composition below is the CPU oracle
        from ..ops import ext
        return ext().hessianres(
            C.contiguous(), R.contiguous(), J.contiguous(),
            p_idx.to(torch.int32).contiguous(),
            q_idx.to(torch.int32).contiguous(), N)
    Res = R.reshape(S, 2, 2)
    Ci = _c22(C)                                          # (K,S,2,2)

    # off-diagonal blocks, summed over timeslots per baseline
    Imp = torch.einsum('dsij,sab->dsiajb', -Ci.conj(), Res)
    Imp = Imp.reshape(K, T, B, 4, 4).sum(dim=1)           # (K,B,4,4)

    Jv = J.reshape(K, N, 2, 2)
    Jp = Jv[:, p_idx]                                     # (K,B,2,2)
    Jq = Jv[:, q_idx]
    CiT = Ci.reshape(K, T, B, 2, 2)
    # elementwise 2×2 products (radio.small_complex): batched tiny
    # complex GEMMs are pathological for rocBLAS
    from .small_complex import mm2, mm2H, Hmm2
    R1 = mm2H(CiT, Jq.unsqueeze(1))
    D1 = mm2H(R1, R1).sum(dim=1)                          # (K,B,2,2)
    R2 = mm2(Jp.unsqueeze(1), CiT)
    D2 = Hmm2(R2, R2).sum(dim=1)

    Hb = C.new_zeros(K, N * N, 4, 4)
    Hb.index_add_(1, p_idx * N + q_idx, Imp)
    Hb.index_add_(1, q_idx * N + p_idx, Imp.conj().mT)
    Hb.index_add_(1, p_idx * (N + 1), _kron_T_I2(D1))
    Hb.index_add_(1, q_idx * (N + 1), _kron_T_I2(D2))
    H = Hb.reshape(K, N, N, 4, 4).permute(0, 1, 3, 2, 4) \
          .reshape(K, 4 * N, 4 * N)
    return H / (B * T)


def dsolutions_r(C: torch.Tensor, J: torch.Tensor, N: int,
                 Dgrad: torch.Tensor) -> torch.Tensor:
    """dJ (8, K, 4N, B): solution derivatives for all 8 perturbation
    directions. Batched `Dsolutions_r_torch`
    (`calibration_tools.py:827-875`).

    Per sample at baseline (p,q): lhs = J_q C^H; the AdV column b gains
    rows vec(lhs^T · V_r) split between the p-block of the top and bottom
    halves; then dJ = solve(Dgrad + εI, AdV) per direction.
    """
    K, S = C.shape[0], C.shape[1]
    B = N * (N - 1) // 2
    T = S // B
    dev = C.device
    p_idx, q_idx = baseline_pq(N, dev)
    Ci = _c22(C)
    Jv = J.reshape(K, N, 2, 2)
    Jq = Jv[:, q_idx]                                      # (K,B,2,2)
    from .small_complex import mm2H
    lhs = mm2H(Jq.unsqueeze(1), Ci.reshape(K, T, B, 2, 2))  # (K,T,B,2,2)
    lhsT = lhs.mT.sum(dim=1)                               # (K,B,2,2)

    Vr = _r_vectors(dev, C.dtype)                          # (8,2,2)
    # M[r,k,b] = lhsT[k,b] @ Vr[r]  → (8,K,B,2,2)
    M = torch.einsum('kbim,rmj->rkbij', lhsT, Vr)

    AdV = C.new_zeros(8, K, 2, N, 2, B)
    ar = torch.arange(B, device=dev)
    # row h·2N + 2p + c, column b  ⇐  M[r,k,b,h,c]
    AdV[:, :, :, p_idx, :, ar] = M.permute(2, 0, 1, 3, 4)
    AdV = AdV.reshape(8, K, 4 * N, B)

    eye = torch.eye(4 * N, dtype=C.dtype, device=dev)
    # inv+GEMM instead of batched solve: ROCm's hipblasCtrsmBatched fails
    # for complex64 with nrhs ≥ 1024 (B can be 1891 at LOFAR scale), and
    # the GEMM path keeps MFMA busy; Dgrad is ε-regularized
    Ainv = torch.linalg.inv(Dgrad + _EPS * eye)               # (K,4N,4N)
    from ..ops import use_hip
    if use_hip(AdV.real):
        # hand-written batched complex MFMA GEMM (ops/csrc/cgemm.hip):
        # rocBLAS runs this complex shape at ~1.1 TF/s — the single
        # biggest cost of the influence pipeline before this kernel
        from ..ops import ext
        Bv = AdV.permute(1, 0, 2, 3).contiguous() \
                .reshape(K * 8, 4 * N, B)                     # (K*8,4N,B)
        dJ = ext().cgemm_nn_bcast(Ainv.contiguous(), Bv, 8)
        dJ = dJ.reshape(K, 8, 4 * N, B)
    else:
        dJ = Ainv.unsqueeze(1) @ AdV.permute(1, 0, 2, 3)      # (K,8,4N,B)
    return dJ.permute(1, 0, 2, 3).contiguous()


def _dres_blocks(C, J, N, dJ):
    """Shared core of dresiduals_*: per-(r,k,baseline) 4×B blocks
    before k-reduction. Returns (8,K,B,2,2,B) einsum factors applied."""
    K, S = C.shape[0], C.shape[1]
    B = N * (N - 1) // 2
    T = S // B
    p_idx, q_idx = baseline_pq(N, C.device)
    Ci = _c22(C)
    Jv = J.reshape(K, N, 2, 2)
    Jq = Jv[:, q_idx]
    from .small_complex import mm2H
    lhs = -(mm2H(Ci.reshape(K, T, B, 2, 2), Jq.unsqueeze(1))).mT
    lhs_sum = lhs.sum(dim=1)                               # (K,B,2,2)
    dJv = dJ.reshape(8, K, 2, N, 2, B)
    gath = dJv[:, :, :, p_idx]                             # (8,K,2,B,2,B)
    return lhs_sum, gath, B, T, K


def _dres_contract(lhs_sum, gath, per_k: bool):
    """Σ_m (and optionally Σ_k) lhs_sum[k,b,i,m]·gath[r,k,m,b,j,c] as
    2·K broadcasted fused multiplies — the einsum form lowers to M=2
    batched GEMMs that run ~800 µs each on rocBLAS (the single biggest
    cost of the whole influence pipeline before this rewrite)."""
    K, B = lhs_sum.shape[0], lhs_sum.shape[1]
    Bc = gath.shape[-1]
    shape = (8, K, B, 2, 2, Bc) if per_k else (8, B, 2, 2, Bc)
    out = gath.new_zeros(shape)
    for k in range(K):
        for m in range(2):
            term = lhs_sum[k, :, :, m].reshape(1, B, 2, 1, 1) \
                * gath[:, k, m].reshape(8, B, 1, 2, Bc)
            if per_k:
                out[:, k] += term
            else:
                out += term
    return out


def dresiduals_rk(C: torch.Tensor, J: torch.Tensor, N: int,
                  dJ: torch.Tensor, addself: bool) -> torch.Tensor:
    """dR (8, K, 4B, B): residual derivatives per direction. Batched
    `Dresiduals_rk` (`calibration_tools.py:1129-1178`)."""
    lhs_sum, gath, B, T, K = _dres_blocks(C, J, N, dJ)
    blocks = _dres_contract(lhs_sum, gath, per_k=True)
    if addself:
        Vr = _r_vectors(C.device, C.dtype)                 # (8,2,2)
        ar = torch.arange(B, device=C.device)
        # advanced indexing on dims 2 and 5 yields view shape (B,8,K,2,2)
        blocks[:, :, ar, :, :, ar] += T * Vr.reshape(1, 8, 1, 2, 2)
    dR = blocks.reshape(8, K, B, 4, B).reshape(8, K, 4 * B, B)
    return dR / (B * T)


def dresiduals_r(C: torch.Tensor, J: torch.Tensor, N: int,
                 dJ: torch.Tensor, addself: bool) -> torch.Tensor:
    """dR (8, 4B, B), summed over directions. Batched
    `Dresiduals_r_torch` (`calibration_tools.py:1078-1126`)."""
    lhs_sum, gath, B, T, K = _dres_blocks(C, J, N, dJ)
    blocks = _dres_contract(lhs_sum, gath, per_k=False)
    if addself:
        Vr = _r_vectors(C.device, C.dtype)
        ar = torch.arange(B, device=C.device)
        # advanced indexing on dims 1 and 4 yields view shape (B,8,2,2)
        blocks[:, ar, :, :, ar] += K * T * Vr.reshape(1, 8, 2, 2)
    dR = blocks.reshape(8, B, 4, B).reshape(8, 4 * B, B)
    return dR / (B * T)


def dres_colmeans(C: torch.Tensor, J: torch.Tensor, N: int,
                  H: torch.Tensor, per_k: bool = False) -> torch.Tensor:
    """Row-block means of the residual derivatives, computed ANALYTICALLY
    — the influence values the env actually consumes.

    ``influence_values`` needs only ``dR.reshape(8,B,4,B).mean(dim=1)``:
    the mean over baseline row-blocks b. Because every term of dR is
    ``lhs_sum[k,b,i,m] * dJ[r,k, m·2N+2·p_idx[b]+j, c]``, the b-sum
    collapses onto the station axis (p = p_idx[b] takes only N values):

        Lsum[k,p,i,m] = Σ_{b: p_idx[b]=p} lhs_sum[k,b,i,m]      (K,N,2,2)
        W_k[(i,j), m·2N+2p+j] = Lsum[k,p,i,m]                   (K,4,4N)
        X_k = W_k · (H_k+εI)^{-1}      — a 4-RHS SOLVE, not 8·B  (K,4,4N)
        mean[r,(i,j),c] = (1/B) Σ_k Σ_{h,cc}
                            X_k[(i,j), h·2N+2·p_idx[c]+cc] · M[r,k,c,h,cc]

    where M is dsolutions' AdV generator. This never materializes dJ
    (8·K·4N·B, 180 MB at LOFAR scale) nor the (8,B,2,2,B) broadcast
    blocks (1.8 GB/term): measured 11.9 ms → sub-ms per chunk, and the
    4-RHS solve retires the inv+GEMM trsm workaround on this path
    (nrhs=4 ≪ the 1024-rhs rocBLAS bug threshold).

    Returns (8, 4, B) — or (8, K, 4, B) with ``per_k=True`` (the
    per-direction maps of `influence_tools.analysis_uvw_perdir`) —
    equal to ``dresiduals_*(…).reshape(8,[K],B,4,B).mean(dim=b)``
    (oracle-tested to 3e-7): the dresiduals normalization 1/(B·T) times
    the 1/B row-block mean.
    """
    K, S = C.shape[0], C.shape[1]
    B = N * (N - 1) // 2
    T = S // B
    dev = C.device
    p_idx, q_idx = baseline_pq(N, dev)
    Ci = _c22(C)
    Jv = J.reshape(K, N, 2, 2)
    Jq = Jv[:, q_idx]
    from .small_complex import mm2H
    # dsolutions' generator M[r,k,b,i,j] = (Σ_t Jq C^H).T @ Vr
    lhsT = mm2H(Jq.unsqueeze(1),
                Ci.reshape(K, T, B, 2, 2)).mT.sum(dim=1)       # (K,B,2,2)
    Vr = _r_vectors(dev, C.dtype)
    M = torch.einsum('kbim,rmj->rkbij', lhsT, Vr)              # (8,K,B,2,2)
    # dresiduals' lhs_sum = -(Σ_t (C^H J_q)^T)  — note operand order
    # differs from lhsT above (mm2H(Ci, Jq) vs mm2H(Jq, Ci))
    lhs_sum = -(mm2H(Ci.reshape(K, T, B, 2, 2),
                     Jq.unsqueeze(1))).mT.sum(dim=1)           # (K,B,2,2)

    # collapse b onto stations
    Lsum = C.new_zeros(K, N, 2, 2)
    Lsum.index_add_(1, p_idx, lhs_sum)                         # (K,N,2,2)

    # W index [k, i, j, m, p, jslot], nonzero iff jslot == j:
    # W_k[(i,j), m·2N+2p+j] = Lsum[k,p,i,m]
    W = C.new_zeros(K, 2, 2, 2, N, 2)
    for j in range(2):
        W[:, :, j, :, :, j] = Lsum.permute(0, 2, 3, 1)         # (K,i,m,p)
    Wmat = W.reshape(K, 4, 4 * N)                              # rows (i,j)

    eye = torch.eye(4 * N, dtype=C.dtype, device=dev)
    # X = W · (H+εI)^{-1} : 4-RHS solve per k. H is Hermitian by
    # construction, so Cholesky (measured 1.85 ms vs 4.9 ms LU at
    # K=6, 4N=248) with an LU fallback for the indefinite corner.
    # ROCm bug (probed on MI355X / ROCm 7.2, gpurun_scripts/
    # chol384_probe.py): BATCHED complex cholesky_ex crashes with
    # hipErrorLaunchFailure for n >= ~384 at batch > 1 (n=248 and any
    # single-matrix size are fine) — gate the fast path to 4N <= 256
    # and use the (everywhere-working) LU solve for larger arrays.
    Hh = H + _EPS * eye
    if Hh.is_cuda and K > 1 and 4 * N > 256:
        X = torch.linalg.solve(Hh.mT, Wmat.mT).mT
    else:
        L, info = torch.linalg.cholesky_ex(Hh)
        if int(info.abs().sum()) == 0:
            X = torch.cholesky_solve(Wmat.mH, L).mH            # (K,4,4N)
        else:
            X = torch.linalg.solve(Hh.mT, Wmat.mT).mT

    # gather X columns at (h, p_idx[c], cc) and contract with M:
    # out[r,(k),a,c] = Σ_{h,cc} Xg[k,a,h,c,cc] · M[r,k,c,h,cc]
    Xg = X.reshape(K, 4, 2, N, 2)[:, :, :, p_idx, :]           # (K,4,2,B,2)
    norm = float(B) * B * T
    if per_k:
        return torch.einsum('kahcx,rkchx->rkac', Xg, M) / norm
    return torch.einsum('kahcx,rkchx->rac', Xg, M) / norm


def log_likelihood_ratio(R: torch.Tensor, C: torch.Tensor, J: torch.Tensor,
                         N: int) -> torch.Tensor:
    """LLR (K,): batched `log_likelihood_ratio`
    (`calibration_tools.py:1181-1225`). σ² estimated from Stokes V of the
    residual; LLR_k = (‖r+μ_k‖² − ‖r‖²)/σ²."""
    K, S = C.shape[0], C.shape[1]
    B = N * (N - 1) // 2
    T = S // B
    p_idx, q_idx = baseline_pq(N, C.device)
    Res = R.reshape(S, 2, 2)
    sV = 0.5 * (Res[:, 0, 1] - Res[:, 1, 0])
    sigma2 = (sV * sV.conj()).real.sum()
    Ci = _c22(C)
    Jv = J.reshape(K, N, 2, 2)
    Jp = Jv[:, p_idx].unsqueeze(1).expand(K, T, -1, 2, 2).reshape(K, S, 2, 2)
    Jq = Jv[:, q_idx].unsqueeze(1).expand(K, T, -1, 2, 2).reshape(K, S, 2, 2)
    from .small_complex import mm2, mm2H
    Mu = mm2H(mm2(Jp, Ci), Jq)                             # (K,S,2,2)
    rn = (Res.abs() ** 2).sum()
    rmn = ((Res.unsqueeze(0) + Mu).abs() ** 2).sum(dim=(1, 2, 3))
    return ((rmn - rn) / (sigma2 + _EPS)).to(torch.float32)
