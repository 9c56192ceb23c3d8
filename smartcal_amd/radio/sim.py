"""Observation simulation: sky scenarios, corruption, noise — in memory.

Replaces the reference's external simulation pipeline (`makems` + FIELD
rewrite + `sagecal --simulate` + `addnoise.py` + DP3, driven from
`calibration/dosimul.sh` and `calibration/generate_data.py:118-1239`,
sky generation `calibration/simulate.py:6-375`): synthetic visibilities
are produced directly on device as
``V_pq = J_p C_pq J_q^H + AWGN`` with the same statistical recipes
(systematic-error Jones from `radio.solutions`, SNR-scaled noise as in
`addnoise.py:7-16`), with no MS files, shell-outs or casacore.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field

import numpy as np
import torch

from . import array as arr
from .coords import lmtoradec
from .sky import SkyModel, ClusterSet, ClusterDef
from .coherency import predict_coherencies_uvw
from .solutions import simulate_systematic_errors, solutions_to_J
from .hessian import baseline_pq

__all__ = ["ATEAM", "VisData", "make_calibration_sky", "make_demixing_sky",
           "simulate_observation", "apply_jones", "add_noise", "to_R"]

# The five A-team outlier sources (name, ra rad, dec rad, flux Jy at 150MHz)
# — positions as in the reference's base sky model
# (`demixing/base.sky`, `generate_data.py:50-116`).
_h = lambda h, m, s: (h + m / 60 + s / 3600) * math.pi / 12
_d = lambda d, m, s: math.copysign(
    (abs(d) + m / 60 + s / 3600) * math.pi / 180, d)
ATEAM = [
    ("CasA", _h(23, 23, 24.0), _d(58, 48, 54.0), 13000.0),
    ("CygA", _h(19, 59, 28.4), _d(40, 44, 2.0), 10500.0),
    ("HerA", _h(16, 51, 8.0), _d(4, 59, 33.0), 377.0),
    ("TauA", _h(5, 34, 31.9), _d(22, 0, 52.0), 1420.0),
    ("VirA", _h(12, 30, 49.4), _d(12, 23, 28.0), 1050.0),
]


@dataclass
class VisData:
    """One synthetic observation, resident on device.

    Sample ordering is timeslot-major (s = t·B + b), matching the
    reference's MS row order and the `radio.hessian` convention.
    """
    uvw: torch.Tensor            # (S, 3) float32, meters
    freqs: np.ndarray            # (Nf,) Hz
    data: torch.Tensor           # (Nf, S, 4) complex64 — observed
    N: int                       # stations
    ra0: float
    dec0: float
    Ts: int                      # solution intervals
    Tdelta: int                  # timeslots per interval
    noise_sigma: float = 0.0
    model: torch.Tensor | None = None     # (Nf, S, 4) noiseless, optional
    J_true: np.ndarray | None = None      # (Nf, K, 2N·Ts, 2) ground truth

    @property
    def B(self) -> int:
        return self.N * (self.N - 1) // 2

    @property
    def n_time(self) -> int:
        return self.Ts * self.Tdelta

    @property
    def S(self) -> int:
        return self.uvw.shape[0]

    def stokes_i_std(self, which: torch.Tensor) -> float:
        """std of Stokes I over samples — the reference's noise estimate
        `demixingenv.py:241-252` ((XX+YY)/2 over unflagged rows)."""
        sI = 0.5 * (which[..., 0] + which[..., 3])
        return float(torch.std(torch.view_as_real(sI)).item() * math.sqrt(2))


def to_R(v: torch.Tensor) -> torch.Tensor:
    """(S, 4) complex rows [XX,XY,YX,YY] → (2S, 2) residual layout used by
    `radio.hessian` (matches `analysis_torch.py:19-23`)."""
    return v.reshape(-1, 2, 2).reshape(-1, 2)


def make_calibration_sky(K: int, rng: np.random.Generator,
                         nsrc_center: int = 20, nsrc_outlier: int = 8,
                         n_weak: int = 60, f0: float = 150e6):
    """K-direction calibration scenario (center + K−1 outliers + weak
    background), following `simulate.py:6-375`'s statistical recipe at a
    reduced source count (fluxes/spectra/l,m distributions preserved).

    Returns (sky_sim, clusters_sim, sky_cal, clusters_cal, skylmn, rho0):
    *_sim includes the weak background (simulation only); *_cal is the
    calibration model; skylmn is the (K,5) per-direction averaged
    [id, l, m, sI, sP] DQN metadata; rho0 the analytic flux-scaled ADMM
    rho (`simulate.py` admm_rho0).
    """
    ra0, dec0 = 0.0, math.pi / 2.2
    names, ras, decs, sIs, sPs, cl_sim, cl_cal = [], [], [], [], [], [], []
    skylmn = np.zeros((K, 5), np.float32)

    def add_cluster(cid, lm_center, spread, nsrc, flux_lo, flux_hi):
        lnames = []
        l = lm_center[0] + (rng.random(nsrc) - 0.5) * spread
        m = lm_center[1] + (rng.random(nsrc) - 0.5) * spread
        sI = (rng.random(nsrc) * (flux_hi - flux_lo)) + flux_lo
        sP = rng.standard_normal(nsrc)
        ra, dec = lmtoradec(l, m, ra0, dec0)
        for i in range(nsrc):
            nm = f"PC{cid}S{i}"
            lnames.append(nm)
            names.append(nm)
            ras.append(ra[i]); decs.append(dec[i])
            sIs.append(sI[i]); sPs.append(sP[i])
        skylmn[cid] = [cid, -np.mean(l), np.mean(m), np.sum(sI),
                       float(np.mean(sP))]
        return lnames, float(np.sum(sI))

    fluxes = np.zeros(K)
    # center cluster (direction 0)
    ln, fluxes[0] = add_cluster(0, (0.0, 0.0), 0.5, nsrc_center, 0.1, 1.0)
    cl_sim.append(ClusterDef(1, 1, ln)); cl_cal.append(ClusterDef(1, 1, ln))
    # outliers
    for k in range(1, K):
        ang = 2 * math.pi * k / (K - 1) + rng.random() * 0.3
        rad = 1.2 + rng.random() * 0.8
        ln, fluxes[k] = add_cluster(k, (rad * math.cos(ang),
                                        rad * math.sin(ang)), 0.05,
                                    nsrc_outlier, 2.0, 20.0)
        cl_sim.append(ClusterDef(k + 1, 1, ln))
        cl_cal.append(ClusterDef(k + 1, 1, ln))
    # weak background, simulation-only (last sim cluster)
    wk = []
    l = (rng.random(n_weak) - 0.5) * 1.6
    m = (rng.random(n_weak) - 0.5) * 1.6
    ra, dec = lmtoradec(l, m, ra0, dec0)
    for i in range(n_weak):
        nm = f"PW{i}"
        wk.append(nm); names.append(nm)
        ras.append(ra[i]); decs.append(dec[i])
        sIs.append(0.01 + 0.02 * rng.random()); sPs.append(rng.standard_normal())
    cl_sim.append(ClusterDef(K + 1, 1, wk))

    sky = SkyModel.from_arrays(names, ras, decs, sIs, np.asarray(sPs), f0)
    # analytic rho ∝ flux (simulate.py's admm_rho0 recipe)
    rho0 = fluxes / fluxes.min() * 10.0
    return (sky, ClusterSet(cl_sim), sky, ClusterSet(cl_cal),
            skylmn, rho0.astype(np.float32), ra0, dec0)


def make_demixing_sky(rng: np.random.Generator, f0: float = 150e6,
                      n_outliers: int | None = None):
    """Demixing scenario: 5 A-team outliers + target field (target last,
    as in `generate_data.simulate_data` / `demixingenv.py`). Returns
    (sky, clusters (K=6, target last), separation, azimuth, elevation,
    fluxes, ra0, dec0) with sep/az/el in radians (the units
    `simulate_data` returns, `generate_data.py:891`).

    Target selection mirrors `find_valid_target` strategy 1
    (`generate_data.py:50-106`): the pointing is drawn within
    0.5–30.5° of a random A-team source and accepted when the target is
    at least 3° up at the (random) observing epoch — the regime where
    demixing decisions actually matter."""
    # find_valid_target-style rejection sampling
    low_el = math.radians(3.0)
    close_to = int(rng.integers(len(ATEAM)))
    dist_max = math.radians(0.5 + 30 * rng.random())
    while True:
        ra0 = ATEAM[close_to][1] + rng.random() * dist_max
        dec0 = ATEAM[close_to][2] + rng.random() * dist_max
        dec0 = min(max(dec0, -math.pi / 2 + 0.01), math.pi / 2 - 0.01)
        ra0 = ra0 % (2 * math.pi)
        lst = rng.uniform(0, 2 * math.pi)   # random epoch
        _, el0 = arr.azel_of(ra0, dec0, lst)
        if el0 > low_el:
            break
    names, ras, decs, sIs, sPs, clusters = [], [], [], [], [], []
    ateam = ATEAM if n_outliers is None else ATEAM[:n_outliers]
    K = len(ateam) + 1
    sep = np.zeros(K); az = np.zeros(K); el = np.zeros(K)
    fluxes = np.zeros(K)
    for i, (nm, ra, dec, flux) in enumerate(ateam):
        names.append(nm); ras.append(ra); decs.append(dec)
        sIs.append(flux); sPs.append(-0.7)
        clusters.append(ClusterDef(i + 1, 1, [nm]))
        sep[i] = arr.separation(ra, dec, ra0, dec0)
        az[i], el[i] = arr.azel_of(ra, dec, lst)
        fluxes[i] = flux
    # target field: a handful of sources near center
    tn = []
    nsrc = 6
    l = (rng.random(nsrc) - 0.5) * 0.1
    m = (rng.random(nsrc) - 0.5) * 0.1
    ra, dec = lmtoradec(l, m, ra0, dec0)
    for i in range(nsrc):
        nm = f"PT{i}"
        tn.append(nm); names.append(nm)
        ras.append(ra[i]); decs.append(dec[i])
        sIs.append(1.0 + 4.0 * rng.random()); sPs.append(rng.standard_normal())
    clusters.append(ClusterDef(K, 1, tn))
    sep[-1] = 0.0
    az[-1], el[-1] = arr.azel_of(ra0, dec0, lst)
    fluxes[-1] = float(np.sum(sIs[-nsrc:]))
    sky = SkyModel.from_arrays(names, ras, decs, sIs, np.asarray(sPs), f0)
    return sky, ClusterSet(clusters), sep, az, el, fluxes, ra0, dec0


def apply_jones(C: torch.Tensor, J: torch.Tensor, N: int,
                Tdelta: int) -> torch.Tensor:
    """Corrupt coherencies with per-interval Jones and sum directions:
    V_s = Σ_k J_p(t_i) C_k,s J_q(t_i)^H → (S, 4) complex (row-major
    [V00,V01,V10,V11] = [XX,XY,YX,YY]).

    C: (K, S, 4) [XX,XY,YX,YY] (2×2 col-major, as everywhere);
    J: (K, Ts, N, 2, 2) complex.
    """
    K, S = C.shape[0], C.shape[1]
    dev = C.device
    B = N * (N - 1) // 2
    T = S // B
    p_idx, q_idx = baseline_pq(N, dev)
    t_int = (torch.arange(T, device=dev) // Tdelta).clamp_(max=J.shape[1] - 1)
    # row-major 2×2 ([XX,XY],[YX,YY]) — the physical convention used by
    # sim/solver; predicted coherencies have zero off-diagonals so this
    # agrees with the reference's column-major oracle convention
    C22 = C.reshape(K, T, B, 2, 2)
    Jp = J[:, t_int][:, :, p_idx]                     # (K,T,B,2,2)
    Jq = J[:, t_int][:, :, q_idx]
    from .small_complex import mm2, mm2H
    V = mm2H(mm2(Jp, C22), Jq).sum(dim=0)             # (T,B,2,2)
    return V.reshape(S, 4)


def add_noise(V: torch.Tensor, snr: float,
              gen: torch.Generator | None = None) -> tuple[torch.Tensor, float]:
    """AWGN at given SNR, as `addnoise.py:7-16`: noise scaled so
    ‖data‖/‖noise‖ = snr. Returns (noisy, sigma_per_component)."""
    nr = torch.randn(V.shape + (2,), generator=gen, device=V.device)
    noise = torch.view_as_complex(nr)
    scale = torch.linalg.vector_norm(V) / (snr * torch.linalg.vector_norm(noise))
    return V + noise * scale, float(scale)


def simulate_observation(layout: arr.StationLayout, sky: SkyModel,
                         clusters: ClusterSet, freqs: np.ndarray,
                         ra0: float, dec0: float, Ts: int, Tdelta: int,
                         snr: float = 5.0, dt: float = 10.0,
                         device="cpu", rng: np.random.Generator | None = None,
                         smear_bw: float | None = 180e3,
                         torch_seed: int | None = None) -> VisData:
    """Full synthetic observation over Nf frequencies.

    The error Jones follow `simulate.py:386-435` (spatially-smooth random
    seeds × quadratic freq polynomial × cosine time modulation), applied
    per solution interval; thermal noise at ``snr``.
    """
    rng = rng or np.random.default_rng(0)
    Nf = len(freqs)
    N = layout.n_stations
    T = Ts * Tdelta
    times = np.arange(T) * dt
    uvw_t = arr.uvw_synthesis(layout, ra0, dec0, times)     # (T,B,3)
    S = T * uvw_t.shape[1]
    uvw = torch.as_tensor(uvw_t.reshape(S, 3), dtype=torch.float32,
                          device=device)
    gen = None
    if torch_seed is not None:
        gen = torch.Generator(device=device)
        gen.manual_seed(torch_seed)

    # per-direction mean l,m for the spatial error term
    K = len(clusters)
    lm = np.zeros((K, 2))
    idx = sky.index
    l_all, m_all, _ = sky.lmn(ra0, dec0)
    for k, cl in enumerate(clusters):
        sel = [idx[n] for n in cl.names]
        lm[k] = [np.mean(l_all[sel]), np.mean(m_all[sel])]

    f0 = float(np.mean(freqs))
    gs = simulate_systematic_errors(K, N, Ts, freqs, f0, rng, lm=lm)

    data = torch.zeros((Nf, S, 4), dtype=torch.complex64, device=device)
    model = torch.zeros_like(data)
    J_true = np.zeros((Nf, K, 2 * N * Ts, 2), np.complex64)
    sigma = 0.0
    for fi, f in enumerate(freqs):
        C = predict_coherencies_uvw(sky, clusters, uvw, float(f), ra0, dec0,
                                    smear_bw=smear_bw)
        Jf = solutions_to_J(gs[:, :, fi].T, N, Ts)          # (K,2N·Ts,2)
        J_true[fi] = Jf
        J5 = torch.as_tensor(
            Jf.reshape(K, Ts, N, 2, 2), device=device)
        V = apply_jones(C, J5, N, Tdelta)
        model[fi] = V
        data[fi], sig = add_noise(V, snr, gen)
        sigma += sig / Nf
    return VisData(uvw=uvw, freqs=np.asarray(freqs, np.float64), data=data,
                   N=N, ra0=ra0, dec0=dec0, Ts=Ts, Tdelta=Tdelta,
                   noise_sigma=sigma, model=model, J_true=J_true)
