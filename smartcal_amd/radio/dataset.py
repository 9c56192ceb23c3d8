"""Supervised training-example generation for the demixing classifier.

In-memory equivalent of `calibration/generate_data.generate_training_data`
(`generate_data.py:118-622`): one synthetic observation → target-only
calibration → per-direction influence maps + summary stats → feature
vector x (K·(Ninf²+8), per-direction layout identical to
`generate_data.py:591-615`: normalized influence image, separation,
azimuth, elevation, log‖J‖, log‖C‖, log|Inf|, LLR, log f) and 0/1 label
vector y over the K−1 outliers (bright-and-above-horizon criterion,
`generate_data.py:575-589` — with synthetic ground truth instead of
masked-flux photometry).
"""

from __future__ import annotations

import math

import numpy as np
import torch

from . import array as arr
from . import sim as rsim
from . import solver as rsolver
from . import influence as rinf
from . import imaging as rimg
from .coherency import predict_coherencies_uvw
from .sky import ClusterSet

__all__ = ["generate_training_example", "info_from_observation"]


def info_from_observation(vis, Ninf: int = 64, admm_iter: int = 5,
                          lst: float | None = None,
                          sky=None, clusters=None):
    """Real-observation → model input x (K·(Ninf²+8),) — the in-memory
    equivalent of `generate_data.get_info_from_dataset:696-877` used by
    the deployment eval (`demixing/evaluate.py:20-58`).

    ``vis`` is a :class:`radio.sim.VisData` (e.g. loaded from exported
    npz via :func:`radio.ms_io.observation_from_npz`). The direction set
    is the A-team fixture (the reference's ``demixing/base.sky``) plus
    the target field; when no target sky model is supplied a unit point
    source at the phase center is used (the reference reads its model
    from the sky file distributed with the observation — pass
    ``sky``/``clusters`` to use one). Geometry (separation/az/el) comes
    from the observation's own phase center.
    """
    from . import sim as rs
    from .sky import ClusterDef, SkyModel

    ra0, dec0 = vis.ra0, vis.dec0
    if lst is None:
        lst = 0.0
    if sky is None:
        names, ras, decs, sIs, sPs, cls = [], [], [], [], [], []
        for i, (nm, ra, dec, flux) in enumerate(rs.ATEAM):
            names.append(nm)
            ras.append(ra)
            decs.append(dec)
            sIs.append(flux)
            sPs.append(-0.7)
            cls.append(ClusterDef(i + 1, 1, [nm]))
        names.append("TARGET")
        ras.append(ra0)
        decs.append(dec0)
        sIs.append(1.0)
        sPs.append(0.0)
        cls.append(ClusterDef(len(rs.ATEAM) + 1, 1, ["TARGET"]))
        sky = SkyModel.from_arrays(names, ras, decs, sIs,
                                   np.asarray(sPs), float(vis.freqs[0]))
        clusters = ClusterSet(cls)
    K = len(clusters)
    sep = np.zeros(K, np.float32)
    az = np.zeros(K, np.float32)
    el = np.zeros(K, np.float32)
    for i, (nm, ra, dec, _f) in enumerate(rs.ATEAM[:K - 1]):
        sep[i] = arr.separation(ra, dec, ra0, dec0)
        az[i], el[i] = arr.azel_of(ra, dec, lst)
    az[-1], el[-1] = arr.azel_of(ra0, dec0, lst)

    freqs = np.asarray(vis.freqs).reshape(-1)
    cs_t = ClusterSet([clusters[K - 1]])
    C_t = torch.stack([
        predict_coherencies_uvw(sky, cs_t, vis.uvw, float(f), ra0, dec0,
                                smear_bw=180e3) for f in freqs])
    sol = rsolver.calibrate(vis, sky, cs_t, np.ones(1, np.float32),
                            admm_iter=admm_iter, poly_order=2,
                            C_cache=C_t)
    C_all = predict_coherencies_uvw(sky, clusters, vis.uvw,
                                    float(freqs[0]), ra0, dec0,
                                    smear_bw=180e3)
    N = vis.N
    J = torch.zeros((K, 2 * N * vis.Ts, 2), dtype=torch.complex64,
                    device=vis.data.device)
    J[:, 0::2, 0] = 1.0
    J[:, 1::2, 1] = 1.0
    J[K - 1] = sol.J_ref_layout(0)
    vals, Jn, Cn, inf_mean, llr = rinf.influence_per_direction(
        sol.residual[0], C_all, J, N, vis.Tdelta)

    Nout = Ninf * Ninf + 8
    x = np.zeros(K * Nout, np.float32)
    for ck in range(K):
        sI = 0.5 * (vals[ck, :, 0] + vals[ck, :, 3])
        img = rimg.dirty_image(vis.uvw, sI, float(freqs[0]), Ninf)
        flat = img.T.reshape(-1).cpu().numpy()
        nrm = np.linalg.norm(flat)
        x[ck * Nout:ck * Nout + Ninf * Ninf] = flat / max(nrm, 1e-12)
        o = ck * Nout + Ninf * Ninf
        x[o + 0] = sep[ck]
        x[o + 1] = az[ck]
        x[o + 2] = el[ck]
        x[o + 3] = math.log(max(float(Jn[ck]), 1e-12))
        x[o + 4] = math.log(max(float(Cn[ck]), 1e-12))
        x[o + 5] = math.log(max(float(inf_mean[ck]), 1e-12))
        x[o + 6] = float(llr[ck])
        x[o + 7] = math.log(float(freqs[0]))
    return x, K


def generate_training_example(rng: np.random.Generator, Ninf: int = 64,
                              N_stations: int = 26, Nf: int = 3,
                              Ts: int = 2, Tdelta: int = 5,
                              device="cpu", admm_iter: int = 5,
                              el_floor_deg: float = 3.0):
    """→ (x (K·(Ninf²+8),), y (K−1,), K). Target direction is last."""
    sky, cs, sep, az, el, fluxes, ra0, dec0 = rsim.make_demixing_sky(rng)
    K = len(cs)
    layout = arr.lofar_like_layout(N_stations, rng)
    freqs = np.linspace(115e6, 185e6, Nf)
    snr = rng.random() * (0.5 - 0.05) + 0.05
    vis = rsim.simulate_observation(layout, sky, cs, freqs, ra0, dec0,
                                    Ts, Tdelta, snr=snr, device=device,
                                    rng=rng,
                                    torch_seed=int(rng.integers(2 ** 31)))
    # target-only calibration (as the data pipeline does before the
    # per-direction influence analysis)
    cs_t = ClusterSet([cs[K - 1]])
    C_t = torch.stack([
        predict_coherencies_uvw(sky, cs_t, vis.uvw, float(f), ra0, dec0,
                                smear_bw=180e3) for f in freqs])
    sol = rsolver.calibrate(vis, sky, cs_t, np.ones(1, np.float32),
                            admm_iter=admm_iter, poly_order=2,
                            C_cache=C_t)
    # per-direction influence with the FULL direction set in the model
    C_all = predict_coherencies_uvw(sky, cs, vis.uvw, float(freqs[0]),
                                    ra0, dec0, smear_bw=180e3)
    # J for all K directions: target solution for target, identity for
    # the (uncalibrated) outliers
    N = vis.N
    J = torch.zeros((K, 2 * N * vis.Ts, 2), dtype=torch.complex64,
                    device=vis.data.device)
    J[:, 0::2, 0] = 1.0
    J[:, 1::2, 1] = 1.0
    J[K - 1] = sol.J_ref_layout(0)
    vals, Jn, Cn, inf_mean, llr = rinf.influence_per_direction(
        sol.residual[0], C_all, J, N, vis.Tdelta)

    Nout = Ninf * Ninf + 8
    x = np.zeros(K * Nout, np.float32)
    for ck in range(K):
        sI = 0.5 * (vals[ck, :, 0] + vals[ck, :, 3])
        img = rimg.dirty_image(vis.uvw, sI, float(freqs[0]), Ninf)
        flat = img.T.reshape(-1).cpu().numpy()       # order='F' flatten
        nrm = np.linalg.norm(flat)
        x[ck * Nout:ck * Nout + Ninf * Ninf] = flat / max(nrm, 1e-12)
        o = ck * Nout + Ninf * Ninf
        x[o + 0] = sep[ck]
        x[o + 1] = az[ck]
        x[o + 2] = el[ck]
        x[o + 3] = math.log(max(float(Jn[ck]), 1e-12))
        x[o + 4] = math.log(max(float(Cn[ck]), 1e-12))
        x[o + 5] = math.log(max(float(inf_mean[ck]), 1e-12))
        x[o + 6] = float(llr[ck])
        x[o + 7] = math.log(freqs[0])
    # labels: outlier should be demixed iff above the elevation floor
    # and apparently bright (synthetic stand-in for the masked-flux
    # photometry of `generate_data.py:540-589`)
    el_floor = math.radians(el_floor_deg)
    flux_thresh = 10.0    # Jy apparent — A-team are thousands
    y = ((el[:-1] > el_floor)
         & (fluxes[:-1] > flux_thresh)).astype(np.float32)
    return x, y, K
