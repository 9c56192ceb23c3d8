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

__all__ = ["generate_training_example"]


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
