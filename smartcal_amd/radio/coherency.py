"""Coherency (model visibility) prediction.

Replaces `calibration_tools.py:215-464` (`skytocoherencies[_torch|_uvw]`),
which loops over sources in Python. Here the whole prediction is one
batched tensor expression per cluster: ``phase = uvw_scaled @ lmn^T``
(a GEMM shape the GPU likes), fused with the complex exponential, flux
scaling, bandwidth-smearing sinc and the Gaussian-source envelope.

Convention: sample axis T = baselines × timeslots in the same ordering as
the visibility data; output C is (K, T, 4) complex64 with XX = YY = Stokes-I
coherency, XY = YX = 0, exactly like the reference.
"""

from __future__ import annotations

import math

import numpy as np
import torch

from .sky import SkyModel, ClusterSet

C_LIGHT = 2.99792458e8

__all__ = ["predict_coherencies", "predict_coherencies_uvw"]


def _cluster_source_tensors(sky: SkyModel, cluster, ra0, dec0, freq, device):
    """Gather per-source tensors for one cluster at ``freq``."""
    idx = sky.index
    sel = np.asarray([idx[n] for n in cluster.names], dtype=np.int64)
    l, m, n = sky.lmn(ra0, dec0)
    lmn = np.stack([l[sel], m[sel], n[sel]], axis=1)          # (S,3)
    flux = sky.flux_at(freq)[sel]                              # (S,)
    t = lambda a, dt=torch.float32: torch.as_tensor(
        np.ascontiguousarray(a), dtype=dt, device=device)
    return (t(lmn, torch.float64), t(flux, torch.float64),
            t(sky.eX[sel], torch.float64), t(sky.eY[sel], torch.float64),
            t(sky.eP[sel], torch.float64),
            torch.as_tensor(sky.gaussian[sel], device=device))


def predict_coherencies_uvw(sky: SkyModel, clusters: ClusterSet,
                            uvw: torch.Tensor, freq: float,
                            ra0: float, dec0: float,
                            smear_bw: float | None = None,
                            chunk: int = 512) -> torch.Tensor:
    """Predict C (K, T, 4) complex64 for raw uvw (T, 3) in meters.

    ``smear_bw``: channel bandwidth in Hz for the sinc smearing factor
    (the reference hardcodes 180e3 in `calibration_tools.py:380`); None
    disables smearing (matching `skytocoherencies[_torch]`).
    Gaussian sources (name 'G…') get the projected-envelope treatment of
    `calibration_tools.py:424-448`.
    """
    device = uvw.device
    T = uvw.shape[0]
    scale = 2.0 * math.pi / C_LIGHT * freq
    u = uvw[:, 0].double() * scale
    v = uvw[:, 1].double() * scale
    w = uvw[:, 2].double() * scale
    fdelta = (smear_bw / freq) if smear_bw is not None else None

    from ..ops import use_hip
    if use_hip(uvw.float() if uvw.is_cuda else uvw):
        # ONE kernel launch for the whole sky (ops/csrc/coherency.hip);
        # the torch composition below is the CPU oracle
        return _predict_hip(sky, clusters, torch.stack((u, v, w), dim=1)
                            .contiguous(), freq, ra0, dec0,
                            fdelta if fdelta is not None else 0.0)

    K = len(clusters)
    C = torch.zeros((K, T, 4), dtype=torch.complex64, device=device)
    for ck, cluster in enumerate(clusters):
        lmn, flux, eX2, eY2, eP, gflag = _cluster_source_tensors(
            sky, cluster, ra0, dec0, freq, device)
        S = lmn.shape[0]
        acc = torch.zeros(T, dtype=torch.complex128, device=device)
        for s0 in range(0, S, chunk):
            s1 = min(s0 + chunk, S)
            lm = lmn[s0:s1]                                      # (s,3)
            # (T,s) phase — GEMM shape
            ph = u.unsqueeze(1) * lm[:, 0] + v.unsqueeze(1) * lm[:, 1] \
                + w.unsqueeze(1) * lm[:, 2]
            amp = flux[s0:s1].expand(T, s1 - s0).clone()
            if fdelta is not None:
                amp = amp * torch.abs(torch.sinc(ph * (0.5 * fdelta / math.pi)))
            g = gflag[s0:s1]
            if bool(g.any()):
                gi = torch.nonzero(g, as_tuple=True)[0]
                ll, mm, nn = lm[gi, 0], lm[gi, 1], lm[gi, 2]
                # projection of uv onto the source tangent plane
                # (`calibration_tools.py:425-441`)
                # note: nn is (n-1) as returned by radectolm; the reference
                # takes acos of that value directly (`calibration_tools.py:425`)
                phi = -torch.acos(torch.clamp(nn, -1.0, 1.0))
                xi = -torch.atan2(-ll, mm)
                cxi, sxi = torch.cos(xi), torch.sin(xi)
                cphi, sphi = torch.cos(phi), torch.sin(phi)
                uup = u.unsqueeze(1) * cxi - v.unsqueeze(1) * cphi * sxi \
                    + w.unsqueeze(1) * sphi * sxi
                vvp = u.unsqueeze(1) * sxi + v.unsqueeze(1) * cphi * cxi \
                    - w.unsqueeze(1) * sphi * cxi
                cpa, spa = torch.cos(eP[gi]), torch.sin(eP[gi])
                uut = 2.0 * eX2[gi] * (cpa * uup - spa * vvp)
                vvt = 2.0 * eY2[gi] * (spa * uup + cpa * vvp)
                amp[:, gi] = amp[:, gi] * (0.5 * math.pi
                                           * torch.exp(-(uut * uut + vvt * vvt)))
            acc = acc + (torch.polar(amp, ph)).sum(dim=1)
        C[ck, :, 0] = acc.to(torch.complex64)
        C[ck, :, 3] = C[ck, :, 0]
    return C


def _predict_hip(sky, clusters, uvw_scaled, freq, ra0, dec0, fdelta):
    """Build the per-source coefficient table and run the HIP kernel.

    Table row: [l, m, n, flux, gflag, gu1, gv1, gw1, gu2, gv2, gw2] —
    the Gaussian projected-envelope rotation is linear in (u, v, w), so
    it collapses to 6 per-source coefficients (see coherency.hip)."""
    from ..ops import ext

    device = uvw_scaled.device
    rows = []
    offs = [0]
    for cluster in clusters:
        lmn, flux, eX2, eY2, eP, gflag = _cluster_source_tensors(
            sky, cluster, ra0, dec0, freq, device)
        S = lmn.shape[0]
        tab = torch.zeros(S, 11, dtype=torch.float64, device=device)
        tab[:, 0:3] = lmn
        tab[:, 3] = flux
        g = gflag.to(torch.bool)
        if bool(g.any()):
            gi = torch.nonzero(g, as_tuple=True)[0]
            ll, mm, nn = lmn[gi, 0], lmn[gi, 1], lmn[gi, 2]
            phi = -torch.acos(torch.clamp(nn, -1.0, 1.0))
            xi = -torch.atan2(-ll, mm)
            cxi, sxi = torch.cos(xi), torch.sin(xi)
            cphi, sphi = torch.cos(phi), torch.sin(phi)
            cpa, spa = torch.cos(eP[gi]), torch.sin(eP[gi])
            # uup = (cxi)u + (-cphi sxi)v + (sphi sxi)w; vvp likewise
            up = torch.stack((cxi, -cphi * sxi, sphi * sxi), dim=1)
            vp = torch.stack((sxi, cphi * cxi, -sphi * cxi), dim=1)
            tab[gi, 4] = 1.0
            tab[gi, 5:8] = 2.0 * eX2[gi, None] * (cpa[:, None] * up
                                                  - spa[:, None] * vp)
            tab[gi, 8:11] = 2.0 * eY2[gi, None] * (spa[:, None] * up
                                                   + cpa[:, None] * vp)
        rows.append(tab)
        offs.append(offs[-1] + S)
    src = torch.cat(rows).contiguous()
    off = torch.tensor(offs, dtype=torch.int32, device=device)
    return ext().coherency_predict(uvw_scaled, src, off, float(fdelta))


def predict_coherencies(sky: SkyModel, clusters: ClusterSet,
                        uvw: torch.Tensor, freq: float, ra0: float,
                        dec0: float) -> torch.Tensor:
    """No-smearing variant matching `skytocoherencies_torch`
    (`calibration_tools.py:298-368`)."""
    return predict_coherencies_uvw(sky, clusters, uvw, freq, ra0, dec0,
                                   smear_bw=None)
