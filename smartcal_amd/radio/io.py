"""Visibility data I/O — text dump/restore and npz checkpoints.

The reference round-trips visibilities between MeasurementSets and text
files (`calibration/readcorr.py:5-43`, `writecorr.py:3-51`,
`calibration_tools.readuvw/writeuvw:505-522`). Here the canonical store
is the in-memory :class:`radio.sim.VisData`; these helpers provide the
same text format (``u v w xx_re xx_im xy_re xy_im yx_re yx_im yy_re
yy_im`` per cross-correlation row) plus compact npz save/load so
observations can be checkpointed and inspected.
"""

from __future__ import annotations

import numpy as np
import torch

from .sim import VisData

__all__ = ["write_corr_text", "read_corr_text", "save_visdata",
           "load_visdata"]


def write_corr_text(uvw: torch.Tensor, vis4: torch.Tensor) -> str:
    """(S,3) uvw + (S,4) visibilities → reference text rows
    (`readcorr.py` output format)."""
    u = uvw.detach().cpu().numpy()
    v = vis4.detach().cpu().numpy()
    lines = []
    for i in range(u.shape[0]):
        row = [u[i, 0], u[i, 1], u[i, 2]]
        for p in range(4):
            row += [v[i, p].real, v[i, p].imag]
        lines.append(" ".join(f"{x:.8e}" for x in row))
    return "\n".join(lines) + "\n"


def read_corr_text(text: str):
    """Inverse of :func:`write_corr_text` → (uvw (S,3) f32,
    vis (S,4) c64) — also parses the reference's `smalluvw.txt` format
    (`calibration_tools.readuvw:505-513`)."""
    a = np.loadtxt(text.splitlines(), dtype=np.float64)
    a = np.atleast_2d(a)
    uvw = a[:, :3].astype(np.float32)
    vis = (a[:, 3::2] + 1j * a[:, 4::2]).astype(np.complex64)
    return torch.from_numpy(uvw), torch.from_numpy(vis)


def save_visdata(vis: VisData, path: str):
    np.savez_compressed(
        path, uvw=vis.uvw.cpu().numpy(), freqs=vis.freqs,
        data=vis.data.cpu().numpy(), N=vis.N, ra0=vis.ra0, dec0=vis.dec0,
        Ts=vis.Ts, Tdelta=vis.Tdelta, noise_sigma=vis.noise_sigma)


def load_visdata(path: str, device="cpu") -> VisData:
    d = np.load(path)
    return VisData(
        uvw=torch.as_tensor(d["uvw"], device=device),
        freqs=d["freqs"],
        data=torch.as_tensor(d["data"], device=device),
        N=int(d["N"]), ra0=float(d["ra0"]), dec0=float(d["dec0"]),
        Ts=int(d["Ts"]), Tdelta=int(d["Tdelta"]),
        noise_sigma=float(d["noise_sigma"]))
