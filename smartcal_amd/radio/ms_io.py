"""Real-observation ingestion: MeasurementSet reader + npz bridge.

The reference consumes real LOFAR data through python-casacore
(`calibration/casa_io.py:9-72` read/write of MS data columns,
`generate_data.py:623-877` ``extract_dataset``/``get_info_from_dataset``,
`demixing/evaluate.py:20-58` real-MS deployment eval). This module gives
the MI355X framework the same capability with two entry points:

* :func:`read_ms` — direct MS → :class:`~smartcal_amd.radio.sim.VisData`
  when ``python-casacore`` is importable (it is an optional dependency:
  this image ships without it, real LOFAR reduction nodes have it);
* the **npz bridge** — :func:`ms_to_npz` exports an MS to the
  ``save_visdata`` npz schema on any casacore-equipped machine, and
  :func:`observation_from_npz` merges one-or-more per-sub-band npz files
  into a single multi-frequency ``VisData`` here. The npz schema is the
  documented interchange format: arrays ``uvw (S,3) f32``,
  ``freqs (Nf,) f64``, ``data (Nf,S,4) c64``, scalars ``N, ra0, dec0,
  Ts, Tdelta, noise_sigma``, with S ordered timeslot-major over the
  p<q lexicographic cross-correlation baselines (autocorrelations
  excluded — `casa_io.py:20`).
"""

from __future__ import annotations

from typing import Sequence

import numpy as np
import torch

from .io import load_visdata, save_visdata
from .sim import VisData

__all__ = ["read_ms", "ms_to_npz", "observation_from_npz",
           "merge_visdata", "average_visdata"]


def _casacore_tables():
    try:
        from casacore import tables  # type: ignore
        return tables
    except ImportError as e:  # pragma: no cover - casacore not in image
        raise ImportError(
            "python-casacore is not installed. Read MeasurementSets on a "
            "casacore-equipped machine and export them with "
            "radio.ms_io.ms_to_npz(ms, out.npz); then load the npz here "
            "with radio.ms_io.observation_from_npz([...]).") from e


def read_ms(ms_path: str, col: str = "DATA", Ts: int = 1,
            Tdelta: int | None = None, device="cpu") -> VisData:
    """One MS → VisData (casacore required; see the npz bridge otherwise).

    Mirrors `casa_io.read_corr:9-72`: reads uvw + the 4-pol data column
    excluding autocorrelations, the channel frequencies from
    SPECTRAL_WINDOW (averaged to one band value, as the reference's
    per-sub-band MSs carry one effective frequency each) and the phase
    center from FIELD.
    """
    tables = _casacore_tables()
    t = tables.table(ms_path, readonly=True, ack=False)
    a1 = t.getcol("ANTENNA1")
    a2 = t.getcol("ANTENNA2")
    sel = a1 != a2                       # exclude autocorrelations
    uvw = t.getcol("UVW")[sel].astype(np.float32)
    data = t.getcol(col)[sel]            # (rows, nchan, 4)
    t.close()
    tf = tables.table(f"{ms_path}/SPECTRAL_WINDOW", readonly=True,
                      ack=False)
    chan_freq = tf.getcol("CHAN_FREQ")[0]
    tf.close()
    tp = tables.table(f"{ms_path}/FIELD", readonly=True, ack=False)
    ra0, dec0 = tp.getcol("PHASE_DIR")[0][0]
    tp.close()
    N = int(max(a1.max(), a2.max())) + 1
    B = N * (N - 1) // 2
    vis4 = data.mean(axis=1).astype(np.complex64)   # channel-average
    T = vis4.shape[0] // B
    if Tdelta is None:
        Tdelta = max(T // max(Ts, 1), 1)
    return VisData(
        uvw=torch.as_tensor(uvw, device=device),
        freqs=np.asarray([float(chan_freq.mean())]),
        data=torch.as_tensor(vis4, device=device).unsqueeze(0),
        N=N, ra0=float(ra0), dec0=float(dec0), Ts=Ts, Tdelta=Tdelta,
        noise_sigma=0.0)


def ms_to_npz(ms_path: str, out_path: str, col: str = "DATA",
              Ts: int = 1) -> None:
    """Export one MS to the npz interchange schema (casacore machine)."""
    save_visdata(read_ms(ms_path, col=col, Ts=Ts), out_path)


def merge_visdata(parts: Sequence[VisData]) -> VisData:
    """Stack per-sub-band VisData (same geometry) into one
    multi-frequency observation — the in-memory analogue of the
    reference's `L_SB[1-8].MS` globs (`docal.sh:12`)."""
    v0 = parts[0]
    for v in parts[1:]:
        if v.data.shape[1:] != v0.data.shape[1:] or v.N != v0.N:
            raise ValueError("sub-band geometry mismatch")
    freqs = np.concatenate([np.asarray(v.freqs).reshape(-1)
                            for v in parts])
    data = torch.cat([v.data for v in parts], dim=0)
    order = np.argsort(freqs)
    return VisData(uvw=v0.uvw, freqs=freqs[order],
                   data=data[list(order)], N=v0.N, ra0=v0.ra0,
                   dec0=v0.dec0, Ts=v0.Ts, Tdelta=v0.Tdelta,
                   noise_sigma=v0.noise_sigma)


def average_visdata(vis: VisData, time_factor: int = 1,
                    freq_factor: int = 1) -> VisData:
    """Time/frequency averaging of an observation — the in-memory
    equivalent of the DP3 averaging step in the reference's
    ``extract_dataset`` (`generate_data.py:623-694`: real MS → averaged
    working set). Visibilities are averaged over ``time_factor``
    consecutive timeslots (uvw taken from the first slot of each group)
    and ``freq_factor`` consecutive bands (frequencies averaged).
    """
    B = vis.B
    T = vis.n_time
    Nf = vis.data.shape[0]
    tf = max(int(time_factor), 1)
    ff = max(int(freq_factor), 1)
    Tn = T // tf
    Fn = Nf // ff
    if Tn < 1 or Fn < 1:
        raise ValueError("averaging factors exceed the data extent")
    d = vis.data[:Fn * ff, :Tn * tf * B]
    d = d.reshape(Fn, ff, Tn, tf, B, 4).mean(dim=(1, 3))
    uvw = vis.uvw[:Tn * tf * B].reshape(Tn, tf, B, 3)[:, 0] \
        .reshape(Tn * B, 3)
    freqs = np.asarray(vis.freqs).reshape(-1)[:Fn * ff] \
        .reshape(Fn, ff).mean(axis=1)
    return VisData(uvw=uvw, freqs=freqs,
                   data=d.reshape(Fn, Tn * B, 4), N=vis.N, ra0=vis.ra0,
                   dec0=vis.dec0, Ts=max(vis.Ts // tf, 1),
                   Tdelta=max(vis.Tdelta // tf, 1),
                   noise_sigma=vis.noise_sigma)


def observation_from_npz(paths: Sequence[str], device="cpu") -> VisData:
    """Load + merge exported sub-band npz files into one observation."""
    if not paths:
        raise ValueError("no npz files given")
    return merge_visdata([load_visdata(p, device=device)
                          for p in sorted(paths)])
