"""BBS/makesourcedb sky-model conversion → SAGECal sky/cluster/rho.

Parity with `calibration/convertmodel.py:16-88` (which uses lsmtool, not
in this image): parse the BBS `makesourcedb` text format — the same
format the reference's own simulator emits (`simulate.py:140` writes
`(Name, Type, Patch, Ra, Dec, I, Q, U, V, ReferenceFrequency,
SpectralIndex, MajorAxis, MinorAxis, Orientation) = format`) — and emit
the SAGECal sky/cluster/rho texts via `radio.sky`'s writers, grouping
sources by patch into clusters (patch = cluster, Gaussians prefixed
'G', start_cluster offset supported as in the reference).
"""

from __future__ import annotations

import math

import numpy as np

from .sky import SkyModel, ClusterDef, ClusterSet, write_sky_text, \
    write_cluster_text, write_rho_text

__all__ = ["parse_bbs_text", "convert_model"]


def _parse_hms(v: str) -> float:
    """BBS RA 'hh:mm:ss.s' or decimal degrees → radians."""
    if ":" in v:
        h, m, s = v.split(":")
        return (float(h) + float(m) / 60 + float(s) / 3600) * math.pi / 12
    return float(v) * math.pi / 180


def _parse_dms(v: str) -> float:
    """BBS Dec 'dd.mm.ss.s' or decimal degrees → radians."""
    if v.count(".") >= 2:
        parts = v.split(".")
        d = float(parts[0])
        m = float(parts[1])
        s = float(".".join(parts[2:]))
        sign = -1.0 if v.strip().startswith("-") else 1.0
        return sign * (abs(d) + m / 60 + s / 3600) * math.pi / 180
    return float(v) * math.pi / 180


def parse_bbs_text(text: str):
    """→ (SkyModel, ClusterSet) grouping by Patch. Expects a leading
    `(...) = format` header naming the columns."""
    lines = [l for l in text.splitlines() if l.strip()]
    header = None
    rows = []
    for l in lines:
        if "= format" in l:
            cols = l.split("=")[0].strip().lstrip("#( ").rstrip(") ")
            header = [c.strip().split("=")[0].strip("'\"")
                      for c in cols.split(",")]
            continue
        if l.startswith("#") or header is None:
            continue
        rows.append([v.strip() for v in l.split(",")])
    assert header is not None, "no '(...) = format' header found"
    idx = {name: i for i, name in enumerate(header)}

    def get(row, name, default=""):
        i = idx.get(name)
        if i is None or i >= len(row):
            return default
        return row[i] or default

    names, ras, decs, sIs, sPs, f0s, eXs, eYs, ePs, gflags = \
        ([] for _ in range(10))
    patches: dict = {}
    for ci, row in enumerate(rows):
        patch = get(row, "Patch", "patch0")
        stype = get(row, "Type", "POINT").upper()
        gaussian = stype == "GAUSSIAN"
        nm = ("G" if gaussian else "P") + patch + str(ci)
        names.append(nm)
        ras.append(_parse_hms(get(row, "Ra", "0")))
        decs.append(_parse_dms(get(row, "Dec", "0")))
        sIs.append(float(get(row, "I", "1") or 1))
        si = get(row, "SpectralIndex", "[]").strip("[]").split(";")[0]
        sPs.append([float(si) if si else 0.0, 0.0, 0.0])
        f0 = float(get(row, "ReferenceFrequency", "0") or 0)
        f0s.append(f0 if f0 > 0 else 100e6)
        # arcsec → rad; the reference halves axes into eX/eY
        asec = math.pi / (180 * 3600)
        eXs.append(0.5 * float(get(row, "MajorAxis", "0") or 0) * asec)
        eYs.append(0.5 * float(get(row, "MinorAxis", "0") or 0) * asec)
        ori = float(get(row, "Orientation", "0") or 0)
        ePs.append(math.pi / 2 - (math.pi - ori * math.pi / 180))
        gflags.append(gaussian)
        patches.setdefault(patch, []).append(nm)

    sky = SkyModel.from_arrays(names, ras, decs, sIs, np.asarray(sPs),
                               np.asarray(f0s), eXs, eYs, ePs, gflags)
    clusters = ClusterSet([ClusterDef(i + 1, 1, nms)
                           for i, (p, nms) in enumerate(patches.items())])
    return sky, clusters


def convert_model(bbs_text: str, start_cluster: int = 1,
                  num_patches: int = 0):
    """→ (sagecal_sky_text, cluster_text, rho_text), the three files
    `convertmodel.read_skymodel` writes."""
    sky, clusters = parse_bbs_text(bbs_text)
    if num_patches > 0:
        clusters = ClusterSet(clusters.clusters[:num_patches])
    cl = ClusterSet([ClusterDef(start_cluster + i, 1, c.names)
                     for i, c in enumerate(clusters)])
    K = len(cl)
    return (write_sky_text(sky), write_cluster_text(cl),
            write_rho_text(np.ones(K), np.ones(K)))
