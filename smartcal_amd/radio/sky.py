"""Sky / cluster model containers and text-format I/O.

The reference drives everything through SAGECal-format text files
(`sky.txt`, `cluster.txt`, `admm_rho.txt`) parsed ad hoc in
`calibration/calibration_tools.py:243-268,470-502,1228-1249`. Here the
canonical representation is in-memory structured arrays (ready to upload
to HBM once and keep resident); the text formats are supported for
compatibility.

Sky line format (after the source name; see
`calibration/simulate.py:132` and the index use in
`calibration_tools.py:275-285`)::

    name ra_h ra_m ra_s dec_d dec_m dec_s sI sQ sU sV sp1 sp2 sp3 RM eX eY eP f0

Sources whose name starts with ``G`` are Gaussians (eX/eY/eP used,
`calibration_tools.py:383-385`).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import numpy as np

from .coords import hms_to_rad, dms_to_rad, radectolm

__all__ = ["SkyModel", "ClusterDef", "ClusterSet", "parse_sky_text",
           "parse_cluster_text", "parse_rho_text", "write_sky_text",
           "write_cluster_text", "write_rho_text", "read_skycluster"]


@dataclass
class SkyModel:
    """Columnar sky model: parallel numpy arrays over all sources."""

    names: list = field(default_factory=list)
    ra: np.ndarray = field(default_factory=lambda: np.zeros(0))       # rad
    dec: np.ndarray = field(default_factory=lambda: np.zeros(0))      # rad
    sI: np.ndarray = field(default_factory=lambda: np.zeros(0))       # Jy
    sP: np.ndarray = field(default_factory=lambda: np.zeros((0, 3)))  # sp1..3
    f0: np.ndarray = field(default_factory=lambda: np.zeros(0))       # Hz
    # Gaussian shape params (zeros for point sources)
    eX: np.ndarray = field(default_factory=lambda: np.zeros(0))
    eY: np.ndarray = field(default_factory=lambda: np.zeros(0))
    eP: np.ndarray = field(default_factory=lambda: np.zeros(0))
    gaussian: np.ndarray = field(default_factory=lambda: np.zeros(0, bool))

    def __len__(self):
        return len(self.names)

    @property
    def index(self):
        return {n: i for i, n in enumerate(self.names)}

    @staticmethod
    def from_arrays(names, ra, dec, sI, sP=None, f0=1e6, eX=None, eY=None,
                    eP=None, gaussian=None):
        n = len(names)
        sm = SkyModel()
        sm.names = list(names)
        sm.ra = np.asarray(ra, np.float64)
        sm.dec = np.asarray(dec, np.float64)
        sm.sI = np.asarray(sI, np.float64)
        sP = np.zeros((n, 3)) if sP is None else np.asarray(sP, np.float64)
        if sP.ndim == 1:
            sP = np.stack([sP, np.zeros(n), np.zeros(n)], axis=1)
        sm.sP = sP
        sm.f0 = np.full(n, f0, np.float64) if np.isscalar(f0) \
            else np.asarray(f0, np.float64)
        sm.eX = np.zeros(n) if eX is None else np.asarray(eX, np.float64)
        sm.eY = np.zeros(n) if eY is None else np.asarray(eY, np.float64)
        sm.eP = np.zeros(n) if eP is None else np.asarray(eP, np.float64)
        if gaussian is None:
            gaussian = [nm.startswith("G") for nm in sm.names]
        sm.gaussian = np.asarray(gaussian, bool)
        return sm

    def flux_at(self, freq: float) -> np.ndarray:
        """Extrapolate intensity to ``freq`` with the reference's
        log-polynomial spectral model (`calibration_tools.py:283-285`)."""
        fr = np.log(freq / self.f0)
        return np.exp(np.log(self.sI) + self.sP[:, 0] * fr
                      + self.sP[:, 1] * fr ** 2 + self.sP[:, 2] * fr ** 3)

    def lmn(self, ra0: float, dec0: float):
        return radectolm(self.ra, self.dec, ra0, dec0)


@dataclass
class ClusterDef:
    cid: int          # cluster id as written in the file
    hybrid: int       # hybrid solve count (passed through, unused here)
    names: list       # source names belonging to this cluster


@dataclass
class ClusterSet:
    clusters: list    # list[ClusterDef], file order == direction order

    def __len__(self):
        return len(self.clusters)

    def __iter__(self):
        return iter(self.clusters)

    def __getitem__(self, i):
        return self.clusters[i]

    def subset(self, keep) -> "ClusterSet":
        """New ClusterSet with only the flagged clusters (demixing
        direction selection, `demixing_rl/demixingenv.py:254-279`)."""
        return ClusterSet([c for i, c in enumerate(self.clusters)
                           if keep[i]])


# --------------------------------------------------------------------------
# text-format I/O (SAGECal formats)
# --------------------------------------------------------------------------

def _data_lines(text: str):
    for line in text.splitlines():
        if line.startswith("#") or len(line) <= 1:
            continue
        yield line.split()


def parse_sky_text(text: str) -> SkyModel:
    names, ra, dec, sI, sP, f0, eX, eY, eP = ([] for _ in range(9))
    for cl in _data_lines(text):
        names.append(cl[0])
        s = [float(x) for x in cl[1:]]
        ra.append(hms_to_rad(s[0], s[1], s[2]))
        dec.append(dms_to_rad(s[3], s[4], s[5]))
        sI.append(s[6])
        sP.append(s[10:13])
        eX.append(s[14])
        eY.append(s[15])
        eP.append(s[16])
        f0.append(s[17])
    return SkyModel.from_arrays(names, ra, dec, sI, np.asarray(sP),
                                np.asarray(f0), eX, eY, eP)


def parse_cluster_text(text: str) -> ClusterSet:
    out = []
    for cl in _data_lines(text):
        out.append(ClusterDef(int(cl[0]), int(cl[1]), cl[2:]))
    return ClusterSet(out)


def parse_rho_text(text: str, K: int):
    """→ (rho_spectral[K], rho_spatial[K]); `calibration_tools.py:470-485`."""
    rs = np.zeros(K, np.float32)
    ra_ = np.zeros(K, np.float32)
    ci = 0
    for cl in _data_lines(text):
        rs[ci] = float(cl[2])
        ra_[ci] = float(cl[3])
        ci += 1
    return rs, ra_


def write_sky_text(sky: SkyModel) -> str:
    from .coords import rad_to_ra, rad_to_dec
    lines = ["## name h m s d m s sI sQ sU sV sp1 sp2 sp3 RM eX eY eP f0"]
    for i, nm in enumerate(sky.names):
        h, m, s = rad_to_ra(float(sky.ra[i]))
        d, dm, ds = rad_to_dec(float(sky.dec[i]))
        sp = sky.sP[i]
        lines.append(
            f"{nm} {h} {m} {s:.6f} {d} {dm} {ds:.6f} {sky.sI[i]:.6g} 0 0 0 "
            f"{sp[0]:.6g} {sp[1]:.6g} {sp[2]:.6g} 0 {sky.eX[i]:.6g} "
            f"{sky.eY[i]:.6g} {sky.eP[i]:.6g} {sky.f0[i]:.6g}")
    return "\n".join(lines) + "\n"


def write_cluster_text(cs: ClusterSet) -> str:
    lines = ["# cluster_id hybrid source_names"]
    for c in cs:
        lines.append(f"{c.cid} {c.hybrid} " + " ".join(c.names))
    return "\n".join(lines) + "\n"


def write_rho_text(rho_spectral, rho_spatial) -> str:
    """Format of `admm_rho.txt`: id hybrid rho_spectral rho_spatial
    (`calibenv.py:105-114`)."""
    lines = ["# id hybrid rho_spectral rho_spatial"]
    for i, (rs, ra_) in enumerate(zip(rho_spectral, rho_spatial)):
        lines.append(f"{i + 1} 1 {float(rs):.6g} {float(ra_):.6g}")
    return "\n".join(lines) + "\n"


def read_skycluster(text: str, M: int) -> np.ndarray:
    """Parse the `skylmn.txt` DQN metadata file: rows of
    ``cluster_id l m sI sP`` (`calibration_tools.py:488-502`)."""
    skl = np.zeros((M, 5), np.float32)
    ci = 0
    for cl in _data_lines(text):
        skl[ci] = [float(x) for x in cl[:5]]
        ci += 1
    return skl
