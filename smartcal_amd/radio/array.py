"""Synthetic interferometer array: station layout, uvw synthesis, az/el.

Replaces the reference's externally-created MeasurementSets (`makems`,
template MS + casacore in `calibration/generate_data.py:118-224` and the
casacore-measures az/el/separation math in
`calibration/influence_tools.py:16-216`). Everything is computed from a
parametric station layout and standard earth-rotation synthesis; arrays
never touch disk (MI355X-native: uvw for a full observation is a few MB
resident in HBM).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np

__all__ = ["StationLayout", "lofar_like_layout", "uvw_synthesis",
           "azel_of", "separation"]

# LOFAR core latitude (rad) — used for az/el synthesis
SITE_LAT = math.radians(52.915)
EARTH_OMEGA = 2.0 * math.pi / 86164.0905   # sidereal rate (rad/s)


@dataclass
class StationLayout:
    """Station positions in a local East-North-Up frame (meters)."""
    enu: np.ndarray          # (N, 3)
    latitude: float = SITE_LAT

    @property
    def n_stations(self) -> int:
        return self.enu.shape[0]

    def baselines_xyz(self) -> np.ndarray:
        """Per-baseline (p<q lexicographic) coordinate differences in the
        equatorial (X, Y, Z) frame used by uvw synthesis."""
        lat = self.latitude
        e, n, u = self.enu[:, 0], self.enu[:, 1], self.enu[:, 2]
        # ENU → equatorial XYZ at the site (standard geodetic rotation)
        x = -math.sin(lat) * n + math.cos(lat) * u
        y = e
        z = math.cos(lat) * n + math.sin(lat) * u
        xyz = np.stack([x, y, z], axis=1)                  # (N,3)
        N = xyz.shape[0]
        pi, qi = np.triu_indices(N, k=1)
        return xyz[pi] - xyz[qi]                            # (B,3)


def lofar_like_layout(N: int = 62, rng: np.random.Generator | None = None,
                      core_frac: float = 0.6, rmax: float = 40e3,
                      rcore: float = 2e3) -> StationLayout:
    """Dense core + log-spiral remote arms, loosely like LOFAR: N stations,
    ~core_frac in an rcore-radius core, the rest out to rmax."""
    rng = rng or np.random.default_rng(0)
    ncore = int(N * core_frac)
    nrem = N - ncore
    th = rng.uniform(0, 2 * math.pi, ncore)
    rr = rcore * np.sqrt(rng.uniform(0.001, 1.0, ncore))
    core = np.stack([rr * np.cos(th), rr * np.sin(th)], axis=1)
    # remote: 3 spiral arms, logarithmic radius growth
    arm = rng.integers(0, 3, nrem)
    tt = rng.uniform(0.15, 1.0, nrem)
    radius = rcore * (rmax / rcore) ** tt
    ang = arm * 2 * math.pi / 3 + 1.5 * np.log(radius / rcore) \
        + rng.normal(0, 0.08, nrem)
    rem = np.stack([radius * np.cos(ang), radius * np.sin(ang)], axis=1)
    en = np.concatenate([core, rem], axis=0)
    up = rng.normal(0, 2.0, (N, 1))
    return StationLayout(np.concatenate([en, up], axis=1))


def uvw_synthesis(layout: StationLayout, ra0: float, dec0: float,
                  times_s: np.ndarray, ha0: float = 0.0) -> np.ndarray:
    """uvw (Ntime, B, 3) in meters for phase center (ra0, dec0).

    Standard synthesis rotation: hour angle advances at the sidereal
    rate; the baseline XYZ vector maps to uvw via the (H0, dec0)
    projection matrix.
    """
    Lxyz = layout.baselines_xyz()                           # (B,3)
    H = ha0 + EARTH_OMEGA * np.asarray(times_s, np.float64) - ra0
    sH, cH = np.sin(H), np.cos(H)
    sd, cd = math.sin(dec0), math.cos(dec0)
    # rows of the rotation for each time
    zero = np.zeros_like(sH)
    ru = np.stack([sH, cH, zero], axis=1)                   # (Nt,3)
    rv = np.stack([-sd * cH, sd * sH, np.full_like(sH, cd)], axis=1)
    rw = np.stack([cd * cH, -cd * sH, np.full_like(sH, sd)], axis=1)
    u = ru @ Lxyz.T                                          # (Nt,B)
    v = rv @ Lxyz.T
    w = rw @ Lxyz.T
    return np.stack([u, v, w], axis=2)                       # (Nt,B,3)


def azel_of(ra, dec, lst: float, latitude: float = SITE_LAT):
    """Azimuth/elevation (rad) of (ra, dec) at local sidereal time lst.
    Replaces the casacore measures conversion of
    `influence_tools.py:16-74`."""
    ra = np.asarray(ra, np.float64)
    dec = np.asarray(dec, np.float64)
    H = lst - ra
    sin_el = np.sin(latitude) * np.sin(dec) \
        + np.cos(latitude) * np.cos(dec) * np.cos(H)
    el = np.arcsin(np.clip(sin_el, -1, 1))
    az = np.arctan2(-np.sin(H) * np.cos(dec),
                    np.sin(dec) * np.cos(latitude)
                    - np.cos(dec) * np.sin(latitude) * np.cos(H))
    return np.mod(az, 2 * math.pi), el


def separation(ra, dec, ra0: float, dec0: float):
    """Great-circle separation (rad) from (ra0, dec0)."""
    ra = np.asarray(ra, np.float64)
    dec = np.asarray(dec, np.float64)
    cs = np.sin(dec) * math.sin(dec0) \
        + np.cos(dec) * math.cos(dec0) * np.cos(ra - ra0)
    return np.arccos(np.clip(cs, -1, 1))
