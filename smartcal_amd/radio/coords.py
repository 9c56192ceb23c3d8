"""Celestial coordinate transforms.

Behavioral parity with the reference's scalar helpers
(`calibration/calibration_tools.py:6-86`), implemented vectorized over
numpy arrays so whole sky models transform in one call.
"""

from __future__ import annotations

import math

import numpy as np

__all__ = ["radectolm", "lmtoradec", "rad_to_ra", "rad_to_dec", "hms_to_rad",
           "dms_to_rad"]


def radectolm(ra, dec, ra0, dec0):
    """Direction cosines (l, m, n-1) of (ra, dec) w.r.t. phase center.

    Accepts scalars or numpy arrays for ra/dec. Matches
    `calibration_tools.py:6-16` including the dec0 wrap for negative
    phase-center declination.
    """
    ra = np.asarray(ra, dtype=np.float64)
    dec = np.asarray(dec, dtype=np.float64)
    if dec0 < 0.0:
        # reference applies the wrap per-source only when dec >= 0
        dec0 = np.where(dec >= 0.0, dec0 + 2.0 * math.pi, dec0)
    l = np.sin(ra - ra0) * np.cos(dec)
    m = -(np.cos(ra - ra0) * np.cos(dec) * np.sin(dec0)
          - np.cos(dec0) * np.sin(dec))
    n = np.sqrt(1.0 - l * l - m * m) - 1.0
    return l, m, n


def lmtoradec(l, m, ra0, dec0):
    """Inverse of :func:`radectolm` (small-field approximation),
    `calibration_tools.py:19-40`."""
    l = np.asarray(l, dtype=np.float64)
    m = np.asarray(m, dtype=np.float64)
    sind0 = math.sin(dec0)
    cosd0 = math.cos(dec0)
    d0 = m ** 2 * sind0 ** 2 + l ** 2 - 2 * m * cosd0 * sind0
    sind = np.sqrt(np.abs(sind0 ** 2 - d0))
    cosd = np.sqrt(np.abs(cosd0 ** 2 + d0))
    sind = np.abs(sind) if sind0 > 0 else -np.abs(sind)
    dec = np.arctan2(sind, cosd)
    ra = np.arctan2(np.where(l != 0, -l, 1e-10), cosd0 - m * sind0) + ra0
    return ra, dec


def rad_to_ra(rad: float):
    """Radians → (hr, min, sec); `calibration_tools.py:43-62`."""
    if rad < 0:
        rad = rad + 2 * math.pi
    tmp = rad * 12.0 / math.pi
    hr = math.floor(tmp)
    tmp = (tmp - hr) * 60
    mins = math.floor(tmp)
    sec = (tmp - mins) * 60
    return hr % 24, mins % 60, sec


def rad_to_dec(rad: float):
    """Radians → (deg, min, sec); `calibration_tools.py:64-86`."""
    mult = -1 if rad < 0 else 1
    rad = abs(rad)
    tmp = rad * 180.0 / math.pi
    deg = math.floor(tmp)
    tmp = (tmp - deg) * 60
    mins = math.floor(tmp)
    sec = (tmp - mins) * 60
    return mult * (deg % 180), mins % 60, sec


def hms_to_rad(h, m, s):
    """(hr, min, sec) → radians, as parsed by
    `calibration_tools.py:277-278`."""
    return (np.asarray(h, dtype=np.float64) + np.asarray(m) / 60.0
            + np.asarray(s) / 3600.0) * 360.0 / 24.0 * math.pi / 180.0


def dms_to_rad(d, m, s):
    """(deg, min, sec) → radians (no sign handling beyond d's own sign,
    matching the reference parser)."""
    return (np.asarray(d, dtype=np.float64) + np.asarray(m) / 60.0
            + np.asarray(s) / 3600.0) * math.pi / 180.0
