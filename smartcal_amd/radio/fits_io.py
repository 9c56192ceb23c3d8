"""Minimal pure-python FITS 2-D image I/O (no astropy dependency).

The reference exchanges influence/data/residual maps as FITS images
(`calibenv.py:148-158` reads three maps per step with astropy,
`calmean.sh`'s generated script writes the weighted mean image). This
image has no astropy, so the framework carries its own reader/writer for
the simple case it needs: single primary HDU, 2-D (or trailing-degenerate
4-D, as radio imagers write) float32/float64 arrays, big-endian per the
FITS standard, 2880-byte header/data blocks.

Covers: reading real influence/residual maps exported from a LOFAR
pipeline, writing maps other tools (ds9, astropy) can open, and the
weighted-mean-image combination step.
"""

from __future__ import annotations

import numpy as np

__all__ = ["read_image", "write_image", "weighted_mean_fits"]

_BLOCK = 2880


def _card(key: str, value, comment: str = "") -> bytes:
    if isinstance(value, bool):
        v = "T" if value else "F"
        s = f"{key:<8}= {v:>20}"
    elif isinstance(value, (int, np.integer)):
        s = f"{key:<8}= {value:>20d}"
    elif isinstance(value, float):
        s = f"{key:<8}= {value:>20.12G}"
    else:
        s = f"{key:<8}= '{value:<8}'"
    if comment:
        s += f" / {comment}"
    return s[:80].ljust(80).encode("ascii")


def write_image(path: str, img: np.ndarray, extra_cards=()) -> None:
    """img (H, W) float → minimal single-HDU FITS file."""
    a = np.asarray(img)
    if a.ndim != 2:
        raise ValueError("write_image expects a 2-D array")
    a32 = a.astype(">f4")
    hdr = [
        _card("SIMPLE", True, "minimal FITS (smartcal_amd)"),
        _card("BITPIX", -32),
        _card("NAXIS", 2),
        _card("NAXIS1", a.shape[1]),
        _card("NAXIS2", a.shape[0]),
    ]
    for k, v in extra_cards:
        hdr.append(_card(k, v))
    hdr.append(b"END".ljust(80))
    h = b"".join(hdr)
    h += b" " * (-len(h) % _BLOCK)
    d = a32.tobytes()
    d += b"\x00" * (-len(d) % _BLOCK)
    with open(path, "wb") as f:
        f.write(h + d)


def read_image(path: str) -> np.ndarray:
    """FITS primary HDU → 2-D float32 array. Accepts BITPIX -32/-64 and
    trailing degenerate axes (NAXIS3/4 == 1, the radio-imager layout)."""
    with open(path, "rb") as f:
        raw = f.read()
    # parse header cards until END
    cards = {}
    pos = 0
    while True:
        block = raw[pos:pos + _BLOCK]
        if len(block) < _BLOCK:
            raise ValueError("truncated FITS header")
        done = False
        for i in range(0, _BLOCK, 80):
            card = block[i:i + 80].decode("ascii", "replace")
            key = card[:8].strip()
            if key == "END":
                done = True
                break
            if "=" in card[8:10]:
                val = card[10:].split("/")[0].strip()
                cards[key] = val
        pos += _BLOCK
        if done:
            break
    bitpix = int(cards["BITPIX"])
    naxis = int(cards["NAXIS"])
    shape = [int(cards[f"NAXIS{i}"]) for i in range(1, naxis + 1)]
    for extra in shape[2:]:
        if extra != 1:
            raise ValueError(f"non-degenerate axis {extra}: only 2-D "
                             "images supported")
    if bitpix == -32:
        dt, isz = ">f4", 4
    elif bitpix == -64:
        dt, isz = ">f8", 8
    else:
        raise ValueError(f"BITPIX {bitpix} unsupported (float images only)")
    n = int(np.prod(shape))
    a = np.frombuffer(raw[pos:pos + n * isz], dtype=dt).astype(np.float32)
    # FITS is Fortran-ordered over (NAXIS1, NAXIS2) = (W, H)
    return a.reshape(shape[1::-1][0], shape[0])


def weighted_mean_fits(paths, weights=None) -> np.ndarray:
    """Weighted mean of FITS images — the `calmean.sh` generated-script
    behavior (weights default 1/variance of each map)."""
    imgs = [read_image(p) for p in paths]
    if weights is None:
        weights = [1.0 / max(float(np.var(im)), 1e-30) for im in imgs]
    w = np.asarray(weights, np.float64)
    stack = np.stack(imgs).astype(np.float64)
    return (np.tensordot(w, stack, axes=1) / w.sum()).astype(np.float32)
