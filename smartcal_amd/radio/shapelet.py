"""Random shapelet-mode models (diffuse-sky components).

Parity with `calibration_tools.generate_random_shapelet_model`
(`calibration_tools.py:1254-1296`): the mode-file text format (position
line, n0/beta line, n0² coefficients with 1/|order|^1.2 attenuation,
linear-transform trailer), optional 10% perturbed twin for calibration,
plus a parser and a basis evaluator so diffuse components can be
rendered into images without external tools.
"""

from __future__ import annotations

import math

import numpy as np

__all__ = ["generate_random_shapelet_model", "parse_shapelet_model",
           "shapelet_basis"]


def generate_random_shapelet_model(rng: np.random.Generator,
                                   ra_hms=(1, 2, 3.0), dec_dms=(4, 5, 6.0),
                                   perturbed: bool = False):
    """→ (text, n0, beta, coeff[, perturbed_text])."""
    n0 = int(rng.integers(10, 20))
    beta = float(rng.random()) + 0.1
    if beta * n0 > 2:
        beta = (2 + float(rng.random()) * 0.001) / n0
    coeff = rng.standard_normal((n0, n0))
    x = np.arange(1, n0 + 1)
    coeff = (coeff / (np.abs(np.outer(x, x)) ** 1.2)).reshape(-1)

    def fmt(b, c):
        lines = [" ".join(str(v) for v in (*ra_hms, *dec_dms)),
                 f"{n0} {b}"]
        lines += [f"{i} {c[i]}" for i in range(n0 * n0)]
        lines.append(f"L 1.0 1.0 {math.pi / 2}")
        lines.append("#model created by smartcal_amd")
        return "\n".join(lines) + "\n"

    text = fmt(beta, coeff)
    if not perturbed:
        return text, n0, beta, coeff
    beta_p = beta + 0.1 * beta * float(rng.random())
    noise = rng.standard_normal((n0, n0)).reshape(-1)
    noise = noise / np.linalg.norm(noise) * 0.1 * np.linalg.norm(coeff)
    return text, n0, beta, coeff, fmt(beta_p, coeff + noise)


def parse_shapelet_model(text: str):
    """→ (position_tuple, n0, beta, coeff (n0²,))."""
    lines = [l for l in text.splitlines() if l and not l.startswith("#")]
    pos = tuple(float(v) for v in lines[0].split())
    n0_s, beta_s = lines[1].split()
    n0, beta = int(n0_s), float(beta_s)
    coeff = np.zeros(n0 * n0)
    for ln in lines[2:2 + n0 * n0]:
        i, v = ln.split()
        coeff[int(i)] = float(v)
    return pos, n0, beta, coeff


def _hermite(n: int, x: np.ndarray) -> np.ndarray:
    h0 = np.ones_like(x)
    if n == 0:
        return h0
    h1 = 2 * x
    for k in range(1, n):
        h0, h1 = h1, 2 * x * h1 - 2 * k * h0
    return h1


def shapelet_basis(n0: int, beta: float, l: np.ndarray,
                   m: np.ndarray) -> np.ndarray:
    """Gauss-Hermite shapelet basis evaluated at direction cosines
    (l, m): (n0², P) for P sample points."""
    xl = l / beta
    xm = m / beta
    gl = np.exp(-0.5 * xl ** 2)
    gm = np.exp(-0.5 * xm ** 2)
    out = np.zeros((n0 * n0, l.size))
    for a in range(n0):
        na = 1.0 / math.sqrt((2 ** a) * math.factorial(a)
                             * math.sqrt(math.pi) * beta)
        ha = _hermite(a, xl) * gl * na
        for b in range(n0):
            nb = 1.0 / math.sqrt((2 ** b) * math.factorial(b)
                                 * math.sqrt(math.pi) * beta)
            out[a * n0 + b] = ha * _hermite(b, xm) * gm * nb
    return out


def correct_shapelet_modes(text: str) -> str:
    """Rescale shapelet coefficients by i!/(i+1)! per row block — parity
    with `calibration/correct_shapelet_modes.py:4-30` (mode-file format
    conversion between SAGECal versions)."""
    lines = text.splitlines()
    out = [lines[0], lines[1]]
    n0 = int(lines[1].split()[0])
    idx = 2
    for ci in range(n0):
        scale = 1.0 / (ci + 1)          # i!/(i+1)!
        for cj in range(n0):
            num, val = lines[idx].split()
            out.append(f"{num} {float(val) * scale}")
            idx += 1
    out.extend(lines[idx:])
    return "\n".join(out) + "\n"
