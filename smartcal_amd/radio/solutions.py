"""Jones-solution containers, SAGECal text formats, and synthetic
systematic-error generation.

Replaces `calibration_tools.py:88-211` (solution parsers) and the error
synthesis of `calibration/simulate.py:386-464`. The canonical form is an
in-memory tensor J (K, 2N·Nt, 2) complex64 — the same layout the parsers
return in the reference — plus per-station real coefficient storage for
the simulator (8 reals per station per direction: re/im of the 2×2 Jones
entries).
"""

from __future__ import annotations

import io

import numpy as np

__all__ = ["solutions_to_J", "J_to_solutions", "parse_solutions_text",
           "format_solutions_text", "parse_global_solutions_text",
           "simulate_systematic_errors"]


def solutions_to_J(a: np.ndarray, N: int, Nto: int) -> np.ndarray:
    """(8N·Nto, K) real coefficient table → J (K, 2N·Nto, 2) complex64.

    Row layout per timeslot per station: 8 reals =
    [re J00, im J00, re J01, im J01, re J10, im J10, re J11, im J11],
    matching `calibration_tools.py:105-119`.
    """
    K = a.shape[1]
    J = np.zeros((K, 2 * N * Nto, 2), dtype=np.complex64)
    b = a.reshape(Nto, N, 8, K)
    c = b[..., 0::2, :] + 1j * b[..., 1::2, :]   # (Nto,N,4,K)
    c = np.moveaxis(c, -1, 0)                    # (K,Nto,N,4)
    J2 = c.reshape(K, Nto, N, 2, 2)              # row-major 2×2 per station
    J[:, 0::2, :] = J2[:, :, :, 0, :].reshape(K, -1, 2)
    J[:, 1::2, :] = J2[:, :, :, 1, :].reshape(K, -1, 2)
    return J


def J_to_solutions(J: np.ndarray, N: int) -> np.ndarray:
    """Inverse of :func:`solutions_to_J` → (8N·Nto, K) float32."""
    K = J.shape[0]
    Nto = J.shape[1] // (2 * N)
    J2 = np.zeros((K, Nto, N, 2, 2), dtype=np.complex64)
    J2[:, :, :, 0, :] = J[:, 0::2, :].reshape(K, Nto, N, 2)
    J2[:, :, :, 1, :] = J[:, 1::2, :].reshape(K, Nto, N, 2)
    c = np.moveaxis(J2.reshape(K, Nto, N, 4), 0, -1)   # (Nto,N,4,K)
    a = np.zeros((Nto, N, 8, K), dtype=np.float32)
    a[..., 0::2, :] = c.real
    a[..., 1::2, :] = c.imag
    return a.reshape(8 * N * Nto, K)


def parse_solutions_text(text: str):
    """SAGECal per-frequency solutions file → (freq, J (K,2N·Nt,2)).
    Parity with `readsolutions` (`calibration_tools.py:88-119`)."""
    fh = io.StringIO(text)
    next(fh); next(fh)
    cl = next(fh).split()
    freq = float(cl[0]) * 1e6
    Ns = int(cl[3])
    K = int(cl[5])
    rows = [ln.split() for ln in fh if ln.strip()]
    Nt = len(rows)
    Nto = Nt // (8 * Ns)
    a = np.zeros((Nt, K), dtype=np.float32)
    for ci, cl1 in enumerate(rows):
        for cj in range(len(cl1) - 1):
            a[ci, cj] = float(cl1[cj + 1])
    return freq, solutions_to_J(a[:Nto * 8 * Ns], Ns, Nto)


def format_solutions_text(freq_hz: float, N: int, a: np.ndarray,
                          bw_mhz: float = 0.183105,
                          tint_min: float = 20.027802) -> str:
    """(8N·Nto, K) coefficient table → SAGECal solutions text
    (`simulate.py:439-464`): appends the trailing unit column."""
    K = a.shape[1]
    Nto = a.shape[0] // (8 * N)
    out = ["#solution file created by smartcal_amd",
           "#freq(MHz) bandwidth(MHz) time_interval(min) stations clusters "
           "effective_clusters",
           f"{freq_hz / 1e6} {bw_mhz} {tint_min} {N} {K + 1} {K + 1}"]
    for ct in range(Nto):
        for ci in range(8 * N):
            off = ci % 8
            vals = " ".join(str(a[ct * 8 * N + ci, ck]) for ck in range(K))
            unit = "1" if off in (0, 6) else "0"
            out.append(f"{ci} {vals} {unit}")
    return "\n".join(out) + "\n"


def parse_global_solutions_text(text: str):
    """Global-Z solutions file → (N, f0, P, K, Z (Nto,K,2PN,2));
    parity with `read_global_solutions` (`calibration_tools.py:122-160`)."""
    fh = io.StringIO(text)
    next(fh); next(fh)
    cl = next(fh).split()
    freq = float(cl[0]) * 1e6
    P = int(cl[1])
    Ns = int(cl[2])
    K = int(cl[4])
    rows = [ln.split() for ln in fh if ln.strip()]
    Nt = len(rows)
    Nto = Nt // (8 * P * Ns)
    a = np.zeros((Nt, K), dtype=np.float32)
    for ci, cl1 in enumerate(rows):
        for cj in range(len(cl1) - 1):
            a[ci, cj] = float(cl1[cj + 1])
    Z = np.zeros((Nto, K, 2 * P * Ns, 2), dtype=np.complex64)
    for ci in range(Nto):
        for cj in range(K):
            b = a[ci * 8 * P * Ns:(ci + 1) * 8 * P * Ns, cj]
            c = b[0::2] + 1j * b[1::2]
            Z[ci, cj] = c.reshape((2 * P * Ns, 2), order="F")
    return Ns, freq, P, K, Z


def simulate_systematic_errors(K: int, N: int, Ts: int, freqs: np.ndarray,
                               f0: float, rng: np.random.Generator,
                               spatial_term: bool = True,
                               spalpha: float = 0.95,
                               lm: np.ndarray | None = None) -> np.ndarray:
    """Synthesize systematic-error Jones coefficients gs (K, 8N·Ts, Nf).

    Same statistical recipe as `simulate.py:386-435`: per direction a
    random (optionally spatially-smooth plane a0·l+a1·m+a2) 8N seed with
    +1 on the diagonal real parts, modulated by a random quadratic
    polynomial over normalized frequency and a random cosine over time.

    lm: (K, 2) per-direction (l, m) used when spatial_term is set.
    """
    Nf = len(freqs)
    gs = np.zeros((K, 8 * N * Ts, Nf), dtype=np.float32)
    ff = (np.asarray(freqs, np.float64) - f0) / f0

    if spatial_term:
        if lm is None:
            lm = rng.standard_normal((K, 2)) * 0.1
        a0 = rng.standard_normal(8 * N)
        a1 = rng.standard_normal(8 * N)
        a2 = rng.standard_normal(8 * N)
        a0 /= np.linalg.norm(a0)
        a1 /= np.linalg.norm(a1)
        a2 /= np.linalg.norm(a2)

    for ck in range(K):
        if not spatial_term:
            seed = rng.standard_normal(8 * N)
        else:
            randpart = rng.standard_normal(8 * N)
            seed = (1 - spalpha) * randpart / np.linalg.norm(randpart) \
                + spalpha * (a0 * lm[ck, 0] + a1 * lm[ck, 1] + a2)
            seed /= np.linalg.norm(seed)
        seed = seed.copy()
        seed[0::8] += 1.0
        seed[6::8] += 1.0
        # quadratic frequency polynomial per coefficient
        beta = rng.standard_normal((8 * N, 3))
        freqpol = seed[:, None] * (beta[:, 0:1] + beta[:, 1:2] * ff[None, :]
                                   + beta[:, 2:3] * ff[None, :] ** 2)
        gs[ck, 0:8 * N, :] = freqpol
        for ct in range(1, Ts):
            gs[ck, ct * 8 * N:(ct + 1) * 8 * N] = gs[ck, 0:8 * N]

    # cosine time modulation per coefficient
    timerange = np.arange(Ts) / Ts
    for ck in range(K):
        b = rng.standard_normal((8 * N, 4))
        b /= np.linalg.norm(b, axis=1, keepdims=True)
        timepol = 1 + b[:, 0:1] + b[:, 1:2] \
            * np.cos(timerange[None, :] * b[:, 2:3] + b[:, 3:4])  # (8N,Ts)
        mod = timepol.T.reshape(Ts * 8 * N)   # mod[ct·8N+cn] = timepol[cn,ct]
        gs[ck] *= mod[:, None].astype(np.float32)
    return gs
