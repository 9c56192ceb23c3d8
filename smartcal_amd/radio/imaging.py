"""uv gridding + FFT imaging.

Replaces the external ``excon`` imager (`calibration/doinfluence.sh:8`,
`generate_data.py:461`) for the paths the framework
needs: Stokes-I dirty images of visibility-sampled values (data maps,
residual maps, 128² influence maps) and the weighted mean over sub-bands
(`calmean.sh`). Gridding is a scatter-add into the uv plane (index_put_
accumulate → hipBLAS-free, bandwidth-bound) followed by a centered
inverse FFT; conjugate symmetry is enforced so images are real.
"""

from __future__ import annotations


import torch

C_LIGHT = 2.99792458e8

__all__ = ["dirty_image", "image_std", "weighted_mean_image"]


def dirty_image(uvw: torch.Tensor, values: torch.Tensor, freq: float,
                npix: int = 128, fov: float | None = None) -> torch.Tensor:
    """Dirty image (npix, npix) float32 of complex per-sample ``values``.

    uvw: (S, 3) meters; values: (S,) complex (e.g. Stokes I);
    fov: field of view in direction-cosine units (image spans ±fov/2);
    uv cell follows from du = 1/fov. With fov=None the cell auto-scales
    so the longest sampled baseline lands on the grid edge (what excon's
    default pixel scaling effectively does for our synthetic layouts).
    """
    dev = uvw.device
    lam = C_LIGHT / freq
    u = uvw[:, 0] / lam
    v = uvw[:, 1] / lam
    if fov is None:
        umax = float(torch.maximum(u.abs().max(), v.abs().max()).item())
        du = max(umax, 1.0) / (npix // 2 - 1)
    else:
        du = 1.0 / fov
    iu = torch.round(u / du).long() + npix // 2
    iv = torch.round(v / du).long() + npix // 2
    ok = (iu >= 0) & (iu < npix) & (iv >= 0) & (iv < npix)
    grid = torch.zeros((npix, npix), dtype=torch.complex64, device=dev)
    wsum = torch.zeros((), dtype=torch.float32, device=dev)
    vals = values[ok]
    grid.index_put_((iv[ok], iu[ok]), vals, accumulate=True)
    # conjugate symmetry: V(-u,-v) = V*(u,v)
    iu2 = npix - iu[ok]
    iv2 = npix - iv[ok]
    ok2 = (iu2 >= 0) & (iu2 < npix) & (iv2 >= 0) & (iv2 < npix)
    grid.index_put_((iv2[ok2], iu2[ok2]), vals[ok2].conj(), accumulate=True)
    nvis = float(ok.sum()) + float(ok2.sum())
    img = torch.fft.fftshift(torch.fft.ifft2(torch.fft.ifftshift(grid)))
    return (img.real * (npix * npix / max(nvis, 1.0))).to(torch.float32)


def image_std(uvw: torch.Tensor, vis4: torch.Tensor, freq: float,
              npix: int = 256, fov: float = 2.0) -> float:
    """std of the Stokes-I dirty image of (S,4) visibilities — the
    reference's image-noise estimate (`demixingenv.py:221-231`,
    `calibenv.py:148-158`)."""
    sI = 0.5 * (vis4[:, 0] + vis4[:, 3])
    return float(dirty_image(uvw, sI, freq, npix, fov).std().item())


def weighted_mean_image(images: list[torch.Tensor],
                        freqs) -> torch.Tensor:
    """Frequency-weighted mean image à la `calmean.sh` (weights ∝ f²
    normalized)."""
    w = torch.as_tensor([float(f) ** 2 for f in freqs],
                        device=images[0].device)
    w = w / w.sum()
    out = torch.zeros_like(images[0])
    for wi, im in zip(w, images):
        out += wi * im
    return out
