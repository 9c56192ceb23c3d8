"""Batched small-complex matrix products as elementwise arithmetic.

rocBLAS/hipBLASLt dispatch batched complex GEMMs with m=n=k=2 (Jones
chains) or k=2 contractions (normal equations) to 128×64 macro-tile
kernels that run ~800 µs per call at LOFAR scale — tiny-GEMM shapes are
pathological for the library. These helpers compute the same products
as broadcasted fused elementwise complex arithmetic (bandwidth-bound,
microseconds), which is the right CDNA4 mapping for 2×2-block math.
All take/return (..., 2, 2) or (..., m, 2) complex tensors.
"""

from __future__ import annotations

import torch

__all__ = ["mm2", "mm2H", "Hmm2", "abH_k2", "outer_k2"]


def mm2(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A @ B for batched 2×2."""
    a00, a01 = A[..., 0, 0], A[..., 0, 1]
    a10, a11 = A[..., 1, 0], A[..., 1, 1]
    b00, b01 = B[..., 0, 0], B[..., 0, 1]
    b10, b11 = B[..., 1, 0], B[..., 1, 1]
    return torch.stack((
        torch.stack((a00 * b00 + a01 * b10, a00 * b01 + a01 * b11), -1),
        torch.stack((a10 * b00 + a11 * b10, a10 * b01 + a11 * b11), -1),
    ), -2)


def mm2H(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A @ B^H for batched 2×2."""
    return mm2(A, B.conj().transpose(-1, -2))


def Hmm2(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A^H @ B for batched 2×2."""
    return mm2(A.conj().transpose(-1, -2), B)


def abH_k2(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A @ B^H where the contraction dim is the trailing axis of size 2:
    A (..., m, 2), B (..., n, 2) → (..., m, n)."""
    Bc = B.conj()
    return (A[..., :, 0:1] * Bc[..., None, :, 0]
            + A[..., :, 1:2] * Bc[..., None, :, 1])


def outer_k2(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A @ B where A is (..., m, 2) and B is (..., 2, n) → (..., m, n)."""
    return (A[..., :, 0:1] * B[..., None, 0, :]
            + A[..., :, 1:2] * B[..., None, 1, :])
