"""Soft target updates (N4).

With the flat-parameter pools of ``smartcal_amd.utils.flatten`` the polyak
update is a single fused axpby over one contiguous buffer per network —
the multi-tensor-apply problem the reference solves with a python
state_dict walk (``enet_sac.py:523-542``) disappears by construction.
"""

from __future__ import annotations

import torch


@torch.no_grad()
def polyak_(target_flat: torch.Tensor, online_flat: torch.Tensor,
            tau: float) -> None:
    """target <- tau*online + (1-tau)*target, in place (one kernel)."""
    target_flat.lerp_(online_flat, tau)


@torch.no_grad()
def polyak_modules_(target: torch.nn.Module, online: torch.nn.Module,
                    tau: float) -> None:
    """Per-tensor fallback for modules without flat pools."""
    for tp, op in zip(target.parameters(), online.parameters()):
        tp.lerp_(op, tau)
    for tb, ob in zip(target.buffers(), online.buffers()):
        if tb.dtype.is_floating_point:
            tb.lerp_(ob, tau)
        else:
            tb.copy_(ob)
