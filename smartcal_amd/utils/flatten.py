"""Flat parameter/gradient pools.

MI355X-first runtime choice: every network's parameters live as views into
ONE contiguous fp32 buffer, and gradients accumulate into one contiguous
grad buffer. Consequences:

* the soft target update (polyak) is ONE axpby over the flat pair instead of
  a per-tensor state_dict walk (reference ``enet_sac.py:523-542`` copies the
  whole state_dict every learn step),
* the fused Adam step is ONE kernel over (flat, flat_grad, m, v),
* data-parallel gradient exchange is ONE RCCL all-reduce of the flat grad
  buffer over xGMI (RL gradients are KB-MB sized and latency-bound — fusing
  into a single collective is the right shape for 7×153 GB/s point-to-point
  links; SURVEY.md §7 hard part (v)).
"""

from __future__ import annotations

from typing import List

import torch


class FlatParams:
    """Re-materialize a module's parameters into one flat buffer."""

    def __init__(self, module: torch.nn.Module):
        params: List[torch.nn.Parameter] = [p for p in module.parameters()]
        assert params, "module has no parameters"
        device = params[0].device
        dtype = params[0].dtype
        n = sum(p.numel() for p in params)
        self.flat = torch.empty(n, device=device, dtype=dtype)
        self.flat_grad = torch.zeros(n, device=device, dtype=dtype)
        self.params = params
        self.numel = n
        offset = 0
        for p in params:
            k = p.numel()
            self.flat[offset:offset + k].copy_(p.detach().reshape(-1))
            p.data = self.flat[offset:offset + k].view_as(p)
            # pre-install grad views so autograd accumulates in place
            p.grad = self.flat_grad[offset:offset + k].view_as(p)
            offset += k

    def zero_grad(self):
        self.flat_grad.zero_()

    def rebind_grads(self):
        """Re-install grad views (autograd can replace .grad on some paths)."""
        offset = 0
        for p in self.params:
            k = p.numel()
            g = self.flat_grad[offset:offset + k].view_as(p)
            if p.grad is None or p.grad.data_ptr() != g.data_ptr():
                if p.grad is not None:
                    g.copy_(p.grad.detach().reshape(-1).view_as(p))
                p.grad = g
            offset += k

    @torch.no_grad()
    def polyak_from(self, online: "FlatParams", tau: float):
        """target <- tau * online + (1 - tau) * target — one fused op."""
        self.flat.lerp_(online.flat, tau)

    @torch.no_grad()
    def copy_from(self, online: "FlatParams"):
        self.flat.copy_(online.flat)


class FusedAdam:
    """Adam over a FlatParams pool — one fused HIP kernel per step on GPU.

    Standard Adam (bias-corrected), matching torch.optim.Adam defaults so
    checkpoints and learning curves line up with the reference agents.
    """

    def __init__(self, fp: FlatParams, lr: float, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        self.fp = fp
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.m = torch.zeros_like(fp.flat)
        self.v = torch.zeros_like(fp.flat)
        # device-resident step counter so the whole Adam step (incl. bias
        # correction) is hipGraph-capturable
        self.t_dev = torch.zeros(1, device=fp.flat.device,
                                 dtype=torch.float32)

    def zero_grad(self):
        self.fp.zero_grad()

    @torch.no_grad()
    def step(self):
        from ..ops import use_hip, ext
        self.step_count += 1
        t = self.step_count
        g = self.fp.flat_grad
        if self.weight_decay:
            g = g.add(self.fp.flat, alpha=self.weight_decay)
        if use_hip(self.fp.flat):
            ext().fused_adam(self.fp.flat, g, self.m, self.v, self.t_dev,
                             self.lr, self.beta1, self.beta2, self.eps)
        else:
            self.m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            self.v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            mhat = self.m / (1 - self.beta1 ** t)
            vhat = self.v / (1 - self.beta2 ** t)
            self.fp.flat.addcdiv_(mhat, vhat.sqrt().add_(self.eps),
                                  value=-self.lr)

    def state_dict(self):
        return {"step": self.step_count, "m": self.m, "v": self.v,
                "lr": self.lr, "t_dev": self.t_dev}

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.lr = sd.get("lr", self.lr)
        if "t_dev" in sd:
            self.t_dev.copy_(sd["t_dev"])
        else:
            self.t_dev.fill_(float(self.step_count))
