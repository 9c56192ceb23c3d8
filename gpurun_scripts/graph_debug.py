"""Bisect the CNN learn-graph capture failure: which component breaks
capture — conv/BN, StreamFork, or the sampler? Prints a verdict per
variant with full tracebacks."""

import sys
import traceback
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.rl.sac_cnn import Agent


def mk(use_influence=True):
    torch.manual_seed(3)
    return Agent(gamma=0.99, batch_size=8, n_actions=4, tau=0.005,
                 max_mem_size=64, input_dims=(1, 32, 32), meta_dim=10,
                 lr_a=1e-3, lr_c=1e-3, device=torch.device("cuda"),
                 use_influence=use_influence)


def fill(ag):
    rng = np.random.default_rng(1)
    for i in range(16):
        obs = {"img": rng.standard_normal((1, 32, 32)).astype(np.float32),
               "metadata": rng.standard_normal(10).astype(np.float32)}
        ag.store_transition(obs, rng.standard_normal(4).astype(np.float32),
                            float(i % 3), obs, False)


def attempt(name, ag, pre=None):
    fill(ag)
    if pre:
        pre(ag)
    try:
        ag.enable_cuda_graph()
        for _ in range(3):
            ag.learn()
        torch.cuda.synchronize()
        ok = bool(torch.isfinite(ag.critic_1_fp.flat).all())
        print(f"[{name}] CAPTURE+REPLAY OK finite={ok}", flush=True)
    except Exception:
        print(f"[{name}] FAILED:", flush=True)
        traceback.print_exc()
        torch.cuda.synchronize()


def serial_fork(ag):
    ag._fork = lambda *fns: tuple(f() for f in fns)


def bn_eval(ag):
    for net in (ag.actor, ag.critic_1, ag.critic_2, ag.target_critic_1,
                ag.target_critic_2):
        for m in net.modules():
            if isinstance(m, torch.nn.BatchNorm2d):
                m.eval()


def main():
    attempt("meta-only (no conv/BN)", mk(use_influence=False))
    attempt("cnn serial-fork", mk(), pre=serial_fork)
    attempt("cnn bn-eval", mk(), pre=bn_eval)
    attempt("cnn full", mk())


if __name__ == "__main__":
    main()
