"""Isolate the source of exploding min(EE)/max(EE) rewards on GPU.

Runs SAC episodes; for steps with reward < -100 captures (A, y, rho) and
cross-evaluates:
  A. GPU solver + GPU influence   (the env path — what exploded)
  B. GPU solver pairs -> CPU oracle influence
  C. CPU reference solver + CPU oracle influence (the reference behavior)
Prints a table; writes gpurun_out/explosion_debug.json.
"""

import json
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.ops import enet as enet_ops
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything
import smartcal_amd.ops as ops

N = M = 20
EPISODES = int(sys.argv[1]) if len(sys.argv) > 1 else 120


def main():
    seed_everything(3)
    dev = torch.device("cuda")
    env = ENetEnv(M, N, device=dev)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03, device=dev)
    bad = []
    for ep in range(EPISODES):
        obs = env.reset()
        for _ in range(5):
            a = agent.choose_action(obs)
            obs2, r, done, info = env.step(a)
            if float(r) < -100 and len(bad) < 12:
                bad.append(dict(A=env.A.detach().cpu().clone(),
                                y=env.y.detach().cpu().clone(),
                                rho=env.rho.detach().cpu().clone()
                                if torch.is_tensor(env.rho)
                                else torch.tensor(env.rho),
                                r=float(r)))
            agent.store_transition(obs, a, r, obs2, done,
                                   np.zeros(2, np.float32))
            agent.learn()
            obs = obs2
        if len(bad) >= 12:
            break
    print(f"captured {len(bad)} exploding steps", flush=True)
    out = []
    for i, d in enumerate(bad):
        A, y = d["A"], d["y"]
        rho1, rho2 = float(d["rho"][0]), float(d["rho"][1])
        # A: env path (GPU solve + GPU influence)
        Ag = A.cuda().unsqueeze(0).contiguous()
        yg = y.cuda().unsqueeze(0).contiguous()
        rg = torch.tensor([[rho1, rho2]], device="cuda")
        xg, Yg, Sg, nh = ops.ext().enet_lbfgs_solve(Ag, yg, rg, 20, 10, 7)
        pen = torch.zeros(1, device="cuda")
        EEg, rg_out = ops.ext().enet_influence(Ag, yg, xg, Yg, Sg, nh,
                                               pen, rg)
        k = int(nh[0])
        # B: GPU pairs -> CPU oracle influence
        EEb = enet_ops.influence_eigs_reference(
            A, Yg[0, :k].cpu(), Sg[0, :k].cpu(), rho1=rho1)
        # C: full CPU reference
        xc, opt = enet_ops.lbfgs_solve_reference(A, y, rho1, rho2)
        Yc, Sc = enet_ops.curvature_stacks(opt)
        EEc = enet_ops.influence_eigs_reference(A, Yc, Sc, rho1=rho1)
        # pair quality stats
        ysg = (Yg[0, :k].cpu() * Sg[0, :k].cpu()).sum(-1)
        ssg = (Sg[0, :k].cpu() ** 2).sum(-1)
        rec = dict(
            r_env=d["r"], rho1=rho1, rho2=rho2,
            gpu=dict(reward=float(rg_out[0]),
                     emin=float(EEg[0].min()), emax=float(EEg[0].max())),
            gpu_pairs_cpu_eig=dict(emin=float(EEb.min()),
                                   emax=float(EEb.max()),
                                   ratio=float(EEb.min() / EEb.max())),
            cpu_ref=dict(emin=float(EEc.min()), emax=float(EEc.max()),
                         ratio=float(EEc.min() / EEc.max())),
            gpu_pair_ys_over_ss=[float(v) for v in (ysg / ssg)],
        )
        out.append(rec)
        print(json.dumps(rec), flush=True)
    Path("gpurun_out").mkdir(exist_ok=True)
    with open("gpurun_out/explosion_debug.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
