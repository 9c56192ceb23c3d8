"""Flagship benchmark: elastic-net SAC training, env-steps/sec (whole node).

Measures the BASELINE.json metric — env-steps/sec of the canonical
elastic-net SAC config (N=M=20, input_dims=[420], batch 64, mem 1024,
gamma .99, tau .005, lr 1e-3, reward_scale N, alpha .03 — reference
``elasticnet/main_sac.py:28-40``) — on 1..8 MI355X GPUs, one rank per GPU
over RCCL, weak scaling (each rank runs its own env + agent replica;
gradients all-reduced over xGMI each learn step, i.e. the data-parallel
learner of SURVEY.md §2.3 P2).

One bench "step" = one full training-loop step in steady state:
choose_action -> env.step (20-epoch L-BFGS solve + influence eigens +
reward, fused HIP kernels) -> store_transition -> learn (twin-critic +
actor update + polyak). The replay buffer is pre-filled (untimed) so learn()
is active during the whole timed region — nothing is skipped.

Usage: python bench.py --gpus N --steps K --warmup W
(for N>1 the driver launches via torch.distributed.run, one rank per GPU).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch
import torch.distributed as dist

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything

N = 20
M = 20


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # steady-state default: >=200 timed steps puts the timed region at
    # ~0.3-0.5 s on one MI355X (the workload swings 456-667 steps/s from
    # solver early-stop trajectories — profiles/bench_variance.txt — so a
    # thin region is noise)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=32)
    ap.add_argument("--seed", type=int, default=1)
    # BASELINE config 2: elastic-net TD3 + prioritized replay, bf16
    ap.add_argument("--algo", choices=["sac", "td3"], default="sac")
    ap.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32")
    args = ap.parse_args()

    if args.dtype == "bf16":
        from smartcal_amd.ops import linear as linear_ops
        linear_ops.set_compute_dtype("bf16")

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    have_gpu = torch.cuda.is_available()

    if world > 1:
        backend = "nccl" if have_gpu else "gloo"
        dist.init_process_group(backend=backend)

    if have_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    seed_everything(args.seed + 1000 * rank)

    env = ENetEnv(M, N, provide_hint=False, device=device)

    def grad_hook(fps):
        if world > 1:
            for fp in fps:
                dist.all_reduce(fp.flat_grad, op=dist.ReduceOp.SUM)
                fp.flat_grad.div_(world)

    if args.algo == "td3":
        # reference elasticnet/main_td3.py:12-22 config + PER
        from smartcal_amd.rl.td3 import Agent as TD3Agent
        agent = TD3Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                         max_mem_size=1024, input_dims=[N + N * M],
                         lr_a=1e-3, lr_c=1e-3, warmup=0, noise=0.1,
                         prioritized=True, use_hint=False, device=device,
                         grad_hook=grad_hook)
    else:
        agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                      max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                      lr_c=1e-3, reward_scale=N, alpha=0.03,
                      prioritized=False, use_hint=False, device=device,
                      grad_hook=grad_hook)

    obs = env.reset()
    zeros2 = np.zeros(2, dtype=np.float32)

    if have_gpu:
        # sync-free training step: action, reward and observation stay on
        # the GPU for the whole loop (no host round trips)
        def one_step(o):
            a = agent.choose_action_tensor(o)
            o2, r, done, info = env.step(a)
            agent.store_transition(o, a, r, o2, done, zeros2)
            agent.learn()
            return o2
    else:
        def one_step(o):
            a = agent.choose_action(o)
            o2, r, done, info = env.step(a)
            agent.store_transition(o, a, r, o2, done, zeros2)
            agent.learn()
            return o2

    # ---- pre-fill replay so learn() is live in the timed region ----
    while agent.replaymem.mem_cntr < agent.batch_size:
        obs = one_step(obs)

    if have_gpu:
        # capture the whole learn step into one hipGraph (forwards,
        # backwards, fused Adam, polyak — single replay per step)
        graph_ok = True
        try:
            agent.enable_cuda_graph()
        except Exception as e:  # noqa: BLE001
            import sys as _sys
            graph_ok = False
            print(f"[bench] graph capture unavailable, eager path: {e}",
                  file=_sys.stderr)
        if world > 1:
            # all ranks must agree on graphed vs eager: capture records
            # (does not execute) the in-graph all_reduce, so a rank that
            # replays while another runs eager would desequence RCCL —
            # take the AND over ranks and fall back together
            ok = torch.tensor([1.0 if graph_ok else 0.0], device=device)
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if ok.item() < 1.0 and graph_ok:
                agent.disable_cuda_graph()

    # ---- warmup ----
    for _ in range(args.warmup):
        obs = one_step(obs)

    if world > 1:
        dist.barrier()
    if have_gpu:
        torch.cuda.synchronize()
    # timed region, split into 4 sub-chunks so the printed JSON carries a
    # run-internal variance band alongside the headline value (the value
    # itself is computed over the WHOLE region — every step is timed)
    n_chunks = 4 if args.steps >= 8 else 1
    per = args.steps // n_chunks
    counts = [per] * n_chunks
    counts[-1] += args.steps - per * n_chunks
    chunk_rates = []
    t0 = time.perf_counter()
    for cnt in counts:
        tc = time.perf_counter()
        for _ in range(cnt):
            obs = one_step(obs)
        if have_gpu:
            torch.cuda.synchronize()
        chunk_rates.append(cnt / (time.perf_counter() - tc))
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    el = torch.tensor([elapsed], dtype=torch.float64,
                      device=device if (world > 1 and have_gpu) else "cpu")
    if world > 1:
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
    elapsed_max = float(el.item())

    n_gpus = world if world > 1 else args.gpus
    total_steps = args.steps * max(world, 1)
    value = total_steps / elapsed_max
    ms_per_step = elapsed_max / args.steps * 1000.0

    if rank == 0:
        algo = args.algo.upper()
        print(json.dumps({
            "metric": f"env-steps/sec (whole node), elastic-net {algo}",
            "value": value,
            "unit": "env-steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "band_steps_per_s": [round(min(chunk_rates) * max(world, 1), 1),
                                 round(max(chunk_rates) * max(world, 1), 1)],
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": f"elasticnet-{args.algo}", "N": N, "M": M,
                       "input_dims": N + N * M, "global_batch": 64 * max(world, 1),
                       "replay": 1024, "reward_scale": N, "alpha": 0.03,
                       "prioritized": args.algo == "td3",
                       "parallelism": f"dp{max(world, 1)}"},
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
