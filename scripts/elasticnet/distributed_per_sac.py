"""Distributed PER SAC for elastic-net: learner/actor split.

CLI parity with `elasticnet/distributed_per_sac.py:176-194`
(--rank --world-size --learner-addr --learner-port); rank 0 = learner,
others = actors. Transport is torch.distributed (RCCL over xGMI on an
MI355X node, gloo on CPU) instead of TensorPipe RPC — one flat weight
broadcast + one fixed-shape experience gather per episode round.
"""

import argparse
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.distributed.learner_actor import run_process
from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent

N = M = 20


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rank", type=int, default=0)
    ap.add_argument("--world-size", type=int, default=2)
    ap.add_argument("--learner-addr", type=str, default="localhost")
    ap.add_argument("--learner-port", type=int, default=6985)
    ap.add_argument("--episodes", type=int, default=1000)
    ap.add_argument("--backend", default=None,
                    help="nccl|gloo (default: nccl when each rank can "
                         "own a GPU, else gloo)")
    args = ap.parse_args()

    def agent_factory():
        return Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                     max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                     lr_c=1e-3, reward_scale=N, alpha=0.03,
                     prioritized=True, use_hint=True)

    def env_factory():
        return ENetEnv(M, N, provide_hint=True)

    import torch as _t
    backend = args.backend
    if backend is None:
        gpu_ok = (_t.cuda.is_available()
                  and _t.cuda.device_count() >= args.world_size)
        backend = "nccl" if gpu_ok else "gloo"
    run_process(args.rank, args.world_size, agent_factory, env_factory,
                obs_dim=N + N * M, n_actions=2, episodes=args.episodes,
                epochs=10, steps=10, use_hint=True,
                learner_addr=args.learner_addr,
                learner_port=args.learner_port, max_transitions=100,
                save_every=10, backend=backend)


if __name__ == "__main__":
    main()
