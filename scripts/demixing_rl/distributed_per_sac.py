"""Distributed PER SAC for demixing: learner/actor split.

CLI parity with `demixing_rl/distributed_per_sac.py:176-209`. The
observation is the flattened {infmap, metadata} dict; transport is
torch.distributed collectives (RCCL on GPU, gloo on CPU).
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.distributed.learner_actor import run_process
from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.rl.sac_cnn import Agent

K = 6
NINF = 128
META = 3 * K + 2
OBS_DIM = NINF * NINF + META


class _FlatObsAgent(Agent):
    """Adapter: the learner ingests flat records; split them back into
    {infmap, metadata} for the CNN buffer/nets."""

    def _split_flat(self, flat):
        flat = torch.as_tensor(np.asarray(flat), dtype=torch.float32) \
            .reshape(-1)
        return {"infmap": flat[:NINF * NINF].reshape(1, NINF, NINF),
                "metadata": flat[NINF * NINF:]}

    def store_transition(self, state, action, reward, state_, terminal,
                         hint=None):
        if not isinstance(state, dict):
            state = self._split_flat(state)
            state_ = self._split_flat(state_)
        super().store_transition(state, action, reward, state_, terminal,
                                 hint)


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
    ap.add_argument("--stations", type=int, default=62)
    args = ap.parse_args()

    def agent_factory():
        return _FlatObsAgent(gamma=0.99, batch_size=256, n_actions=K,
                             tau=0.005, max_mem_size=16000,
                             input_dims=(1, NINF, NINF), meta_dim=META,
                             lr_a=3e-4, lr_c=3e-4, prioritized=True,
                             use_hint=True)

    def env_factory():
        return DemixingEnv(K=K, Nf=3, Ninf=NINF, provide_hint=True,
                           provide_influence=False,
                           N_stations=args.stations)

    import torch as _t
    backend = args.backend
    if backend is None:
        gpu_ok = (_t.cuda.is_available()
                  and _t.cuda.device_count() >= args.world_size)
        backend = "nccl" if gpu_ok else "gloo"
    run_process(args.rank, args.world_size, agent_factory, env_factory,
                obs_dim=OBS_DIM, n_actions=K, episodes=args.episodes,
                epochs=10, steps=7, use_hint=True,
                learner_addr=args.learner_addr,
                learner_port=args.learner_port, max_transitions=100,
                save_every=10, backend=backend)


if __name__ == "__main__":
    main()
