"""Whole-iteration hipGraph for the vectorized rollout: actor forward,
batched env solve+influence, batched store, and the SAC learn step as
ONE graph replay per iteration.

With mem_size == E the ring write set is the whole buffer every
iteration (idx = (arange+c) mod E covers all slots for any c), so the
python-side counter freezing under capture is distributionally
irrelevant; all randomness (observation noise, action sampling, replay
indices) is philox-based and graph-safe. Script-level experiment — no
library changes. Writes gpurun_out/vec_graph_bench.json.
"""

import json
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.vec_enet import VecENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything

N = M = 20


def run(E, iters=50, warmup=8):
    seed_everything(1)
    dev = torch.device("cuda")
    env = VecENetEnv(E, M, N, device=dev)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=E, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03, device=dev)
    obs = env.reset()
    state = torch.cat((obs["eig"], obs["A"]), dim=1).contiguous()

    def one_iter():
        with torch.no_grad():
            actions, _ = agent.actor.sample_normal(state,
                                                   reparameterize=False)
        obs_, rewards, done, _ = env.step(actions)
        state_ = torch.cat((obs_["eig"], obs_["A"]), dim=1)
        agent.replaymem.store_batch(state, actions, rewards, state_, done)
        agent.learn()
        state.copy_(state_)

    # eager timing first
    for _ in range(warmup):
        one_iter()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        one_iter()
    torch.cuda.synchronize()
    eager = E * iters / (time.perf_counter() - t0)

    # fill the buffer and freeze the counter at "full"
    agent.replaymem.mem_cntr = E

    # capture
    torch.cuda.synchronize()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            one_iter()
    torch.cuda.current_stream().wait_stream(s)
    agent.replaymem.mem_cntr = E
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        one_iter()
    agent.replaymem.mem_cntr = E

    for _ in range(warmup):
        g.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        g.replay()
    torch.cuda.synchronize()
    graphed = E * iters / (time.perf_counter() - t0)
    assert torch.isfinite(agent.actor_fp.flat).all()
    assert torch.isfinite(env.x).all()
    return eager, graphed


def main():
    out = {}
    for E in (64, 256, 1024):
        try:
            eager, graphed = run(E)
            out[E] = {"eager": round(eager, 1), "graphed": round(graphed, 1)}
            print(f"E={E:5d}: eager {eager:12.1f}  graphed {graphed:12.1f} "
                  f"env-steps/s  ({graphed / eager:.2f}x)", flush=True)
        except Exception as e:  # noqa: BLE001
            print(f"E={E}: FAILED {type(e).__name__}: {e}", flush=True)
            out[E] = {"error": str(e)}
    Path("gpurun_out").mkdir(exist_ok=True)
    with open("gpurun_out/vec_graph_bench.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
