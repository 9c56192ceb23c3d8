"""Learner/actor distributed RL on torch.distributed collectives.

The reference ships actor weights as CPU state_dicts over torch-RPC and
uploads whole pickled replay buffers per episode
(`elasticnet/distributed_per_sac.py:23-174`,
`demixing_rl/distributed_per_sac.py:23-209`). Here, per round:

1. the learner broadcasts the actor's FLAT parameter pool (one
   broadcast of one contiguous tensor — RCCL over xGMI on GPU);
2. every actor runs ``epochs × steps`` env steps locally, packing each
   transition into a fixed-width record tensor;
3. one fixed-shape ``gather`` moves all actors' transition blocks to the
   learner, which ingests them and calls ``learn()`` per transition
   (same ingestion semantics as `Learner.download_replaybuffer:44-57`,
   without the lock — the learner is single-threaded here).

CLI contract preserved: ``--rank --world-size --learner-addr
--learner-port`` (`distributed_per_sac.py:176-190`); rank 0 is the
learner, ranks 1..W−1 actors.
"""

from __future__ import annotations

import datetime
import os
import pickle

import numpy as np
import torch
import torch.distributed as dist

__all__ = ["Learner", "Actor", "run_process"]


def _flat_obs(obs) -> torch.Tensor:
    """Flatten an observation (tensor/array or dict of them) to 1-D.
    Elastic-net dicts use the agent's eig-then-A order
    (`enet_sac.py:548`); other dicts use insertion order."""
    if isinstance(obs, dict):
        if set(obs.keys()) == {"eig", "A"}:
            from ..envs.enet import obs_to_state
            return obs_to_state(obs).reshape(-1)
        parts = [torch.as_tensor(np.asarray(v), dtype=torch.float32)
                 .reshape(-1) for v in obs.values()]
        return torch.cat(parts)
    return torch.as_tensor(np.asarray(obs), dtype=torch.float32).reshape(-1)


class _Codec:
    """Fixed-width transition records: [state | action | reward | state' |
    done | hint]."""

    def __init__(self, obs_dim: int, n_actions: int):
        self.obs_dim = obs_dim
        self.n_actions = n_actions
        self.width = 2 * obs_dim + 2 * n_actions + 2

    def pack(self, state, action, reward, state_, done, hint):
        rec = torch.empty(self.width, dtype=torch.float32)
        o = self.obs_dim
        a = self.n_actions
        rec[:o] = _flat_obs(state)
        rec[o:o + a] = torch.as_tensor(np.asarray(action),
                                       dtype=torch.float32).reshape(-1)
        rec[o + a] = float(reward)
        rec[o + a + 1:2 * o + a + 1] = _flat_obs(state_)
        rec[2 * o + a + 1] = float(done)
        if hint is None:
            rec[2 * o + a + 2:] = 0.0
        else:
            rec[2 * o + a + 2:] = torch.as_tensor(
                np.asarray(hint), dtype=torch.float32).reshape(-1)
        return rec

    def unpack(self, rec: torch.Tensor):
        o = self.obs_dim
        a = self.n_actions
        return (rec[:o], rec[o:o + a], float(rec[o + a]),
                rec[o + a + 1:2 * o + a + 1], bool(rec[2 * o + a + 1] > 0.5),
                rec[2 * o + a + 2:])


class Actor:
    """Env-stepping worker (ranks ≥ 1). Mirrors
    `distributed_per_sac.Actor:104-153`: pulls weights, runs
    epochs×steps, uploads its transition block."""

    def __init__(self, env, agent, codec: _Codec, max_transitions: int,
                 epochs: int = 10, steps: int = 10, use_hint: bool = False,
                 env_factory=None):
        self.env = env
        self.agent = agent
        self.codec = codec
        self.cap = max_transitions
        self.epochs = epochs
        self.steps = steps
        self.use_hint = use_hint
        self.env_factory = env_factory

    def run_round(self):
        # 1. receive current actor weights (flat broadcast from rank 0)
        dist.broadcast(self.agent.actor_fp.flat, src=0)
        # 2. local rollouts — failure-tolerant: an env error (numerical
        # blow-up, bad scenario draw) must not wedge the collective
        # schedule, so the actor reports what it has (possibly 0
        # transitions), rebuilds its env, and stays in the ring. (The
        # reference has no failure handling at all — SURVEY.md §5.)
        block = torch.zeros((self.cap, self.codec.width),
                            dtype=torch.float32)
        count = 0
        try:
            for _ in range(self.epochs):
                obs = self.env.reset()
                hint = getattr(self.env, "hint", None) if self.use_hint \
                    else None
                for _ in range(self.steps):
                    action = self.agent.choose_action(obs)
                    out = self.env.step(action)
                    if len(out) == 5:
                        obs_, reward, done, hint, _ = out
                    else:
                        obs_, reward, done, _ = out
                    if count < self.cap:
                        block[count] = self.codec.pack(obs, action, reward,
                                                       obs_, done, hint)
                        count += 1
                    obs = obs_
                    if done:
                        break
        except Exception as e:  # noqa: BLE001
            print(f"[actor] env failure ({type(e).__name__}: {e}); "
                  f"uploading {count} transitions and rebuilding env")
            if self.env_factory is not None:
                try:
                    self.env = self.env_factory()
                except Exception:  # noqa: BLE001
                    pass
        hdr = torch.tensor([float(count)])
        # 3. upload to the learner
        dist.gather(hdr, dst=0)
        dist.gather(block, dst=0)


class Learner:
    """Rank-0 trainer. Mirrors `distributed_per_sac.Learner:23-103`."""

    def __init__(self, agent, codec: _Codec, world_size: int,
                 max_transitions: int, save_every: int = 10,
                 checkpoint_scores: str = "scores.pkl"):
        self.agent = agent
        self.codec = codec
        self.world = world_size
        self.cap = max_transitions
        self.save_every = save_every
        self.scores = []
        self.checkpoint_scores = checkpoint_scores

    def run_round(self, episode: int):
        dist.broadcast(self.agent.actor_fp.flat, src=0)
        hdrs = [torch.zeros(1) for _ in range(self.world)]
        dist.gather(torch.zeros(1), gather_list=hdrs, dst=0)
        blocks = [torch.zeros((self.cap, self.codec.width))
                  for _ in range(self.world)]
        dist.gather(torch.zeros((self.cap, self.codec.width)),
                    gather_list=blocks, dst=0)
        total_r = 0.0
        n = 0
        for w in range(1, self.world):
            cnt = int(hdrs[w].item())
            for i in range(cnt):
                s, a, r, s_, d, h = self.codec.unpack(blocks[w][i])
                self.agent.store_transition(s, a, r, s_, d, h)
                self.agent.learn()
                total_r += r
                n += 1
        if n:
            self.scores.append(total_r / n)
        if self.save_every and (episode + 1) % self.save_every == 0:
            self.agent.save_models()
            with open(self.checkpoint_scores, "wb") as f:
                pickle.dump(self.scores, f)
        return n

    def run_episodes(self, n_episodes: int):
        for ep in range(n_episodes):
            self.run_round(ep)
        return self.scores


def run_process(rank: int, world_size: int, agent_factory, env_factory,
                obs_dim: int, n_actions: int, episodes: int = 10,
                epochs: int = 10, steps: int = 10, use_hint: bool = False,
                learner_addr: str = "localhost", learner_port: int = 6985,
                max_transitions: int = 100, backend: str | None = None,
                save_every: int = 0):
    """Entry point preserving the reference CLI semantics
    (`distributed_per_sac.py:154-194`)."""
    if backend is None:
        # Composite backend: the flat weight broadcast rides RCCL over
        # xGMI when the agent lives on a GPU, while the hdr/transition
        # gathers are CPU tensors and need gloo — a NCCL-only group
        # cannot carry them.
        backend = "cpu:gloo,cuda:nccl" if torch.cuda.is_available() \
            else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            init_method=f"tcp://{learner_addr}:{learner_port}",
            timeout=datetime.timedelta(minutes=30))
    codec = _Codec(obs_dim, n_actions)
    try:
        if rank == 0:
            agent = agent_factory()
            learner = Learner(agent, codec, world_size, max_transitions,
                              save_every=save_every)
            scores = learner.run_episodes(episodes)
            return scores
        agent = agent_factory()
        env = env_factory()
        actor = Actor(env, agent, codec, max_transitions, epochs=epochs,
                      steps=steps, use_hint=use_hint,
                      env_factory=env_factory)
        for _ in range(episodes):
            actor.run_round()
        return None
    finally:
        dist.barrier()
        dist.destroy_process_group()
