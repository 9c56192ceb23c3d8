"""Multi-process CPU tests (gloo, world_size 2) for the distributed
layer: learner/actor rounds, DP grad all-reduce, solver freq-sharding."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

N = 6
M = 6
OBS_DIM = N + N * M
NACT = 2


def _la_worker(rank, world, port, q):
    from smartcal_amd.distributed.learner_actor import run_process
    from smartcal_amd.envs.enet import ENetEnv
    from smartcal_amd.rl.sac import Agent

    torch.manual_seed(rank)
    np.random.seed(rank)

    def agent_factory():
        return Agent(gamma=0.99, batch_size=4, n_actions=NACT, tau=0.005,
                     max_mem_size=64, input_dims=[OBS_DIM], lr_a=1e-3,
                     lr_c=1e-3, reward_scale=N, alpha=0.03,
                     device=torch.device("cpu"))

    def env_factory():
        return ENetEnv(M, N, device=torch.device("cpu"))

    scores = run_process(rank, world, agent_factory, env_factory,
                         obs_dim=OBS_DIM, n_actions=NACT, episodes=2,
                         epochs=1, steps=3, learner_addr="127.0.0.1",
                         learner_port=port, max_transitions=8,
                         backend="gloo")
    if rank == 0:
        q.put(scores)


@pytest.mark.timeout(300)
def test_learner_actor_round():
    port = 29531
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_la_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    assert all(p.exitcode == 0 for p in procs), \
        [p.exitcode for p in procs]
    scores = q.get()
    assert len(scores) == 2          # one mean-reward entry per episode
    assert all(np.isfinite(s) for s in scores)


def _dp_worker(rank, world, port, q):
    import torch.distributed as dist
    from smartcal_amd.distributed.dp import allreduce_grad_hook
    from smartcal_amd.rl.sac import Agent

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)        # same init on both ranks
    np.random.seed(rank)        # different data
    agent = Agent(gamma=0.99, batch_size=4, n_actions=NACT, tau=0.005,
                  max_mem_size=32, input_dims=[OBS_DIM], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03,
                  device=torch.device("cpu"),
                  grad_hook=allreduce_grad_hook(world))
    rng = np.random.default_rng(rank)
    for _ in range(8):
        s = rng.standard_normal(OBS_DIM).astype(np.float32)
        agent.store_transition(torch.from_numpy(s),
                               rng.standard_normal(NACT).astype(np.float32),
                               float(rng.standard_normal()),
                               torch.from_numpy(s), False,
                               np.zeros(NACT, np.float32))
    torch.manual_seed(123 + 0)  # identical sampling on both ranks? no —
    agent.learn()
    # after one synchronized learn step the actor params must be
    # IDENTICAL across ranks (same init + averaged grads ⇒ same update)
    q.put((rank, agent.actor_fp.flat.clone()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_allreduce_keeps_replicas_synced():
    port = 29541
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(2):
        r, flat = q.get()
        out[r] = flat
    for p in procs:
        p.join(240)
    assert all(p.exitcode == 0 for p in procs)
    # Adam on identical init + identical (averaged) grads ⇒ same params,
    # provided the batch sampling was also identical. Our agents sample
    # on-device with the global torch seed; ranks set the same
    # torch.manual_seed before learn() above.
    torch.testing.assert_close(out[0], out[1])


def _solver_worker(rank, world, port, q):
    import torch.distributed as dist
    from smartcal_amd.radio import array as arr, sim, solver

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    rng = np.random.default_rng(3)     # same scenario on both ranks
    layout = arr.lofar_like_layout(N=6, rng=rng)
    sky, cs, *_, ra0, dec0 = sim.make_demixing_sky(rng, n_outliers=1)
    freqs_all = np.linspace(120e6, 160e6, 4)
    vis = sim.simulate_observation(layout, sky, cs, freqs_all, ra0, dec0,
                                   Ts=1, Tdelta=4, snr=50.0, rng=rng,
                                   torch_seed=0)
    # shard frequencies across the two ranks
    mine = slice(rank * 2, rank * 2 + 2)
    shard = sim.VisData(uvw=vis.uvw, freqs=vis.freqs[mine],
                        data=vis.data[mine], N=vis.N, ra0=ra0, dec0=dec0,
                        Ts=1, Tdelta=4)
    sol = solver.calibrate(shard, sky, cs, np.full(2, 5.0, np.float32),
                           admm_iter=3, poly_order=2, n_sweeps=1,
                           init_sweeps=4)
    q.put((rank, sol.Z.cpu(), float(torch.linalg.vector_norm(sol.residual)),
           float(torch.linalg.vector_norm(shard.data))))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_solver_freq_sharded_consensus():
    port = 29551
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_solver_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(2):
        r, Z, res, dat = q.get()
        out[r] = (Z, res, dat)
    for p in procs:
        p.join(240)
    assert all(p.exitcode == 0 for p in procs)
    # consensus: both ranks agree on the global Z
    torch.testing.assert_close(out[0][0], out[1][0], rtol=1e-4, atol=1e-5)
    # and each rank's local residual is well below its data power
    for r in range(2):
        assert out[r][1] < 0.2 * out[r][2]


def _fault_worker(rank, world, port, q):
    from smartcal_amd.distributed.learner_actor import (run_process,
                                                        Actor, _Codec)
    from smartcal_amd.envs.enet import ENetEnv
    from smartcal_amd.rl.sac import Agent

    torch.manual_seed(rank)
    np.random.seed(rank)

    class FlakyEnv(ENetEnv):
        """Env that blows up on its second reset (then works again)."""
        calls = 0

        def reset(self):
            FlakyEnv.calls += 1
            if FlakyEnv.calls == 2:
                raise RuntimeError("synthetic env failure")
            return super().reset()

    def agent_factory():
        return Agent(gamma=0.99, batch_size=4, n_actions=NACT, tau=0.005,
                     max_mem_size=64, input_dims=[OBS_DIM], lr_a=1e-3,
                     lr_c=1e-3, reward_scale=N, alpha=0.03,
                     device=torch.device("cpu"))

    def env_factory():
        return FlakyEnv(M, N, device=torch.device("cpu"))

    scores = run_process(rank, world, agent_factory, env_factory,
                         obs_dim=OBS_DIM, n_actions=NACT, episodes=3,
                         epochs=1, steps=2, learner_addr="127.0.0.1",
                         learner_port=port, max_transitions=8,
                         backend="gloo")
    if rank == 0:
        q.put(scores)


@pytest.mark.timeout(300)
def test_learner_tolerates_actor_env_failure():
    """A mid-round env crash must not wedge the collective schedule:
    the actor uploads a short round and keeps going."""
    port = 29561
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_fault_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    scores = q.get()
    # 3 episodes ran; the failed round contributed no transitions
    assert len(scores) >= 2


def test_bench_torchrun_world2_cpu(tmp_path):
    """The driver's exact multi-rank launch contract: torchrun world=2
    over gloo on CPU must print one valid JSON line from rank 0."""
    import json as _json
    import subprocess, os, sys
    from pathlib import Path
    ROOT = Path(__file__).resolve().parents[1]
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", str(ROOT / "bench.py"),
         "--gpus", "2", "--steps", "4", "--warmup", "1"],
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=540)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = _json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0


def test_demix_flat_obs_agent_split():
    """The demixing distributed CLI ingests flat records and must split
    them back into {infmap, metadata} for the CNN buffer."""
    import importlib.util
    from pathlib import Path
    ROOT = Path(__file__).resolve().parents[1]
    spec = importlib.util.spec_from_file_location(
        "dps_demix", ROOT / "scripts" / "demixing_rl"
        / "distributed_per_sac.py")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    agent = mod._FlatObsAgent(
        gamma=0.99, batch_size=2, n_actions=mod.K, tau=0.005,
        max_mem_size=8, input_dims=(1, mod.NINF, mod.NINF),
        meta_dim=mod.META, lr_a=3e-4, lr_c=3e-4, prioritized=True,
        use_hint=True)
    flat = np.arange(mod.OBS_DIM, dtype=np.float32)
    agent.store_transition(flat, np.zeros(mod.K, np.float32), 1.0,
                           flat, False, np.zeros(mod.K, np.float32))
    img = agent.replaymem.img_memory[0]
    meta = agent.replaymem.meta_memory[0]
    assert float(img.reshape(-1)[0]) == 0.0
    assert float(img.reshape(-1)[-1]) == mod.NINF * mod.NINF - 1
    assert float(meta[0]) == mod.NINF * mod.NINF
    assert float(meta[-1]) == mod.OBS_DIM - 1


def _la3_worker(rank, world, port, q):
    from smartcal_amd.distributed.learner_actor import run_process
    from smartcal_amd.envs.enet import ENetEnv
    from smartcal_amd.rl.sac import Agent

    torch.manual_seed(rank)
    np.random.seed(rank)

    def agent_factory():
        return Agent(gamma=0.99, batch_size=4, n_actions=NACT, tau=0.005,
                     max_mem_size=64, input_dims=[OBS_DIM], lr_a=1e-3,
                     lr_c=1e-3, reward_scale=N, alpha=0.03,
                     device=torch.device("cpu"))

    def env_factory():
        return ENetEnv(M, N, device=torch.device("cpu"))

    scores = run_process(rank, world, agent_factory, env_factory,
                         obs_dim=OBS_DIM, n_actions=NACT, episodes=2,
                         epochs=1, steps=2, learner_addr="127.0.0.1",
                         learner_port=port, max_transitions=6,
                         backend="gloo")
    if rank == 0:
        q.put(scores)


@pytest.mark.timeout(300)
def test_learner_two_actors():
    """1 learner + 2 actors (the reference's `mpirun -np 3` shape)."""
    port = 29561
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_la3_worker, args=(r, 3, port, q))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    assert all(p.exitcode == 0 for p in procs), \
        [p.exitcode for p in procs]
    scores = q.get()
    assert len(scores) == 2
    assert all(np.isfinite(s) for s in scores)


# ---- world-8 coverage (the driver's 8-GPU shapes, on CPU/gloo) ----

def _la8_worker(rank, world, port, q):
    from smartcal_amd.distributed.learner_actor import run_process
    from smartcal_amd.envs.enet import ENetEnv
    from smartcal_amd.rl.sac import Agent

    torch.manual_seed(rank)
    np.random.seed(rank)

    def agent_factory():
        return Agent(gamma=0.99, batch_size=4, n_actions=NACT, tau=0.005,
                     max_mem_size=128, input_dims=[OBS_DIM], lr_a=1e-3,
                     lr_c=1e-3, reward_scale=N, alpha=0.03,
                     device=torch.device("cpu"))

    def env_factory():
        return ENetEnv(M, N, device=torch.device("cpu"))

    scores = run_process(rank, world, agent_factory, env_factory,
                         obs_dim=OBS_DIM, n_actions=NACT, episodes=1,
                         epochs=1, steps=2, learner_addr="127.0.0.1",
                         learner_port=port, max_transitions=4,
                         backend="gloo")
    if rank == 0:
        q.put(scores)


@pytest.mark.timeout(600)
def test_learner_seven_actors_world8():
    """BASELINE config 4 shape: 1 learner + 7 actors (world 8), one
    episode round — weight broadcast + 7-way transition gather."""
    port = 29561
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_la8_worker, args=(r, 8, port, q))
             for r in range(8)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(540)
    assert all(p.exitcode == 0 for p in procs), \
        [p.exitcode for p in procs]
    scores = q.get()
    assert len(scores) == 1 and np.isfinite(scores[0])


@pytest.mark.timeout(900)
def test_bench_torchrun_world8_cpu(tmp_path):
    """The driver's 8-GPU bench launch (BASELINE config 4/5 shape) on
    CPU/gloo: 8 ranks, DP grad all-reduce, one JSON line from rank 0."""
    import json as _json
    import subprocess, os, sys
    from pathlib import Path
    ROOT = Path(__file__).resolve().parents[1]
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29518", str(ROOT / "bench.py"),
         "--gpus", "8", "--steps", "2", "--warmup", "0"],
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=840)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = _json.loads(line)
    assert out["n_gpus"] == 8
    assert out["config"]["parallelism"] == "dp8"
    assert out["value"] > 0


def _solver8_worker(rank, world, port, q):
    import torch.distributed as dist
    from smartcal_amd.radio import array as arr, sim, solver

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    rng = np.random.default_rng(3)     # same scenario on all ranks
    layout = arr.lofar_like_layout(N=6, rng=rng)
    sky, cs, *_, ra0, dec0 = sim.make_demixing_sky(rng, n_outliers=1)
    freqs_all = np.linspace(120e6, 160e6, 8)
    vis = sim.simulate_observation(layout, sky, cs, freqs_all, ra0, dec0,
                                   Ts=1, Tdelta=4, snr=50.0, rng=rng,
                                   torch_seed=0)
    mine = slice(rank, rank + 1)       # one frequency per rank
    shard = sim.VisData(uvw=vis.uvw, freqs=vis.freqs[mine],
                        data=vis.data[mine], N=vis.N, ra0=ra0, dec0=dec0,
                        Ts=1, Tdelta=4)
    sol = solver.calibrate(shard, sky, cs, np.full(2, 5.0, np.float32),
                           admm_iter=2, poly_order=2, n_sweeps=1,
                           init_sweeps=3)
    q.put((rank, sol.Z.cpu()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_solver_freq_sharded_world8():
    """Consensus-ADMM with 8 frequency shards at world 8 — the full
    8-GPU Z-exchange shape (BASELINE docal.sh topology, one band per
    rank). All ranks must agree on the global Z."""
    port = 29563
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_solver8_worker, args=(r, 8, port, q))
             for r in range(8)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(8):
        r, Z = q.get()
        out[r] = Z
    for p in procs:
        p.join(540)
    assert all(p.exitcode == 0 for p in procs), \
        [p.exitcode for p in procs]
    for r in range(1, 8):
        torch.testing.assert_close(out[0], out[r], rtol=1e-4, atol=1e-5)
