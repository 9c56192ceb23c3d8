"""GPU tests for the radio layer: CPU↔GPU numerics parity and
reference-scale env steps on the MI355X."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="needs a GPU")


@needs_gpu
def test_hessian_engine_gpu_matches_cpu():
    from smartcal_amd.radio import hessian as hs
    rng = np.random.default_rng(0)
    N, T, K = 6, 3, 3
    B = N * (N - 1) // 2
    S = B * T
    R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                          + 1j * rng.standard_normal((2 * S, 2))
                          ).astype(np.complex64))
    C = torch.from_numpy((rng.standard_normal((K, S, 4))
                          + 1j * rng.standard_normal((K, S, 4))
                          ).astype(np.complex64))
    J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                          + 1j * rng.standard_normal((K, 2 * N, 2))
                          ).astype(np.complex64))
    H_cpu = hs.hessianres(R, C, J, N)
    H_gpu = hs.hessianres(R.cuda(), C.cuda(), J.cuda(), N).cpu()
    torch.testing.assert_close(H_gpu, H_cpu, rtol=1e-4, atol=1e-4)
    Dg = H_cpu + 3.0 * torch.eye(4 * N, dtype=torch.complex64)
    dJ_cpu = hs.dsolutions_r(C, J, N, Dg)
    dJ_gpu = hs.dsolutions_r(C.cuda(), J.cuda(), N, Dg.cuda()).cpu()
    torch.testing.assert_close(dJ_gpu, dJ_cpu, rtol=1e-3, atol=1e-3)
    dR_cpu = hs.dresiduals_r(C, J, N, dJ_cpu, True)
    dR_gpu = hs.dresiduals_r(C.cuda(), J.cuda(), N, dJ_cpu.cuda(),
                             True).cpu()
    torch.testing.assert_close(dR_gpu, dR_cpu, rtol=1e-3, atol=1e-3)


@needs_gpu
def test_calibrate_gpu():
    from smartcal_amd.radio import array as arr, sim, solver
    rng = np.random.default_rng(5)
    layout = arr.lofar_like_layout(N=16, rng=rng)
    sky, cs, *_ , ra0, dec0 = sim.make_demixing_sky(rng)
    freqs = np.linspace(120e6, 160e6, 3)
    vis = sim.simulate_observation(layout, sky, cs, freqs, ra0, dec0,
                                   Ts=2, Tdelta=5, snr=20.0,
                                   device="cuda", rng=rng, torch_seed=0)
    K = len(cs)
    sol = solver.calibrate(vis, sky, cs, np.full(K, 5.0, np.float32),
                           admm_iter=4, poly_order=2, n_sweeps=2,
                           init_sweeps=8)
    res = torch.linalg.vector_norm(sol.residual)
    noise = torch.linalg.vector_norm(vis.data - vis.model)
    data = torch.linalg.vector_norm(vis.data)
    assert res < 0.15 * data
    assert res < 4.0 * noise


@needs_gpu
def test_calib_env_step_gpu():
    from smartcal_amd.envs.calib import CalibEnv
    env = CalibEnv(M=4, N_stations=24, Nf=4, Ts=2, Tdelta=5, Ninf=128,
                   admm_iter=3, poly_order=2, device="cuda", inf_nfreq=1,
                   seed=0)
    obs = env.reset()
    assert obs["img"].shape == (1, 128, 128)
    a = env.action_space.sample()
    obs, r, done, info = env.step(a)
    assert np.isfinite(r)


@needs_gpu
def test_demix_env_step_gpu():
    from smartcal_amd.envs.demix import DemixingEnv
    env = DemixingEnv(K=6, Nf=3, Ninf=128, Tdelta=5, Ts=2,
                      provide_influence=True, N_stations=24,
                      device="cuda", seed=1)
    obs = env.reset()
    a = np.zeros(6, np.float32)
    a[0] = 1.0
    obs, r, done, info = env.step(a)
    assert np.isfinite(r)
    assert obs["infmap"].shape == (1, 128, 128)


@needs_gpu
def test_cnn_sac_gpu_learn():
    from smartcal_amd.rl import sac_cnn
    rng = np.random.default_rng(7)
    agent = sac_cnn.Agent(gamma=0.99, lr_a=3e-4, lr_c=3e-4,
                          input_dims=(1, 128, 128), batch_size=32,
                          n_actions=6, max_mem_size=256, meta_dim=20,
                          prioritized=True, use_hint=True,
                          device=torch.device("cuda"))
    obs = {"infmap": rng.standard_normal((1, 128, 128)).astype(np.float32),
           "metadata": rng.standard_normal(20).astype(np.float32)}
    for _ in range(40):
        a = agent.choose_action(obs)
        agent.store_transition(obs, a, 0.1, obs, False,
                               hint=np.zeros(6, np.float32))
    agent.learn()
    assert torch.isfinite(agent.actor_fp.flat).all()


@needs_gpu
def test_training_example_gpu():
    from smartcal_amd.radio.dataset import generate_training_example
    rng = np.random.default_rng(3)
    x, y, K = generate_training_example(rng, Ninf=64, N_stations=24,
                                        device="cuda")
    assert x.shape == (K * (64 * 64 + 8),)
    assert np.isfinite(x).all()
    assert y.shape == (K - 1,)


@needs_gpu
def test_transformer_gpu():
    from smartcal_amd.models import TransformerEncoder
    torch.manual_seed(0)
    K = 6
    Nout = 64 * 64 + 8
    net = TransformerEncoder(num_layers=1, input_dim=K * Nout,
                             model_dim=K * 66, num_classes=K - 1,
                             num_heads=K, dropout=0.0).cuda()
    x = torch.randn(8, K * Nout, device="cuda")
    y = net(x)
    loss = torch.nn.functional.binary_cross_entropy(
        y, torch.rand(8, K - 1, device="cuda"))
    loss.backward()
    assert torch.isfinite(y).all()


@needs_gpu
def test_als_sweep_kernel_matches_torch():
    """The fused ALS-sweep HIP kernel must match the pure-torch path."""
    from smartcal_amd.radio import solver as rs
    from smartcal_amd.radio.hessian import baseline_pq
    torch.manual_seed(0)
    F, K, Ts, Td, N = 2, 3, 2, 3, 7
    B = N * (N - 1) // 2
    T = Ts * Td
    C22 = (torch.randn(F, K, T, B, 2, 2) + 1j * torch.randn(F, K, T, B, 2, 2)
           ).to(torch.complex64).cuda()
    data22 = (torch.randn(F, T, B, 2, 2) + 1j * torch.randn(F, T, B, 2, 2)
              ).to(torch.complex64).cuda()
    p_idx, q_idx = baseline_pq(N, torch.device("cuda"))
    rho_t = torch.ones(K, device="cuda")
    J0 = torch.zeros((F, Ts, K, N, 2, 2), dtype=torch.complex64,
                     device="cuda")
    J0[..., 0, 0] = 1.0
    J0[..., 1, 1] = 1.0
    # one sweep with the kernel path (device cuda => use_kernel)
    Jk = J0.clone()
    rs._solve_sweeps(data22, C22, Jk, rho_t, None, p_idx, q_idx, N, Td, 2)
    # same on CPU (torch path)
    Jc = J0.cpu().clone()
    rs._solve_sweeps(data22.cpu(), C22.cpu(), Jc, rho_t.cpu(), None,
                     p_idx.cpu(), q_idx.cpu(), N, Td, 2)
    torch.testing.assert_close(Jk.cpu(), Jc, rtol=2e-4, atol=2e-4)


@needs_gpu
def test_cgemm_nn_bcast_matches_torch():
    """Hand-written batched complex MFMA GEMM vs torch matmul, incl. the
    A-broadcast-over-8 pattern and edge shapes."""
    from smartcal_amd import ops
    torch.manual_seed(0)
    for (KA, rep, M, Kd, N) in [(3, 8, 248, 248, 200), (1, 1, 64, 64, 64),
                                (2, 8, 68, 68, 1891), (1, 2, 17, 33, 65)]:
        A = (torch.randn(KA, M, Kd) + 1j * torch.randn(KA, M, Kd)) \
            .to(torch.complex64).cuda() / Kd ** 0.5
        B = (torch.randn(KA * rep, Kd, N) + 1j * torch.randn(KA * rep, Kd, N)) \
            .to(torch.complex64).cuda() / Kd ** 0.5
        C = ops.ext().cgemm_nn_bcast(A.contiguous(), B.contiguous(), rep)
        ref = torch.repeat_interleave(A, rep, dim=0) @ B
        err = (C - ref).abs().max()
        scale = ref.abs().max().clamp(min=1.0)
        assert float(err / scale) < 1e-4, (KA, rep, M, Kd, N, float(err))


@needs_gpu
def test_dsolutions_gpu_uses_cgemm_and_matches_cpu():
    from smartcal_amd.radio import hessian as hs
    rng = np.random.default_rng(3)
    N, T, K = 8, 2, 2
    B = N * (N - 1) // 2
    S = B * T
    C = torch.from_numpy((rng.standard_normal((K, S, 4))
                          + 1j * rng.standard_normal((K, S, 4))
                          ).astype(np.complex64))
    J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                          + 1j * rng.standard_normal((K, 2 * N, 2))
                          ).astype(np.complex64))
    R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                          + 1j * rng.standard_normal((2 * S, 2))
                          ).astype(np.complex64))
    H = hs.hessianres(R, C, J, N)
    dJ_cpu = hs.dsolutions_r(C, J, N, H)
    dJ_gpu = hs.dsolutions_r(C.cuda(), J.cuda(), N, H.cuda()).cpu()
    scale = dJ_cpu.abs().max().clamp(min=1.0)
    assert float((dJ_cpu - dJ_gpu).abs().max() / scale) < 2e-3


@needs_gpu
def test_cnn_sac_graph_capture_learns():
    """CNN SAC learn step captured into one hipGraph: replays train, stay
    finite, and track the eager path's critic outputs from the same
    start (BatchNorm running stats update inside the graph)."""
    from smartcal_amd.rl.sac_cnn import Agent
    torch.manual_seed(0)
    np.random.seed(0)

    def mk():
        torch.manual_seed(3)
        return Agent(gamma=0.99, batch_size=8, n_actions=4, tau=0.005,
                     max_mem_size=64, input_dims=(1, 32, 32), meta_dim=10,
                     lr_a=1e-3, lr_c=1e-3, device=torch.device("cuda"))

    def fill(ag):
        rng = np.random.default_rng(1)
        for i in range(16):
            obs = {"img": rng.standard_normal((1, 32, 32)).astype(
                       np.float32),
                   "metadata": rng.standard_normal(10).astype(np.float32)}
            ag.store_transition(obs, rng.standard_normal(4).astype(
                np.float32), float(i % 3), obs, False)

    ag = mk()
    fill(ag)
    ag.enable_cuda_graph()
    for _ in range(5):
        ag.learn()
    torch.cuda.synchronize()
    assert ag.learn_counter == 5
    assert torch.isfinite(ag.critic_1_fp.flat).all()
    assert torch.isfinite(ag.actor_fp.flat).all()
    # BN running stats must have moved inside the graph
    bn = [m for m in ag.critic_1.modules()
          if isinstance(m, torch.nn.BatchNorm2d)]
    if bn:
        assert float(bn[0].running_mean.abs().sum()) != 0.0


@needs_gpu
def test_coherency_kernel_matches_torch_oracle():
    """coherency_kernel (one launch, f64 phase accumulation) vs the torch
    composition on a sky with point + Gaussian sources and smearing."""
    import os
    from smartcal_amd.radio import array as arr, sim
    from smartcal_amd.radio.coherency import predict_coherencies_uvw
    rng = np.random.default_rng(2)
    layout = arr.lofar_like_layout(N=10, rng=rng)
    (sky, cs_sim, _sky_cal, cs_cal, _lmn, _rho, ra0,
     dec0) = sim.make_calibration_sky(3, rng)
    times = np.arange(4) * 600.0
    uvw_t = arr.uvw_synthesis(layout, ra0, dec0, times)
    vis_uvw = torch.as_tensor(uvw_t.reshape(-1, 3), dtype=torch.float32)
    for smear in (None, 180e3):
        os.environ["SMARTCAL_FORCE_EAGER"] = "1"
        try:
            ref = predict_coherencies_uvw(sky, cs_sim, vis_uvw.cuda(),
                                          150e6, ra0, dec0,
                                          smear_bw=smear)
        finally:
            del os.environ["SMARTCAL_FORCE_EAGER"]
        out = predict_coherencies_uvw(sky, cs_sim, vis_uvw.cuda(), 150e6,
                                      ra0, dec0, smear_bw=smear)
        scale = ref.abs().max().clamp(min=1e-6)
        err = (out - ref).abs().max() / scale
        assert float(err) < 1e-5, (smear, float(err))


@needs_gpu
def test_hessianres_kernel_matches_torch_oracle():
    """hessianres_kernel (one launch) vs the torch einsum/index_add
    oracle across sizes, incl. non-multiple-of-4 baseline counts."""
    import os
    from smartcal_amd.radio import hessian as hs
    rng = np.random.default_rng(1)
    for (N, T, K) in [(6, 3, 3), (8, 2, 2), (10, 5, 1)]:
        B = N * (N - 1) // 2
        S = B * T
        C = torch.from_numpy((rng.standard_normal((K, S, 4))
                              + 1j * rng.standard_normal((K, S, 4))
                              ).astype(np.complex64)).cuda()
        J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                              + 1j * rng.standard_normal((K, 2 * N, 2))
                              ).astype(np.complex64)).cuda()
        R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                              + 1j * rng.standard_normal((2 * S, 2))
                              ).astype(np.complex64)).cuda()
        H_k = hs.hessianres(R, C, J, N)
        os.environ["SMARTCAL_FORCE_EAGER"] = "1"
        try:
            H_ref = hs.hessianres(R, C, J, N)
        finally:
            del os.environ["SMARTCAL_FORCE_EAGER"]
        scale = H_ref.abs().max().clamp(min=1e-6)
        err = (H_k - H_ref).abs().max() / scale
        assert float(err) < 1e-5, (N, T, K, float(err))


@needs_gpu
def test_gather_sum_kernel_matches_torch():
    from smartcal_amd import ops
    torch.manual_seed(0)
    F, X, S, G, Cnt = 3, 24, 4000, 50, 80
    inp = (torch.randn(F, X, S) + 1j * torch.randn(F, X, S)) \
        .to(torch.complex64).cuda()
    gidx = torch.randint(0, S, (G, Cnt)).long().cuda()
    out = ops.ext().gather_sum(inp.contiguous(), gidx.contiguous())
    ref = inp[:, :, gidx.reshape(-1)].reshape(F, X, G, Cnt).sum(dim=3)
    scale = ref.abs().max().clamp(min=1.0)
    assert float((out - ref).abs().max() / scale) < 1e-5


@needs_gpu
def test_influence_core_large_array():
    """Large-array influence core (N=96, 4N=384): regression for the
    ROCm batched-complex-potrf crash (hessian.py gates Cholesky to
    4N<=256; larger arrays take the LU path)."""
    from smartcal_amd.radio import hessian as hs
    rng = np.random.default_rng(0)
    N, T, K = 96, 4, 3
    B = N * (N - 1) // 2
    S = B * T
    C = torch.from_numpy((rng.standard_normal((K, S, 4))
                          + 1j * rng.standard_normal((K, S, 4))
                          ).astype(np.complex64)).cuda() * 0.1
    J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                          + 1j * rng.standard_normal((K, 2 * N, 2))
                          ).astype(np.complex64)).cuda()
    R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                          + 1j * rng.standard_normal((2 * S, 2))
                          ).astype(np.complex64)).cuda()
    H = hs.hessianres(R, C, J, N)
    m = hs.dres_colmeans(C, J, N, H)
    torch.cuda.synchronize()
    assert m.shape == (8, 4, B)
    assert torch.isfinite(m.real).all() and torch.isfinite(m.imag).all()
