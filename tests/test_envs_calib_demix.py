"""Smoke + semantics tests for CalibEnv and DemixingEnv (small sizes)."""

import numpy as np
import pytest

from smartcal_amd.envs.calib import CalibEnv
from smartcal_amd.envs.demix import DemixingEnv


@pytest.fixture(scope="module")
def calib_env():
    return CalibEnv(M=3, provide_hint=True, N_stations=8, Nf=3, Ts=1,
                    Tdelta=4, Ninf=32, admm_iter=2, poly_order=2,
                    device="cpu", inf_nfreq=1, seed=0)


def test_calib_reset_step(calib_env):
    env = calib_env
    obs = env.reset()
    assert obs["img"].shape == (1, 32, 32)
    assert obs["sky"].shape == (4, 7)
    assert np.isfinite(obs["img"]).all() and np.isfinite(obs["sky"]).all()
    assert env.hint is not None and env.hint.shape == (6,)
    action = env.action_space.sample()
    obs, reward, done, hint, info = env.step(action)
    assert obs["img"].shape == (1, 32, 32)
    assert np.isfinite(reward)
    assert not done
    # out-of-range rho incurred penalty bookkeeping without crashing
    obs2, r2, _, _, _ = env.step(np.ones(6, np.float32) * 2.0)
    assert np.isfinite(r2)
    # rho clipped to HIGH
    assert (env.rho_spectral[:env.K] <= 1000.0).all()


@pytest.fixture(scope="module")
def demix_env():
    return DemixingEnv(K=6, Nf=2, Ninf=32, Tdelta=4, Ts=1,
                       provide_hint=False, provide_influence=True,
                       N_stations=8, poly_order=2, device="cpu", seed=1)


def test_demix_reset_step(demix_env):
    env = demix_env
    obs = env.reset()
    assert obs["infmap"].shape == (1, 32, 32)
    assert obs["metadata"].shape == (20,)
    md = obs["metadata"] / 1e-3
    assert md[-1] == 8                      # stations
    assert np.isclose(md[env.K - 1], 0.0)   # target separation 0
    action = np.zeros(6, np.float32)
    action[0] = 1.0    # select first outlier
    action[-1] = 0.0   # mid-range maxiter
    obs, reward, done, info = env.step(action)
    assert np.isfinite(reward)
    assert env.clus_id[0] == 0 and env.clus_id[-1] == env.K - 1
    assert 5 <= env.maxiter <= 30
    # selected directions' separations zeroed in the obs
    md2 = obs["metadata"] / 1e-3
    assert md2[0] == 0.0


def test_demix_scalar_to_kvec():
    v = DemixingEnv.scalar_to_kvec(5, 5)
    np.testing.assert_array_equal(v, [0, 0, 1, 0, 1])


def test_demix_hint_small():
    env = DemixingEnv(K=3, Nf=2, Ninf=16, Tdelta=4, Ts=1,
                      provide_hint=True, provide_influence=False,
                      N_stations=6, poly_order=2, device="cpu", seed=2)
    env.reset()
    hint = env.get_hint()
    assert hint.shape == (3,)
    assert np.isfinite(hint).all()
    assert (hint >= -1.001).all() and (hint <= 1.001).all()
