"""Fuzzy-controller demixing environment.

Re-design of `demixing_fuzzy/demixingenv.py:30-375`: the action is the
fuzzy membership breakpoints (24 per outlier + 8 shared for the target
= 24(K−1)+8), a per-direction `DemixController` evaluation decides the
selection (priority ≥ 'high' cutoff), maxiter is fixed at 15, metadata
is 5K+2 (sep/az/el + log-fluxes + selection flags + log f + N), and the
hint is the default fuzzy configuration mapped back to action space.
The calibration pipeline is shared with :class:`DemixingEnv`.
"""

from __future__ import annotations

import numpy as np

from ..fuzzy import DemixController
from ..gymapi import Box, Dict as DictSpace
from .demix import DemixingEnv, INF_SCALE, META_SCALE


class FuzzyDemixingEnv(DemixingEnv):
    def __init__(self, K: int = 6, Nf: int = 3, Ninf: int = 128,
                 Npix: int = 1024, Tdelta: int = 10, provide_hint=False,
                 provide_influence=False, N_stations: int = 26,
                 Ts: int = 2, poly_order: int = 2, device=None,
                 seed: int | None = None):
        super().__init__(K=K, Nf=Nf, Ninf=Ninf, Npix=Npix, Tdelta=Tdelta,
                         provide_hint=provide_hint,
                         provide_influence=provide_influence,
                         N_stations=N_stations, Ts=Ts,
                         poly_order=poly_order, device=device, seed=seed)
        self.n_fuzzy = 32
        self.n_action = 24 * (K - 1) + 8
        self.n_metadata = 5 * K + 2
        self.action_space = Box(low=-1.0, high=1.0, shape=(self.n_action,))
        self.observation_space = DictSpace({
            "infmap": Box(low=-np.inf, high=np.inf, shape=(Ninf, Ninf)),
            "metadata": Box(low=-np.inf, high=np.inf,
                            shape=(self.n_metadata, 1)),
        })
        self.ctrl = DemixController(n_action=self.n_fuzzy)
        self.log_fluxes = None
        self.target_flux = 0.0
        self.azimuth = None
        self.separation = None

    def reset(self):
        sep, az, el, freqs, fluxes = self._simulate_episode()
        vis = self._scenario["vis"]
        self.N = vis.N
        self.separation = sep
        self.azimuth = az
        self.elevation = el
        self.log_fluxes = np.log(np.maximum(fluxes[:-1], 1e-12))
        self.target_flux = float(max(fluxes[-1], 1e-12))
        self.freq_low = freqs[0] / 1e6
        self.freq_high = freqs[-1] / 1e6
        self.rho = np.ones(self.K, np.float32)
        self.clus_id = [self.K - 1]
        self.maxiter = 15
        self.std_data = np.sqrt(np.mean(np.square(
            [vis.stokes_i_std(vis.data[fi]) for fi in range(self.Nf)])))
        self.std_residual, sol = self._calibrate_subset(self.clus_id,
                                                        self.maxiter)
        self.reward0 = self.calculate_reward_(1)
        md = np.zeros(self.n_metadata, np.float32)
        md[:self.K] = sep
        md[self.K:2 * self.K] = az
        md[2 * self.K:3 * self.K] = el
        md[3 * self.K:4 * self.K - 1] = self.log_fluxes
        md[4 * self.K - 1] = np.log(self.target_flux)
        md[4 * self.K:5 * self.K] = 0
        md[5 * self.K - 1] = 1          # target always selected
        md[-2] = np.log(freqs[0])
        md[-1] = self.N
        self.metadata_vec = md
        infdata = self._influence_map(sol)
        self.hint = None
        return {"infmap": infdata * INF_SCALE,
                "metadata": md * META_SCALE}

    def step(self, action):
        action = np.asarray(action, np.float32).squeeze()
        assert action.shape[0] == self.n_action
        action_scaled = action * 0.5 + 0.5
        done = False
        flux_ratio = np.exp(self.log_fluxes) / self.target_flux
        priority = np.zeros(self.K - 1)
        cutoff = np.zeros(self.K - 1)
        for ndir in range(self.K - 1):
            a = np.zeros(self.n_fuzzy)
            a[:24] = action_scaled[ndir * 24:(ndir + 1) * 24]
            a[-8:] = action_scaled[-8:]
            self.ctrl.update_limits(a)
            self.ctrl.create_controller()
            priority[ndir] = self.ctrl.evaluate(
                self.azimuth[ndir], self.azimuth[-1],
                self.elevation[ndir], self.elevation[-1],
                self.separation[ndir], self.log_fluxes[ndir],
                flux_ratio[ndir])
            cutoff[ndir] = self.ctrl.get_high_priority()
        indices = np.where(priority >= cutoff)
        self.clus_id = np.unique(indices[0]).tolist() if len(indices) else []
        self.clus_id.append(self.K - 1)
        Kselected = len(self.clus_id)
        self.std_residual, sol = self._calibrate_subset(self.clus_id,
                                                        self.maxiter)
        infdata = self._influence_map(sol)
        md = self.metadata_vec.copy()
        md[4 * self.K:5 * self.K] = 0
        for ci in self.clus_id:
            md[4 * self.K + ci] = 1
        self.metadata_vec = md
        observation = {"infmap": infdata * INF_SCALE,
                       "metadata": md * META_SCALE}
        reward = self.calculate_reward_(Kselected) - self.reward0
        info = {}
        if self.provide_hint:
            if self.hint is None:
                self.hint = self.get_hint()
            return observation, reward, done, self.hint, info
        return observation, reward, done, info

    def get_hint(self) -> np.ndarray:
        """Default fuzzy config as the action
        (`demixing_fuzzy/demixingenv.py:324-333`)."""
        hint_full = np.zeros(self.n_action)
        hint = DemixController(n_action=self.n_fuzzy).update_action()
        for ndir in range(self.K - 1):
            hint_full[24 * ndir:24 * (ndir + 1)] = hint[:24]
        hint_full[-8:] = hint[-8:]
        return 2.0 * (hint_full - 0.5)
