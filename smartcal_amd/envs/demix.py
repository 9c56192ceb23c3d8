"""DemixingEnv — direction-selection environment.

Re-design of the reference `demixing_rl/demixingenv.py:36-390`: same
action/observation/reward semantics (direction selection by probability
> 0.5 with the target always included, max-ADMM-iteration knob, −AIC
reward with the reference's normalization constants, exhaustive
2^(K−1) softmin hint), with the external LOFAR simulation + sagecal-mpi
pipeline replaced by the in-memory device pipeline (`radio.sim`,
`radio.solver`, `radio.influence`). Coherencies for all K directions are
predicted once per episode; selecting a direction subset is a tensor
index — where the reference rewrites cluster text files and re-runs
`mpirun`, a step here is a pure GPU compute graph.
"""

from __future__ import annotations

import itertools

import numpy as np
import torch

from .. import gymapi
from ..gymapi import Box, Dict as DictSpace
from ..radio import array as arr
from ..radio import sim as rsim
from ..radio import solver as rsolver
from ..radio import influence as rinf
from ..radio import imaging as rimg
from ..radio.coherency import predict_coherencies_uvw
from ..radio.sky import ClusterSet

LOW, HIGH = 0.0, 1.0
LOW_ITER, HIGH_ITER = 5, 30
INF_SCALE = 1e-3
META_SCALE = 1e-3
EPS = 0.01


class DemixingEnv(gymapi.Env):
    """In-memory demixing environment (`demixing_rl/demixingenv.py:36`)."""

    metadata = {"render.modes": ["human"]}

    def __init__(self, K: int = 6, Nf: int = 3, Ninf: int = 128,
                 Npix: int = 1024, Tdelta: int = 10, provide_hint=False,
                 provide_influence=False, N_stations: int = 26,
                 Ts: int = 2, poly_order: int = 2, device=None,
                 seed: int | None = None):
        super().__init__()
        self.K = K
        self.Nf = Nf
        self.Ninf = Ninf
        self.Npix = Npix
        self.Tdelta = Tdelta
        self.Ts = Ts
        self.poly_order = poly_order
        self.provide_hint = provide_hint
        self.provide_influence = provide_influence
        self.Nst = N_stations
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        self.rng = np.random.default_rng(seed)
        self.action_space = Box(low=-1.0, high=1.0, shape=(K,))
        self.observation_space = DictSpace({
            "infmap": Box(low=-np.inf, high=np.inf, shape=(Ninf, Ninf)),
            "metadata": Box(low=-np.inf, high=np.inf,
                            shape=(3 * K + 2, 1)),
        })
        self.rho = np.ones(K, np.float32)
        self.elevation = None
        self.metadata_vec = np.zeros(3 * K + 2, np.float32)
        self.N = N_stations
        self.prev_clus_id = None
        self.reward0 = 0.0
        self.std_data = 0.0
        self.std_residual = 0.0
        self.maxiter = 10
        self.hint = None
        self.tau = 100.0
        self.clus_id = [K - 1]
        self._scenario = None

    # -- internal pipeline -------------------------------------------------

    def _simulate_episode(self):
        sky, cs, sep, az, el, fluxes, ra0, dec0 = \
            rsim.make_demixing_sky(self.rng, n_outliers=self.K - 1)
        assert len(cs) == self.K
        layout = arr.lofar_like_layout(self.Nst, self.rng)
        freqs = np.linspace(115e6, 185e6, self.Nf)
        snr = self.rng.random() * (0.5 - 0.05) + 0.05  # generate_data.py:1221
        vis = rsim.simulate_observation(
            layout, sky, cs, freqs, ra0, dec0, self.Ts, self.Tdelta,
            snr=snr, device=self.device, rng=self.rng,
            torch_seed=int(self.rng.integers(2 ** 31)))
        C = torch.stack([
            predict_coherencies_uvw(sky, cs, vis.uvw, float(f), ra0, dec0,
                                    smear_bw=180e3) for f in freqs])
        self._scenario = dict(sky=sky, clusters=cs, vis=vis, C=C)
        return sep, az, el, freqs, fluxes

    def _calibrate_subset(self, clus_id, maxiter):
        """Calibrate with only the selected directions; returns
        (std_residual, solution)."""
        sc = self._scenario
        vis: rsim.VisData = sc["vis"]
        sel = sorted(clus_id)
        cs_sub = ClusterSet([sc["clusters"][i] for i in sel])
        C_sub = sc["C"][:, sel]
        sol = rsolver.calibrate(
            vis, sc["sky"], cs_sub, self.rho[sel],
            admm_iter=int(maxiter), poly_order=self.poly_order,
            C_cache=C_sub)
        # std of residual Stokes I over samples — the reference estimates
        # this directly from the MODEL_DATA column (`demixingenv.py:241-252`)
        stds = [vis.stokes_i_std(sol.residual[fi]) for fi in range(self.Nf)]
        return float(np.sqrt(np.mean(np.square(stds)))), sol

    def _influence_map(self, sol):
        sc = self._scenario
        vis: rsim.VisData = sc["vis"]
        if not self.provide_influence:
            return np.zeros((1, self.Ninf, self.Ninf), np.float32)
        fi = self.Nf // 2
        sel = sorted(self.clus_id)
        Ksub = len(sel)
        Hadd = rinf.hadd_for(Ksub, vis.N, self.poly_order, vis.freqs,
                             float(np.mean(vis.freqs)), fi,
                             self.rho[sel], None, self.device)
        vals = rinf.influence_values(sol.residual[fi], sc["C"][fi, sel],
                                     sol.J_ref_layout(fi), vis.N,
                                     vis.Tdelta, Hadd)
        sI = 0.5 * (vals[:, 0] + vals[:, 3])
        img = rimg.dirty_image(vis.uvw, sI, float(vis.freqs[fi]), self.Ninf)
        return img.unsqueeze(0).cpu().numpy().astype(np.float32)

    def calculate_reward_(self, Kselected: int) -> float:
        """−AIC reward with the reference's normalization
        (`demixingenv.py:338-355`)."""
        data_var = self.std_data ** 2
        noise_var = self.std_residual ** 2
        reward = -self.N * self.N * noise_var / (data_var + EPS) \
            - Kselected * self.N
        reward = (reward - (-859)) / 3559.0
        penalty = -self.maxiter / 100.0
        return reward + penalty

    # -- gym API -----------------------------------------------------------

    def reset(self):
        sep, az, el, freqs, fluxes = self._simulate_episode()
        vis = self._scenario["vis"]
        self.N = vis.N
        self.elevation = el
        self.freq_low = freqs[0] / 1e6
        self.freq_high = freqs[-1] / 1e6
        self.rho = np.ones(self.K, np.float32)
        self.clus_id = [self.K - 1]
        self.maxiter = 10
        self.std_data = np.sqrt(np.mean(np.square(
            [vis.stokes_i_std(vis.data[fi]) for fi in range(self.Nf)])))
        self.std_residual, sol = self._calibrate_subset(self.clus_id,
                                                        self.maxiter)
        self.reward0 = self.calculate_reward_(1)
        md = np.zeros(3 * self.K + 2, np.float32)
        md[:self.K] = sep
        md[self.K:2 * self.K] = az
        md[2 * self.K:3 * self.K] = el
        md[-2] = np.log(freqs[0])
        md[-1] = self.N
        self.metadata_vec = md
        infdata = self._influence_map(sol)
        self.prev_clus_id = list(self.clus_id)
        self.hint = None
        return {"infmap": infdata * INF_SCALE,
                "metadata": md * META_SCALE}

    def step(self, action):
        action = np.asarray(action, np.float32).squeeze()
        action_rho = action[:self.K - 1]
        action_maxiter = float(action[self.K - 1])
        done = False
        rho = action_rho * (HIGH - LOW) / 2 + (HIGH + LOW) / 2
        self.maxiter = int(action_maxiter * (HIGH_ITER - LOW_ITER) / 2
                           + (HIGH_ITER + LOW_ITER) / 2)
        indices = np.where(rho > 0.5)
        self.clus_id = np.unique(indices[0]).tolist()
        self.clus_id.append(self.K - 1)
        if self.prev_clus_id != self.clus_id:
            self.prev_clus_id = list(self.clus_id)
        Kselected = len(self.clus_id)
        self.std_residual, sol = self._calibrate_subset(self.clus_id,
                                                        self.maxiter)
        infdata = self._influence_map(sol)
        metadata_update = self.metadata_vec.copy()
        metadata_update[self.clus_id] = 0
        observation = {"infmap": infdata * INF_SCALE,
                       "metadata": metadata_update * META_SCALE}
        reward = self.calculate_reward_(Kselected) - self.reward0
        info = {}
        if self.provide_hint:
            if self.hint is None:
                self.hint = self.get_hint()
            return observation, reward, done, self.hint, info
        return observation, reward, done, info

    @staticmethod
    def scalar_to_kvec(n: int, K: int = 5) -> np.ndarray:
        """Integer → K-bit selection vector (`demixingenv.py:293-299`)."""
        ll = [1 if digit == "1" else 0 for digit in bin(n)[2:]]
        a = np.zeros(K)
        a[-len(ll):] = ll
        return a

    def get_hint(self) -> np.ndarray:
        """Exhaustive 2^(K−1) sweep → AIC softmin expectation
        (`demixingenv.py:301-336`)."""
        AIC = np.zeros(2 ** (self.K - 1))
        for index in range(2 ** (self.K - 1)):
            action = self.scalar_to_kvec(index, self.K - 1)
            chosen_el = itertools.compress(self.elevation[:-1], action)
            if any(x < 1 for x in chosen_el):
                AIC[index] = 1e5
                continue
            clus_id = np.unique(np.where(action > 0)[0]).tolist()
            clus_id.append(self.K - 1)
            Kselected = len(clus_id)
            std_residual, _ = self._calibrate_subset(clus_id, self.maxiter)
            AIC[index] = (self.N * std_residual
                          / max(self.std_data, 1e-12)) ** 2 \
                + Kselected * self.N
        probs = np.exp(-AIC / self.tau)
        probs = probs / probs.sum()
        hint = np.zeros(self.K - 1)
        for ci in range(2 ** (self.K - 1)):
            hint += probs[ci] * self.scalar_to_kvec(ci, self.K - 1)
        hint = (hint - (HIGH + LOW) / 2) * (2 / (HIGH - LOW))
        hint_full = np.zeros(self.K)
        hint_full[:self.K - 1] = hint
        hint_full[self.K - 1] = (self.maxiter - (HIGH_ITER + LOW_ITER) / 2) \
            * (2 / (HIGH_ITER - LOW_ITER))
        return hint_full

    def render(self, mode="human"):
        print("clusters:", self.clus_id, "maxiter:", self.maxiter)

    def close(self):
        pass
