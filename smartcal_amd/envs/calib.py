"""CalibEnv — ADMM-regularization tuning environment.

Re-design of the reference `calibration/calibenv.py:30-243`: identical
observation/action/reward semantics, but the whole pipeline that the
reference drives through shell scripts and external binaries
(`dosimul.sh` → `docal.sh` (mpirun sagecal-mpi_gpu) → `doinfluence.sh` →
FITS files) runs in-process on device: `radio.sim` simulates the
observation, `radio.solver.calibrate` is the consensus-ADMM solver,
`radio.influence` + `radio.imaging` produce the 128² influence map and
data/residual images. Coherencies are predicted once per episode and
cached in HBM; a step is pure GPU compute with no file or process
boundary.

Action:  (2M,) in [-1,1] → K spectral + K spatial ADMM rho in
         [LOW, HIGH] (`calibenv.py:21-22,121`).
Obs:     {'img': (1, Ninf, Ninf) influence map ×1e-3,
          'sky': (M+1, 7) metadata ×1e-3} (`calibenv.py:53-56,164-166`).
Reward:  σ_data/σ_res + 1e-4/(σ_inf+EPS) + bound-penalty
         (`calibenv.py:170`).
"""

from __future__ import annotations


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

LOW = 0.01
HIGH = 1000.0
INF_SCALE = 1e-3
META_SCALE = 1e-3
EPS = 0.01


class CalibEnv(gymapi.Env):
    """In-memory calibration-tuning environment (`calibenv.py:30`)."""

    metadata = {"render.modes": ["human"]}

    def __init__(self, M: int = 5, provide_hint: bool = False,
                 N_stations: int = 26, Nf: int = 8, Ts: int = 2,
                 Tdelta: int = 10, Ninf: int = 128, admm_iter: int = 6,
                 poly_order: int = 3, snr: float = 5.0, device=None,
                 inf_nfreq: int = 2, seed: int | None = None):
        super().__init__()
        self.M = M
        self.K = 0
        self.provide_hint = provide_hint
        self.hint = None
        self.Ninf = Ninf
        self.Nf = Nf
        self.Ts = Ts
        self.Tdelta = Tdelta
        self.N = N_stations
        self.admm_iter = admm_iter
        self.poly_order = poly_order
        self.snr = snr
        self.inf_nfreq = min(inf_nfreq, Nf)
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        self.rng = np.random.default_rng(seed)
        self.action_space = Box(low=-1.0, high=1.0, shape=(2 * M,))
        self.observation_space = DictSpace({
            "img": Box(low=-HIGH, high=HIGH, shape=(1, Ninf, Ninf)),
            "sky": Box(low=-HIGH, high=HIGH, shape=(M + 1, 7)),
        })
        self.rho_spectral = np.ones(M, np.float32)
        self.rho_spatial = np.ones(M, np.float32)
        self.sky_meta = None
        self._scenario = None

    # -- internal pipeline -------------------------------------------------

    def _simulate_episode(self):
        (sky, cs_sim, sky_cal, cs_cal, skylmn, rho0, ra0, dec0) = \
            rsim.make_calibration_sky(self.K, self.rng)
        layout = arr.lofar_like_layout(self.N, self.rng)
        freqs = np.linspace(115e6, 185e6, self.Nf)
        vis = rsim.simulate_observation(
            layout, sky, cs_sim, freqs, ra0, dec0, self.Ts, self.Tdelta,
            snr=self.snr, device=self.device, rng=self.rng,
            torch_seed=int(self.rng.integers(2 ** 31)))
        # cache calibration-model coherencies per freq (constant per episode)
        C = torch.stack([
            predict_coherencies_uvw(sky_cal, cs_cal, vis.uvw, float(f),
                                    ra0, dec0, smear_bw=180e3)
            for f in freqs])
        self._scenario = dict(sky=sky_cal, clusters=cs_cal, vis=vis,
                              C=C, skylmn=skylmn, rho0=rho0,
                              f_low=freqs[0] / 1e6, f_high=freqs[-1] / 1e6)

    def _calibrate_and_observe(self):
        sc = self._scenario
        vis: rsim.VisData = sc["vis"]
        K = self.K
        sol = rsolver.calibrate(
            vis, sc["sky"], sc["clusters"], self.rho_spectral[:K],
            admm_iter=self.admm_iter, poly_order=self.poly_order,
            alpha=float(np.mean(self.rho_spatial[:K])),
            C_cache=sc["C"])
        # data / residual dirty images (σ over the mean image, à la
        # calmean + fits std in `calibenv.py:148-158`)
        imgs_d, imgs_r, imgs_i = [], [], []
        inf_idx = np.linspace(0, self.Nf - 1, self.inf_nfreq).astype(int)
        f0 = float(np.mean(vis.freqs))
        for fi in range(self.Nf):
            f = float(vis.freqs[fi])
            sI_d = 0.5 * (vis.data[fi][:, 0] + vis.data[fi][:, 3])
            sI_r = 0.5 * (sol.residual[fi][:, 0] + sol.residual[fi][:, 3])
            imgs_d.append(rimg.dirty_image(vis.uvw, sI_d, f, self.Ninf))
            imgs_r.append(rimg.dirty_image(vis.uvw, sI_r, f, self.Ninf))
        for fi in inf_idx:
            f = float(vis.freqs[fi])
            Hadd = rinf.hadd_for(K, vis.N, self.poly_order, vis.freqs, f0,
                                 int(fi), self.rho_spectral[:K],
                                 self.rho_spatial[:K], self.device)
            vals = rinf.influence_values(sol.residual[fi], sc["C"][fi],
                                         sol.J_ref_layout(int(fi)), vis.N,
                                         vis.Tdelta, Hadd)
            sI_i = 0.5 * (vals[:, 0] + vals[:, 3])
            imgs_i.append(rimg.dirty_image(vis.uvw, sI_i, f, self.Ninf))
        data_img = rimg.weighted_mean_image(imgs_d, vis.freqs)
        res_img = rimg.weighted_mean_image(imgs_r, vis.freqs)
        inf_img = rimg.weighted_mean_image(imgs_i, vis.freqs[inf_idx])
        return data_img, res_img, inf_img

    def _obs(self, inf_img) -> dict:
        img = inf_img.unsqueeze(0).cpu().numpy().astype(np.float32)
        return {"img": img * INF_SCALE,
                "sky": self.sky_meta * META_SCALE}

    def _update_meta_rho(self):
        self.sky_meta[:self.K, 5] = (self.rho_spectral[:self.K]
                                     - (HIGH + LOW) / 2) * (2 / (HIGH - LOW))
        self.sky_meta[:self.K, 6] = (self.rho_spatial[:self.K]
                                     - (HIGH + LOW) / 2) * (2 / (HIGH - LOW))

    # -- gym API -----------------------------------------------------------

    def reset(self):
        self.K = int(self.rng.integers(2, self.M + 1))
        self._simulate_episode()
        sc = self._scenario
        rho0 = sc["rho0"]
        self.rho_spectral[:self.K] = np.clip(rho0, LOW, HIGH)
        self.rho_spatial[:self.K] = np.clip(0.05 * rho0, LOW, HIGH)
        self.sky_meta = np.zeros((self.M + 1, 7), np.float32)
        self.sky_meta[:self.K, :5] = sc["skylmn"][:self.K]
        self.sky_meta[-1, :5] = [sc["vis"].ra0, sc["vis"].dec0, self.K,
                                 sc["f_low"] / 1000.0, sc["f_high"] / 1000.0]
        self._update_meta_rho()
        _, _, inf_img = self._calibrate_and_observe()
        if self.provide_hint:
            self.hint = np.zeros(2 * self.M, np.float32)
            self.hint[:self.K] = (self.rho_spectral[:self.K]
                                  - (HIGH + LOW) / 2) * (2 / (HIGH - LOW))
            self.hint[self.M:self.M + self.K] = \
                (0.05 * self.rho_spectral[:self.K]
                 - (HIGH + LOW) / 2) * (2 / (HIGH - LOW))
        return self._obs(inf_img)

    def step(self, action):
        action = np.asarray(action, np.float32).squeeze()
        assert action.shape[0] == 2 * self.M
        done = False
        rho = action * (HIGH - LOW) / 2 + (HIGH + LOW) / 2
        self.rho_spectral[:self.K] = rho[:self.K]
        self.rho_spatial[:self.K] = rho[self.M:self.M + self.K]
        penalty = 0.0
        for ci in range(self.K):
            for arr_ in (self.rho_spectral, self.rho_spatial):
                if arr_[ci] < LOW:
                    arr_[ci] = LOW
                    penalty += -0.1
                if arr_[ci] > HIGH:
                    arr_[ci] = HIGH
                    penalty += -0.1
        data_img, res_img, inf_img = self._calibrate_and_observe()
        sigma0 = float(data_img.std())
        sigma1 = float(res_img.std())
        sigma_inf = float(inf_img.std())
        self._update_meta_rho()
        observation = self._obs(inf_img)
        reward = sigma0 / max(sigma1, 1e-12) + 1e-4 / (sigma_inf + EPS) \
            + penalty
        info = {}
        if self.provide_hint:
            return observation, reward, done, self.hint, info
        return observation, reward, done, info

    def render(self, mode="human"):
        print(self.rho_spectral, self.rho_spatial)

    def close(self):
        pass
