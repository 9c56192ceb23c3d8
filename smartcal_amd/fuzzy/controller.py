"""Trapezoidal fuzzy controller for demixing priority — from scratch.

Same controller as the reference (`demixing_fuzzy/demix_controller.py:
6-263`): 7 antecedents (azimuth/elevation × {outlier, target},
separation, log flux, flux ratio), 1 consequent (priority 0..100), the
same default trapezoid breakpoints, the same action→breakpoint morphing
(`update_set_:95-111` and inverse `update_action_:113-122`), and the
same 13 rules. The reference delegates inference to scikit-fuzzy
(not in this image); here Mamdani inference (min-AND / max-OR, max
aggregation, centroid defuzzification — skfuzzy's defaults) is
implemented directly, with the reference's fallback priority=50 when no
rule fires (`demix_controller.py:240-246`).
"""

from __future__ import annotations

import copy
import json

import numpy as np

__all__ = ["DemixController", "trapmf"]


def trapmf(x: np.ndarray, abcd) -> np.ndarray:
    """Trapezoidal membership function (skfuzzy.trapmf semantics)."""
    a, b, c, d = [float(v) for v in abcd]
    x = np.asarray(x, dtype=np.float64)
    y = np.zeros_like(x)
    if b > a:
        idx = (x > a) & (x < b)
        y[idx] = (x[idx] - a) / (b - a)
    y[(x >= b) & (x <= c)] = 1.0
    if d > c:
        idx = (x > c) & (x < d)
        y[idx] = (d - x[idx]) / (d - c)
    return y


def _default_config():
    """Default membership limits (`demix_controller.py:19-93`)."""
    n_var = 0

    def mk(range_, low, medium, high):
        return {"range": list(range_), "low": list(low),
                "medium": list(medium), "high": list(high)}

    azimuth = mk([-180, 180, 1], [-180, -180, -65, -55],
                 [-65, -55, 55, 65], [55, 65, 180, 180]); n_var += 4
    azimuth_target = copy.deepcopy(azimuth); n_var += 4
    elevation = mk([-90, 90, 1], [-90, -90, -5, 5],
                   [-5, 5, 50, 60], [50, 60, 90, 90]); n_var += 4
    elevation_target = copy.deepcopy(elevation); n_var += 4
    separation = mk([0, 180, 1], [0, 0, 10, 15],
                    [10, 15, 45, 50], [45, 50, 180, 180]); n_var += 4
    logI = mk([0, 100, 1], [0, 0, 1.0, 2.0],
              [1.0, 2.0, 5.0, 10], [5.0, 10, 100, 100]); n_var += 4
    ratI = mk([0, 100, 1], [0, 0, 0.5, 1.0],
              [0.5, 1.0, 50, 55], [50, 55, 100, 100]); n_var += 4
    priority = mk([0, 100, 1], [0, 0, 40, 50],
                  [40, 50, 70, 75], [70, 75, 100, 100]); n_var += 4

    cfg = {"inputs": {"_azimuth": azimuth,
                      "_azimuth_target": azimuth_target,
                      "_elevation": elevation,
                      "_elevation_target": elevation_target,
                      "_separation": separation,
                      "_log_intensity": logI,
                      "_intensity_ratio": ratI},
           "outputs": {"_priority": priority},
           "_comment": "membership limits (auto-generated)"}
    return cfg, n_var


# the 13 rules (`demix_controller.py:201-224`): each is
# (list of (var, term) AND-ed | "OR" marker, consequent term)
_RULES = [
    ("and", [("azimuth", "low"), ("azimuth_target", "low")], "medium"),
    ("and", [("azimuth", "medium"), ("azimuth_target", "medium")], "medium"),
    ("and", [("azimuth", "high"), ("azimuth_target", "high")], "medium"),
    ("and", [("separation", "low")], "high"),
    ("and", [("elevation", "low")], "low"),
    ("and", [("elevation", "low"), ("separation", "high"),
             ("log_intensity", "low"), ("intensity_ratio", "low")], "low"),
    ("and", [("elevation", "medium"), ("separation", "medium"),
             ("intensity_ratio", "high")], "medium"),
    ("and", [("elevation", "high"), ("separation", "medium"),
             ("intensity_ratio", "high")], "high"),
    ("and", [("elevation", "high"), ("log_intensity", "high"),
             ("intensity_ratio", "high")], "high"),
    ("or", [("elevation", "medium"), ("separation", "medium"),
            ("log_intensity", "medium"), ("intensity_ratio", "medium")],
     "medium"),
    ("and", [("elevation_target", "low"), ("elevation", "high")], "high"),
    ("and", [("elevation_target", "high"), ("elevation", "low")], "low"),
    ("and", [("elevation_target", "medium"), ("elevation", "high")],
     "medium"),
]


class DemixController:
    """Fuzzy demixing-priority controller (32 tunable breakpoints)."""

    def __init__(self, n_action: int = 32):
        self.n_action = n_action
        self.config, self.n_var = _default_config()
        assert self.n_action == self.n_var
        self._universe = None
        self._cons_mfs = None

    # -- action ↔ breakpoint morphing --------------------------------------
    @staticmethod
    def update_set_(fuzzy_set, action):
        upper = fuzzy_set["range"][1]
        fuzzy_set["low"][2] = fuzzy_set["low"][1] \
            + action[0] * (upper - fuzzy_set["low"][1])
        fuzzy_set["low"][3] = fuzzy_set["low"][2] \
            + action[1] * (upper - fuzzy_set["low"][2])
        fuzzy_set["medium"][0] = fuzzy_set["low"][2]
        fuzzy_set["medium"][1] = fuzzy_set["low"][3]
        fuzzy_set["medium"][2] = fuzzy_set["medium"][1] \
            + action[2] * (upper - fuzzy_set["medium"][1])
        fuzzy_set["medium"][3] = fuzzy_set["medium"][2] \
            + action[3] * (upper - fuzzy_set["medium"][2])
        fuzzy_set["high"][0] = fuzzy_set["medium"][2]
        fuzzy_set["high"][1] = fuzzy_set["medium"][3]

    @staticmethod
    def update_action_(fuzzy_set, action):
        upper = fuzzy_set["range"][1]
        action[0] = (fuzzy_set["low"][2] - fuzzy_set["low"][1]) \
            / (upper - fuzzy_set["low"][1])
        action[1] = (fuzzy_set["low"][3] - fuzzy_set["low"][2]) \
            / (upper - fuzzy_set["low"][2])
        action[2] = (fuzzy_set["medium"][2] - fuzzy_set["medium"][1]) \
            / (upper - fuzzy_set["medium"][1])
        action[3] = (fuzzy_set["medium"][3] - fuzzy_set["medium"][2]) \
            / (upper - fuzzy_set["medium"][2])

    _ORDER = ["_azimuth", "_elevation", "_separation", "_log_intensity",
              "_intensity_ratio", "_priority", "_azimuth_target",
              "_elevation_target"]

    def update_limits(self, action: np.ndarray):
        assert action.size == self.n_var
        io = {**self.config["inputs"], **self.config["outputs"]}
        for i, name in enumerate(self._ORDER):
            self.update_set_(io[name], action[4 * i:4 * i + 4])

    def update_action(self) -> np.ndarray:
        action = np.zeros(self.n_var)
        io = {**self.config["inputs"], **self.config["outputs"]}
        for i, name in enumerate(self._ORDER):
            self.update_action_(io[name], action[4 * i:4 * i + 4])
        return action

    # -- inference ---------------------------------------------------------
    def create_controller(self):
        pr = self.config["outputs"]["_priority"]
        self._universe = np.arange(*pr["range"], dtype=np.float64)
        self._cons_mfs = {t: trapmf(self._universe, pr[t])
                          for t in ("low", "medium", "high")}

    def _membership(self, var: str, term: str, value: float) -> float:
        fs = self.config["inputs"]["_" + var]
        return float(trapmf(np.array([value]), fs[term])[0])

    def evaluate(self, azimuth, azimuth_target, elevation, elevation_target,
                 separation, log_intensity, intensity_ratio) -> float:
        if self._cons_mfs is None:
            self.create_controller()
        vals = {"azimuth": azimuth, "azimuth_target": azimuth_target,
                "elevation": elevation, "elevation_target": elevation_target,
                "separation": separation, "log_intensity": log_intensity,
                "intensity_ratio": intensity_ratio}
        agg = np.zeros_like(self._universe)
        for op, terms, cons in _RULES:
            ms = [self._membership(v, t, vals[v]) for v, t in terms]
            strength = min(ms) if op == "and" else max(ms)
            if strength > 0:
                np.maximum(agg, np.minimum(strength, self._cons_mfs[cons]),
                           out=agg)
        s = agg.sum()
        if s <= 0:
            # no rule fired — reference fallback (`demix_controller.py:246`)
            return 50.0
        return float((agg * self._universe).sum() / s)

    def get_high_priority(self) -> float:
        return self.config["outputs"]["_priority"]["high"][0]

    def print_config(self, filename=None):
        if filename:
            with open(filename, "w+") as f:
                json.dump(self.config, f)
        else:
            print(self.config)
