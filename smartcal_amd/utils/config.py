"""Typed configuration (SURVEY.md §5 'config/flag system').

The reference scatters knobs across argparse, module constants and
hardcoded shell strings (`generate_data.py:13-30`, `docal.sh:12`,
`enetenv.py:21-22`). Here one dataclass tree holds them, loadable from
YAML/dict and overridable from CLI-style ``key=value`` pairs.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional

__all__ = ["EnvConfig", "AgentConfig", "SolverConfig", "TrainConfig",
           "load_config"]


@dataclass
class SolverConfig:
    admm_iter: int = 10          # -A
    poly_order: int = 2          # -P
    polytype: int = 1            # Bernstein
    alpha: float = 0.0           # spatial / federated regularization
    n_sweeps: int = 3
    init_sweeps: int = 6
    smear_bw: Optional[float] = 180e3


@dataclass
class EnvConfig:
    workload: str = "elasticnet"     # elasticnet|calibration|demixing|fuzzy
    N: int = 20
    M: int = 20
    K: int = 6
    Nf: int = 3
    Ninf: int = 128
    Tdelta: int = 10
    Ts: int = 2
    stations: int = 62
    provide_hint: bool = False
    provide_influence: bool = False
    snr: float = 5.0
    solver: SolverConfig = field(default_factory=SolverConfig)


@dataclass
class AgentConfig:
    algo: str = "sac"                # sac|td3|ddpg
    gamma: float = 0.99
    tau: float = 0.005
    lr_a: float = 1e-3
    lr_c: float = 1e-3
    batch_size: int = 64
    max_mem_size: int = 1024
    reward_scale: float = 20.0
    alpha: float = 0.03
    prioritized: bool = False
    use_hint: bool = False
    warmup: int = 100
    noise: float = 0.1
    update_actor_interval: int = 2


@dataclass
class TrainConfig:
    episodes: int = 1000
    steps: int = 5
    seed: int = 0
    save_every: int = 10
    env: EnvConfig = field(default_factory=EnvConfig)
    agent: AgentConfig = field(default_factory=AgentConfig)


def _from_dict(cls, d: dict):
    kwargs = {}
    for f in dataclasses.fields(cls):
        if f.name not in d:
            continue
        v = d[f.name]
        if dataclasses.is_dataclass(f.type) or f.name in ("env", "agent",
                                                          "solver"):
            sub = {"env": EnvConfig, "agent": AgentConfig,
                   "solver": SolverConfig}[f.name]
            v = _from_dict(sub, v)
        kwargs[f.name] = v
    return cls(**kwargs)


def load_config(path: str | None = None, overrides: list[str] | None = None
                ) -> TrainConfig:
    """Load a TrainConfig from YAML (optional) + dotted key=value
    overrides, e.g. ``env.stations=26 agent.lr_a=3e-4``."""
    d: dict = {}
    if path is not None:
        import yaml
        with open(path) as f:
            d = yaml.safe_load(f) or {}
    cfg = _from_dict(TrainConfig, d)
    for ov in overrides or []:
        key, _, val = ov.partition("=")
        obj = cfg
        parts = key.split(".")
        for p in parts[:-1]:
            obj = getattr(obj, p)
        cur = getattr(obj, parts[-1])
        typ = type(cur) if cur is not None else str
        if typ is bool:
            setattr(obj, parts[-1], val.lower() in ("1", "true", "yes"))
        else:
            setattr(obj, parts[-1], typ(val))
    return cfg
