"""Smoke tests for the CLI entry-point scripts (tiny configs)."""

import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

ROOT = Path(__file__).resolve().parents[1]


def _run(script, *args, cwd):
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    return subprocess.run([sys.executable, str(ROOT / script), *args],
                          cwd=cwd, env=env, capture_output=True, text=True,
                          timeout=600)


@pytest.mark.parametrize("script", [
    "scripts/elasticnet/main_sac.py",
    "scripts/elasticnet/main_td3.py",
    "scripts/elasticnet/main_ddpg.py",
    "scripts/calibration/main_sac.py",
    "scripts/calibration/main_td3.py",
    "scripts/calibration/main_ddpg.py",
    "scripts/demixing_rl/main_sac.py",
    "scripts/demixing_rl/main_td3.py",
    "scripts/demixing_fuzzy/main_sac.py",
    "scripts/demixing/simulate_data.py",
    "scripts/demixing/train_model.py",
    "scripts/demixing_rl/makedata.py",
    "scripts/demixing_rl/train_regressor.py",
    "scripts/demixing_rl/train_tsk.py",
    "scripts/demixing_rl/evaluate_models.py",
    "scripts/elasticnet/distributed_per_sac.py",
    "scripts/demixing_rl/distributed_per_sac.py",
])
def test_script_help(script, tmp_path):
    r = _run(script, "--help", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]


def test_enet_sac_short(tmp_path):
    r = _run("scripts/elasticnet/main_sac.py", "--episodes", "2",
             "--steps", "2", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "episode 1" in r.stdout
    assert (tmp_path / "scores.pkl").exists()


def test_calib_sac_short(tmp_path):
    r = _run("scripts/calibration/main_sac.py", "--episodes", "1",
             "--steps", "1", "--M", "3", "--stations", "8",
             cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]


def test_demix_fuzzy_short(tmp_path):
    r = _run("scripts/demixing_fuzzy/main_sac.py", "--episodes", "1",
             "--steps", "1", "--stations", "6", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]


def test_supervised_pipeline(tmp_path):
    """simulate_data → train_model end to end at toy size."""
    r = _run("scripts/demixing/simulate_data.py", "--samples", "2",
             "--ninf", "16", "--stations", "6", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    r = _run("scripts/demixing/train_model.py", "--iters", "3",
             "--batch", "2", "--ninf", "16", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    assert (tmp_path / "transformer.model").exists()


def test_distill_pipeline(tmp_path):
    """makedata → regressor + tsk training at toy size."""
    r = _run("scripts/demixing_rl/makedata.py", "--samples", "2",
             "--stations", "6", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    r = _run("scripts/demixing_rl/train_regressor.py", "--iters", "5",
             "--batch", "2", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    r = _run("scripts/demixing_rl/train_tsk.py", "--iters", "5",
             "--batch", "2", cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
