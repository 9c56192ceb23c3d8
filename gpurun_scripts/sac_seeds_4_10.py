"""Extend the SAC no-hint curve set to 10 seeds (reference do.sh runs
10 seeds x {hint,nohint}); merges into gpurun_out/r2_curves.json."""
import json, sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
sys.argv = ["r2_curves.py", "1000"]
import importlib.util
spec = importlib.util.spec_from_file_location(
    "r2c", Path(__file__).parent / "r2_curves.py")
r2c = importlib.util.module_from_spec(spec)
spec.loader.exec_module(r2c)
r2c.out = {"nan_episodes": {}}
r2c.run_arm("sac", "sac", False, (4, 5, 6, 7, 8, 9, 10))
r2c.run_arm("sac_hint", "sac", True, (4, 5, 6, 7))
print("EXTRA SEEDS DONE", flush=True)
