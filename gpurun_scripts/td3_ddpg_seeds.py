"""Extend TD3/DDPG no-hint curves (seeds 4-6 each)."""
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
r2c.run_arm("td3", "td3", False, (4, 5, 6))
r2c.run_arm("ddpg", "ddpg", False, (4, 5, 6))
print("TD3/DDPG SEEDS DONE", flush=True)
