"""Re-record the hint arms (+1 sac seed) with the EE projection in."""
import json, sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
sys.argv = ["r2_curves.py", sys.argv[1] if len(sys.argv) > 1 else "1000"]
import importlib
spec = importlib.util.spec_from_file_location(
    "r2c", Path(__file__).parent / "r2_curves.py")
r2c = importlib.util.module_from_spec(spec)
spec.loader.exec_module(r2c)
r2c.out = {"nan_episodes": {}}          # fresh: do not resume
r2c.run_arm("sac_hint", "sac", True, (1, 2, 3))
r2c.run_arm("td3_hint", "td3", True, (1, 2, 3))
r2c.run_arm("sac", "sac", False, (1,))
r2c.OUT.write_text(json.dumps(r2c.out))
print("HINT ARMS DONE", flush=True)
