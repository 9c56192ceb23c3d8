"""Plot demix influence-on hint vs no-hint curves (local, on the JSON)."""
import json, sys
import matplotlib
matplotlib.use("Agg")
import matplotlib.pyplot as plt
import numpy as np

SRC = sys.argv[1] if len(sys.argv) > 1 else "profiles/demix_influence_curves.json"
OUT = sys.argv[2] if len(sys.argv) > 2 else "profiles/demix_influence_curves.png"
d = json.load(open(SRC))
plt.figure(figsize=(7, 4))
W = 25
for arm, c in (("hint", "C0"), ("nohint", "C1")):
    if arm not in d:
        continue
    a = np.asarray(d[arm], float)
    sm = np.convolve(a, np.ones(W) / W, mode="valid")
    plt.plot(sm, color=c, label=f"{arm} ({len(a)} eps)")
    print(f"{arm}: n={len(a)} first50={np.mean(a[:50]):.4f} "
          f"last50={np.mean(a[-50:]):.4f}")
plt.xlabel("episode")
plt.ylabel(f"reward ({W}-episode moving average)")
plt.title("Demixing SAC, influence maps ON (CNN), 1x MI355X, N=62")
plt.legend(); plt.grid(alpha=0.3); plt.tight_layout()
plt.savefig(OUT, dpi=130)
