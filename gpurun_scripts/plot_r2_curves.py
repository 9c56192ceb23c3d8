"""Plot the round-2 all-arms curves (run locally on the recorded JSON).

Usage: python gpurun_scripts/plot_r2_curves.py [in.json] [out_prefix]
Produces <out_prefix>.png (no-hint algo comparison, the reference
figures/comparison.png semantics) and <out_prefix>_hint.png (hint vs
no-hint + bf16 parity).
"""

import json
import sys

import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402
import numpy as np  # noqa: E402

SRC = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/r2_curves.json"
OUT = sys.argv[2] if len(sys.argv) > 2 else "profiles/r2_curves"
W = 25


def smooth(runs):
    sm = np.stack([np.convolve(np.asarray(s, float), np.ones(W) / W,
                               mode="valid") for s in runs.values()])
    return np.median(sm, axis=0), sm.min(0), sm.max(0)


def main():
    data = json.load(open(SRC))
    fig, ax = plt.subplots(figsize=(7, 4.5))
    for algo, c in (("sac", "C0"), ("td3", "C1"), ("ddpg", "C2")):
        med, lo, hi = smooth(data[algo])
        ax.plot(med, color=c, label=algo.upper())
        ax.fill_between(np.arange(len(med)), lo, hi, color=c, alpha=0.15)
    ax.set_xlabel("episode")
    ax.set_ylabel(f"score ({W}-episode moving average)")
    ax.set_title("Elastic-net tuning, 1x MI355X (median of 3 seeds, "
                 "round 2: analytic pair band)")
    ax.legend()
    ax.grid(alpha=0.3)
    lo_all = min(np.min(v) for a in ("sac", "td3", "ddpg")
                 for v in [smooth(data[a])[1]])
    ax.set_ylim(max(lo_all, -3), 3)
    fig.tight_layout()
    fig.savefig(f"{OUT}.png", dpi=130)

    fig, ax = plt.subplots(figsize=(7, 4.5))
    arms = (("sac", "C0", "SAC"), ("sac_hint", "C3", "SAC+hint(PER)"),
            ("td3", "C1", "TD3+PER"), ("td3_hint", "C4", "TD3+hint"),
            ("td3_per_bf16", "C5", "TD3+PER bf16"))
    for arm, c, lbl in arms:
        if arm not in data:
            continue
        med, lo, hi = smooth(data[arm])
        ax.plot(med, color=c, label=lbl)
        ax.fill_between(np.arange(len(med)), lo, hi, color=c, alpha=0.12)
    ax.set_xlabel("episode")
    ax.set_ylabel(f"score ({W}-episode moving average)")
    ax.set_title("Hint arms + bf16 parity (1x MI355X)")
    ax.legend(fontsize=8)
    ax.grid(alpha=0.3)
    ax.set_ylim(-3, 3)
    fig.tight_layout()
    fig.savefig(f"{OUT}_hint.png", dpi=130)

    # summary table
    for arm in data:
        if arm == "nan_episodes":
            continue
        for seed, s in data[arm].items():
            a = np.asarray(s, float)
            print(f"{arm:14s} seed {seed}: last100 mean "
                  f"{np.mean(a[-100:]):9.3f} median "
                  f"{np.median(a[-100:]):7.3f} min {a.min():10.1f} "
                  f"frac<-100 {(a < -100).mean():.3f}")


if __name__ == "__main__":
    main()
