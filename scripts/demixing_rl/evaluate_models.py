"""Compare hint-trained / no-hint / untrained agents and the hint itself.

Reference `demixing_rl/evaluate_models.py:12-86`: run live env episodes
with each policy and report mean rewards.
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.rl.sac_cnn import Agent


def run_policy(env, policy, episodes, steps):
    rewards = []
    for _ in range(episodes):
        obs = env.reset()
        for _ in range(steps):
            a = policy(obs)
            obs, r, done, *_ = env.step(a)
            rewards.append(float(r))
            if done:
                break
    return float(np.mean(rewards))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--episodes", default=3, type=int)
    ap.add_argument("--steps", default=5, type=int)
    ap.add_argument("--stations", default=26, type=int)
    ap.add_argument("--hint_dir", default=None,
                    help="checkpoint dir of a hint-trained agent")
    ap.add_argument("--nohint_dir", default=None)
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()

    K = 6
    env = DemixingEnv(K=K, Nf=3, Ninf=128, provide_hint=True,
                      provide_influence=False, N_stations=args.stations,
                      seed=args.seed)

    def make_agent(ckpt_dir):
        a = Agent(gamma=0.99, batch_size=256, n_actions=K, tau=0.005,
                  max_mem_size=16, input_dims=(1, 128, 128),
                  meta_dim=3 * K + 2, lr_a=3e-4, lr_c=3e-4,
                  checkpoint_dir=ckpt_dir or "./")
        if ckpt_dir:
            a.load_models_for_eval()
        return a

    results = {}
    results["untrained"] = run_policy(
        env, make_agent(None).choose_action, args.episodes, args.steps)
    if args.hint_dir:
        results["hint-trained"] = run_policy(
            env, make_agent(args.hint_dir).choose_action,
            args.episodes, args.steps)
    if args.nohint_dir:
        results["nohint-trained"] = run_policy(
            env, make_agent(args.nohint_dir).choose_action,
            args.episodes, args.steps)

    def hint_policy(obs):
        h = env.get_hint() if env.hint is None else env.hint
        return np.asarray(h, np.float32)

    results["hint-itself"] = run_policy(env, hint_policy,
                                        args.episodes, args.steps)
    for k, v in results.items():
        print(f"{k}: mean reward {v:.4f}")


if __name__ == "__main__":
    main()
