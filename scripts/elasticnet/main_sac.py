"""Elastic-net SAC training entry point.

CLI-compatible with reference ``elasticnet/main_sac.py`` (``--seed
--episodes --steps --use_hint``), same canonical config (N=M=20,
input_dims=[N+N*M], batch 64, mem 1024, gamma .99, tau .005, lr 1e-3,
reward_scale=N, alpha .03).
"""

import argparse
import pickle
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything


def main():
    parser = argparse.ArgumentParser(
        description="Elastic net regression hyperparameter tuning (SAC)",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("--seed", default=0, type=int)
    parser.add_argument("--episodes", default=1000, type=int)
    parser.add_argument("--steps", default=5, type=int)
    parser.add_argument("--use_hint", action="store_true", default=False)
    args = parser.parse_args()

    seed_everything(args.seed)

    N = 20  # rows = data points
    M = 20  # columns = parameters
    provide_hint = args.use_hint
    env = ENetEnv(M, N, provide_hint=provide_hint)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03, prioritized=False,
                  use_hint=provide_hint)

    scores = []
    for i in range(args.episodes):
        score = 0.0
        done = False
        observation = env.reset()
        loop = 0
        while (not done) and loop < args.steps:
            action = agent.choose_action(observation)
            if provide_hint:
                observation_, reward, done, hint, info = env.step(action)
                agent.store_transition(observation, action, reward,
                                       observation_, done, hint)
            else:
                observation_, reward, done, info = env.step(action)
                agent.store_transition(observation, action, reward,
                                       observation_, done,
                                       np.zeros_like(action))
            score += float(reward)
            agent.learn()
            observation = observation_
            loop += 1
        score /= loop
        scores.append(score)
        avg_score = np.mean(scores[-100:])
        print(f"episode {i} score {score:.2f} average score {avg_score:.2f}")
        if i % 500 == 0:
            agent.save_models()

    with open("scores.pkl", "wb") as f:
        pickle.dump(scores, f)


if __name__ == "__main__":
    main()
