"""Shared training-loop helper for the entry-point scripts.

Implements the episode loop every reference main uses
(`elasticnet/main_sac.py:47-76`, `demixing_rl/main_sac.py:54-98`): roll
episodes, store transitions (hint-aware), learn per step, track scores,
periodic model saves, scores.pkl.
"""

from __future__ import annotations

import pickle

import numpy as np


def _try_enable_graph(agent):
    """Capture the learn step into one hipGraph when the agent/config
    supports it (GPU, plain replay, no hint/learnable-alpha). Safe no-op
    otherwise."""
    if getattr(agent, "_graph", None) is not None:
        return True
    enable = getattr(agent, "enable_cuda_graph", None)
    if enable is None:
        return False
    try:
        enable()
        return True
    except (AssertionError, RuntimeError, AttributeError):
        return False


def run_training(env, agent, episodes: int, steps: int,
                 provide_hint: bool = False, save_every: int = 10,
                 scores_file: str = "scores.pkl",
                 reward_shaping=None, warmup_episodes: int = 0,
                 verbose: bool = True, use_graph: bool = True):
    scores = []
    graph_tried = False
    for i in range(episodes):
        score = 0.0
        done = False
        observation = env.reset()
        loop = 0
        while (not done) and loop < steps:
            if i < warmup_episodes:
                action = env.action_space.sample().reshape(-1)
            else:
                action = agent.choose_action(observation)
            if provide_hint:
                observation_, reward, done, hint, info = env.step(action)
            else:
                observation_, reward, done, info = env.step(action)
                hint = np.zeros_like(action)
            if reward_shaping is not None:
                reward = reward_shaping(reward)
            agent.store_transition(observation, action, reward,
                                   observation_, done, hint)
            score += float(reward)
            if use_graph and not graph_tried \
                    and len(getattr(agent, "replaymem", ())) \
                    >= getattr(agent, "batch_size", 1):
                graph_tried = True
                _try_enable_graph(agent)
            agent.learn()
            observation = observation_
            loop += 1
        scores.append(score / max(loop, 1))
        if verbose:
            avg = np.mean(scores[-100:])
            print(f"episode {i} score {scores[-1]:.2f} "
                  f"average score {avg:.2f}")
        if save_every and i % save_every == 0:
            agent.save_models()
    with open(scores_file, "wb") as f:
        pickle.dump(scores, f)
    return scores
