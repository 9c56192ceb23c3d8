"""Agent-level tests: SAC/TD3/DDPG learn steps, checkpoints, polyak."""

import numpy as np
import pytest
import torch

from smartcal_amd.rl import ddpg, sac, td3

CPU = torch.device("cpu")


def _fake_obs(n_state, n=8):
    # state layout: eig (n) + A (n_state - n)
    return {"eig": torch.randn(n), "A": torch.randn(n_state - n)}


def _fill_and_learn(agent, n_state, steps=3):
    obs = _fake_obs(n_state)
    for _ in range(agent.batch_size + 2):
        a = agent.choose_action(obs)
        obs2 = _fake_obs(n_state)
        agent.store_transition(obs, a, float(np.random.randn()), obs2,
                               False, np.zeros_like(a))
        obs = obs2
    before = agent.actor_fp.flat.clone()
    for _ in range(steps):
        agent.learn()
    return before


def test_sac_learn_updates_params(tmp_path):
    agent = sac.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                      batch_size=8, n_actions=2, max_mem_size=64,
                      tau=0.005, reward_scale=2, alpha=0.03, device=CPU,
                      checkpoint_dir=str(tmp_path))
    before = _fill_and_learn(agent, 24)
    assert not torch.allclose(before, agent.actor_fp.flat)
    # target critics track online critics
    assert not torch.allclose(agent.target_critic_1_fp.flat,
                              agent.critic_1_fp.flat)
    agent.save_models()
    agent2 = sac.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                       batch_size=8, n_actions=2, max_mem_size=64,
                       tau=0.005, reward_scale=2, alpha=0.03, device=CPU,
                       checkpoint_dir=str(tmp_path))
    agent2.load_models()
    assert torch.allclose(agent2.actor_fp.flat, agent.actor_fp.flat)
    # hard target sync after load (tau=1)
    assert torch.allclose(agent2.target_critic_1_fp.flat,
                          agent2.critic_1_fp.flat)


def test_sac_prioritized_and_hint(tmp_path):
    agent = sac.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                      batch_size=8, n_actions=2, max_mem_size=64,
                      tau=0.005, reward_scale=2, alpha=0.03,
                      prioritized=True, use_hint=True, device=CPU,
                      checkpoint_dir=str(tmp_path))
    _fill_and_learn(agent, 24, steps=11)
    assert agent.learn_counter == 11
    assert float(agent.rho) >= 0.0


def test_td3_learn_and_warmup():
    agent = td3.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                      batch_size=8, n_actions=2, max_mem_size=64,
                      tau=0.005, warmup=5, noise=0.1, prioritized=True,
                      device=CPU)
    obs = _fake_obs(24)
    acts = [agent.choose_action(obs) for _ in range(6)]
    assert all(a.shape == (2,) for a in acts)
    before = _fill_and_learn(agent, 24, steps=4)
    assert not torch.allclose(before, agent.actor_fp.flat)


def test_td3_hint_admm():
    agent = td3.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                      batch_size=8, n_actions=2, max_mem_size=64,
                      tau=0.005, warmup=0, noise=0.1, use_hint=True,
                      device=CPU)
    _fill_and_learn(agent, 24, steps=4)


def test_ddpg_learn():
    agent = ddpg.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                       batch_size=8, n_actions=2, max_mem_size=64,
                       tau=0.005, device=CPU)
    before = _fill_and_learn(agent, 24, steps=3)
    assert not torch.allclose(before, agent.actor_fp.flat)


def test_polyak_flat_matches_reference_rule():
    agent = ddpg.Agent(gamma=0.99, lr_a=1e-3, lr_c=1e-3, input_dims=[24],
                       batch_size=8, n_actions=2, max_mem_size=64,
                       tau=0.3, device=CPU)
    online = agent.critic_fp.flat.clone()
    target0 = torch.randn_like(online)
    agent.target_critic_fp.flat.copy_(target0)
    agent.update_network_parameters()  # tau = 0.3
    expected = 0.3 * online + 0.7 * target0
    assert torch.allclose(agent.target_critic_fp.flat, expected, atol=1e-6)


def test_seeded_determinism():
    """Same seed → identical scores (golden-run regression guard)."""
    import numpy as np
    import torch
    from smartcal_amd.envs.enet import ENetEnv
    from smartcal_amd.rl.sac import Agent
    from smartcal_amd.utils.device import seed_everything

    def run():
        seed_everything(7)
        env = ENetEnv(6, 6, device=torch.device("cpu"))
        agent = Agent(gamma=0.99, batch_size=4, n_actions=2, tau=0.005,
                      max_mem_size=32, input_dims=[6 + 36], lr_a=1e-3,
                      lr_c=1e-3, reward_scale=6, alpha=0.03,
                      device=torch.device("cpu"))
        scores = []
        for _ in range(3):
            obs = env.reset()
            tot = 0.0
            for _ in range(3):
                a = agent.choose_action(obs)
                obs2, r, done, info = env.step(a)
                agent.store_transition(obs, a, r, obs2, done,
                                       np.zeros(2, np.float32))
                agent.learn()
                tot += float(r)
                obs = obs2
            scores.append(tot)
        return scores

    s1 = run()
    s2 = run()
    np.testing.assert_allclose(s1, s2, rtol=0, atol=0)


def test_checkpoint_file_layout(tmp_path):
    """The saved-model layout matches the reference's filenames exactly
    (`enet_sac.py:396-403`, `enet_td3.py:151-159`, `enet_ddpg.py`), so a
    reference user's resume scripts keep working."""
    import os
    from smartcal_amd.rl.sac import Agent as SAC
    from smartcal_amd.rl.td3 import Agent as TD3
    from smartcal_amd.rl.ddpg import Agent as DDPG
    kw = dict(gamma=0.99, batch_size=4, n_actions=2, tau=0.005,
              max_mem_size=16, input_dims=[12], lr_a=1e-3, lr_c=1e-3,
              checkpoint_dir=str(tmp_path))
    SAC(reward_scale=2, alpha=0.03, **kw).save_models()
    TD3(**kw).save_models()
    DDPG(**kw).save_models()
    files = set(os.listdir(tmp_path))
    expect = {
        "a_eval_sac_actor.model", "q_eval_1_sac_critic.model",
        "q_eval_2_sac_critic.model", "replaymem_sac.model",
        "a_eval_td3_actor.model", "a_target_td3_actor.model",
        "q_eval_1_td3_critic.model", "q_eval_2_td3_critic.model",
        "q_target_1_td3_critic.model", "q_target_2_td3_critic.model",
        "a_eval_ddpg_actor.model", "a_target_ddpg_actor.model",
        "q_eval_ddpg_critic.model", "q_target_ddpg_critic.model",
        "replaymem_ddpg.model",
    }
    missing = expect - files
    assert not missing, f"missing reference-layout files: {missing}"
