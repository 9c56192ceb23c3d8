"""Repro of the hint-path NaN: seed 2, ENetEnv(20,20,hint), SAC agent."""
import sys, torch
sys.path.insert(0, ".")
from smartcal_amd.utils.device import seed_everything
from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd import ops

seed_everything(2)
env = ENetEnv(20, 20, provide_hint=True)
# direct kernel check on the exact failing instance
obs, hint = env.reset()
x, EE, r = ops.enet.solve_and_influence(env.A, env.y0, 0.0696, 0.0621, 0.0)
print("direct: r finite", bool(torch.isfinite(torch.as_tensor(r)).all()),
      "x finite", bool(torch.isfinite(x).all()),
      "EE finite", bool(torch.isfinite(EE).all()), flush=True)

agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
              max_mem_size=1024, input_dims=[20 + 400], lr_a=1e-3,
              lr_c=1e-3, reward_scale=20, alpha=0.03, prioritized=True,
              use_hint=True)
bad = 0
for ep in range(120):
    obs, hint = env.reset()
    done = False; step = 0
    while not done and step < 10:
        a = agent.choose_action(obs)
        if not torch.isfinite(torch.as_tensor(a)).all():
            print(f"ep{ep} st{step}: NaN ACTION", flush=True); bad += 1
        obs_, r, done, info = env.step(a)
        rv = float(r)
        if rv != rv:
            print(f"ep{ep} st{step}: NaN REWARD a={a}", flush=True); bad += 1
        agent.store_transition(obs, a, rv, obs_, done, hint)
        agent.learn()
        obs = obs_; step += 1
print("soak done, bad =", bad, flush=True)
assert bad == 0
print("PASS", flush=True)
