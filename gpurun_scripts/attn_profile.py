"""Learn-heavy transformer-agent run for attention-kernel trace
evidence (batch 8 so learn() fires from step 8)."""
import sys
from pathlib import Path
import numpy as np, torch
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from smartcal_amd.rl.sac_cnn import Agent
torch.manual_seed(0)
ag = Agent(gamma=0.99, batch_size=8, n_actions=4, tau=0.005,
           max_mem_size=64, input_dims=(1, 32, 32), meta_dim=14,
           lr_a=1e-3, lr_c=1e-3, arch="transformer",
           device=torch.device("cuda"))
rng = np.random.default_rng(0)
for i in range(40):
    obs = {"img": rng.standard_normal((1, 32, 32)).astype(np.float32),
           "metadata": rng.standard_normal(14).astype(np.float32)}
    ag.store_transition(obs, rng.standard_normal(4).astype(np.float32),
                        float(i % 3), obs, False)
    ag.learn()
torch.cuda.synchronize()
print("attn profile run done")
