"""smartcal_amd.distributed — multi-GPU training layers.

Re-designs the reference's torch-RPC learner/actor split
(`elasticnet/distributed_per_sac.py`, `demixing_rl/distributed_per_sac.py`,
SURVEY.md §2.3 P1) on torch.distributed collectives: on an 8×MI355X node
the backend is RCCL over xGMI (``"nccl"`` on ROCm) — weight distribution
is ONE flat broadcast and experience upload ONE fixed-shape gather per
round, replacing the reference's CPU-state_dict RPC pulls and pickled
buffer pushes. On CPU (tests) the same code runs on gloo.

`dp` provides the gradient all-reduce hook for the symmetric
data-parallel learner (P2 — the bench's 1..8 GPU scaling mode).
"""

from .learner_actor import Learner, Actor, run_process  # noqa: F401
from .dp import allreduce_grad_hook, init_from_env  # noqa: F401
