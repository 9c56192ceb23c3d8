"""smartcal_amd — MI355X-native framework for RL-driven calibration tuning.

A from-scratch, MI355X-first (gfx950 / CDNA4) re-design of the capabilities of
SarodYatawatta/smart-calibration: gym environments for elastic-net regression,
direction-dependent calibration regularization tuning and demixing direction
selection; DDPG/TD3/SAC agents with plain and prioritized (sum-tree) replay;
an L-BFGS optimizer with strong-Wolfe line search; influence-map machinery
(Jacobian / HVP / inverse-HVP); an in-repo HIP calibration solver, gridder and
radio-astronomy math layer replacing the reference's external SAGECal/excon
pipeline; and RCCL-over-xGMI distributed actor/learner + data-parallel
training.

Compute path: PyTorch-ROCm for composition + hand-written HIP/CDNA4 kernels
(``smartcal_amd.ops``) for the hot ops. No CUDA compatibility layers.
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
