"""smartcal_amd.models — supervised/auxiliary models.

Transformer encoder classifier (`calibration/transformer_models.py`),
MLP regressor (`demixing_rl/regressor_net.py`), TSK fuzzy network
(`demixing_rl/train_tsk.py`, re-implemented from scratch — no pytsk
dependency), and the supervised data buffers
(`transformer_models.ReplayBuffer`, `demixing_rl/training_buffer.py`).
"""

from .transformer import SupervisedBuffer, TransformerEncoder  # noqa: F401
from .regressor import RegressorNet  # noqa: F401
from .tsk import TSKModel, center_difference_loss, sigma_loss  # noqa: F401
from .buffers import TrainingBuffer  # noqa: F401
