from .device import default_device, seed_everything  # noqa: F401
from .config import TrainConfig, EnvConfig, AgentConfig, SolverConfig, \
    load_config  # noqa: F401
from .metrics import MetricsLogger, trace_range  # noqa: F401
