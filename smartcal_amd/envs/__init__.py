from .enet import ENetEnv  # noqa: F401
from .calib import CalibEnv  # noqa: F401
from .demix import DemixingEnv  # noqa: F401
from .demix_fuzzy import FuzzyDemixingEnv  # noqa: F401
from .vec_enet import VecENetEnv  # noqa: F401
