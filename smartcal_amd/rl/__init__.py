from .buffers import ReplayBuffer, PERBuffer, SumTree  # noqa: F401
from .buffers_dict import DictReplayBuffer, DictPERBuffer  # noqa: F401
