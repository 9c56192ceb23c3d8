from .buffers import ReplayBuffer, PERBuffer, SumTree  # noqa: F401
