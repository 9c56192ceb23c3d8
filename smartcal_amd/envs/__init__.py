from .enet import ENetEnv  # noqa: F401
