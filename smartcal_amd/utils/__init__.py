from .device import default_device, seed_everything  # noqa: F401
