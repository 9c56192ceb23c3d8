from .controller import DemixController, trapmf  # noqa: F401
