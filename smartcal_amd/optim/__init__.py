from .lbfgs import LBFGSNew  # noqa: F401
