from .lbfgs import LBFGS  # noqa: F401
