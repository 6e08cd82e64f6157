from . import philox, logfmt  # noqa: F401
