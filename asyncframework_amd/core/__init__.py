from .context import ASYNCcontext, RDDPartialRes, workerState  # noqa: F401
