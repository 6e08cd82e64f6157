from .context import ASYNCcontext, RDDPartialRes, workerState  # noqa: F401
from .rdd import AsyncRDD, ASYNCbroadcast  # noqa: F401
