from .config import EngineConfig  # noqa: F401
from .local import AsyncEngine, RunResult  # noqa: F401
