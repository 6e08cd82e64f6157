from .config import EngineConfig  # noqa: F401
from .local import AsyncEngine, RunResult, SyncEngine  # noqa: F401
from .server import Server  # noqa: F401
from .worker import Shard, Worker  # noqa: F401

# Distributed engines import torch.distributed; keep them one hop deeper
# (asyncframework_amd.engine.dist / .dist_native / .dist_sync) so this
# package import stays light.
