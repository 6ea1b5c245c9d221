from .dist import DistContext, init_from_env
from .tile_queue import TileQueue, TileScheduler

__all__ = ["DistContext", "init_from_env", "TileQueue", "TileScheduler"]
