from . import comm
from .sharded import PullHandle, ShardedVariable

__all__ = ["comm", "PullHandle", "ShardedVariable"]
