from .sharded_fm import ShardedFMModel
from .ring import RingDataParallel, broadcast_params, allreduce_gradients

__all__ = ["ShardedFMModel", "RingDataParallel", "broadcast_params",
           "allreduce_gradients"]
