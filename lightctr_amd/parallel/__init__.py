from .sharded_fm import ShardedFMModel
from .ring import RingDataParallel, broadcast_params, allreduce_gradients
from .sharded_widedeep import ShardedWideDeepModel
from .sharded_ffm import ShardedFFMModel
from .ps import PSConfig, PSShard, PSWorker, ps_train_fm, setup_pair_groups

__all__ = ["ShardedFMModel", "ShardedWideDeepModel", "ShardedFFMModel", "RingDataParallel", "broadcast_params",
           "allreduce_gradients", "PSConfig", "PSShard", "PSWorker",
           "ps_train_fm", "setup_pair_groups"]
