from .annotation import NoShardDim, ShardAnnotation, ShardDim
from .combination import CombinationFunc, HaloHint, try_combination
from .metaop import MetaOp, shard_tensor
from .metair import (SPMD, ClusterStrategy, MetaGraph, MetaNode,
                     MetaNodeCluster, MetaVar, NodeSPMDStrategy, P, R, S)
from .view_propagation import (local_view_shape, view_dim_map,
                               view_propagation)

__all__ = [
    "ShardDim", "ShardAnnotation", "NoShardDim",
    "CombinationFunc", "try_combination", "HaloHint",
    "MetaOp", "shard_tensor",
    "SPMD", "R", "S", "P", "MetaVar", "MetaNode", "MetaGraph",
    "MetaNodeCluster", "ClusterStrategy", "NodeSPMDStrategy",
    "view_propagation", "view_dim_map", "local_view_shape",
]
