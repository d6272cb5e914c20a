from .diloco import DilocoOptimizer
from .flat import FlatParamSpace, FusedAdamW
from .fsdp import ShardedParamSpace
from .mesh import ElasticDeviceMesh, MeshConfig
from .ring import allreduce_fp32, ring_allreduce_int8

__all__ = [
    "DilocoOptimizer",
    "FlatParamSpace",
    "FusedAdamW",
    "ShardedParamSpace",
    "ElasticDeviceMesh",
    "MeshConfig",
    "allreduce_fp32",
    "ring_allreduce_int8",
]
