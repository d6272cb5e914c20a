from .diloco import DilocoOptimizer
from .flat import FlatParamSpace, FusedAdamW
from .mesh import ElasticDeviceMesh, MeshConfig
from .ring import allreduce_fp32, ring_allreduce_int8

__all__ = [
    "DilocoOptimizer",
    "FlatParamSpace",
    "FusedAdamW",
    "ElasticDeviceMesh",
    "MeshConfig",
    "allreduce_fp32",
    "ring_allreduce_int8",
]
