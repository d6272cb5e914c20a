"""prime_amd — MI355X-native low-communication (DiLoCo) training engine.

Built from scratch for gfx950/CDNA4: hand-written HIP kernels for the Llama
hot path, RCCL-over-xGMI collectives for inner sharding, an int8-quantized
ring all-reduce for the outer pseudo-gradient sync, ElasticDeviceMesh fault
tolerance, and async distributed checkpointing. See SURVEY.md for the
capability map against the reference (PrimeIntellect-ai/prime).
"""
__version__ = "0.1.0"

from . import models, ops  # noqa: F401
