"""prime_amd.ops — hand-written gfx950 HIP kernels with autograd wrappers.

Kernel inventory: flash attention fwd/bwd, RMSNorm, RoPE,
SwiGLU, fused AdamW, DiLoCo outer Nesterov step, blockwise int8
quant/dequant, fused cross-entropy. GEMMs go through hipBLASLt via
torch.nn.functional.linear (library GEMMs; everything fused is ours).
"""
from .functional import (
    apply_rope,
    invalidate_wt_cache,
    set_linear_fp8,
    set_linear_tuned,
    tuned_linear,
    attn_decode,
    fused_add_rmsnorm,
    cross_entropy,
    dequant_int8,
    flash_attention,
    fused_adamw,
    grad_sqnorm,
    mfma_probe,
    nesterov_outer,
    pseudograd,
    quant_int8,
    rmsnorm,
    swiglu,
    QBLK,
)
from . import reference
from ._lib import have_lib
from .build import build, LIB_PATH

__all__ = [
    "apply_rope",
    "attn_decode",
    "fused_add_rmsnorm",
    "cross_entropy",
    "dequant_int8",
    "flash_attention",
    "fused_adamw",
    "grad_sqnorm",
    "mfma_probe",
    "nesterov_outer",
    "pseudograd",
    "quant_int8",
    "rmsnorm",
    "swiglu",
    "QBLK",
    "reference",
    "have_lib",
    "build",
    "LIB_PATH",
]
