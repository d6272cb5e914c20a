"""ctypes binding to the in-tree gfx950 kernel library (libprime_hip.so).

Policy (matches the build contract): on a machine WITH a GPU the HIP library
is mandatory — a missing/unbuildable .so raises instead of silently falling
back to eager PyTorch. On CPU-only machines the ops use their PyTorch
reference paths and the library is never required.
"""
from __future__ import annotations

import ctypes
from typing import Optional

import torch

from .build import LIB_PATH, build, needs_build

_LIB: Optional[ctypes.CDLL] = None
_TRIED = False

_SIGS = {
    # name -> argtypes (all return c_int hipError_t)
    "prime_rmsnorm_fwd": [ctypes.c_void_p] * 5 + [ctypes.c_int64] * 2 + [ctypes.c_double],
    "prime_rmsnorm_bwd": [ctypes.c_void_p] * 7 + [ctypes.c_int64] * 2 + [ctypes.c_double],
    "prime_add_rmsnorm_fwd": [ctypes.c_void_p] * 7 + [ctypes.c_int64] * 2 + [ctypes.c_double],
    "prime_add_rmsnorm_bwd": [ctypes.c_void_p] * 8 + [ctypes.c_int64] * 2 + [ctypes.c_double],
    "prime_rope": [ctypes.c_void_p] * 5 + [ctypes.c_int64] * 4 + [ctypes.c_int, ctypes.c_int64, ctypes.c_void_p],
    "prime_swiglu_fwd": [ctypes.c_void_p] * 3 + [ctypes.c_int64] * 2,
    "prime_swiglu_bwd": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 2,
    "prime_adamw": [ctypes.c_void_p] * 7 + [ctypes.c_int64] + [ctypes.c_double] * 5 + [ctypes.c_int64],
    "prime_grad_sqnorm": [ctypes.c_void_p] * 3 + [ctypes.c_int64],
    "prime_pseudograd": [ctypes.c_void_p] * 4 + [ctypes.c_int64],
    "prime_quant_int8": [ctypes.c_void_p] * 4 + [ctypes.c_int64],
    "prime_dequant_int8": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 2,
    "prime_nesterov_outer": [ctypes.c_void_p] * 6 + [ctypes.c_int64] + [ctypes.c_double] * 2,
    "prime_cross_entropy": [ctypes.c_void_p] * 5 + [ctypes.c_int64] * 2 + [ctypes.c_double] + [ctypes.c_int64] * 2,
    "prime_flash_fwd": [ctypes.c_void_p] * 6 + [ctypes.c_int64] * 5 + [ctypes.c_double] + [ctypes.c_int64] * 7,
    "prime_attn_delta": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 2,
    "prime_flash_bwd_dq": [ctypes.c_void_p] * 9 + [ctypes.c_int64] * 5 + [ctypes.c_double] + [ctypes.c_int64] * 10,
    "prime_flash_bwd_dkv": [ctypes.c_void_p] * 13 + [ctypes.c_int64] * 6 + [ctypes.c_double] + [ctypes.c_int64] * 10,
    "prime_mfma_probe": [ctypes.c_void_p] * 4,
    "prime_mfma_probe32": [ctypes.c_void_p] * 4,
    "prime_transpose_bshd": [ctypes.c_void_p] * 3 + [ctypes.c_int64] * 7,
    "prime_attn_decode": [ctypes.c_void_p] * 5 + [ctypes.c_int64] * 6 + [ctypes.c_double, ctypes.c_void_p],
    "prime_gemm_nt": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 4,
    "prime_gemm_nt8": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 4,
    "prime_quant_fp8": [ctypes.c_void_p] * 6 + [ctypes.c_int64] * 2,
    "prime_rowwise_quant_fp8": [ctypes.c_void_p] * 4 + [ctypes.c_int64] * 3,
    "prime_transpose_fp8": [ctypes.c_void_p] * 3 + [ctypes.c_int64] * 7 + [ctypes.c_void_p] * 3 + [ctypes.c_int64],
}


def _load() -> ctypes.CDLL:
    import os

    override = os.environ.get("PRIME_AMD_LIB_PATH")  # A/B testing of builds
    path = override or str(LIB_PATH)
    if override is None and needs_build():
        # stale (source-hash stamp mismatch) or missing: rebuild — hipcc
        # cross-compiles without a GPU and exists on every target image
        build(verbose=True)
    lib = ctypes.CDLL(path)
    for name, argtypes in _SIGS.items():
        try:
            fn = getattr(lib, name)
        except AttributeError:
            continue  # older A/B variant builds may lack newest entry points
        fn.argtypes = argtypes
        fn.restype = ctypes.c_int
    return lib


def lib() -> ctypes.CDLL:
    """Return the kernel library, loading (and if needed building) it."""
    global _LIB, _TRIED
    if _LIB is None:
        if _TRIED:
            raise RuntimeError("prime_amd HIP kernel library failed to load earlier")
        _TRIED = True
        try:
            _LIB = _load()
        except Exception as e:  # noqa: BLE001
            raise RuntimeError(
                "prime_amd requires its HIP kernel library on GPU machines; "
                f"build failed or .so missing at {LIB_PATH}: {e}"
            ) from e
    return _LIB


def have_lib() -> bool:
    try:
        lib()
        return True
    except RuntimeError:
        return False


def stream_of(t: torch.Tensor) -> ctypes.c_void_p:
    """Current HIP stream for the tensor's device, as a raw handle."""
    s = torch.cuda.current_stream(t.device).cuda_stream
    return ctypes.c_void_p(s)


def check(ret: int, what: str) -> None:
    if ret != 0:
        raise RuntimeError(f"HIP kernel {what} failed with hipError_t={ret}")


def ptr(t: Optional[torch.Tensor]) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    return ctypes.c_void_p(t.data_ptr())
