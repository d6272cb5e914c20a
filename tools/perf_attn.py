"""Standalone flash-attention kernel timing at the bench shape.

Usage (GPU box): python tools/perf_attn.py [iters]
Prints achieved TF/s for fwd / bwd-dq / bwd-dkv with causal FLOP counting.
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from prime_amd import ops
from prime_amd.ops.functional import _FlashAttention


def main():
    nums = [a for a in sys.argv[1:] if a.isdigit()]
    iters = int(nums[0]) if nums else 20
    causal = "--non-causal" not in sys.argv
    B, S, H, Hkv, D = 8, 2048, 32, 8, 128
    torch.manual_seed(0)
    dev = "cuda:0"
    q = torch.randn(B, S, H, D, device=dev).bfloat16().requires_grad_(True)
    k = torch.randn(B, S, Hkv, D, device=dev).bfloat16().requires_grad_(True)
    v = torch.randn(B, S, Hkv, D, device=dev).bfloat16().requires_grad_(True)

    # attention FLOPs (fwd): 2 matmuls * S^2(/2 causal) * D * H * B * 2
    fwd_flops = 2 * 2 * (S * S / (2 if causal else 1)) * D * H * B
    bwd_flops = 2.5 * fwd_flops  # 5 matmuls in bwd vs 2 in fwd

    def timeit(fn, n):
        fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    # forward only
    t_fwd = timeit(lambda: _FlashAttention.apply(q, k, v, causal, D**-0.5), iters)
    print(f"fwd : {t_fwd*1e3:7.3f} ms  {fwd_flops/t_fwd/1e12:7.1f} TF/s")

    # fwd+bwd (isolates bwd by subtraction)
    o = _FlashAttention.apply(q, k, v, causal, D**-0.5)
    do = torch.randn_like(o)

    def fb():
        out = _FlashAttention.apply(q, k, v, causal, D**-0.5)
        out.backward(do)
        q.grad = k.grad = v.grad = None

    t_fb = timeit(fb, iters)
    t_bwd = t_fb - t_fwd
    print(f"bwd : {t_bwd*1e3:7.3f} ms  {bwd_flops/t_bwd/1e12:7.1f} TF/s")
    print(f"f+b : {t_fb*1e3:7.3f} ms  {(fwd_flops+bwd_flops)/t_fb/1e12:7.1f} TF/s")


if __name__ == "__main__":
    main()
