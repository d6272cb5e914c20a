"""Per-shape hipBLASLt GEMM throughput at the 10B bench shapes (fwd, dX,
dW for each projection + lm_head). Identifies which shapes drag the 52%
GEMM share below hipBLASLt's ~1.6-2.0 PF capability."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def bench_mm(name, a, b, iters=20, transpose_a=False):
    f = (lambda: a.T @ b) if transpose_a else (lambda: a @ b)
    f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        f()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    m = a.shape[1] if transpose_a else a.shape[0]
    k = a.shape[0] if transpose_a else a.shape[1]
    n = b.shape[1]
    tf = 2 * m * k * n / dt / 1e12
    print(f"{name:28s} M={m:6d} K={k:6d} N={n:6d}  {dt*1e3:7.3f} ms  {tf:7.0f} TF/s")
    return tf


def main():
    torch.manual_seed(0)
    dev = "cuda:0"
    M = 16384  # tokens (mb8 x 2048)
    dim, inter, vocab = 4096, 14336, 128256
    x = torch.randn(M, dim, device=dev).bfloat16()
    g = torch.randn(M, dim, device=dev).bfloat16()

    shapes = [
        ("qkv fwd  (x @ Wqkv^T)", (M, dim), (dim, 6144)),
        ("gateup fwd", (M, dim), (dim, 2 * inter)),
        ("down fwd", (M, inter), (inter, dim)),
        ("o fwd", (M, dim), (dim, dim)),
        ("lm_head fwd", (M, dim), (dim, vocab)),
    ]
    for name, (m, k), (_, n) in shapes:
        a = torch.randn(m, k, device=dev).bfloat16()
        b = torch.randn(k, n, device=dev).bfloat16()
        bench_mm(name, a, b)
    # backward dW shapes: [K_tokens reduction] dW = x^T @ grad
    for name, n_out in [("qkv dW (x^T @ dqkv)", 6144), ("gateup dW", 2 * inter),
                        ("o dW", dim)]:
        gg = torch.randn(M, n_out, device=dev).bfloat16()
        bench_mm(name, x, gg, transpose_a=True)
    gi = torch.randn(M, inter, device=dev).bfloat16()
    bench_mm("down dW (act^T @ g)", gi, g, transpose_a=True)
    # dX shapes: grad @ W
    for name, k_in in [("qkv dX", 6144), ("gateup dX", 2 * inter)]:
        gg = torch.randn(M, k_in, device=dev).bfloat16()
        w = torch.randn(k_in, dim, device=dev).bfloat16()
        bench_mm(name, gg, w)
    gl = torch.randn(M, vocab, device=dev).bfloat16()
    wl = torch.randn(vocab, dim, device=dev).bfloat16()
    bench_mm("lm_head dX", gl, wl)


if __name__ == "__main__" and "--custom" not in sys.argv:
    main()


def bench_custom_nt(name, m, n, k, iters=20, v2=False, setprio=0):
    """Custom gemm_nt kernel vs hipBLASLt on the same NT shape."""
    from prime_amd.ops._lib import check, lib, ptr, stream_of

    a = torch.randn(m, k, device="cuda:0").bfloat16()
    w = torch.randn(n, k, device="cuda:0").bfloat16()
    c = torch.empty(m, n, device="cuda:0", dtype=torch.bfloat16)
    fn = lib().prime_gemm_nt8 if v2 else lib().prime_gemm_nt

    def custom():
        check(fn(stream_of(a), ptr(a), ptr(w), ptr(c), m, n, k, setprio),
              "gemm_nt")

    # refcheck vs library
    custom()
    torch.cuda.synchronize()
    refc = a @ w.T
    rel = (c.float() - refc.float()).abs().max() / refc.float().abs().max()
    custom()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        custom()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2 * m * k * n / dt / 1e12

    def libmm():
        torch.matmul(a, w.T, out=c)
    libmm()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        libmm()
    torch.cuda.synchronize()
    dtl = (time.perf_counter() - t0) / iters
    tfl = 2 * m * k * n / dtl / 1e12
    print(f"{name:24s} M={m:6d} N={n:6d} K={k:6d}  custom {tf:7.0f} TF/s"
          f"  lib {tfl:7.0f} TF/s  ratio {tf/tfl:5.2f}  relerr {rel:.2e}")


def main_custom():
    torch.manual_seed(1)
    M, dim, inter, vocab = 16384, 4096, 14336, 128256
    v2 = "--v2" in sys.argv
    sp = 1 if "--setprio" in sys.argv else 0
    if v2:
        print(f"== custom NT v2 (counted-vmcnt, setprio={sp}) vs hipBLASLt ==")
        for name, m, n, k in [("qkv fwd", M, 6144, dim),
                              ("gateup fwd", M, 2 * inter, dim),
                              ("down fwd", M, dim, inter),
                              ("square 4k", 4096, 4096, 4096)]:
            bench_custom_nt(name, m, n, k, v2=True, setprio=sp)
        return
    print("== custom NT kernel vs hipBLASLt ==")
    bench_custom_nt("qkv fwd", M, 6144, dim)
    bench_custom_nt("gateup fwd", M, 2 * inter, dim)
    bench_custom_nt("down fwd", M, dim, inter)
    bench_custom_nt("o fwd", M, dim, dim)
    bench_custom_nt("lm_head fwd", M, vocab, dim)
    bench_custom_nt("dX-as-NT (qkv)", M, dim, 6144)
    bench_custom_nt("dW-as-NT (qkv)", 6144, dim, M)
    bench_custom_nt("square 4k", 4096, 4096, 4096)


if __name__ == "__main__" and "--custom" in sys.argv:
    main_custom()
    sys.exit(0)
