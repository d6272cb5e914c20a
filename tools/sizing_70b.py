"""70B-dims sizing run on one MI355X: full-dim Llama-70B blocks (dim 8192,
64 heads / 8 KV, intermediate 28672, vocab 128256) at a reduced layer
count through real fwd/bwd/fused-AdamW steps. Validates the block shapes,
memory math and kernel coverage that the 8-GPU FSDP config
(configs/llama70b_fsdp8.toml) relies on, using hardware we can lease.

Usage: python tools/sizing_70b.py [n_layers] [steps]
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from prime_amd.models import build_model
from prime_amd.parallel.flat import FlatParamSpace, FusedAdamW
from prime_amd import ops


def main():
    n_layers = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 3
    dev = "cuda:0"
    torch.manual_seed(0)
    with torch.device(dev):
        m = build_model("llama_70b", n_layers=n_layers,
                        activation_checkpointing=True)
    m = m.to(dtype=torch.bfloat16)
    m.reset_rope(torch.device(dev))
    import os

    flat = FlatParamSpace(m)
    opt = FusedAdamW(flat, lr=1e-4)
    ops.set_linear_tuned(True)
    fp8 = os.environ.get("PRIME_AMD_FP8", "0") == "1"
    ops.set_linear_fp8(fp8, dgrad=fp8, wgrad=fp8)
    if fp8:
        print("fp8 mode: fwd+dgrad+wgrad")
    n_params = flat.numel_padded
    print(f"layers={n_layers} params={n_params/1e9:.2f}B "
          f"(full 80-layer model: ~{(n_params + 0) / n_layers * 80 / 1e9:.0f}B-ish core)")
    B, S = 2, 2048
    x = torch.randint(0, m.cfg.vocab_size, (B, S), device=dev)
    y = torch.randint(0, m.cfg.vocab_size, (B, S), device=dev)

    def step():
        flat.zero_grad()
        loss = m.loss(x, y)
        loss.backward()
        opt.step()
        return loss

    loss = step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        loss = step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    alloc = torch.cuda.max_memory_allocated() / 2**30
    toks = B * S
    print(f"loss={float(loss):.4f} ms/step={dt*1e3:.0f} tok/s={toks/dt:,.0f} "
          f"peak_mem={alloc:.1f} GiB")
    # memory extrapolation for the FSDP(8) full model: per-GPU bf16 shard
    # + fp32 master/m/v shard + one gathered block
    per_layer = n_params / n_layers
    full_params = per_layer * 80 + 2 * 128256 * 8192
    shard = full_params / 8
    est = (shard * (2 + 4 * 3) + per_layer * 2 * 2) / 2**30
    print(f"FSDP(8) est. per-GPU state: {est:.0f} GiB of 288 GiB "
          f"(shard bf16+fp32 master/m/v + 2 gathered blocks)")


if __name__ == "__main__":
    main()
