"""Decode (serving) throughput: tokens/sec for KV-cache generation."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from prime_amd.models import build_model
from prime_amd.models.generate import generate


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else "llama_1b"
    bs = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    new = int(sys.argv[3]) if len(sys.argv) > 3 else 128
    torch.manual_seed(0)
    with torch.device("cuda"):
        m = build_model(model_name)
    m = m.to(dtype=torch.bfloat16)
    m.reset_rope("cuda")
    prompt = torch.randint(0, m.cfg.vocab_size, (bs, 128), device="cuda")
    for graphed in (False, True):
        generate(m, prompt, 8, use_graph=graphed)  # warmup
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = generate(m, prompt, new, use_graph=graphed)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        tps = bs * new / dt
        mode = "graph" if graphed else "eager"
        print(f"{model_name} bs={bs} {mode}: {tps:,.0f} decode tok/s "
              f"({1e3*dt/new:.2f} ms/tok, {out.shape[1]} total len)")


if __name__ == "__main__":
    main()
