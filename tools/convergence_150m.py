"""GPU convergence evidence: train llama_150m on a synthetic Zipf-distributed
bigram corpus (learnable structure, unlike uniform random) and log the loss
curve to gpurun_out/."""
import json
import os
import sys

import numpy as np


sys.path.insert(0, "/root/repo")
from prime_amd.utils.config import (DilocoConfig, DataSection, MetricsConfig,
                                    ModelConfig, TrainConfig)
from prime_amd.train import Trainer

# corpus: first-order Markov chain over 2048 symbols with Zipf marginals
rng = np.random.default_rng(0)
V = 2048
n_tok = 2_000_000
probs = 1.0 / np.arange(1, V + 1) ** 1.1
probs /= probs.sum()
# per-state transition: mixture of global zipf + strong successor preference
toks = np.empty(n_tok, dtype=np.uint16)
cur = 0
jump = rng.random(n_tok) < 0.3
zipf_draw = rng.choice(V, size=n_tok, p=probs)
for i in range(n_tok):
    cur = zipf_draw[i] if jump[i] else (cur * 31 + 7) % V
    toks[i] = cur
toks.tofile("/tmp/zipf.bin")

cfg = TrainConfig(
    run_name="zipf150m", steps=300,
    model=ModelConfig(name="llama_150m", seq_len=512,
                      fp8=os.environ.get("PRIME_AMD_FP8", "0") == "1",
                      fp8_dgrad=os.environ.get("PRIME_AMD_FP8_DGRAD", "0") == "1",
                      fp8_wgrad=os.environ.get("PRIME_AMD_FP8_WGRAD", "0") == "1",
                      overrides={"vocab_size": V, "max_seq": 1024}),
    data=DataSection(kind="token_file", path="/tmp/zipf.bin",
                     micro_batch_size=16),
    diloco=DilocoConfig(H=100),
    metrics=MetricsConfig(log_interval=20),
)
cfg.optim.warmup_steps = 30
tr = Trainer(cfg, run_dir="gpurun_out/zipf_run")
res = tr.run()
tr.close()
print(json.dumps({"final_loss": res["loss"], "tokens_per_sec": res["tokens_per_sec"]}))
