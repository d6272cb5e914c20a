import sys, time
sys.path.insert(0, "/root/repo")
import torch
from prime_amd.utils.config import TrainConfig, ModelConfig, DilocoConfig, MetricsConfig
from prime_amd.train import Trainer

cfg = TrainConfig(run_name="outercost", steps=1,
                  model=ModelConfig(name="intellect_10b", seq_len=2048),
                  diloco=DilocoConfig(H=100, outer_device="host"),
                  metrics=MetricsConfig(log_interval=1000))
cfg.data.micro_batch_size = 8
tr = Trainer(cfg, run_dir="/tmp/outercost")
tr.train_step()
torch.cuda.synchronize()
for i in range(3):
    t0 = time.perf_counter()
    tr.diloco.outer_step()
    torch.cuda.synchronize()
    print(f"outer step {i}: {time.perf_counter()-t0:.2f}s (pinned={tr.diloco.theta_outer.is_pinned()})")
tr.close()
