"""Torch-profiler attribution of ONE stacked DP step (config 5)."""
import os
import sys

sys.path.insert(0, os.getcwd())
import torch

from colearn_federated_learning_amd import FederatedServer, driver_config

cfg = driver_config(5).override(
    rounds=1, data__samples_per_client=128, num_evaluators=0,
)
s = FederatedServer(cfg, device=torch.device("cuda", 0))
s.run_round(0)  # warm
eng = s.stacked.full
G = eng.G
x = torch.randn(G * 64, 3, 32, 32, device="cuda")
y = torch.randint(0, 10, (G * 64,), device="cuda")
torch.cuda.synchronize()
from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as p:
    eng._dp_step(x, y, use_bf16=True)
    torch.cuda.synchronize()
print(p.key_averages().table(sort_by="self_cuda_time_total",
                             row_limit=30, max_src_column_width=50))
