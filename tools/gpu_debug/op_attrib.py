"""Torch-profiler attribution of the flagship round (op-level, shapes)."""
import sys, os; sys.path.insert(0, os.getcwd())
import torch
from colearn_federated_learning_amd import FederatedServer, driver_config

cfg = driver_config(2).override(rounds=1, samples_per_client=2048)
s = FederatedServer(cfg)
s.run_round(0)  # warm (captures graphs)
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    s.run_round(1)
print(prof.key_averages(group_by_input_shape=True)
      .table(sort_by="self_cuda_time_total", row_limit=28, max_src_column_width=60))
