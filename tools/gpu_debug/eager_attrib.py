"""Attribute the remaining torch-native kernels in the stacked step.

Runs ONE eager (non-graphed) stacked flagship step under torch.profiler
with shapes, printing ops that launch FillFunctor / elementwise /
copy kernels — the residual small-kernel tail after the head fusion.
"""
import os
import sys

sys.path.insert(0, os.getcwd())

import torch

from colearn_federated_learning_amd import FederatedServer, driver_config

cfg = driver_config(2).override(
    rounds=1, data__samples_per_client=256, train__use_graphs=False,
    num_evaluators=0,
)
s = FederatedServer(cfg, device=torch.device("cuda", 0))
s.run_round(0)  # warm

# trace who calls Tensor.zero_ / torch.zeros during the step
import traceback
from collections import Counter

_zero_callers = Counter()
_orig_zero = torch.Tensor.zero_


def _traced_zero(self):
    _zero_callers["".join(traceback.format_stack()[-4:-1])] += 1
    return _orig_zero(self)


torch.Tensor.zero_ = _traced_zero
from torch.profiler import ProfilerActivity, profile

eng = s.stacked.full
x = torch.randn(8 * 64, 3, 32, 32, device="cuda")
y = torch.randint(0, 10, (8 * 64,), device="cuda")
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    eng.optimizer.zero_grad()
    with torch.autocast("cuda", torch.bfloat16):
        lv = eng.model.loss_terms(x, y)
    lv.sum().backward()
    eng.optimizer.step()
    eng.sflat.sync_shadow()
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True)
      .table(sort_by="self_cuda_time_total", row_limit=40,
             max_src_column_width=60))
# kernel-level counts for the tail kernels
ka = prof.key_averages()
torch.Tensor.zero_ = _orig_zero
print("\n---- zero_ call sites ----")
for stack, n in _zero_callers.most_common(10):
    print(f"x{n}:\n{stack}")

# dispatch-level: catch C++-originated fills/zeros with shapes
from torch.utils._python_dispatch import TorchDispatchMode


class _FillSpy(TorchDispatchMode):
    def __init__(self):
        super().__init__()
        self.hits = Counter()

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if "zero_" in name or "fill" in name or "copy_" in name:
            shape = tuple(args[0].shape) if args and hasattr(
                args[0], "shape") else None
            self.hits[(name, shape)] += 1
        return func(*args, **(kwargs or {}))


spy = _FillSpy()
with spy:
    eng.optimizer.zero_grad()
    with torch.autocast("cuda", torch.bfloat16):
        lv = eng.model.loss_terms(x, y)
    lv.sum().backward()
    eng.optimizer.step()
    eng.sflat.sync_shadow()
    torch.cuda.synchronize()
print("\n---- dispatch-level fills/zeros/copies ----")
for (name, shape), n in spy.hits.most_common(25):
    print(f"x{n:4d}  {name}  {shape}")

print("\n---- per-op CUDA kernel counts (fills/copies/elementwise) ----")
for ev in sorted(ka, key=lambda e: -e.self_device_time_total)[:60]:
    n = ev.key
    if any(t in n for t in ("fill", "Fill", "copy", "add", "elementwise",
                            "zero", "div", "mul", "mean", "sum")):
        print(f"{ev.self_device_time_total/1000.0:9.3f} ms  n={ev.count:5d}  {n[:100]}")
