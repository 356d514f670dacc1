"""Bandwidth of the DP clip pipeline kernels on config-5 shapes."""
import os
import sys
import time

sys.path.insert(0, os.getcwd())
import torch

from colearn_federated_learning_amd.ops import require_ext

m = require_ext()
B, D, G = 512, 2_359_296, 64


def t(fn, n=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


g = torch.randn(B, D, device="cuda")
sq = torch.zeros(B, device="cuda")
scale = torch.rand(B, device="cuda")
out = torch.empty(G, D, device="cuda")
outf = torch.empty(D, device="cuda")

sec = t(lambda: m.sq_norm_per_row(g, sq))
print(f"sq_norm      : {sec*1e3:8.2f} ms  {B*D*4/sec/1e12:6.2f} TB/s")
sec = t(lambda: m.scale_rows_mean(g, scale, out.reshape(-1), G))
print(f"scale_mean G : {sec*1e3:8.2f} ms  {B*D*4/sec/1e12:6.2f} TB/s")
sec = t(lambda: m.scale_rows_mean(g, scale, outf, 1))
print(f"scale_mean 1 : {sec*1e3:8.2f} ms  {B*D*4/sec/1e12:6.2f} TB/s")
# torch reference reductions for context
sec = t(lambda: g.pow(2).sum(dim=1))
print(f"torch sqsum  : {sec*1e3:8.2f} ms  {B*D*4/sec/1e12:6.2f} TB/s")
sec = t(lambda: g.view(G, B // G, D).mean(dim=1))
print(f"torch gmean  : {sec*1e3:8.2f} ms  {B*D*4/sec/1e12:6.2f} TB/s")

# full config-5-like list: 20 conv tensors + small params, ~23 GB live
print("\n-- full-list clip_and_reduce (23 GB live) --")
from colearn_federated_learning_amd.ops.dp import clip_and_reduce

shapes = ([2_359_296] * 4 + [1_179_648] * 3 + [589_824] * 4 +
          [294_912] * 3 + [147_456] * 4 + [36864, 8192, 5120] +
          [512] * 20 + [64] * 20)
per = [torch.randn(B, d, device="cuda") for d in shapes]
print(f"total {sum(s for s in shapes)*B*4/1e9:.1f} GB")
sec = t(lambda: clip_and_reduce(per, 1.0, pre_scale=64.0, groups=G), n=5)
tot = sum(shapes) * B * 4 * 2  # read twice (norms + scale)
print(f"clip_and_reduce: {sec*1e3:8.2f} ms  {tot/sec/1e12:6.2f} TB/s eff")
