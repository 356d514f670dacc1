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
