"""Within-probe A/B of wgrad LDS layout variants on ResNet layer shapes."""
import sys, os; sys.path.insert(0, os.getcwd())
import os, time, torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
dev = "cuda"
G = 8

def t(fn, n=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

# (B, C, H, K, ks, stride) resnet18 layer shapes
shapes = [(64, 64, 32, 64, 3, 1), (64, 128, 16, 128, 3, 1),
          (64, 256, 8, 256, 3, 1), (64, 512, 4, 512, 3, 1)]
for B, C, H, K, ks, st in shapes:
    P = H // st
    x = torch.randn(G * B, C, H, H).bfloat16().to(dev).contiguous(memory_format=torch.channels_last)
    gy = torch.randn(G * B, K, P, P).bfloat16().to(dev).contiguous(memory_format=torch.channels_last)
    res = {}
    outs = {}
    for v in [0, 1, 2, 0, 1, 2]:  # interleaved
        os.environ["COLEARN_WGRAD_VARIANT"] = str(v)
        us = t(lambda: m.conv2d_wgrad(gy, x, ks, ks, st, 1, G))
        res[v] = min(res.get(v, 1e18), us)
        outs[v] = m.conv2d_wgrad(gy, x, ks, ks, st, 1, G)
    ok1 = torch.allclose(outs[0].float(), outs[1].float(), rtol=1e-2, atol=1e-2)
    ok2 = torch.allclose(outs[0].float(), outs[2].float(), rtol=1e-2, atol=1e-2)
    print(f"C{C} H{H} K{K}: v0={res[0]:7.1f}us v1={res[1]:7.1f}us "
          f"v2={res[2]:7.1f}us  match v1={ok1} v2={ok2}")
os.environ.pop("COLEARN_WGRAD_VARIANT", None)
