import sys, os; sys.path.insert(0, os.getcwd())
"""A/B conv BK=32 vs BK=64 on ResNet layer shapes (within-probe)."""
import time, torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
dev = "cuda"; G = 8

def t(fn, n=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

shapes = [(64, 64, 32, 64, 3, 1), (64, 128, 16, 128, 3, 1),
          (64, 256, 8, 256, 3, 1), (64, 512, 4, 512, 3, 1)]
for B, C, H, K, ks, st in shapes:
    x = torch.randn(G*B, C, H, H).bfloat16().to(dev).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(G, K, C, ks, ks)*0.1).bfloat16().to(dev)
    gy = torch.randn(G*B, K, H, H).bfloat16().to(dev).contiguous(memory_format=torch.channels_last)
    res = {}
    outs = {}
    for bk in ["32", "64", "32", "64"]:
        os.environ["COLEARN_CONV_BK"] = bk
        f = t(lambda: m.conv2d_fwd(x, w, st, 1))
        d = t(lambda: m.conv2d_dgrad(gy, w, H, H, st, 1))
        prev = res.get(bk, (1e18, 1e18))
        res[bk] = (min(prev[0], f), min(prev[1], d))
        outs[bk] = (m.conv2d_fwd(x, w, st, 1), m.conv2d_dgrad(gy, w, H, H, st, 1))
    okf = torch.allclose(outs["32"][0].float(), outs["64"][0].float(), rtol=1e-2, atol=1e-2)
    okd = torch.allclose(outs["32"][1].float(), outs["64"][1].float(), rtol=1e-2, atol=1e-2)
    print(f"C{C} K{K}: fwd32={res['32'][0]:7.1f} fwd64={res['64'][0]:7.1f} "
          f"dg32={res['32'][1]:7.1f} dg64={res['64'][1]:7.1f} match={okf and okd}")
os.environ.pop("COLEARN_CONV_BK", None)
