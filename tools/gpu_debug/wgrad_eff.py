"""Per-layer wgrad efficiency: useful TFLOP/s vs tile occupancy.

If achieved/padded-TFLOPs is flat across layers, the kernel is
MFMA-issue-bound and the K<128 tile padding is the loss; if it sags on
large-RSC layers, the X gather is the bound.
"""
import sys, os; sys.path.insert(0, os.getcwd())
import time, torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
G = 8


def t(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


# resnet18 CIFAR wgrad shapes: (B, C, H, K, ks, stride, pad)
shapes = [
    ("stem", 64, 3, 32, 64, 3, 1, 1),
    ("l1", 64, 64, 32, 64, 3, 1, 1),
    ("l2a", 64, 64, 32, 128, 3, 2, 1),
    ("l2", 64, 128, 16, 128, 3, 1, 1),
    ("l2d", 64, 64, 32, 128, 1, 2, 0),
    ("l3a", 64, 128, 16, 256, 3, 2, 1),
    ("l3", 64, 256, 8, 256, 3, 1, 1),
    ("l4a", 64, 256, 8, 512, 3, 2, 1),
    ("l4", 64, 512, 4, 512, 3, 1, 1),
]
tot = 0.0
for name, B, C, H, K, ks, st, pad in shapes:
    P = (H + 2 * pad - ks) // st + 1
    x = (torch.randn(G * B, C, H, H).bfloat16().cuda()
         .contiguous(memory_format=torch.channels_last))
    gy = (torch.randn(G * B, K, P, P).bfloat16().cuda()
          .contiguous(memory_format=torch.channels_last))
    us = t(lambda: m.conv2d_wgrad_km(gy, x, ks, ks, st, pad, G))
    # numerics guard vs the untouched non-km entry
    a = m.conv2d_wgrad_km(gy, x, ks, ks, st, pad, G)
    b = m.conv2d_wgrad(gy, x, ks, ks, st, pad, G).permute(0, 1, 3, 4, 2)
    ok = torch.allclose(a.float(), b.float(), rtol=1e-2, atol=1e-2)
    NPQ, RSC = B * P * P, ks * ks * C
    flops = 2.0 * NPQ * K * RSC * G
    pk = -(-K // 128) * 128
    prc = -(-RSC // 128) * 128
    occ = flops / (2.0 * NPQ * pk * prc * G)
    gather_gb = (NPQ * RSC + NPQ * K) * 2.0 * G / 1e9
    tf = flops / us / 1e6
    tot += us
    print(f"{name:5s} C{C:4d} H{H:2d} K{K:4d} ks{ks} s{st}: {us:7.1f}us "
          f"useful={tf:7.1f}TF/s tile-occ={occ:4.2f} "
          f"padded={tf/occ:7.1f}TF/s gather={gather_gb*1e3/us:5.2f}TB/s ok={ok}")
print(f"total {tot:.0f}us")
