"""bn_bwd grid-cap sweep on the layer1 BN shape (the dominant bn_bwd_gx)."""
import sys, os; sys.path.insert(0, os.getcwd())
import time, torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
G, B, C, H = 8, 64, 64, 32
R = B * H * H
x = torch.randn(G * B, C, H, H).bfloat16().cuda().contiguous(memory_format=torch.channels_last)
w = torch.ones(G * C, device="cuda"); bias = torch.zeros(G * C, device="cuda")
rm = torch.zeros(G * C, device="cuda"); rv = torch.ones(G * C, device="cuda")
empty = torch.empty(0, device="cuda", dtype=torch.bfloat16)
y, mean, invstd = m.batchnorm_fwd(x, w, bias, rm, rv, empty, True, 0.1, 1e-5, True, G)
gy = torch.randn_like(y)

def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

n_el = G * R * C
# full bwd traffic: reduce reads gy,y,x (6B/el) + gx pass reads gym,x writes gx,gym(write in reduce) ~ 13B/el
for cap in ["512", "2048", "8192", "1048576"]:
    os.environ["COLEARN_BN_GX_GRID"] = cap
    us = t(lambda: m.batchnorm_bwd(gy, y, x, w, mean, invstd, True, True, G))
    print(f"cap={cap:>8}: full bwd {us:7.1f}us  ~{n_el*13/us/1e3:5.2f}TB/s")
os.environ.pop("COLEARN_BN_GX_GRID", None)
