"""A/B: native K-strided staging vs transpose+NT, BERT backward shapes."""
import torch, time
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
dev = "cuda"

def t(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6  # us

shapes = [  # (M, N, K) for C[M,N] = A[M,K] @ B[K,N]  (nn flavor: gx = gy @ w)
    (4096, 768, 2304), (4096, 2304, 768), (4096, 768, 3072),
    (4096, 3072, 768), (1024, 768, 30522),
]
print("== gemm_nn: gx = gy[M,K] @ w[K,N] ==")
for M, N, K in shapes:
    gy = torch.randn(M, K).bfloat16().to(dev)
    w = torch.randn(K, N).bfloat16().to(dev)
    nat = t(lambda: m.gemm_nn(gy, w))
    via = t(lambda: m.gemm_nt(gy, w.transpose(0, 1).contiguous()))
    print(f"M{M} N{N} K{K}: native={nat:7.1f}us  transposeNT={via:7.1f}us")

print("== gemm_tn: gw = gy[K,M]^T @ x[K,N] (K=batch) ==")
for M, N, K in [(2304, 768, 4096), (768, 3072, 4096), (30522, 768, 1024)]:
    gy = torch.randn(K, M).bfloat16().to(dev)
    x = torch.randn(K, N).bfloat16().to(dev)
    nat = t(lambda: m.gemm_tn(gy, x))
    via = t(lambda: m.gemm_nt(gy.transpose(0,1).contiguous(), x.transpose(0,1).contiguous()))
    print(f"M{M} N{N} K{K}: native={nat:7.1f}us  transposeNT={via:7.1f}us")

print("== bmm (attention) B=384 S=128 dh=64 ==")
q = torch.randn(384, 128, 64).bfloat16().to(dev)
k = torch.randn(384, 128, 64).bfloat16().to(dev)
p = torch.randn(384, 128, 128).bfloat16().to(dev)
v = torch.randn(384, 128, 64).bfloat16().to(dev)
print("scores nt:", t(lambda: m.bmm_nt(q, k)), "us")
print("ctx nn native:", t(lambda: m.bmm_nn(p, v)), "us")
print("ctx nn viaT:", t(lambda: m.bmm_nt(p, v.transpose(1,2).contiguous())), "us")
