"""Per-op run-to-run determinism bisect."""
import torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
torch.manual_seed(0)
dev = "cuda"

def check(name, fn, n=3):
    outs = [fn() for _ in range(n)]
    ok = all(torch.equal(outs[0], o) for o in outs[1:])
    mx = max(float((outs[0]-o).abs().max()) for o in outs[1:]) if not ok else 0.0
    print(f"{name:28s} deterministic={ok} maxdiff={mx:g}")

x = torch.randn(64, 64, 16, 16).bfloat16().to(dev)
w3 = (torch.randn(64, 64, 3, 3)*0.1).bfloat16().to(dev)
gy = torch.randn(64, 64, 16, 16).bfloat16().to(dev)
a = torch.randn(512, 576).bfloat16().to(dev)
b = torch.randn(256, 576).bfloat16().to(dev)
wbn = torch.ones(64, device=dev); bbn = torch.zeros(64, device=dev)

check("conv_fwd", lambda: m.conv2d_fwd(x, w3, 1, 1))
check("conv_dgrad", lambda: m.conv2d_dgrad(gy, w3, 16, 16, 1, 1))
check("conv_wgrad", lambda: m.conv2d_wgrad(gy, x, 3, 3, 1, 1))
check("gemm_nt", lambda: m.gemm_nt(a, b))
def bnf():
    rm = torch.zeros(64, device=dev); rv = torch.ones(64, device=dev)
    y, mean, inv = m.batchnorm_fwd(x, wbn, bbn, rm, rv, torch.empty(0, device=dev, dtype=torch.bfloat16), True, 0.1, 1e-5, True)
    return y
check("batchnorm_fwd", bnf)
rm = torch.zeros(64, device=dev); rv = torch.ones(64, device=dev)
y, mean, inv = m.batchnorm_fwd(x, wbn, bbn, rm, rv, torch.empty(0, device=dev, dtype=torch.bfloat16), True, 0.1, 1e-5, True)
check("batchnorm_bwd_gx", lambda: m.batchnorm_bwd(gy, y, x, wbn, mean, inv, True, True)[0])
check("batchnorm_bwd_gw", lambda: m.batchnorm_bwd(gy, y, x, wbn, mean, inv, True, True)[1])
# torch ops in the path
xf = torch.randn(64, 512, 4, 4).bfloat16().to(dev).contiguous(memory_format=torch.channels_last)
check("adaptive_avg_pool", lambda: torch.nn.functional.adaptive_avg_pool2d(xf, 1))
lg = torch.randn(512, 10, device=dev, requires_grad=True)
yy = torch.randint(0, 10, (512,), device=dev)
def ce():
    l = torch.nn.functional.cross_entropy(lg.float(), yy)
    g = torch.autograd.grad(l, lg)[0]
    return g
check("cross_entropy_bwd", ce)
# full model fwd
from colearn_federated_learning_amd.config import DataConfig, ModelConfig
from colearn_federated_learning_amd.models import build_model
torch.manual_seed(1)
net = build_model(ModelConfig(name="resnet18"), DataConfig()).to(dev)
xi = torch.randn(32, 3, 32, 32, device=dev)
yi = torch.randint(0, 10, (32,), device=dev)
def fwd():
    net.zero_grad(set_to_none=False)
    with torch.autocast("cuda", torch.bfloat16):
        l = net.loss(xi, yi)
    return l.detach().clone()
check("resnet18_fwd_loss", fwd)
def fwdbwd():
    net.zero_grad(set_to_none=False)
    with torch.autocast("cuda", torch.bfloat16):
        l = net.loss(xi, yi)
    l.backward()
    return torch.cat([p.grad.reshape(-1) for p in net.parameters()])
check("resnet18_fwd_bwd", fwdbwd)
