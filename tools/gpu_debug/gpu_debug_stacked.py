"""Stacked kernel numerics bisect on GPU."""
import torch
import torch.nn.functional as F
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
torch.manual_seed(0)
G, B, C, H, W = 3, 8, 64, 16, 16
CO, ks = 64, 3

x = torch.randn(G*B, C, H, W).bfloat16().cuda()
w = (torch.randn(G, CO, C, ks, ks)*0.1).bfloat16().cuda()
# stacked conv fwd vs per-client loop
y = m.conv2d_fwd(x, w, 1, 1)
refs = [F.conv2d(x[g*B:(g+1)*B].float().cpu(), w[g].float().cpu(), None, 1, 1) for g in range(G)]
ref = torch.cat(refs)
print("conv fwd stacked maxdiff:", float((y.float().cpu()-ref).abs().max()))

gy = torch.randn(G*B, CO, H, W).bfloat16().cuda()
gx = m.conv2d_dgrad(gy, w, H, W, 1, 1)
refs = [torch.nn.grad.conv2d_input((B,C,H,W), w[g].float().cpu(), gy[g*B:(g+1)*B].float().cpu(), 1, 1) for g in range(G)]
print("conv dgrad stacked maxdiff:", float((gx.float().cpu()-torch.cat(refs)).abs().max()))

gw = m.conv2d_wgrad(gy, x, ks, ks, 1, 1, G)
refs = [torch.nn.grad.conv2d_weight(x[g*B:(g+1)*B].float().cpu(), (CO,C,ks,ks), gy[g*B:(g+1)*B].float().cpu(), 1, 1) for g in range(G)]
print("conv wgrad stacked maxdiff:", float((gw.float().cpu()-torch.stack(refs)).abs().max()))

# stacked BN
wbn = torch.rand(G*C, device="cuda") + 0.5
bbn = torch.randn(G*C, device="cuda") * 0.1
rm = torch.zeros(G*C, device="cuda"); rv = torch.ones(G*C, device="cuda")
yb, mean, inv = m.batchnorm_fwd(x, wbn, bbn, rm, rv, torch.empty(0, device="cuda", dtype=torch.bfloat16), True, 0.1, 1e-5, False, G)
refs = []
for g in range(G):
    xs = x[g*B:(g+1)*B].float().cpu()
    mu = xs.mean(dim=(0,2,3)); var = xs.var(dim=(0,2,3), unbiased=False)
    xhat = (xs - mu[None,:,None,None]) / (var[None,:,None,None]+1e-5).sqrt()
    refs.append(xhat * wbn[g*C:(g+1)*C].cpu()[None,:,None,None] + bbn[g*C:(g+1)*C].cpu()[None,:,None,None])
print("bn fwd stacked maxdiff:", float((yb.float().cpu()-torch.cat(refs)).abs().max()))

gyb = torch.randn(G*B, C, H, W).bfloat16().cuda()
gxb, gwb, gbb, _ = m.batchnorm_bwd(gyb, yb, x, wbn, mean, inv, True, False, G)
print("bn bwd gx finite:", bool(torch.isfinite(gxb.float()).all()), "gw finite:", bool(torch.isfinite(gwb).all()))

# bmm (SLinear path)
a = torch.randn(G, B, 512).bfloat16().cuda()
wl = torch.randn(G, 10, 512).bfloat16().cuda()
out = m.bmm_nn(a, wl.transpose(1,2).contiguous())
ref = torch.matmul(a.float().cpu(), wl.float().cpu().transpose(1,2))
print("bmm maxdiff:", float((out.float().cpu()-ref).abs().max()))
