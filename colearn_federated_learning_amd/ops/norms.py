"""BatchNorm2d and LayerNorm with fused HIP kernels on the GPU path.

SURVEY.md §2b N6 (BatchNorm, ResNet path) and N7 (LayerNorm, BERT path).
Both are memory-bound: the HIP kernels (csrc/norms.hip) do one fused sweep
per pass with vectorized bf16 loads (G13) and wave-level reductions; stats
accumulate in fp32.  CPU path = stock PyTorch math (oracle).

Autocast note: like PyTorch's own norms, statistics are computed in fp32;
inputs may be bf16, parameters stay fp32.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from . import is_vmapped, native_enabled, require_ext


def _native(x: torch.Tensor) -> bool:
    return (x.is_cuda and native_enabled() and not is_vmapped(x)
            and x.dtype == torch.bfloat16)


class _BatchNormFn(torch.autograd.Function):
    """BatchNorm with optional fused residual-add + ReLU (the ResNet block
    epilogue runs inside the BN sweeps on gfx950 — see csrc/norms.hip)."""

    generate_vmap_rule = True

    @staticmethod
    def forward(x, weight, bias, running_mean, running_var, residual,
                training: bool, momentum: float, eps: float, relu: bool,
                groups: int = 1, gslot_w=None, gslot_b=None):
        # gslot_w/gslot_b: flat-grad views; when set, the backward writes
        # the param grads straight into them and returns None for w/b, so
        # no AccumulateGrad add kernels run (stacked engine fast path).
        if _native(x):
            m = require_ext()
            y, save_mean, save_invstd = m.batchnorm_fwd(
                x, weight, bias, running_mean, running_var,
                residual if residual is not None
                else torch.empty(0, device=x.device, dtype=x.dtype),
                training, momentum, eps, relu, groups,
            )
            return y, save_mean, save_invstd
        assert groups == 1, "stacked BN has no CPU fast path here"

        if training:
            dims = (0, 2, 3)
            xf = x.float()
            mean = xf.mean(dim=dims)
            var = xf.var(dim=dims, unbiased=False)
            n = x.numel() / x.shape[1]
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(
                    var * n / max(n - 1, 1), alpha=momentum
                )
            invstd = (var + eps).rsqrt()
        else:
            # clone: an input returned as an output may not be saved for
            # backward under the setup_context protocol
            mean = running_mean.clone()
            invstd = (running_var + eps).rsqrt()
            xf = x.float()
        y = (xf - mean[None, :, None, None]) * invstd[None, :, None, None] \
            * weight[None, :, None, None] + bias[None, :, None, None]
        if residual is not None:
            y = y + residual.float()
        if relu:
            y = y.relu()
        return y.to(x.dtype), mean, invstd

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, weight = inputs[0], inputs[1]
        y, save_mean, save_invstd = output
        ctx.save_for_backward(x, weight, save_mean, save_invstd, y)
        ctx.training = inputs[6]
        ctx.relu = inputs[9]
        ctx.groups = inputs[10] if len(inputs) > 10 else 1
        ctx.gslot_w = inputs[11] if len(inputs) > 11 else None
        ctx.gslot_b = inputs[12] if len(inputs) > 12 else None
        ctx.n_inputs = len(inputs)
        ctx.has_res = inputs[5] is not None
        ctx.mark_non_differentiable(output[1], output[2])
        # without this the engine materializes ZERO tensors for the two
        # non-differentiable outputs' grads — 2 fills per BN per step
        # (40/step on ResNet-18, measured: the FillFunctor tail)
        ctx.set_materialize_grads(False)

    @staticmethod
    def backward(ctx, gy, _gmean, _ginvstd):
        x, weight, mean, invstd, y = ctx.saved_tensors
        direct = ctx.gslot_w is not None and _native(x)
        if _native(x):
            e = torch.empty(0, device=x.device, dtype=torch.float32)
            gx, gw, gb, gres = require_ext().batchnorm_bwd(
                gy, y, x, weight, mean, invstd, ctx.training, ctx.relu,
                ctx.groups,
                ctx.gslot_w if direct else e,
                ctx.gslot_b if direct else e,
            )
        else:
            gyf = gy.float()
            if ctx.relu:
                gyf = gyf * (y > 0).float()
            gres = gyf.to(x.dtype)
            xf = x.float()
            xhat = (xf - mean[None, :, None, None]) * invstd[None, :, None, None]
            n = x.numel() / x.shape[1]
            gb = gyf.sum(dim=(0, 2, 3))
            gw = (gyf * xhat).sum(dim=(0, 2, 3))
            if ctx.training:
                gx = (weight * invstd)[None, :, None, None] * (
                    gyf - gb[None, :, None, None] / n
                    - xhat * gw[None, :, None, None] / n
                )
            else:
                gx = (weight * invstd)[None, :, None, None] * gyf
            gx = gx.to(x.dtype)
        out = [gx, None if direct else gw, None if direct else gb,
               None, None, gres if ctx.has_res else None,
               None, None, None, None, None, None, None]
        return tuple(out[: ctx.n_inputs])


class ColearnBatchNorm2d(nn.Module):
    """BatchNorm2d with an optional fused ``+ residual`` / ReLU epilogue."""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = False) -> None:
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x: torch.Tensor, residual=None) -> torch.Tensor:
        # num_batches_tracked is kept for state-dict compatibility but NOT
        # incremented: torch only consults it when momentum is None (ours
        # is fixed), and the on-device `+= 1` cost one extra launch per BN
        # layer per captured step (20/step on ResNet-18 — measured in
        # profiles/final_resnet18_kernel_stats.txt as the 1980-launch
        # CUDAFunctorOnSelf_add<long> row).
        y, _, _ = _BatchNormFn.apply(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            residual, self.training, self.momentum, self.eps, self.relu, 1,
        )
        return y


class _LayerNormFn(torch.autograd.Function):
    """LayerNorm over the last dim; ``groups`` > 1 = stacked-client LN
    (weight/bias [G, H], client-major rows — the BERT stacking path)."""

    generate_vmap_rule = True

    @staticmethod
    def forward(x, weight, bias, eps: float, groups: int = 1,
                gslot_w=None, gslot_b=None):
        if _native(x):
            y, mean, invstd = require_ext().layernorm_fwd(
                x.contiguous(), weight.view(-1), bias.view(-1), eps,
                groups,
            )
            return y, mean, invstd
        if groups > 1:
            # CPU oracle: per-client loop
            G = groups
            rows = x.numel() // x.shape[-1]
            x3 = x.reshape(G, rows // G, x.shape[-1])
            w2, b2 = weight.view(G, -1), bias.view(G, -1)
            xf = x3.float()
            mean = xf.mean(dim=-1)
            var = xf.var(dim=-1, unbiased=False)
            invstd = (var + eps).rsqrt()
            y = ((xf - mean[..., None]) * invstd[..., None]
                 * w2[:, None, :] + b2[:, None, :])
            return (y.to(x.dtype).reshape(x.shape),
                    mean.reshape(x.shape[:-1]),
                    invstd.reshape(x.shape[:-1]))
        xf = x.float()
        mean = xf.mean(dim=-1)
        var = xf.var(dim=-1, unbiased=False)
        invstd = (var + eps).rsqrt()
        y = ((xf - mean[..., None]) * invstd[..., None] * weight
             + bias).to(x.dtype)
        return y, mean, invstd

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, weight = inputs[0], inputs[1]
        _, mean, invstd = output
        ctx.save_for_backward(x, weight, mean, invstd)
        ctx.groups = inputs[4] if len(inputs) > 4 else 1
        ctx.gslot_w = inputs[5] if len(inputs) > 5 else None
        ctx.gslot_b = inputs[6] if len(inputs) > 6 else None
        ctx.n_inputs = len(inputs)
        ctx.mark_non_differentiable(output[1], output[2])
        ctx.set_materialize_grads(False)  # see _BatchNormFn

    @staticmethod
    def backward(ctx, gy, _gmean, _ginvstd):
        x, weight, mean, invstd = ctx.saved_tensors
        G = ctx.groups
        gy = gy.contiguous()
        if _native(x) and ctx.gslot_w is not None:
            # direct-slot path: gw/gb land straight in the flat-grad
            # views (no clones, no AccumulateGrad adds)
            gx = require_ext().layernorm_bwd_slots(
                gy, x.contiguous(), weight.view(-1), mean, invstd, G,
                ctx.gslot_w.view(-1), ctx.gslot_b.view(-1),
            )
            gw = gb = None
        elif _native(x):
            gx, gw, gb = require_ext().layernorm_bwd(
                gy, x.contiguous(), weight.view(-1), mean, invstd, G
            )
            # weight.shape covers both plain [H] and stacked [G, H] —
            # including the G==1 stack ([1, H]), whose grad must keep the
            # leading slot dim (8-rank runs stack ONE client per rank)
            gw = gw.view(weight.shape)
            gb = gb.view(weight.shape)
        elif G > 1:
            h = x.shape[-1]
            rows = x.numel() // h
            x3 = x.reshape(G, rows // G, h).float()
            gy3 = gy.reshape(G, rows // G, h).float()
            m3 = mean.reshape(G, rows // G)
            i3 = invstd.reshape(G, rows // G)
            w2 = weight.view(G, h)
            xhat = (x3 - m3[..., None]) * i3[..., None]
            gw = (gy3 * xhat).sum(dim=1).view(weight.shape)
            gb = gy3.sum(dim=1).view(weight.shape)
            wg = gy3 * w2[:, None, :]
            gx = (i3[..., None] * (
                wg - wg.mean(dim=-1, keepdim=True)
                - xhat * (wg * xhat).mean(dim=-1, keepdim=True)
            )).to(x.dtype).reshape(x.shape)
        else:
            xf, gyf = x.float(), gy.float()
            xhat = (xf - mean[..., None]) * invstd[..., None]
            h = x.shape[-1]
            gw = (gyf * xhat).reshape(-1, h).sum(dim=0).view(weight.shape)
            gb = gyf.reshape(-1, h).sum(dim=0).view(weight.shape)
            wg = gyf * weight
            gx = invstd[..., None] * (
                wg - wg.mean(dim=-1, keepdim=True)
                - xhat * (wg * xhat).mean(dim=-1, keepdim=True)
            )
            gx = gx.to(x.dtype)
        out = [gx, gw, gb, None, None, None, None]
        return tuple(out[: ctx.n_inputs])


class ColearnLayerNorm(nn.Module):
    def __init__(self, hidden: int, eps: float = 1e-12) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y, _, _ = _LayerNormFn.apply(x, self.weight, self.bias, self.eps)
        return y
