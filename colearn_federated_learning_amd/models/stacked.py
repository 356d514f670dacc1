"""Stacked-client models: G virtual clients as one set of batched kernels.

The MI355X answer to FL multiplexing: sequentially
training G small-model clients leaves the 256-CU chip ~12% occupied on
deep ResNet layers; stacking gives every kernel G x the workgroups and
per-client weights, with bit-compatible per-client semantics (each
client's forward/backward/optimizer touches only its own parameter
slice — gradients of a summed per-client loss separate exactly).

Parameter layout: every stacked parameter is [G, *template_shape] and the
module tree mirrors the template model, so parameter ORDER matches the
template's FlatParams layout — per-client delta extraction is a pure
reshape (federated/stacked.py).

CPU fallbacks use the grouped-conv / per-client-loop oracles, so the
stacked engine is fully testable without a GPU.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import native_enabled, require_ext
from ..ops.gemm import colearn_matmul
from ..ops.norms import _BatchNormFn
from .base import ColearnModel


def _native(x: torch.Tensor) -> bool:
    return x.is_cuda and native_enabled() and x.dtype == torch.bfloat16


class _SConv2dFn(torch.autograd.Function):
    # stacked weight is K-MAJOR: [G, CO, KH, KW, CI] (see ops.conv._Conv2dFn)
    @staticmethod
    def forward(x, w, stride: int, padding: int, G: int):
        if _native(x):
            return require_ext().conv2d_fwd_km(x, w, stride, padding)
        # CPU oracle: grouped conv with the client dim folded into channels
        Gb, C, H, W = x.shape
        B = Gb // G
        wstd = w.permute(0, 1, 4, 2, 3)  # [G, CO, CI, KH, KW]
        co, ci, kh, kw = wstd.shape[1:]
        xr = (x.view(G, B, C, H, W).transpose(0, 1)
              .reshape(B, G * C, H, W))
        yr = F.conv2d(xr, wstd.reshape(G * co, ci, kh, kw), None, stride,
                      padding, groups=G)
        P, Q = yr.shape[-2:]
        return (yr.view(B, G, co, P, Q).transpose(0, 1)
                .reshape(G * B, co, P, Q))

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, w, stride, padding, G = inputs
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.padding, ctx.G = stride, padding, G

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        s, p, G = ctx.stride, ctx.padding, ctx.G
        if _native(gy):
            m = require_ext()
            gx = m.conv2d_dgrad_km(gy, w, x.shape[2], x.shape[3], s, p)
            gw = m.conv2d_wgrad_km(gy, x, w.shape[2], w.shape[3], s, p, G)
        else:
            Gb, C, H, W = x.shape
            B = Gb // G
            wstd = w.permute(0, 1, 4, 2, 3)
            co, ci, kh, kw = wstd.shape[1:]
            xr = (x.view(G, B, C, H, W).transpose(0, 1)
                  .reshape(B, G * C, H, W))
            gyr = (gy.view(G, B, co, gy.shape[2], gy.shape[3])
                   .transpose(0, 1)
                   .reshape(B, G * co, gy.shape[2], gy.shape[3]))
            wg = wstd.reshape(G * co, ci, kh, kw)
            gxr = torch.nn.grad.conv2d_input(xr.shape, wg, gyr, s, p,
                                             groups=G)
            gwr = torch.nn.grad.conv2d_weight(xr, wg.shape, gyr, s, p,
                                              groups=G)
            gx = (gxr.view(B, G, C, H, W).transpose(0, 1)
                  .reshape(G * B, C, H, W))
            gw = gwr.view(G, co, ci, kh, kw).permute(0, 1, 3, 4, 2)
        return gx, gw, None, None, None


class _SConv2dShadowFn(torch.autograd.Function):
    """Shadow-weight conv: compute reads the persistent bf16 shadow view
    (one fused flat cast per step instead of a per-layer autocast cast),
    and the weight gradient is emitted in fp32 straight into the autograd
    slot of the fp32 master parameter (no bf16 grad round trip)."""

    @staticmethod
    def forward(x, w, shadow, shadow_t, gslot, stride: int, padding: int,
                G: int, skip_wgrad: bool = False):
        return require_ext().conv2d_fwd_km(x, shadow, stride, padding)

    @staticmethod
    def setup_context(ctx, inputs, output):
        (x, w, shadow, shadow_t, gslot, stride, padding, G,
         *rest) = inputs
        ctx.save_for_backward(x, shadow, shadow_t)
        ctx.gslot = gslot
        ctx.stride, ctx.padding, ctx.G = stride, padding, G
        # DP: the per-sample hook rule produces the weight grads; the
        # autograd mean-wgrad would be thrown away — skip the kernel
        ctx.skip_wgrad = rest[0] if rest else False

    @staticmethod
    def backward(ctx, gy):
        x, shadow, shadow_t = ctx.saved_tensors
        s, p, G = ctx.stride, ctx.padding, ctx.G
        m = require_ext()
        if not ctx.needs_input_grad[0]:
            # stem conv: the input batch never needs a gradient — skip
            # the whole dgrad kernel
            gx = None
        elif shadow_t is not None:
            # pre-transposed dgrad weight (refreshed once per step by
            # shadow_sync) — no per-call permute+contiguous copy
            gx = m.conv2d_dgrad_km_pre(gy, shadow_t, x.shape[2],
                                       x.shape[3], s, p)
        else:
            gx = m.conv2d_dgrad_km(gy, shadow, x.shape[2], x.shape[3],
                                   s, p)
        if ctx.skip_wgrad:
            gw = None
        elif ctx.gslot is not None:
            # write the fp32 grad straight into the flat-grad view; no
            # AccumulateGrad add kernel runs for this parameter
            m.conv2d_wgrad_km32(gy, x, shadow.shape[2], shadow.shape[3],
                                s, p, G, ctx.gslot)
            gw = None
        else:
            e = torch.empty(0, device=x.device, dtype=torch.float32)
            gw = m.conv2d_wgrad_km32(gy, x, shadow.shape[2],
                                     shadow.shape[3], s, p, G, e)
        return gx, gw, None, None, None, None, None, None, None


class SConv2d(nn.Module):
    def __init__(self, G: int, cin: int, cout: int, k: int,
                 stride: int = 1, padding: int = 0) -> None:
        super().__init__()
        w = torch.empty(G, cout, cin, k, k)
        nn.init.kaiming_normal_(w, mode="fan_out", nonlinearity="relu")
        self.weight = nn.Parameter(w.permute(0, 1, 3, 4, 2).contiguous())
        self.wshadow = None    # bf16 flat-buffer view (StackedFlatParams)
        self.wshadow_t = None  # pre-transposed dgrad layout [G,CI,R,S,CO]
        self.wgslot = None     # fp32 flat-grad view (direct wgrad target)
        self.dp_mode = False   # per-sample rule owns the weight grads
        self.G, self.stride, self.padding = G, stride, padding

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.wshadow is not None and x.is_cuda and native_enabled():
            if x.dtype != torch.bfloat16:
                x = x.bfloat16()
            return _SConv2dShadowFn.apply(x, self.weight, self.wshadow,
                                          self.wshadow_t, self.wgslot,
                                          self.stride, self.padding,
                                          self.G, self.dp_mode)
        w = self.weight
        if torch.is_autocast_enabled("cuda") and x.is_cuda:
            dt = torch.get_autocast_dtype("cuda")
            x, w = x.to(dt), w.to(dt)
        return _SConv2dFn.apply(x, w, self.stride, self.padding, self.G)


class SBatchNorm2d(nn.Module):
    """Per-client BatchNorm: stats over each client's own batch slice."""

    def __init__(self, G: int, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = False) -> None:
        super().__init__()
        self.G, self.C = G, num_features
        self.eps, self.momentum, self.relu = eps, momentum, relu
        self.weight = nn.Parameter(torch.ones(G, num_features))
        self.bias = nn.Parameter(torch.zeros(G, num_features))
        self.register_buffer("running_mean", torch.zeros(G, num_features))
        self.register_buffer("running_var", torch.ones(G, num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))
        self.gslot_w = None  # fp32 flat-grad views (direct grad targets)
        self.gslot_b = None

    def forward(self, x: torch.Tensor, residual=None) -> torch.Tensor:
        # no num_batches_tracked increment: unused with fixed momentum and
        # it cost one launch per BN layer per step (see ops/norms.py)
        if _native(x):
            y, _, _ = _BatchNormFn.apply(
                x, self.weight.view(-1), self.bias.view(-1),
                self.running_mean.view(-1), self.running_var.view(-1),
                residual, self.training, self.momentum, self.eps,
                self.relu, self.G, self.gslot_w, self.gslot_b,
            )
            return y
        # CPU oracle: per-client F.batch_norm
        G, C = self.G, self.C
        B = x.shape[0] // G
        outs = []
        for g in range(G):
            xs = x[g * B:(g + 1) * B]
            ys = F.batch_norm(
                xs.float(), self.running_mean[g], self.running_var[g],
                self.weight[g], self.bias[g], self.training, self.momentum,
                self.eps,
            )
            outs.append(ys)
        y = torch.cat(outs, dim=0)
        if residual is not None:
            y = y + residual.float()
        if self.relu:
            y = y.relu()
        return y.to(x.dtype)


class _SLinearShadowFn(torch.autograd.Function):
    """Shadow-weight stacked linear: forward reads the persistent bf16
    shadow (no per-layer autocast weight cast — BERT's weights are 85M x
    G, the casts were ~2.8 ms/step), bias is fused into the GEMM epilogue
    from the fp32 master, and the weight/bias grads are emitted in FP32
    straight into the flat-grad slots (no bf16 grad round trip, no
    AccumulateGrad adds)."""

    @staticmethod
    def forward(x3, w, b, wsh, gslot_w, gslot_b, skip_wgrad: bool = False):
        return require_ext().bmm_nt_bias(x3, wsh, b)

    @staticmethod
    def setup_context(ctx, inputs, output):
        x3, w, b, wsh, gslot_w, gslot_b, *rest = inputs
        ctx.save_for_backward(x3, wsh)
        # grad slots stash on ctx directly: they are written in place by
        # every layer's backward, so the saved-tensor version check would
        # (correctly for normal tensors, wrongly here) reject them
        ctx.gslot_w, ctx.gslot_b = gslot_w, gslot_b
        ctx.skip_wgrad = rest[0] if rest else False

    @staticmethod
    def backward(ctx, gy):
        x3, wsh = ctx.saved_tensors
        gslot_w, gslot_b = ctx.gslot_w, ctx.gslot_b
        m = require_ext()
        gy = gy.contiguous()
        gx = m.bmm_nn(gy, wsh)
        if not ctx.skip_wgrad:  # DP: per-sample rule owns these
            m.bmm_tn_f32(gy, x3, gslot_w)
            m.colsum_f32(gy, gslot_b)  # deterministic bias grads
        return gx, None, None, None, None, None, None


class SLinear(nn.Module):
    def __init__(self, G: int, cin: int, cout: int) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.empty(G, cout, cin))
        self.bias = nn.Parameter(torch.zeros(G, cout))
        nn.init.normal_(self.weight, std=0.02)
        self.wshadow = None  # bf16 flat-buffer view (StackedFlatParams)
        self.dp_mode = False
        self.G = G

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x: [G*B, F] -> per-client GEMM via batched MFMA
        G = self.G
        x3 = x.view(G, -1, x.shape[-1])
        if (self.wshadow is not None and x.is_cuda and native_enabled()
                and self.weight.grad is not None
                and (x.dtype == torch.bfloat16
                     or torch.is_autocast_enabled("cuda"))):
            # bf16 compute only: fp32 configs (logreg) keep the master path
            if x3.dtype != torch.bfloat16:
                x3 = x3.bfloat16()
            out = _SLinearShadowFn.apply(
                x3.contiguous(), self.weight, self.bias, self.wshadow,
                self.weight.grad, self.bias.grad, self.dp_mode,
            )
            return out.reshape(-1, out.shape[-1])
        w, b = self.weight, self.bias
        if torch.is_autocast_enabled("cuda") and x.is_cuda:
            dt = torch.get_autocast_dtype("cuda")
            x3, w, b = x3.to(dt), w.to(dt), b.to(dt)
        out = colearn_matmul(x3, w.transpose(1, 2)) + b[:, None, :]
        return out.reshape(-1, out.shape[-1])


class _SGNActFn(torch.autograd.Function):
    """Native stacked GroupNorm + per-client affine + optional residual
    add + ReLU, fwd and gx/gres backward in two kernels (elemwise.hip).
    The per-client weight/bias grads come from the per-sample hook rule
    (DP mode), so this Function returns None for them."""

    @staticmethod
    def forward(x, w, b, residual, groups: int, B_per_client: int,
                relu: bool):
        y, mean, invstd = require_ext().sgn_fwd(
            x, w, b,
            residual if residual is not None
            else torch.empty(0, device=x.device, dtype=x.dtype),
            groups, B_per_client, 1e-5, relu)
        return y, mean, invstd

    @staticmethod
    def setup_context(ctx, inputs, output):
        x, w = inputs[0], inputs[1]
        y, mean, invstd = output
        ctx.save_for_backward(x, w, y, mean, invstd)
        ctx.groups, ctx.bpc, ctx.relu = inputs[4], inputs[5], inputs[6]
        ctx.has_res = inputs[3] is not None
        ctx.mark_non_differentiable(output[1], output[2])
        ctx.set_materialize_grads(False)

    @staticmethod
    def backward(ctx, gy, _gm, _gi):
        x, w, y, mean, invstd = ctx.saved_tensors
        gx, gres = require_ext().sgn_bwd(
            gy.contiguous(), x.contiguous(), y, w, mean, invstd,
            ctx.groups, ctx.bpc, ctx.relu, ctx.has_res)
        return (gx, None, None, gres if ctx.has_res else None, None,
                None, None)


class _SGNAct(nn.Module):
    """Stacked GroupNorm substitute (DP mode): per-SAMPLE stats,
    per-CLIENT affine, fused residual/ReLU epilogue on the native path.
    Params live directly on this module (the per-sample grad engine
    hooks it and masks grad_output by the saved ReLU output)."""

    def __init__(self, G: int, planes: int, relu: bool) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(G, planes))
        self.bias = nn.Parameter(torch.zeros(G, planes))
        self.G = G
        self.groups = min(32, planes)
        self.do_relu = relu

    def forward(self, x: torch.Tensor, residual=None) -> torch.Tensor:
        B = x.shape[0] // self.G
        if _native(x) and self.weight.shape[1] % 8 == 0:
            y, _, _ = _SGNActFn.apply(x, self.weight, self.bias,
                                      residual, self.groups, B,
                                      self.do_relu)
            return y
        xn = F.group_norm(x, self.groups)
        # DETACHED affine in the input dtype: the per-sample hook rule
        # owns the weight/bias grads (autograd's would be overwritten),
        # and matching dtype avoids promoting the whole activation to
        # fp32 through the broadcast
        w = self.weight.detach().to(x.dtype).repeat_interleave(B, dim=0)
        b = self.bias.detach().to(x.dtype).repeat_interleave(B, dim=0)
        y = xn * w[:, :, None, None] + b[:, :, None, None]
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.do_relu else y


def _snorm(G: int, planes: int, group_norm: bool,
           relu: bool = False) -> nn.Module:
    if group_norm:
        return _SGNAct(G, planes, relu)
    return SBatchNorm2d(G, planes, relu=relu)


class SBasicBlock(nn.Module):
    expansion = 1

    def __init__(self, G: int, inplanes: int, planes: int, stride: int,
                 group_norm: bool) -> None:
        super().__init__()
        self.conv1 = SConv2d(G, inplanes, planes, 3, stride, 1)
        self.bn1 = _snorm(G, planes, group_norm, relu=True)
        self.conv2 = SConv2d(G, planes, planes, 3, 1, 1)
        self.bn2 = _snorm(G, planes, group_norm, relu=True)
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or inplanes != planes:
            self.downsample = nn.Sequential(
                SConv2d(G, inplanes, planes, 1, stride, 0),
                _snorm(G, planes, group_norm),
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=idt)


class SBottleneck(nn.Module):
    expansion = 4

    def __init__(self, G: int, inplanes: int, planes: int, stride: int,
                 group_norm: bool) -> None:
        super().__init__()
        self.conv1 = SConv2d(G, inplanes, planes, 1, 1, 0)
        self.bn1 = _snorm(G, planes, group_norm, relu=True)
        self.conv2 = SConv2d(G, planes, planes, 3, stride, 1)
        self.bn2 = _snorm(G, planes, group_norm, relu=True)
        self.conv3 = SConv2d(G, planes, planes * 4, 1, 1, 0)
        self.bn3 = _snorm(G, planes * 4, group_norm, relu=True)
        self.downsample: Optional[nn.Module] = None
        if stride != 1 or inplanes != planes * 4:
            self.downsample = nn.Sequential(
                SConv2d(G, inplanes, planes * 4, 1, stride, 0),
                _snorm(G, planes * 4, group_norm),
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        idt = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=idt)


class StackedResNet(ColearnModel):
    """G independent ResNets sharing kernels.  Input [G*B, C, H, W]
    client-major; ``loss_terms`` returns the per-client loss vector [G]."""

    def __init__(self, G: int, block, layers: List[int],
                 num_classes: int = 10, in_channels: int = 3,
                 small_input: bool = True, group_norm: bool = False) -> None:
        super().__init__()
        self.G = G
        self.group_norm = group_norm  # DP mode: per-sample hooks need
        self.inplanes = 64            # the SLinear module call (no fused head)
        if small_input:
            self.stem = nn.Sequential(
                SConv2d(G, in_channels, 64, 3, 1, 1),
                _snorm(G, 64, group_norm, relu=True),
            )
            self.pool = nn.Identity()
        else:
            self.stem = nn.Sequential(
                SConv2d(G, in_channels, 64, 7, 2, 3),
                _snorm(G, 64, group_norm, relu=True),
            )
            self.pool = nn.MaxPool2d(3, 2, 1)
        self.layer1 = self._make_layer(G, block, 64, layers[0], 1,
                                       group_norm)
        self.layer2 = self._make_layer(G, block, 128, layers[1], 2,
                                       group_norm)
        self.layer3 = self._make_layer(G, block, 256, layers[2], 2,
                                       group_norm)
        self.layer4 = self._make_layer(G, block, 512, layers[3], 2,
                                       group_norm)
        self.fc = SLinear(G, 512 * block.expansion, num_classes)

    def _make_layer(self, G, block, planes, n, stride, group_norm):
        blocks = [block(G, self.inplanes, planes, stride, group_norm)]
        self.inplanes = planes * block.expansion
        for _ in range(1, n):
            blocks.append(block(G, self.inplanes, planes, 1, group_norm))
        return nn.Sequential(*blocks)

    def features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool(self.stem(x))
        return self.layer4(self.layer3(self.layer2(self.layer1(x))))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.adaptive_avg_pool2d(self.features(x), 1).flatten(1)
        return self.fc(x)  # [G*B, classes]

    def loss_terms(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        from ..ops.head import fused_head_applicable, fused_head_loss

        feat = self.features(x)
        if not self.group_norm and fused_head_applicable(
                feat, self.fc.weight, self.fc.weight.shape[1]):
            # fused avgpool+FC+CE (4 launches fwd+bwd, fp32 head math,
            # direct grad writes) — csrc/head.hip
            return fused_head_loss(feat, self.fc.weight, self.fc.bias,
                                   y, self.G)
        logits = self.fc(F.adaptive_avg_pool2d(feat, 1).flatten(1))
        per_sample = F.cross_entropy(logits.float(), y, reduction="none")
        return per_sample.view(self.G, -1).mean(dim=1)  # [G]

    def loss(self, x, y, params=None, buffers=None):
        return self.loss_terms(x, y).sum()


class StackedLogReg(ColearnModel):
    def __init__(self, G: int, in_features: int, num_classes: int) -> None:
        super().__init__()
        self.G = G
        self.fc = SLinear(G, in_features, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(x.flatten(1))

    def loss_terms(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        logits = self(x)
        per_sample = F.cross_entropy(logits.float(), y, reduction="none")
        return per_sample.view(self.G, -1).mean(dim=1)

    def loss(self, x, y, params=None, buffers=None):
        return self.loss_terms(x, y).sum()


def build_stacked_model(model_cfg, data_cfg, G: int,
                        dp_mode: bool = False) -> Optional[ColearnModel]:
    """Stacked equivalent of models.registry.build_model; None when the
    family has no stacked implementation (falls back to sequential)."""
    name = model_cfg.name
    if name == "logreg":
        nf = (model_cfg.in_channels * model_cfg.image_size
              * model_cfg.image_size)
        return StackedLogReg(G, nf, model_cfg.num_classes)
    if name in ("resnet18", "resnet50"):
        block = SBasicBlock if name == "resnet18" else SBottleneck
        layers = [2, 2, 2, 2] if name == "resnet18" else [3, 4, 6, 3]
        return StackedResNet(
            G, block, layers, num_classes=model_cfg.num_classes,
            in_channels=model_cfg.in_channels,
            small_input=model_cfg.image_size <= 64, group_norm=dp_mode,
        )
    if name == "bert_base":
        from .registry import bert_config
        from .stacked_bert import StackedBert

        return StackedBert(G, bert_config(model_cfg, data_cfg))
    return None
