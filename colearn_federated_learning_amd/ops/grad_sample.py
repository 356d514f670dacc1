"""Per-sample gradients via module hooks (hard part #2).

DP-SGD needs the gradient of EVERY sample's loss w.r.t. every parameter.
The naive route (torch.func vmap of per-sample grad) can't see through
functorch wrappers into the native kernels, so it falls back to eager
composites and runs ~10x slower.  This engine computes per-sample grads
the Opacus way — from quantities the normal backward already produces:

  conv          gw[b] = wgrad(gy[b], x[b])  ==  the STACKED wgrad kernel
                with G = batch and per-client batch 1 (conv_bf16.hip) —
                per-sample conv gradients are literally one launch here.
  linear        gw[b] = gy[b] (outer) x[b]        (einsum)
  group norm    gw[b,c] = sum_hw gy*xhat, gb[b,c] = sum_hw gy

A forward hook records each module's input; a full-backward hook computes
``param.grad_sample`` ([B, *param.shape]).  The regular backward still
fills the flat grad buffer — DP overwrites it with the clipped mean
(ops.dp.clip_and_reduce, the fused norm/clip kernels).

The rules return grads w.r.t. the MEAN loss, unscaled: the batch factor
folds into ops.dp.clip_and_reduce(pre_scale=B) so no extra sweep ever
touches the [B, *param] tensors.
"""

from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import native_enabled, require_ext


def _conv_grad_sample(module, x: torch.Tensor,
                      gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    """NOTE: rules return UNSCALED per-sample grads; the loss-mean factor
    folds into the clip pass (ops.dp.clip_and_reduce pre_scale) so no
    extra elementwise sweep touches the [B, *param] tensors."""
    b = x.shape[0]
    if x.is_cuda and gy.dtype == torch.bfloat16 and x.dtype != torch.bfloat16:
        x = x.bfloat16()  # stem input arrives fp32 under autocast
    # module.weight is K-major [CO, KH, KW, CI]
    kh, kw = module.weight.shape[1], module.weight.shape[2]
    if (x.is_cuda and native_enabled() and x.dtype == torch.bfloat16
            and gy.dtype == torch.bfloat16):
        gw = require_ext().conv2d_wgrad_fp32_km(
            gy.contiguous(), x.contiguous(), kh, kw,
            module.stride, module.padding, b,
        )
        return {"weight": gw}
    std_shape = (module.weight.shape[0], module.weight.shape[3], kh, kw)
    gws = [
        torch.nn.grad.conv2d_weight(
            x[i : i + 1].float(), std_shape, gy[i : i + 1].float(),
            module.stride, module.padding,
        ).permute(0, 2, 3, 1)
        for i in range(b)
    ]
    return {"weight": torch.stack(gws).float()}


def _linear_grad_sample(module, x: torch.Tensor,
                        gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    x2 = x.reshape(x.shape[0], -1, x.shape[-1]).float()
    g2 = gy.reshape(gy.shape[0], -1, gy.shape[-1]).float()
    out = {"weight": torch.einsum("bsi,bsj->bij", g2, x2)}
    if getattr(module, "bias", None) is not None:
        out["bias"] = g2.sum(dim=1)
    return out


def _groupnorm_grad_sample(module, x: torch.Tensor,
                           gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    xf, gf = x.float(), gy.float()
    xhat = F.group_norm(xf, module.num_groups, eps=module.eps)
    dims = tuple(range(2, x.dim()))
    return {
        "weight": (gf * xhat).sum(dim=dims),
        "bias": gf.sum(dim=dims),
    }


# ---- stacked-module rules (per-sample grads are weight-free, so the
# template rules apply with batch = G*B; shapes come out per-TEMPLATE) ----
def _sconv_grad_sample(module, x: torch.Tensor,
                       gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    b = x.shape[0]
    if x.is_cuda and gy.dtype == torch.bfloat16 and x.dtype != torch.bfloat16:
        x = x.bfloat16()
    # stacked weight is [G, CO, KH, KW, CI]
    kh, kw = module.weight.shape[2], module.weight.shape[3]
    if (x.is_cuda and native_enabled() and x.dtype == torch.bfloat16
            and gy.dtype == torch.bfloat16):
        # BF16 per-sample grads: at G*B = 4096 samples the per-sample
        # grad slabs are ~180 GB in fp32 — bf16 halves both the wgrad
        # write and the clip pipeline's two read sweeps.  Clipping is
        # applied to exactly these bf16 values (the quantity summed), so
        # the sensitivity bound is unaffected.
        gw = require_ext().conv2d_wgrad_km(
            gy.contiguous(), x.contiguous(), kh, kw,
            module.stride, module.padding, b,
        )
        return {"weight": gw}
    co, ci = module.weight.shape[1], module.weight.shape[4]
    gws = [
        torch.nn.grad.conv2d_weight(
            x[i : i + 1].float(), (co, ci, kh, kw),
            gy[i : i + 1].float(), module.stride, module.padding,
        ).permute(0, 2, 3, 1)
        for i in range(b)
    ]
    return {"weight": torch.stack(gws).float()}


def _sgnact_grad_sample(module, x: torch.Tensor,
                        gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    # grad_output is wrt the POST-ReLU output: the kernel masks by the
    # saved output in-flight (no gy*(y>0) temporary)
    if (x.is_cuda and native_enabled() and x.dtype == torch.bfloat16
            and gy.dtype == torch.bfloat16):
        yout = (module._gs_output if module.do_relu
                else torch.empty(0, device=x.device, dtype=x.dtype))
        gw, gb = require_ext().gn_grad_sample(
            x, gy, yout, module.groups, 1e-5, module.do_relu)
        return {"weight": gw, "bias": gb}
    if module.do_relu:
        gy = gy * (module._gs_output > 0).to(gy.dtype)
    xf, gf = x.float(), gy.float()
    xhat = F.group_norm(xf, module.groups, eps=1e-5)
    dims = tuple(range(2, x.dim()))
    return {
        "weight": (gf * xhat).sum(dim=dims),
        "bias": gf.sum(dim=dims),
    }


def _slinear_grad_sample(module, x: torch.Tensor,
                         gy: torch.Tensor) -> Dict[str, torch.Tensor]:
    # per-sample grads vs the sample's OWN client slice = plain outer
    # product (a sample only touches its client's weights)
    x2 = x.reshape(x.shape[0], -1, x.shape[-1]).float()
    g2 = gy.reshape(gy.shape[0], -1, gy.shape[-1]).float()
    return {
        "weight": torch.einsum("bsi,bsj->bij", g2, x2),
        "bias": g2.sum(dim=1),
    }


class GradSampleEngine:
    """Attach to a model; after ``loss.backward()`` on a MEAN loss, every
    supported parameter has ``.grad_sample`` ([B, *shape], fp32, unscaled
    — multiply by B for per-sample-loss gradients)."""

    def __init__(self, model: nn.Module) -> None:
        from ..models.base import CLinear
        from ..models.bert import _Linear as BertLinear
        from ..models.resnet import _Conv2d
        from ..models.stacked import SConv2d, SLinear, _SGNAct

        self.model = model
        self._handles = []
        self._rules = []
        for mod in model.modules():
            if isinstance(mod, _Conv2d):
                self._rules.append((mod, _conv_grad_sample))
            elif isinstance(mod, (nn.Linear, BertLinear, CLinear)):
                self._rules.append((mod, _linear_grad_sample))
            elif isinstance(mod, nn.GroupNorm):
                self._rules.append((mod, _groupnorm_grad_sample))
            elif isinstance(mod, SConv2d):
                self._rules.append((mod, _sconv_grad_sample))
            elif isinstance(mod, SLinear):
                self._rules.append((mod, _slinear_grad_sample))
            elif isinstance(mod, _SGNAct):
                self._rules.append((mod, _sgnact_grad_sample))
        covered = {
            p for mod, _ in self._rules for p in mod.parameters(recurse=False)
        }
        uncovered = [
            n for n, p in model.named_parameters()
            if p.requires_grad and p not in covered
        ]
        if uncovered:
            raise NotImplementedError(
                f"no per-sample gradient rule for parameters: {uncovered}"
            )

    # ------------------------------------------------------------------
    def __enter__(self) -> "GradSampleEngine":
        # TENSOR grad hooks, not register_full_backward_hook: the module
        # backward-hook machinery (BackwardHookFunction) COPIES every
        # hooked module's grads — measured ~60 ms/step of pure copy
        # kernels at config 5.  A hook on the output tensor fires with
        # the grad itself, copy-free, with identical semantics (grad wrt
        # the module output, summed over all consumers).
        for mod, rule in self._rules:
            self._handles.append(
                mod.register_forward_hook(self._make_fwd(rule))
            )
        return self

    def __exit__(self, *exc) -> None:
        for h in self._handles:
            h.remove()
        self._handles.clear()
        for mod, _ in self._rules:
            if hasattr(mod, "_gs_input"):
                del mod._gs_input
            if hasattr(mod, "_gs_output"):
                del mod._gs_output

    def _make_fwd(self, rule):
        def fwd_hook(mod, inputs, output):
            # per-CALL saves live in this closure (not on the module), so
            # a weight-shared module invoked twice in one forward gets the
            # right input paired with each call's grad; per-call grads
            # ACCUMULATE into grad_sample (cleared each step).
            x_saved = inputs[0].detach()
            out_saved = (output.detach()
                         if isinstance(output, torch.Tensor) else None)
            if isinstance(output, torch.Tensor) and output.requires_grad:
                def grad_cb(gy, mod=mod):
                    # the GN rule masks grad_output by the post-ReLU output
                    mod._gs_input = x_saved
                    mod._gs_output = out_saved
                    gs = rule(mod, x_saved, gy.detach())
                    for name, g in gs.items():
                        p = getattr(mod, name)
                        prev = getattr(p, "grad_sample", None)
                        p.grad_sample = g if prev is None else prev + g

                output.register_hook(grad_cb)

        return fwd_hook

    # ------------------------------------------------------------------
    def grad_samples(self) -> List[torch.Tensor]:
        """Per-sample grads in named_parameters() order."""
        return [
            p.grad_sample for _, p in self.model.named_parameters()
            if p.requires_grad
        ]

    def clear(self) -> None:
        for _, p in self.model.named_parameters():
            if hasattr(p, "grad_sample"):
                del p.grad_sample
