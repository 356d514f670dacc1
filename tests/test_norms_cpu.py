"""Norm layers vs stock PyTorch (the CPU oracle used for GPU kernels too)."""

import pytest
import torch

from colearn_federated_learning_amd.ops.norms import (
    ColearnBatchNorm2d,
    ColearnLayerNorm,
)


def test_batchnorm_matches_torch_train():
    torch.manual_seed(0)
    ours, ref = ColearnBatchNorm2d(8), torch.nn.BatchNorm2d(8)
    x = torch.randn(4, 8, 5, 5)
    x2 = x.clone().requires_grad_(True)
    x1 = x.clone().requires_grad_(True)
    y1, y2 = ours(x1), ref(x2)
    torch.testing.assert_close(y1, y2, rtol=1e-5, atol=1e-5)
    gy = torch.randn_like(y1)
    y1.backward(gy)
    y2.backward(gy)
    torch.testing.assert_close(x1.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(ours.weight.grad, ref.weight.grad,
                               rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(ours.running_mean, ref.running_mean,
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(ours.running_var, ref.running_var,
                               rtol=1e-5, atol=1e-6)


def test_batchnorm_eval_mode():
    ours, ref = ColearnBatchNorm2d(4), torch.nn.BatchNorm2d(4)
    with torch.no_grad():
        for m in (ours, ref):
            m.running_mean.copy_(torch.tensor([1.0, 2.0, 3.0, 4.0]))
            m.running_var.copy_(torch.tensor([1.0, 4.0, 9.0, 16.0]))
    ours.eval(), ref.eval()
    x = torch.randn(2, 4, 3, 3)
    torch.testing.assert_close(ours(x), ref(x), rtol=1e-5, atol=1e-5)


def test_layernorm_matches_torch():
    torch.manual_seed(1)
    h = 96
    ours = ColearnLayerNorm(h, eps=1e-5)
    ref = torch.nn.LayerNorm(h, eps=1e-5)
    x = torch.randn(3, 7, h)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1, y2 = ours(x1), ref(x2)
    torch.testing.assert_close(y1, y2, rtol=1e-5, atol=1e-5)
    gy = torch.randn_like(y1)
    y1.backward(gy)
    y2.backward(gy)
    torch.testing.assert_close(x1.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(ours.weight.grad, ref.weight.grad,
                               rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(ours.bias.grad, ref.bias.grad,
                               rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("G", [1, 3])
def test_grouped_layernorm_slot_dim(G):
    """Grouped (stacked-client) LN: weight [G,H] must get a [G,H] grad —
    including G==1, where the slot dim once got squeezed away (the 8-rank
    layout stacks one client per rank)."""
    from colearn_federated_learning_amd.ops.norms import _LayerNormFn

    torch.manual_seed(0)
    B, S, H = 2, 5, 16
    x = torch.randn(G * B, S, H, requires_grad=True)
    w = torch.randn(G, H, requires_grad=True)
    b = torch.randn(G, H, requires_grad=True)
    y, _, _ = _LayerNormFn.apply(x, w, b, 1e-5, G)
    y.backward(torch.randn_like(y))
    assert w.grad.shape == (G, H) and b.grad.shape == (G, H)
    # per-slot oracle
    x3 = x.detach().reshape(G, B, S, H)
    for g in range(G):
        xr = x3[g].clone().requires_grad_(True)
        wr = w.detach()[g].clone().requires_grad_(True)
        br = b.detach()[g].clone().requires_grad_(True)
        yr = torch.nn.functional.layer_norm(xr, (H,), wr, br, 1e-5)
        torch.testing.assert_close(
            yr, y.detach().reshape(G, B, S, H)[g], rtol=1e-4, atol=1e-5)
