"""CPU paths of the op wrappers (the oracles the GPU kernels are tested
against) vs plain torch compositions, forward AND backward."""

import torch

from colearn_federated_learning_amd.ops.elemwise import colearn_gelu
from colearn_federated_learning_amd.ops.gemm import (
    colearn_linear,
    colearn_matmul,
    colearn_matmul_nt,
)


def _both(fn_ours, fn_ref, *shapes, seed=0):
    torch.manual_seed(seed)
    ts = [torch.randn(s, dtype=torch.float64) for s in shapes]
    a1 = [t.clone().requires_grad_(True) for t in ts]
    a2 = [t.clone().requires_grad_(True) for t in ts]
    y1, y2 = fn_ours(*a1), fn_ref(*a2)
    torch.testing.assert_close(y1, y2)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    for t1, t2 in zip(a1, a2):
        torch.testing.assert_close(t1.grad, t2.grad)


def test_gelu_matches_torch_exact():
    _both(colearn_gelu, torch.nn.functional.gelu, (5, 33))


def test_gelu_gradcheck():
    x = torch.randn(3, 7, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(colearn_gelu, (x,))


def test_linear_matches_torch():
    _both(lambda x, w, b: colearn_linear(x, w, b),
          torch.nn.functional.linear, (6, 10), (4, 10), (4,))


def test_linear_no_bias_3d():
    _both(lambda x, w: colearn_linear(x, w),
          lambda x, w: torch.nn.functional.linear(x, w),
          (2, 5, 10), (4, 10))


def test_matmul_matches_torch():
    _both(colearn_matmul, torch.matmul, (3, 6, 8), (3, 8, 5))


def test_matmul_nt_matches_torch():
    _both(colearn_matmul_nt, lambda a, b: a @ b.transpose(-1, -2),
          (3, 6, 8), (3, 5, 8))


def test_gelu_vmap_rule():
    """ops declare generate_vmap_rule: the DP per-sample fallback relies
    on vmapping them."""
    from torch.func import grad, vmap

    x = torch.randn(4, 9)

    def f(xi):
        return colearn_gelu(xi).sum()

    g = vmap(grad(f))(x)
    xf = x.clone().requires_grad_(True)
    colearn_gelu(xf).sum().backward()
    torch.testing.assert_close(g, xf.grad)


def test_grad_sample_shared_module():
    """A weight-shared module applied twice must produce the SUM of both
    calls' per-sample grads (per-call closure saves + accumulation)."""
    import torch.nn as nn

    from colearn_federated_learning_amd.ops.grad_sample import (
        GradSampleEngine,
    )

    torch.manual_seed(0)
    lin = nn.Linear(4, 4, bias=False)

    class Twice(nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = lin

        def forward(self, x):
            return self.lin(torch.relu(self.lin(x)))

    m = Twice()
    x = torch.randn(3, 4)
    eng = GradSampleEngine(m)
    with eng:
        m(x).sum().backward()
        gs = lin.weight.grad_sample.clone()
        eng.clear()
    # oracle: per-sample autograd
    for i in range(3):
        m.zero_grad()
        m(x[i : i + 1]).sum().backward()
        torch.testing.assert_close(gs[i], lin.weight.grad, rtol=1e-5,
                                   atol=1e-6)


def test_linear_and_matmul_vmap_rules():
    from torch.func import grad, vmap

    from colearn_federated_learning_amd.ops.gemm import (
        colearn_linear,
        colearn_matmul,
    )

    torch.manual_seed(1)
    w = torch.randn(3, 5)
    xs = torch.randn(4, 5)

    def f(xi, wi):
        return colearn_linear(xi.unsqueeze(0), wi).sum()

    gw = vmap(grad(f, argnums=1), in_dims=(0, None))(xs, w)
    for i in range(4):
        wi = w.clone().requires_grad_(True)
        colearn_linear(xs[i : i + 1], wi).sum().backward()
        torch.testing.assert_close(gw[i], wi.grad)

    a = torch.randn(4, 2, 6)
    b = torch.randn(4, 6, 3)
    out = vmap(colearn_matmul)(a, b)
    torch.testing.assert_close(out, a @ b)
