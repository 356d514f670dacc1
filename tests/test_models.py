import pytest
import torch

from colearn_federated_learning_amd.config import DataConfig, ModelConfig
from colearn_federated_learning_amd.models import build_model


def test_logreg_forward_backward():
    m = build_model(ModelConfig(name="logreg", in_channels=1, image_size=28,
                                num_classes=10), DataConfig())
    x = torch.randn(4, 1, 28, 28)
    y = torch.randint(0, 10, (4,))
    loss = m.loss(x, y)
    loss.backward()
    assert m.fc.weight.grad is not None


@pytest.mark.parametrize("name", ["resnet18", "resnet50"])
def test_resnet_forward_backward(name):
    m = build_model(ModelConfig(name=name, num_classes=10), DataConfig())
    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 10, (2,))
    out = m(x)
    assert out.shape == (2, 10)
    m.loss(x, y).backward()
    grads = [p.grad for p in m.parameters()]
    assert all(g is not None for g in grads)
    assert sum(float(g.abs().sum()) for g in grads) > 0


def test_resnet_group_norm_mode():
    m = build_model(ModelConfig(name="resnet18"), DataConfig(), dp_mode=True)
    names = [type(mod).__name__ for mod in m.modules()]
    assert "GroupNorm" in names
    assert "ColearnBatchNorm2d" not in names


def test_resnet18_param_count():
    # CIFAR-style ResNet-18: ~11.2M params (ImageNet stem would be 11.7M)
    m = build_model(ModelConfig(name="resnet18", num_classes=10), DataConfig())
    n = sum(p.numel() for p in m.parameters())
    assert 10_500_000 < n < 12_000_000


def test_bert_forward_backward_tiny():
    from colearn_federated_learning_amd.models.bert import (
        BertConfig,
        BertForMaskedLM,
    )

    cfg = BertConfig(vocab_size=500, hidden_size=64, num_layers=2,
                     num_heads=4, intermediate_size=128,
                     max_position_embeddings=64)
    m = BertForMaskedLM(cfg)
    ids = torch.randint(0, 500, (2, 16))
    labels = torch.full((2, 16), -100)
    labels[:, 3] = ids[:, 3]
    loss = m.loss(ids, labels)
    loss.backward()
    assert m.tok_emb.weight.grad is not None
    logits = m(ids)
    assert logits.shape == (2, 16, 500)


def test_bert_base_param_count():
    from colearn_federated_learning_amd.models.bert import (
        BertConfig,
        BertForMaskedLM,
    )

    m = BertForMaskedLM(BertConfig())
    n = sum(p.numel() for p in m.parameters())
    # BERT-base ~110M params (tied decoder)
    assert 100_000_000 < n < 120_000_000


def test_functional_call_per_sample_path():
    """The torch.func path used by DP must work on every model family."""
    from torch.func import functional_call, grad, vmap

    m = build_model(ModelConfig(name="logreg", in_channels=1, image_size=8,
                                num_classes=3), DataConfig())
    params = {k: v.detach() for k, v in m.named_parameters()}
    x = torch.randn(5, 1, 8, 8)
    y = torch.randint(0, 3, (5,))

    def sample_loss(p, xi, yi):
        return m.loss(xi.unsqueeze(0), yi.unsqueeze(0), params=p, buffers={})

    g = vmap(grad(sample_loss), in_dims=(None, 0, 0))(params, x, y)
    assert g["fc.weight"].shape == (5, 3, 64)


def test_bert_masked_head_matches_full_logits():
    """loss() (masked-position head) == full-logits cross-entropy."""
    import torch.nn.functional as F

    from colearn_federated_learning_amd.models.bert import (
        BertConfig,
        BertForMaskedLM,
    )

    torch.manual_seed(3)
    cfg = BertConfig(vocab_size=300, hidden_size=64, num_layers=1,
                     num_heads=4, intermediate_size=128,
                     max_position_embeddings=32, dropout=0.0)
    m = BertForMaskedLM(cfg)
    m.eval()
    ids = torch.randint(0, 300, (4, 32))
    labels = torch.full((4, 32), -100)
    mask = torch.rand(4, 32) < 0.15
    mask[0, 0] = True  # at least one
    labels[mask] = ids[mask]
    loss_masked = m.loss(ids, labels)
    logits = m(ids)
    loss_full = F.cross_entropy(
        logits.float().view(-1, 300), labels.reshape(-1), ignore_index=-100
    )
    torch.testing.assert_close(loss_masked, loss_full, rtol=1e-5, atol=1e-6)


def test_kmajor_conv_cpu_fallback_matches_torch():
    """_Conv2dFn stores weights K-major [CO, KH, KW, CI]; the CPU path
    permutes for F.conv2d and permutes the weight grad back.  Guard the
    permute logic (fwd + both grads) against the plain torch oracle."""
    import torch.nn.functional as F

    from colearn_federated_learning_amd.ops.conv import colearn_conv2d

    torch.manual_seed(0)
    x = torch.randn(2, 8, 9, 9, requires_grad=True)
    w_std = torch.randn(16, 8, 3, 3)
    w_km = w_std.permute(0, 2, 3, 1).contiguous().requires_grad_()
    x2 = x.detach().clone().requires_grad_()
    w2 = w_std.clone().requires_grad_()

    y = colearn_conv2d(x, w_km, stride=2, padding=1)
    ref = F.conv2d(x2, w2, None, 2, 1)
    torch.testing.assert_close(y, ref)

    gy = torch.randn_like(y)
    y.backward(gy)
    ref.backward(gy)
    torch.testing.assert_close(x.grad, x2.grad)
    torch.testing.assert_close(w_km.grad, w2.grad.permute(0, 2, 3, 1))


def test_mnist_cnn_forward_backward_and_federated(tmp_path):
    """The reference's own model family (small MNIST CNN): shapes, grads,
    and a 2-round federated session on MNIST-shaped synthetic data."""
    from colearn_federated_learning_amd import driver_config, run
    from colearn_federated_learning_amd.models.registry import build_model

    cfg = driver_config(1).override(
        model__name="mnist_cnn", model__in_channels=1,
        model__image_size=28, rounds=2, num_clients=2,
        data__samples_per_client=16, train__batch_size=8,
        checkpoint_dir=str(tmp_path / "ck"),
    )
    m = build_model(cfg.model, cfg.data)
    x = torch.randn(4, 1, 28, 28)
    y = torch.randint(0, 10, (4,))
    loss = m.loss(x, y)
    assert m(x).shape == (4, 10)
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())
    hist = run(cfg)
    assert len(hist) == 2
    assert hist[-1]["mean_client_loss"] == hist[-1]["mean_client_loss"]
