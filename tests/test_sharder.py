import torch

from colearn_federated_learning_amd.config import DataConfig, ModelConfig
from colearn_federated_learning_amd.data import (
    dirichlet_shards,
    iid_shards,
    make_client_dataset,
    make_shards,
)


def test_iid_deterministic():
    cfg = DataConfig(samples_per_client=256, seed=7)
    a = iid_shards(cfg, 4)
    b = iid_shards(cfg, 4)
    for x, y in zip(a, b):
        assert torch.equal(x, y)


def test_shards_independent_of_num_clients_prefix():
    # client k's shard must not depend on how many other clients exist
    cfg = DataConfig(samples_per_client=128, seed=3)
    a = iid_shards(cfg, 2)
    b = iid_shards(cfg, 8)
    assert torch.equal(a[0], b[0])
    assert torch.equal(a[1], b[1])


def test_dirichlet_skew():
    cfg = DataConfig(samples_per_client=2048, seed=5, dirichlet_alpha=0.1,
                     partition="dirichlet", num_classes=10)
    shards = dirichlet_shards(cfg, 8)
    # with alpha=0.1, shards should be heavily skewed: the top class of each
    # client should hold far more than the IID 10% share
    top_fracs = []
    for s in shards:
        counts = torch.bincount(s, minlength=10).float()
        top_fracs.append((counts.max() / counts.sum()).item())
    assert sum(top_fracs) / len(top_fracs) > 0.35


def test_dirichlet_vs_iid_differ():
    cfg_i = DataConfig(samples_per_client=512, seed=5)
    cfg_d = DataConfig(samples_per_client=512, seed=5, partition="dirichlet",
                       dirichlet_alpha=0.3)
    assert not torch.equal(make_shards(cfg_i, 2)[0], make_shards(cfg_d, 2)[0])


def test_client_dataset_shapes():
    dcfg = DataConfig(dataset="cifar10_synth", samples_per_client=64)
    mcfg = ModelConfig(in_channels=3, image_size=32)
    labels = iid_shards(dcfg, 1)[0]
    ds = make_client_dataset(dcfg, mcfg, 0, labels)
    assert ds.x.shape == (64, 3, 32, 32)
    assert ds.y.shape == (64,)
    # batches cover everything exactly once
    seen = sum(x.shape[0] for x, _ in ds.batches(16))
    assert seen == 64


def test_text_dataset_mlm():
    dcfg = DataConfig(dataset="text_synth", samples_per_client=32,
                      seq_len=64, vocab_size=1000, mlm_probability=0.15)
    mcfg = ModelConfig(name="bert_base")
    labels = make_shards(dcfg, 1)[0]
    ds = make_client_dataset(dcfg, mcfg, 0, labels)
    assert ds.x.shape == (32, 64)
    masked = (ds.y != -100)
    frac = masked.float().mean().item()
    assert 0.05 < frac < 0.3
    # masked positions carry the mask token in inputs
    assert (ds.x[masked] == 103).all()


def test_text_synth_vocab_floor():
    """vocab_size must cover the [MASK] token id (103) — loud error, not
    out-of-range embedding indices downstream."""
    import pytest

    from colearn_federated_learning_amd.config import driver_config
    from colearn_federated_learning_amd.data.synthetic import (
        make_client_dataset,
    )

    cfg = driver_config(4).override(data__vocab_size=50,
                                    data__samples_per_client=4)
    with pytest.raises(ValueError, match="vocab_size"):
        make_client_dataset(cfg.data, cfg.model, 0,
                            torch.zeros(4, dtype=torch.long))
