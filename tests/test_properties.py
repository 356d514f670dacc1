"""Property-based tests (hypothesis) for sharding and compression."""

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from colearn_federated_learning_amd.config import DataConfig
from colearn_federated_learning_amd.data import make_shards
from colearn_federated_learning_amd.ops.topk import (
    topk_compress,
    topk_decompress,
)
from colearn_federated_learning_amd.utils.seed import stable_hash


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 12), st.integers(1, 64), st.integers(0, 10_000),
       st.sampled_from(["iid", "dirichlet"]))
def test_shards_shape_and_range(num_clients, spc, seed, partition):
    cfg = DataConfig(samples_per_client=spc, seed=seed, partition=partition,
                     num_classes=7)
    shards = make_shards(cfg, num_clients)
    assert len(shards) == num_clients
    for s in shards:
        assert s.shape == (spc,)
        assert int(s.min()) >= 0 and int(s.max()) < 7
    # determinism
    again = make_shards(cfg, num_clients)
    assert all(torch.equal(a, b) for a, b in zip(shards, again))


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 500), st.integers(1, 600), st.integers(0, 2**31 - 1))
def test_topk_roundtrip_properties(k, n, seed):
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(n, generator=g)
    idx, val = topk_compress(v, k)
    kk = min(k, n)
    assert idx.shape == (kk,)
    dense = topk_decompress(idx, val, n)
    # kept entries exact, dropped entries zero
    torch.testing.assert_close(dense[idx.long()], v[idx.long()])
    assert int((dense != 0).sum()) <= kk
    # kept set dominates dropped set in |value|
    if kk < n:
        mask = torch.ones(n, dtype=torch.bool)
        mask[idx.long()] = False
        if mask.any() and kk > 0:
            assert v[idx.long()].abs().min() >= v[mask].abs().max() - 1e-6


@settings(max_examples=50, deadline=None)
@given(st.tuples(st.integers(), st.text(max_size=20), st.integers()))
def test_stable_hash_range(parts):
    h = stable_hash(*parts)
    assert 0 <= h < 2**31
    assert h == stable_hash(*parts)


@settings(max_examples=25, deadline=None)
@given(st.floats(1e-5, 10.0), st.integers(1, 64), st.integers(1, 1024))
def test_config_override_yaml_roundtrip(lr, bs, spc):
    from colearn_federated_learning_amd.config import (
        FederatedConfig,
        driver_config,
    )

    cfg = driver_config(2).override(
        train__lr=lr, train__batch_size=bs, data__samples_per_client=spc)
    assert cfg.train.lr == lr and cfg.train.batch_size == bs
    assert cfg.data.samples_per_client == spc
    back = FederatedConfig.from_dict(cfg.to_dict())
    assert back.to_dict() == cfg.to_dict()


def test_config_override_unknown_key_raises():
    import pytest

    from colearn_federated_learning_amd.config import driver_config

    with pytest.raises(KeyError):
        driver_config(2).override(samples_per_client=8)  # missing data__
    with pytest.raises(KeyError):
        driver_config(2).override(train__learning_rate=0.1)


def test_load_yaml_file(tmp_path):
    """load_yaml: a config written as YAML to disk loads back identically
    (the CLI's --config file path)."""
    import yaml

    from colearn_federated_learning_amd.config import (
        driver_config,
        load_yaml,
    )

    cfg = driver_config(4).override(rounds=7, train__lr=0.125,
                                    model__num_layers=2)
    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump(cfg.to_dict()))
    back = load_yaml(str(p))
    assert back.to_dict() == cfg.to_dict()
    assert back.rounds == 7 and back.model.num_layers == 2
