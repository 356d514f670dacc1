"""dataset="file": real-data loading + federated partitioning."""

import torch

from colearn_federated_learning_amd import driver_config, run
from colearn_federated_learning_amd.data.files import (
    load_file_dataset,
    partition_file_indices,
)
from colearn_federated_learning_amd.data.sharder import make_shards
from colearn_federated_learning_amd.data.synthetic import make_client_dataset


def _write_dataset(path, n=200, classes=10):
    g = torch.Generator().manual_seed(7)
    x = torch.randn(n, 1, 28, 28, generator=g)
    y = torch.randint(0, classes, (n,), generator=g)
    torch.save({"x": x, "y": y}, path)
    return x, y


def _cfg(path, **over):
    base = dict(
        data__dataset="file", data__data_path=str(path),
        data__samples_per_client=40, model__in_channels=1,
        model__image_size=28, rounds=2, num_clients=4,
        train__batch_size=8,
    )
    base.update(over)
    return driver_config(1).override(**base)


def test_iid_partition_disjoint_and_deterministic(tmp_path):
    p = tmp_path / "d.pt"
    _write_dataset(p)
    cfg = _cfg(p)
    parts = partition_file_indices(cfg.data, 4)
    again = partition_file_indices(cfg.data, 4)
    allidx = torch.cat(parts)
    assert allidx.numel() == 160 and torch.unique(allidx).numel() == 160
    for a, b in zip(parts, again):
        assert torch.equal(a, b)


def test_dirichlet_partition_skews_and_disjoint(tmp_path):
    p = tmp_path / "d.pt"
    _, y = _write_dataset(p, n=2000)
    cfg = _cfg(p, data__partition="dirichlet", data__dirichlet_alpha=0.2)
    parts = partition_file_indices(cfg.data, 4)
    allidx = torch.cat(parts)
    assert torch.unique(allidx).numel() == allidx.numel()  # no reuse
    # heavy skew: some client's top class should dominate its shard
    tops = []
    for idx in parts:
        counts = torch.bincount(y[idx], minlength=10).float()
        tops.append(float(counts.max() / counts.sum()))
    assert max(tops) > 0.5  # alpha=0.2 concentrates mass


def test_shard_slices_real_samples(tmp_path):
    p = tmp_path / "d.pt"
    x, y = _write_dataset(p)
    cfg = _cfg(p)
    parts = make_shards(cfg.data, 4)
    shard = make_client_dataset(cfg.data, cfg.model, 2, parts[2])
    assert torch.equal(shard.x, x[parts[2]])
    assert torch.equal(shard.y, y[parts[2]])


def test_end_to_end_federated_run_on_file(tmp_path):
    p = tmp_path / "d.pt"
    _write_dataset(p, n=400)
    cfg = _cfg(p, model__name="mnist_cnn",
               checkpoint_dir=str(tmp_path / "ck"))
    hist = run(cfg)
    assert len(hist) == 2
    losses = [r["mean_client_loss"] for r in hist]
    assert all(l == l for l in losses) and losses[-1] < losses[0]


def test_file_dataset_world2_equals_world1(tmp_path):
    """Layout invariance holds for file-loaded data too: the partition is
    global-seeded, each rank slices only its own clients' samples."""
    import os

    import tests.test_equivalence as te

    p = tmp_path / "d.pt"
    _write_dataset(p, n=400)
    over = {"data__dataset": "file", "data__data_path": str(p),
            "data__samples_per_client": 40, "model__in_channels": 1,
            "model__image_size": 28, "rounds": 2}
    d1, d2 = str(tmp_path / "w1"), str(tmp_path / "w2")
    os.makedirs(d1), os.makedirs(d2)
    te._run_world(1, d1, over=over)
    te._run_world(2, d2, over=over)
    torch.testing.assert_close(te._final_weights(d1),
                               te._final_weights(d2),
                               rtol=1e-4, atol=1e-5)


def test_whole_dataset_split(tmp_path):
    """samples_per_client=0 => split the whole file evenly."""
    p = tmp_path / "d.pt"
    _write_dataset(p, n=100)
    cfg = _cfg(p, data__samples_per_client=0)
    parts = partition_file_indices(cfg.data, 4)
    assert all(len(q) == 25 for q in parts)
    allidx = torch.cat(parts)
    assert torch.unique(allidx).numel() == 100
