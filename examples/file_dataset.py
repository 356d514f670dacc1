"""Federate over a REAL dataset loaded from a local file.

Writes a toy dataset to disk in the expected format (a torch.save'd
{"x": [N,C,H,W] float, "y": [N] int64} dict — e.g. pre-exported MNIST),
then runs a Dirichlet-partitioned federated session on it with the
reference's MNIST-CNN model.
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from colearn_federated_learning_amd import driver_config, run

path = os.path.join(tempfile.gettempdir(), "colearn_demo_mnist.pt")
g = torch.Generator().manual_seed(0)
torch.save({"x": torch.randn(2000, 1, 28, 28, generator=g),
            "y": torch.randint(0, 10, (2000,), generator=g)}, path)

cfg = driver_config(1).override(
    data__dataset="file", data__data_path=path,
    data__samples_per_client=256, data__partition="dirichlet",
    data__dirichlet_alpha=0.3,
    model__name="mnist_cnn", model__in_channels=1, model__image_size=28,
    rounds=3, num_clients=4,
)
for rec in run(cfg):
    print(f"round {rec['round']}: loss={rec['mean_client_loss']:.4f}")
