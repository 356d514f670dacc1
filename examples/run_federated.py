"""Minimal federated session with the CoLearn-shaped Python API.

Runs entirely on CPU (driver config 1: 2-client logistic regression on
synthetic MNIST-shape data) — the same call works on MI355X GPUs and
under torchrun for one-rank-per-GPU execution.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from colearn_federated_learning_amd import driver_config, run

cfg = driver_config(1).override(rounds=5, num_evaluators=1)
history = run(cfg)
for rec in history:
    print(f"round {rec['round']}: loss={rec['mean_client_loss']:.4f} "
          f"eval_acc={rec.get('eval_accuracy', float('nan')):.3f} "
          f"samples/s={rec['samples_per_sec']:.0f}")
