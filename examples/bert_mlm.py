"""BERT-base masked-LM federated session (driver config 4), small CPU
version.  On MI355X this runs the stacked BERT engine: per-client
batched MFMA GEMMs, fused multi-head attention, grouped LayerNorm.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from colearn_federated_learning_amd import driver_config, run

cfg = driver_config(4).override(
    rounds=2, num_clients=2, data__samples_per_client=8,
    data__seq_len=16, data__vocab_size=256, train__batch_size=4,
    train__dtype="float32",
)
history = run(cfg)
for rec in history:
    print(f"round {rec['round']}: mlm_loss={rec['mean_client_loss']:.4f}")
