"""DP-FedAvg session (per-sample clipping + Gaussian noise) with the
RDP accountant's (epsilon, delta) report per round.

CPU-runnable small version of driver config 5; on MI355X the same
config trains all virtual clients' per-sample-clipped steps in a single
stacked backward (federated/stacked.py _dp_step).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from colearn_federated_learning_amd import FederatedServer, driver_config

cfg = driver_config(5).override(
    rounds=3, num_clients=4, data__samples_per_client=64,
    train__batch_size=16, train__dtype="float32",
)
server = FederatedServer(cfg)
for r in range(cfg.rounds):
    rec = server.run_round(r)
    print(f"round {r}: loss={rec['mean_client_loss']:.4f} "
          f"epsilon={rec['dp_epsilon']:.2f} (delta={cfg.dp.delta})")
