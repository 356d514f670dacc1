"""Determinism + graph-vs-eager localization (debug)."""
import torch
from colearn_federated_learning_amd.config import driver_config
from colearn_federated_learning_amd.federated.coordinator import FederatedServer

dev = torch.device("cuda", 0)
base = driver_config(2).override(rounds=1, num_clients=1,
                                 data__samples_per_client=64,
                                 train__batch_size=32)

def run_fit(cfg):
    s = FederatedServer(cfg, device=dev)
    c = list(s.clients.values())[0]
    s.flat.load_flat(s.global_flat); s.optimizer.reset()
    r = c.fit(s.model, s.optimizer, 0, runner=s.step_runner)
    torch.cuda.synchronize()
    return s.flat.flat_w.clone(), s.flat.flat_g.clone(), r.loss

w1, g1, l1 = run_fit(base)                                   # graphed
w2, g2, l2 = run_fit(base)                                   # graphed again
w3, g3, l3 = run_fit(base.override(train__use_graphs=False)) # eager
w4, g4, l4 = run_fit(base.override(train__use_graphs=False)) # eager again
print("graphed repeat identical:", torch.equal(w1, w2), l1, l2)
print("eager repeat identical:", torch.equal(w3, w4), l3, l4)
print("graph vs eager: wdiff", float((w1-w3).abs().max()),
      "gdiff", float((g1-g3).abs().max()), "loss", l1, l3)
