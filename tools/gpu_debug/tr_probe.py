"""Dump ds_read_b64_tr_b16 (lane, j) <- LDS element mapping."""
import sys, os; sys.path.insert(0, os.getcwd())
import torch
from colearn_federated_learning_amd.ops import require_ext
m = require_ext()
for scheme in (0, 1, 2):
    out = m.tr_probe(scheme).cpu()
    print(f"== scheme {scheme} ==")
    for l in range(0, 64, 1):
        print(f"lane {l:2d}: " + " ".join(f"{int(v):4d}" for v in out[l]))
