#!/usr/bin/env bash
# The reference's deployment shape: one coordinator ("server") process +
# client worker processes, discovered over the control plane.  Here the
# processes are torch.distributed ranks; availability/role/session topics
# flow over the bus + cross-rank relay (federated/bus.py).
# PORT env var overrides the rendezvous port (default 29711).
set -e
PORT="${PORT:-29711}"
python -m colearn_federated_learning_amd.cli server \
    --driver-config 1 --world 2 --master-port "$PORT" &
SERVER=$!
python -m colearn_federated_learning_amd.cli client \
    --rank 1 --world 2 --driver-config 1 --master-port "$PORT"
wait $SERVER
