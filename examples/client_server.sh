#!/usr/bin/env bash
# The reference's deployment shape: one coordinator ("server") process +
# client worker processes, discovered over the control plane.  Here the
# processes are torch.distributed ranks; availability/role/session topics
# flow over the bus + cross-rank relay (federated/bus.py).
set -e
python -m colearn_federated_learning_amd.cli server \
    --driver-config 1 --world 2 &
SERVER=$!
python -m colearn_federated_learning_amd.cli client \
    --rank 1 --world 2 --driver-config 1
wait $SERVER
