"""CoLearn-AMD: an MI355X-native federated-learning simulation engine.

A from-scratch rebuild of the capabilities of the CoLearn federated-learning
prototype (reference: aferaudo/CoLearn_Federated_Learning — EdgeSys '20,
"CoLearn: Enabling Federated Learning in MUD-compliant IoT Edge Networks").
The reference is a pure-Python PySyft/MQTT prototype targeting Raspberry
Pi-class edge devices; this package re-architects the same capability stack
for a single 8x AMD Instinct MI355X node:

* virtual federated clients are multiplexed onto GPUs (one RCCL rank per GPU),
* round aggregation (FedAvg / FedProx / DP-FedAvg) is a weighted RCCL
  all-reduce of model deltas over xGMI with bucket sizes chosen for the
  7 point-to-point links per GPU,
* the client hot path (conv/GEMM fwd+bwd, BatchNorm/LayerNorm, fused SGD/Adam,
  per-sample grad clipping + DP noise, top-k gradient sparsification) is
  hand-written HIP for CDNA4 (gfx950) with MFMA/LDS tiling,
* CoLearn's MQTT-style control plane (availability announcements,
  trainer/evaluator role assignment, session lifecycle) is preserved at API
  level by an in-process pub/sub bus.

Because the reference checkout was unavailable, the
capability contract implemented here is the one mandated by BASELINE.json.
"""

__version__ = "0.1.0"

from .config import (  # noqa: F401
    AggregationConfig,
    DataConfig,
    DPConfig,
    FederatedConfig,
    ModelConfig,
    TrainingConfig,
    driver_config,
)
from .federated.bus import MessageBus  # noqa: F401
from .federated.client import FederatedClient  # noqa: F401
from .federated.coordinator import FederatedServer  # noqa: F401
from .federated.run import run  # noqa: F401
