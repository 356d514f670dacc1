"""Typed configuration for the CoLearn-AMD engine.

The reference (aferaudo/CoLearn_Federated_Learning) configures each script via
argparse flags (host/port/rounds/epochs/lr/batch/MQTT broker — SURVEY.md §2a
row 10).  Here one dataclass tree expresses every run, including the five
driver configs from BASELINE.json verbatim (``driver_config(1..5)``), and can
be overridden from YAML or CLI key=value pairs.
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class DataConfig:
    """Synthetic dataset shape + federated sharding."""

    dataset: str = "mnist_synth"  # mnist_synth | cifar10_synth | text_synth | file
    # dataset="file": torch.save'd {"x": [N,C,H,W] float, "y": [N] int64}
    # partitioned across clients by the same IID/Dirichlet axes (data/files.py)
    data_path: Optional[str] = None
    num_classes: int = 10
    samples_per_client: int = 1024
    # sharding: "iid" or "dirichlet" (non-IID label skew, alpha controls skew)
    partition: str = "iid"
    dirichlet_alpha: float = 0.3
    seed: int = 1234
    # text-only
    seq_len: int = 128
    vocab_size: int = 30522
    mlm_probability: float = 0.15


@dataclass
class ModelConfig:
    name: str = "logreg"  # logreg | mnist_cnn | resnet18 | resnet50 | bert_base
    num_classes: int = 10
    in_channels: int = 3
    image_size: int = 32
    dropout: float = 0.1  # bert_base only (0 => deterministic paths)
    # bert_base architecture overrides (None => BertConfig defaults,
    # i.e. the real 110M bert-base).  Small values give a CPU-testable
    # transformer with the full stacked/MLM machinery.
    hidden_size: Optional[int] = None
    num_layers: Optional[int] = None
    num_heads: Optional[int] = None
    intermediate_size: Optional[int] = None


@dataclass
class TrainingConfig:
    local_epochs: int = 1
    batch_size: int = 32
    lr: float = 0.05
    momentum: float = 0.9
    weight_decay: float = 0.0
    optimizer: str = "sgd"  # sgd | adam
    dtype: str = "float32"  # float32 | bfloat16 (autocast compute dtype)
    # FedProx proximal term strength (0 => plain FedAvg local objective)
    fedprox_mu: float = 0.0
    max_local_steps: Optional[int] = None  # cap steps/epoch (bench + tests)
    # capture the client train step in a hipGraph on the GPU (steprunner.py)
    use_graphs: bool = True
    # stacked-client execution (federated/stacked.py); sequential if False
    use_stacked: bool = True


@dataclass
class DPConfig:
    """DP-FedAvg: per-sample clip to C, Gaussian noise sigma*C at aggregation."""

    enabled: bool = False
    clip_norm: float = 1.0
    noise_multiplier: float = 0.0  # sigma
    delta: float = 1e-5


@dataclass
class AggregationConfig:
    algorithm: str = "fedavg"  # fedavg | fedprox (server-side both = weighted mean)
    backend: str = "auto"  # auto | gloo | nccl (nccl == RCCL on ROCm)
    # bucket size for the all-reduce of flattened model deltas; tuned for the
    # 7 p2p xGMI links per MI355X GPU (SURVEY.md §5 comm backend analysis).
    bucket_bytes: int = 8 * 1024 * 1024
    # top-k gradient sparsification: fraction of entries kept (0 disables).
    # Applied per-CLIENT to each client's delta (semantically identical on
    # any rank layout); "sparse" transport additionally ships the rank-level
    # aggregate as (idx, val) pairs over an all-gather instead of the dense
    # all-reduce.
    topk_ratio: float = 0.0
    transport: str = "dense"  # dense | sparse
    # >0: split the delta into bucket_bytes chunks all-reduced on this many
    # extra communicators/streams (xGMI link-overlap experiment).  0 = one
    # RCCL all-reduce — RCCL already drives multiple channels/links for a
    # single large collective, so this is the default.
    comm_streams: int = 0


@dataclass
class FederatedConfig:
    rounds: int = 10
    num_clients: int = 8  # total virtual clients across the job
    clients_per_round: Optional[int] = None  # None => all clients
    num_evaluators: int = 0  # clients holding the evaluator role each round
    straggler_timeout_s: Optional[float] = None
    drop_client_prob: float = 0.0  # fault injection: client fails to report
    data: DataConfig = field(default_factory=DataConfig)
    model: ModelConfig = field(default_factory=ModelConfig)
    train: TrainingConfig = field(default_factory=TrainingConfig)
    dp: DPConfig = field(default_factory=DPConfig)
    agg: AggregationConfig = field(default_factory=AggregationConfig)
    checkpoint_dir: Optional[str] = None
    checkpoint_every: int = 0  # rounds; 0 => only at end when dir set
    resume: bool = False
    metrics_path: Optional[str] = None  # JSONL per-round metrics
    seed: int = 0
    target_accuracy: Optional[float] = None  # early stop for rounds-to-target

    # ------------------------------------------------------------------
    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), indent=2, sort_keys=True)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "FederatedConfig":
        def build(tp, val):
            if dataclasses.is_dataclass(tp) and isinstance(val, dict):
                kw = {}
                for f in dataclasses.fields(tp):
                    if f.name in val:
                        kw[f.name] = build(_resolve(f), val[f.name])
                return tp(**kw)
            return val

        def _resolve(f):
            m = {
                "data": DataConfig, "model": ModelConfig, "train": TrainingConfig,
                "dp": DPConfig, "agg": AggregationConfig,
            }
            return m.get(f.name, type(None))

        kw = {}
        for f in dataclasses.fields(cls):
            if f.name in d:
                sub = {"data": DataConfig, "model": ModelConfig,
                       "train": TrainingConfig, "dp": DPConfig,
                       "agg": AggregationConfig}.get(f.name)
                kw[f.name] = build(sub, d[f.name]) if sub else d[f.name]
        return cls(**kw)

    def override(self, **kv: Any) -> "FederatedConfig":
        """Return a copy with dotted-key overrides, e.g. train__lr=0.1.

        Unknown keys raise: a typo'd override (``samples_per_client=`` for
        ``data__samples_per_client=``) silently configuring nothing cost a
        64x-too-long test before this check existed.
        """
        d = self.to_dict()
        for key, val in kv.items():
            parts = key.split("__")
            node = d
            for p in parts[:-1]:
                if p not in node:
                    raise KeyError(f"unknown config group {p!r} in {key!r}")
                node = node[p]
            if parts[-1] not in node:
                raise KeyError(f"unknown config key {key!r}")
            node[parts[-1]] = val
        return FederatedConfig.from_dict(d)


def driver_config(n: int) -> FederatedConfig:
    """The five BASELINE.json configs, by 1-based index."""
    if n == 1:
        # 2-client FedAvg logistic regression, synthetic MNIST-shape, CPU/gloo
        return FederatedConfig(
            rounds=5, num_clients=2,
            data=DataConfig(dataset="mnist_synth", samples_per_client=512),
            model=ModelConfig(name="logreg", num_classes=10, in_channels=1,
                              image_size=28),
            train=TrainingConfig(local_epochs=1, batch_size=32, lr=0.05,
                                 dtype="float32"),
            agg=AggregationConfig(backend="gloo"),
        )
    if n == 2:
        # ResNet-18 CIFAR-10-shape FedAvg, 8 IID clients = 8 GPUs, bf16
        return FederatedConfig(
            rounds=10, num_clients=8,
            data=DataConfig(dataset="cifar10_synth", samples_per_client=4096),
            model=ModelConfig(name="resnet18", num_classes=10, in_channels=3,
                              image_size=32),
            train=TrainingConfig(local_epochs=1, batch_size=64, lr=0.1,
                                 momentum=0.9, dtype="bfloat16"),
            agg=AggregationConfig(backend="auto"),
        )
    if n == 3:
        # ResNet-50 FedProx mu=0.01, Dirichlet(0.3) shards, 8 clients
        return FederatedConfig(
            rounds=10, num_clients=8,
            data=DataConfig(dataset="cifar10_synth", samples_per_client=4096,
                            partition="dirichlet", dirichlet_alpha=0.3),
            model=ModelConfig(name="resnet50", num_classes=10, in_channels=3,
                              image_size=32),
            train=TrainingConfig(local_epochs=1, batch_size=64, lr=0.1,
                                 momentum=0.9, dtype="bfloat16",
                                 fedprox_mu=0.01),
            agg=AggregationConfig(algorithm="fedprox", backend="auto"),
        )
    if n == 4:
        # BERT-base MLM FedAvg on synthetic seq_len=128 text, 8 clients, bf16
        return FederatedConfig(
            rounds=10, num_clients=8,
            data=DataConfig(dataset="text_synth", samples_per_client=2048,
                            seq_len=128, vocab_size=30522),
            model=ModelConfig(name="bert_base"),
            train=TrainingConfig(local_epochs=1, batch_size=32, lr=1e-4,
                                 optimizer="adam", dtype="bfloat16"),
            agg=AggregationConfig(backend="auto"),
        )
    if n == 5:
        # DP-FedAvg + top-k compression, ResNet-18, 64 virtual clients / 8 GPUs
        return FederatedConfig(
            rounds=10, num_clients=64,
            data=DataConfig(dataset="cifar10_synth", samples_per_client=512),
            model=ModelConfig(name="resnet18", num_classes=10, in_channels=3,
                              image_size=32),
            train=TrainingConfig(local_epochs=1, batch_size=64, lr=0.1,
                                 momentum=0.9, dtype="bfloat16"),
            dp=DPConfig(enabled=True, clip_norm=1.0, noise_multiplier=0.8),
            agg=AggregationConfig(backend="auto", topk_ratio=0.01),
        )
    raise ValueError(f"driver config index must be 1..5, got {n}")


def load_yaml(path: str) -> FederatedConfig:
    import yaml

    with open(path) as f:
        return FederatedConfig.from_dict(yaml.safe_load(f))


def parse_overrides(pairs: List[str]) -> Dict[str, Any]:
    """Parse CLI ``a.b.c=value`` pairs into override kwargs."""
    out: Dict[str, Any] = {}
    for p in pairs:
        if "=" not in p:
            raise ValueError(f"override must be key=value, got {p!r}")
        k, v = p.split("=", 1)
        try:
            val: Any = json.loads(v)
        except json.JSONDecodeError:
            val = v
        out[k.replace(".", "__")] = val
    return out
