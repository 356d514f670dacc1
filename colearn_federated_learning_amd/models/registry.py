"""Model factory for the driver configs."""

from __future__ import annotations

from ..config import DataConfig, ModelConfig
from .base import ColearnModel, LogisticRegression


def build_model(model_cfg: ModelConfig, data_cfg: DataConfig,
                dp_mode: bool = False) -> ColearnModel:
    name = model_cfg.name
    if name == "logreg":
        in_features = (
            model_cfg.in_channels * model_cfg.image_size * model_cfg.image_size
        )
        return LogisticRegression(in_features, model_cfg.num_classes)
    if name == "mnist_cnn":
        from .cnn import MnistCNN

        return MnistCNN(
            in_channels=model_cfg.in_channels,
            num_classes=model_cfg.num_classes,
            image_size=model_cfg.image_size,
        )
    if name in ("resnet18", "resnet50"):
        from .resnet import resnet18, resnet50

        fn = resnet18 if name == "resnet18" else resnet50
        return fn(
            num_classes=model_cfg.num_classes,
            in_channels=model_cfg.in_channels,
            small_input=model_cfg.image_size <= 64,
            group_norm=dp_mode,  # DP-SGD is incompatible with batch stats
        )
    if name == "bert_base":
        from .bert import BertForMaskedLM

        return BertForMaskedLM(bert_config(model_cfg, data_cfg))
    raise ValueError(f"unknown model {name!r}")


def bert_config(model_cfg: ModelConfig, data_cfg: DataConfig):
    """One BertConfig for BOTH the sequential and stacked builders, so the
    two engines always agree on architecture (equivalence tests depend on
    this).  ModelConfig's optional dims override the bert-base defaults."""
    from .bert import BertConfig

    kw = dict(
        vocab_size=data_cfg.vocab_size,
        max_position_embeddings=max(data_cfg.seq_len, 512),
        dropout=model_cfg.dropout,
    )
    for f in ("hidden_size", "num_layers", "num_heads", "intermediate_size"):
        v = getattr(model_cfg, f)
        if v is not None:
            kw[f] = v
    return BertConfig(**kw)
