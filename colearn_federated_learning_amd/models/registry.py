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
        from .bert import BertConfig, BertForMaskedLM

        return BertForMaskedLM(BertConfig(
            vocab_size=data_cfg.vocab_size,
            max_position_embeddings=max(data_cfg.seq_len, 512),
            dropout=model_cfg.dropout,
        ))
    raise ValueError(f"unknown model {name!r}")
