"""Checkpoint / round-state persistence.

The reference checkpoints by ``torch.save`` of the global model per round
(SURVEY.md §2a row 9).  The on-disk format here keeps the model state dict
under the conventional ``model_state_dict`` key (loadable by plain
``model.load_state_dict(torch.load(p)["model_state_dict"])``) and adds the
round counter + sampler/RNG state so a run can resume mid-session.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch


CKPT_NAME = "colearn_round_{round:05d}.pt"
LATEST = "colearn_latest.pt"


def save_checkpoint(
    path_dir: str,
    round_idx: int,
    model_state: Dict[str, torch.Tensor],
    sampler_state: Dict[str, Any],
    config_json: str,
    extra: Optional[Dict[str, Any]] = None,
) -> str:
    os.makedirs(path_dir, exist_ok=True)
    payload = {
        "model_state_dict": {k: v.cpu() for k, v in model_state.items()},
        "round": round_idx,
        "sampler_state": sampler_state,
        "config": config_json,
        "format_version": 1,
    }
    if extra:
        payload.update(extra)
    path = os.path.join(path_dir, CKPT_NAME.format(round=round_idx))
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)
    # LATEST is a byte copy of the round file (serialize once), written
    # with the same tmp+rename dance so readers never see a torn file
    import shutil

    latest = os.path.join(path_dir, LATEST)
    tmp = latest + ".tmp"
    shutil.copyfile(path, tmp)
    os.replace(tmp, latest)
    return path


def load_checkpoint(path_dir_or_file: str) -> Dict[str, Any]:
    path = path_dir_or_file
    if os.path.isdir(path):
        path = os.path.join(path, LATEST)
    return torch.load(path, map_location="cpu", weights_only=False)
