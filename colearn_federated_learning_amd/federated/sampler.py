"""Client sampling + trainer/evaluator role assignment.

The reference's coordinator selects a participant subset each session and
assigns trainer vs evaluator roles over MQTT.
Determinism contract: the sample for a round depends only on
(seed, round_idx, registry) — every rank computes the identical subset with
no communication, which is what makes the multi-GPU simulation equivalent
to the 1-GPU one.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Sequence

import torch

from ..utils.seed import stable_hash


@dataclass
class RoundPlan:
    trainers: List[int]
    evaluators: List[int]


class ClientSampler:
    def __init__(self, seed: int, clients_per_round: int = None,
                 num_evaluators: int = 0) -> None:
        self.seed = seed
        self.clients_per_round = clients_per_round
        self.num_evaluators = num_evaluators

    def state_dict(self) -> dict:
        return {"seed": self.seed,
                "clients_per_round": self.clients_per_round,
                "num_evaluators": self.num_evaluators}

    def load_state_dict(self, d: dict) -> None:
        self.seed = d["seed"]
        self.clients_per_round = d["clients_per_round"]
        self.num_evaluators = d["num_evaluators"]

    def plan(self, round_idx: int, registry: Sequence[int]) -> RoundPlan:
        """Pick trainers + evaluators for a round from the client registry."""
        ids = sorted(registry)
        g = torch.Generator().manual_seed(
            stable_hash(self.seed, "sampler", round_idx)
        )
        perm = torch.randperm(len(ids), generator=g).tolist()
        n_eval = min(self.num_evaluators, max(len(ids) - 1, 0))
        k = self.clients_per_round or (len(ids) - n_eval)
        k = min(k, len(ids) - n_eval)
        trainers = sorted(ids[i] for i in perm[:k])
        evaluators = sorted(ids[i] for i in perm[k : k + n_eval])
        return RoundPlan(trainers=trainers, evaluators=evaluators)
