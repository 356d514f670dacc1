"""The FL round engine / coordinator.

Reference behavior preserved: discover participants via the control plane,
select a subset per round, assign trainer/evaluator roles, dispatch local
training, gather results with straggler/drop handling, weighted FedAvg into
the global model, evaluate, checkpoint, log — repeat.

MI355X re-architecture: clients are virtual and sharded across RCCL ranks
(client cid lives on rank cid % world).  Each rank trains its sampled
clients SIMULTANEOUSLY in a stacked engine (per-client weight slices,
grid.z = client — federated/stacked.py; sequential fallback for
unstackable configs), folds each client's delta into a flat fp32
accumulator (optionally top-k compressed per client), and a single
bucketed weighted all-reduce replaces the reference's per-client
websocket model shipping.  Everything that feeds
cross-rank state (sampling, DP noise, drop injection) is computed from
rank-consistent RNG so a run is equivalent on any GPU count
(SURVEY.md §4b.3 — tests/test_equivalence.py).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch

from ..utils.seed import stable_hash

from ..config import FederatedConfig
from ..data.sharder import make_shards
from ..data.synthetic import make_client_dataset
from ..models.registry import build_model
from ..ops.dp import PrivacyAccountant, add_dp_noise
from ..ops.fused_optim import make_flat_optimizer
from ..ops.topk import topk_compress, topk_decompress
from ..parallel.aggregator import Aggregator
from ..parallel.flat import FlatBuffers, FlatParams
from .bus import MessageBus, Topics
from .checkpoint import load_checkpoint, save_checkpoint
from .client import FederatedClient
from .metrics import MetricsLogger, PhaseTimer
from .sampler import ClientSampler


class FederatedServer:
    """Coordinator + the rank-local slice of the client pool."""

    def __init__(
        self,
        cfg: FederatedConfig,
        rank: int = 0,
        world: int = 1,
        device: Optional[torch.device] = None,
        bus: Optional[MessageBus] = None,
    ) -> None:
        self.cfg = cfg
        self.rank = rank
        self.world = world
        self.device = device or torch.device("cpu")
        self.bus = bus or MessageBus()

        torch.manual_seed(cfg.seed)
        dp_mode = cfg.dp.enabled
        self.model = build_model(cfg.model, cfg.data, dp_mode=dp_mode).to(
            self.device
        )
        self.flat = FlatParams(self.model, self.device)
        self.global_flat = self.flat.clone_weights()
        self.delta_acc = torch.zeros_like(self.global_flat)
        self.optimizer = make_flat_optimizer(
            cfg.train.optimizer, self.flat.flat_w, self.flat.flat_g,
            cfg.train, flat_ref=self.global_flat,
        )
        self.aggregator = Aggregator(cfg.agg, self.device)
        self.sampler = ClientSampler(
            cfg.seed, cfg.clients_per_round, cfg.num_evaluators
        )
        # --- client pool: this rank owns cid % world == rank --------------
        shards = make_shards(cfg.data, cfg.num_clients)
        self.clients: Dict[int, FederatedClient] = {}
        for cid in range(cfg.num_clients):
            if cid % world != rank:
                continue
            shard = make_client_dataset(cfg.data, cfg.model, cid, shards[cid])
            if self.device.type == "cuda":
                # 288 GB HBM3E per GPU: client shards live device-resident
                # for the whole session instead of streaming from host.
                shard.x = shard.x.to(self.device)
                shard.y = shard.y.to(self.device)
            c = FederatedClient(cid, cfg, shard, self.device, bus=self.bus)
            c.announce()
            self.clients[cid] = c
        # relay availability announcements across ranks (collective): every
        # rank's bus — in particular rank 0's, the coordinator's — now
        # holds the retained availability messages of ALL clients, exactly
        # as CoLearn's MQTT broker would.
        from .bus import BusRelay

        self.bus_relay = BusRelay(self.bus)
        self.bus_relay.sync()
        # registry built FROM the control plane (deterministic: every
        # configured client announces at construction).
        self.registry: List[int] = sorted(
            p["client_id"]
            for p in self.bus.collect(Topics.AVAILABILITY + "/#").values()
        ) or list(range(cfg.num_clients))
        self.start_round = 0
        self.accountant = (
            PrivacyAccountant(
                cfg.dp.noise_multiplier,
                sample_rate=(cfg.clients_per_round or cfg.num_clients)
                / cfg.num_clients,
                delta=cfg.dp.delta,
            )
            if dp_mode else None
        )
        self.metrics = MetricsLogger(
            cfg.metrics_path if rank == 0 else None
        )
        self.timer = PhaseTimer(self.device)
        # hipGraph-captured client step (see steprunner.py); one capture
        # shared by every virtual client on this rank.
        sample_shape = (
            (cfg.model.in_channels, cfg.model.image_size,
             cfg.model.image_size)
            if cfg.data.dataset != "text_synth" else None
        )
        # FedAvg aggregates full model state: float buffers (BN running
        # stats) are averaged alongside the weight deltas.
        self.flatbuf = FlatBuffers(self.model)
        self.global_buf = self.flatbuf.gather()
        self.buf_acc = torch.zeros_like(self.global_buf)
        # stacked-client engine: all active local clients train
        # simultaneously with per-client weights (federated/stacked.py)
        from .stacked import StackedEnginePool, stackable

        local_clients = [self.clients[c] for c in sorted(self.clients)]
        self.stacked = (
            StackedEnginePool(cfg, local_clients, self.device)
            if stackable(cfg, local_clients) else None
        )
        self.step_runner = None
        if self.stacked is None:
            from .steprunner import make_step_runner

            self.step_runner = make_step_runner(
                self.model, self.optimizer, self.flat, cfg, self.device,
                sample_shape,
            )
        if cfg.resume and cfg.checkpoint_dir:
            self._resume()

    # ------------------------------------------------------------------
    def _resume(self) -> None:
        try:
            ck = load_checkpoint(self.cfg.checkpoint_dir)
        except FileNotFoundError:
            return
        state = {
            k: v.to(self.device) for k, v in ck["model_state_dict"].items()
        }
        # restore only trainable parameters into the flat buffer; buffers
        # (BN running stats) load through the module.  copy_ in place: the
        # optimizer's FedProx reference and any captured hipGraph alias
        # this exact tensor.
        self.model.load_state_dict(state, strict=False)
        self.global_flat.copy_(self.flat.flat_from_state_dict(state))
        self.flat.load_flat(self.global_flat)
        self.sampler.load_state_dict(ck["sampler_state"])
        self.global_buf = self.flatbuf.gather()
        self.start_round = ck["round"] + 1
        if self.accountant is not None and "dp_steps" in ck:
            self.accountant.steps = ck["dp_steps"]

    # ------------------------------------------------------------------
    def _dropped(self, round_idx: int, cid: int) -> bool:
        p = self.cfg.drop_client_prob
        if p <= 0:
            return False
        g = torch.Generator().manual_seed(
            stable_hash(self.cfg.seed, "drop", round_idx, cid)
        )
        return bool(torch.rand((), generator=g) < p)

    # ------------------------------------------------------------------
    def run_round(self, round_idx: int) -> Dict[str, float]:
        cfg = self.cfg
        t_round = time.perf_counter()
        self.timer.reset()
        plan = self.sampler.plan(round_idx, self.registry)
        self.bus.publish(
            Topics.ROUND, {"round": round_idx, "trainers": plan.trainers,
                           "evaluators": plan.evaluators},
        )
        for cid in plan.trainers:
            if cid in self.clients:
                self.bus.publish(Topics.ROLE.format(client_id=cid),
                                 {"role": "trainer", "round": round_idx})
        for cid in plan.evaluators:
            if cid in self.clients:
                self.bus.publish(Topics.ROLE.format(client_id=cid),
                                 {"role": "evaluator", "round": round_idx})

        # --- local training of this rank's sampled clients -------------
        self.delta_acc.zero_()
        self.buf_acc.zero_()
        local_weight = 0.0
        local_losses: Dict[int, float] = {}
        n_samples_round = 0
        topk_ratio = cfg.agg.topk_ratio

        def fold_client(cid, delta, bufs, loss, n_samples):
            nonlocal local_weight, n_samples_round
            w = 1.0 if cfg.dp.enabled else float(n_samples)
            if topk_ratio > 0:
                k = max(int(delta.numel() * topk_ratio), 1)
                idx, val = topk_compress(delta.contiguous(), k)
                delta = topk_decompress(idx, val, delta.numel())
            if cfg.dp.enabled:
                # Client-level sensitivity: the noise is sigma*C on the
                # SUMMED deltas, so each client's contribution must be
                # bounded by C.  Per-sample clipping inside local steps does
                # NOT bound the S-step delta (it can reach ~S*lr*C), so
                # clip the whole delta here (McMahan et al. 2018 DP-FedAvg).
                nrm = float(delta.norm())
                if nrm > cfg.dp.clip_norm:
                    delta = delta * (cfg.dp.clip_norm / nrm)
            self.delta_acc.add_(delta, alpha=w)
            if bufs is not None and bufs.numel():
                self.buf_acc.add_(bufs.to(self.buf_acc.device), alpha=w)
            local_weight += w
            local_losses[cid] = loss
            n_samples_round += n_samples

        with self.timer.phase("train"):
            if self.stacked is not None:
                # train ONLY this round's active clients: the engine pool
                # provides a stacked engine sized to the active count, so
                # clients_per_round < num_clients wastes no compute
                sampled = set(plan.trainers)
                active = [
                    c for c in self.stacked.all_clients
                    if c.client_id in sampled
                    and not self._dropped(round_idx, c.client_id)
                ]
                if active:
                    engine = self.stacked.get(len(active))
                    deltas, bufs, losses, n_per_client = \
                        engine.train_round(
                            active, self.global_flat, self.global_buf,
                            round_idx,
                        )
                    for i, c in enumerate(active):
                        fold_client(c.client_id, deltas[i],
                                    bufs[i] if bufs.numel() else None,
                                    losses[c.client_id], n_per_client)
            else:
                for cid in plan.trainers:
                    client = self.clients.get(cid)
                    if client is None:
                        continue
                    if self._dropped(round_idx, cid):
                        continue
                    deadline = (
                        time.perf_counter() + cfg.straggler_timeout_s
                        if cfg.straggler_timeout_s is not None else None
                    )
                    self.flat.load_flat(self.global_flat)
                    self.flatbuf.load(self.global_buf)
                    self.optimizer.reset()
                    res = client.fit(self.model, self.optimizer, round_idx,
                                     runner=self.step_runner,
                                     deadline=deadline)
                    if res.timed_out:
                        # straggler actually CUT OFF mid-fit (round time is
                        # bounded); its partial result is excluded and the
                        # aggregation reweights over the rest
                        continue
                    fold_client(cid, self.flat.flat_w - self.global_flat,
                                self.flatbuf.gather(), res.loss,
                                res.n_samples)

        # --- aggregation: weighted all-reduce over xGMI -----------------
        with self.timer.phase("aggregate"):
            _, total_weight = self.aggregator.allreduce_weighted(
                self.delta_acc, local_weight
            )
            if cfg.dp.enabled and cfg.dp.noise_multiplier > 0:
                add_dp_noise(self.delta_acc, cfg.dp.clip_norm,
                             cfg.dp.noise_multiplier,
                             key=stable_hash(cfg.seed, "dp", round_idx))
                if self.accountant is not None:
                    self.accountant.step()
            self.aggregator.allreduce_tensor(self.buf_acc)
            if total_weight > 0:
                self.global_flat.add_(self.delta_acc, alpha=1.0 / total_weight)
                if self.buf_acc.numel():
                    self.global_buf.copy_(self.buf_acc / total_weight)
            self.flat.load_flat(self.global_flat)
            self.flatbuf.load(self.global_buf)

        # --- evaluation of the NEW global model -------------------------
        eval_metrics: Dict[str, float] = {}
        with self.timer.phase("evaluate"):
            ev_loss_sum, ev_correct, ev_n = 0.0, 0.0, 0.0
            for cid in plan.evaluators:
                client = self.clients.get(cid)
                if client is None:
                    continue
                m = client.evaluate(self.model, round_idx)
                ev_loss_sum += m["loss"] * m["n_samples"]
                ev_correct += m["accuracy"] * m["n_samples"]
                ev_n += m["n_samples"]
            if plan.evaluators:
                ev = [
                    self.aggregator.allreduce_scalar(v)
                    for v in (ev_loss_sum, ev_correct, ev_n)
                ]
                if ev[2] > 0:
                    eval_metrics = {"eval_loss": ev[0] / ev[2],
                                    "eval_accuracy": ev[1] / ev[2]}

        # --- control-plane relay (collective): results/metrics published
        # by other ranks' clients this round land on every bus, so rank
        # 0's log mirrors the whole node like CoLearn's broker would
        self.bus_relay.sync()

        # --- bookkeeping -------------------------------------------------
        n_total = self.aggregator.allreduce_scalar(float(n_samples_round))
        loss_sum = self.aggregator.allreduce_scalar(
            sum(local_losses.values())
        )
        n_reporting = self.aggregator.allreduce_scalar(
            float(len(local_losses))
        )
        round_time = time.perf_counter() - t_round
        record = {
            "round": round_idx,
            "n_samples": n_total,
            "mean_client_loss": loss_sum / max(n_reporting, 1.0),
            "client_losses": local_losses,
            "samples_per_sec": n_total / max(round_time, 1e-9),
            "round_time_s": round_time,
            **{f"time_{k}_s": v for k, v in self.timer.times.items()},
            **eval_metrics,
        }
        if self.accountant is not None:
            record["dp_epsilon"] = self.accountant.epsilon()
        if self.rank == 0:
            self.metrics.log(record)
            if cfg.checkpoint_dir and (
                cfg.checkpoint_every
                and (round_idx + 1) % cfg.checkpoint_every == 0
            ):
                self._checkpoint(round_idx)
        return record

    # ------------------------------------------------------------------
    def _checkpoint(self, round_idx: int) -> None:
        extra = {}
        if self.accountant is not None:
            extra["dp_steps"] = self.accountant.steps
        save_checkpoint(
            self.cfg.checkpoint_dir,
            round_idx,
            self.model.state_dict(),
            self.sampler.state_dict(),
            self.cfg.to_json(),
            extra=extra,
        )

    # ------------------------------------------------------------------
    def train(self) -> List[Dict[str, float]]:
        """Run the configured number of rounds (the session loop)."""
        self.bus.publish(Topics.SESSION,
                         {"state": "start", "config": self.cfg.to_dict()},
                         retain=True)
        history = []
        r = self.start_round - 1  # resume-at-end: loop may not run at all
        for r in range(self.start_round, self.cfg.rounds):
            rec = self.run_round(r)
            history.append(rec)
            if (self.cfg.target_accuracy is not None
                    and rec.get("eval_accuracy", 0.0)
                    >= self.cfg.target_accuracy):
                break
        if self.rank == 0 and self.cfg.checkpoint_dir and r >= 0:
            self._checkpoint(r)
        self.bus.publish(Topics.SESSION, {"state": "stop"}, retain=True)
        self.metrics.close()
        return history
