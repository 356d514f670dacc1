"""Federated client: availability announcement + local training/evaluation.

Mirrors the reference's client worker: a client
announces availability over the control plane, then serves coordinator-issued
``fit`` / ``evaluate`` commands against its local shard.  The reference hosts
each client in its own process behind a PySyft websocket worker; here clients
are *virtual* — lightweight objects multiplexed onto GPU ranks — and commands
are direct method calls issued by the rank that owns the client.  The API
surface (announce / fit / evaluate, returning the trained weights + sample
count + loss) is preserved.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import torch

from ..utils.seed import stable_hash

from ..config import FederatedConfig
from ..data.synthetic import ClientShard
from .bus import MessageBus, Topics


@dataclass
class FitResult:
    n_samples: int
    loss: float
    timed_out: bool = False  # straggler cut off mid-fit (excluded upstream)
    # new weights live in the shared work model; coordinator extracts deltas


class FederatedClient:
    """One virtual federated participant."""

    def __init__(
        self,
        client_id: int,
        cfg: FederatedConfig,
        shard: ClientShard,
        device: torch.device,
        bus: Optional[MessageBus] = None,
    ) -> None:
        self.client_id = client_id
        self.cfg = cfg
        self.shard = shard
        self.device = device
        self.bus = bus
        self.role = "idle"  # idle | trainer | evaluator
        if bus is not None:
            bus.subscribe(
                Topics.ROLE.format(client_id=client_id), self._on_role
            )

    # -- control plane -------------------------------------------------
    def announce(self) -> None:
        # retained per-client availability subtopic (MQTT idiom) so the
        # coordinator's registry snapshot sees EVERY client, not just the
        # last announcer on a shared topic
        if self.bus is not None:
            self.bus.publish(
                f"{Topics.AVAILABILITY}/{self.client_id}",
                {"client_id": self.client_id, "n_samples": len(self.shard),
                 "device": str(self.device)},
                retain=True,
            )

    def _on_role(self, _topic: str, payload) -> None:
        self.role = payload["role"]

    # -- data plane ----------------------------------------------------
    def _gen(self, round_idx: int, tag: str) -> torch.Generator:
        g = torch.Generator()
        g.manual_seed(
            stable_hash(self.cfg.seed, tag, round_idx, self.client_id)
        )
        return g

    def fit(
        self,
        model: torch.nn.Module,
        optimizer,
        round_idx: int,
        runner=None,
        deadline: Optional[float] = None,
    ) -> FitResult:
        """Local training for ``local_epochs`` over this client's shard.

        ``model`` is the rank's shared work model, already loaded with the
        global weights; the optimizer is the rank's (its state is reset per
        client — federated clients are stateless between rounds, as in the
        reference where a fresh TrainConfig is shipped each round).
        """
        tc = self.cfg.train
        model.train()
        dev = self.device
        use_bf16 = tc.dtype == "bfloat16" and dev.type == "cuda"
        gen = self._gen(round_idx, "batches")
        # accumulate the loss ON DEVICE: a float() per step would force a
        # GPU sync per training step and serialize the whole pipeline.
        loss_acc = torch.zeros((), device=dev)
        total_n, steps = 0, 0
        timed_out = False
        import time as _time
        devices = [dev] if dev.type == "cuda" else []
        with torch.random.fork_rng(devices=devices):
            torch.manual_seed(
                stable_hash(self.cfg.seed, "train", round_idx, self.client_id)
            )
            for _ in range(tc.local_epochs):
                if timed_out:
                    break
                for x, y in self.shard.batches(tc.batch_size, generator=gen):
                    if tc.max_local_steps is not None and steps >= tc.max_local_steps:
                        break
                    if (deadline is not None
                            and _time.perf_counter() > deadline):
                        # straggler bound: abort mid-fit so the ROUND time
                        # is bounded by the timeout, not by this client's
                        # full fit
                        timed_out = True
                        break
                    x = x.to(dev, non_blocking=True)
                    y = y.to(dev, non_blocking=True)
                    if runner is not None and runner.matches(x):
                        # hipGraph fast path (for DP configs this replays
                        # the whole hook+clip+optimizer pipeline)
                        loss = runner.step(x, y)
                    elif self.cfg.dp.enabled:
                        optimizer.zero_grad()
                        loss = _per_sample_clipped_backward(
                            model, x, y, self.cfg.dp.clip_norm, use_bf16
                        )
                        optimizer.step()
                    else:
                        optimizer.zero_grad()
                        with torch.autocast("cuda", torch.bfloat16,
                                            enabled=use_bf16):
                            loss = model.loss(x, y)
                        loss.backward()
                        optimizer.step()
                    bs = x.shape[0]
                    loss_acc += loss.detach() * bs
                    total_n += bs
                    steps += 1
        avg = float(loss_acc) / max(total_n, 1)
        res = FitResult(n_samples=total_n, loss=avg, timed_out=timed_out)
        if self.bus is not None:
            self.bus.publish(
                Topics.RESULT.format(client_id=self.client_id),
                {"round": round_idx, "n_samples": res.n_samples,
                 "loss": res.loss},
            )
        return res

    @torch.no_grad()
    def evaluate(
        self, model: torch.nn.Module, round_idx: int
    ) -> Dict[str, float]:
        """Score the current global model on this client's shard."""
        tc = self.cfg.train
        model.eval()
        dev = self.device
        use_bf16 = tc.dtype == "bfloat16" and dev.type == "cuda"
        total_loss, total_correct, total_pred, total_n = 0.0, 0, 0, 0
        for x, y in self.shard.batches(tc.batch_size):
            x, y = x.to(dev), y.to(dev)
            with torch.autocast("cuda", torch.bfloat16, enabled=use_bf16):
                loss, correct, scored = model.loss_and_correct(x, y)
            bs = x.shape[0]
            total_loss += float(loss) * bs
            total_correct += int(correct)
            total_pred += int(scored)
            total_n += bs
        metrics = {
            "loss": total_loss / max(total_n, 1),
            "accuracy": total_correct / max(total_pred, 1),
            "n_samples": float(total_n),
        }
        if self.bus is not None:
            self.bus.publish(
                Topics.METRICS.format(client_id=self.client_id),
                {"round": round_idx, **metrics},
            )
        return metrics


def _per_sample_clipped_backward(
    model: torch.nn.Module, x: torch.Tensor, y: torch.Tensor,
    clip_norm: float, use_bf16: bool,
) -> torch.Tensor:
    """DP-SGD backward: per-sample gradients clipped to ``clip_norm``.

    Populates ``p.grad`` with the *mean* of clipped per-sample gradients.
    Uses torch.func (vmap of per-sample grad) — works on CPU and GPU; the
    per-sample norm + clip-scale + reduction runs through the fused HIP
    kernel on gfx950 (ops.dp) when gradients live on the GPU.
    """
    from ..ops import dp as dp_ops

    # fast path: hook-based per-sample grads through the native kernels
    # (ops.grad_sample); vmap fallback for unsupported module sets.
    try:
        from ..ops.grad_sample import GradSampleEngine

        eng = GradSampleEngine(model)
        with eng:
            with torch.autocast("cuda", torch.bfloat16, enabled=use_bf16):
                loss_val = model.loss(x, y)
            loss_val.backward()
            per_sample = eng.grad_samples()
            clipped = dp_ops.clip_and_reduce(
                per_sample, clip_norm, pre_scale=float(x.shape[0])
            )
            # grad_samples() covers only requires_grad params — zip against
            # the same filtered sequence so a frozen param can't misalign
            # the copy-back.
            trainable = [p for _, p in model.named_parameters()
                         if p.requires_grad]
            assert len(trainable) == len(clipped)
            for p, g in zip(trainable, clipped):
                p.grad.copy_(g.to(p.dtype))
            eng.clear()
        return loss_val.detach()
    except NotImplementedError:
        pass

    from torch.func import grad, vmap

    params = {k: v.detach() for k, v in model.named_parameters()}
    buffers = {k: v.detach() for k, v in model.named_buffers()}

    def sample_loss(p, b, xi, yi):
        with torch.autocast("cuda", torch.bfloat16, enabled=use_bf16):
            return model.loss(
                xi.unsqueeze(0), yi.unsqueeze(0), params=p, buffers=b
            )

    grads, losses = vmap(
        grad(sample_loss, has_aux=False), in_dims=(None, None, 0, 0),
        randomness="different",
    )(params, buffers, x, y), None
    with torch.no_grad():
        loss_val = model.loss(x, y)
    names = list(grads.keys())
    per_sample = [grads[k] for k in names]  # each [B, *param.shape]
    clipped = dp_ops.clip_and_reduce(per_sample, clip_norm)  # mean over B
    pdict = dict(model.named_parameters())
    for k, g in zip(names, clipped):
        # COPY into the existing grad view: p.grad aliases the flat grad
        # buffer the fused optimizer consumes (parallel.flat) — rebinding
        # would silently disconnect DP grads from the optimizer.
        pdict[k].grad.copy_(g.to(pdict[k].dtype))
    return loss_val
