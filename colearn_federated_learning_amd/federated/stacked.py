"""Stacked-client training engine (SURVEY.md §2b N1 execution backend).

Trains every local virtual client of a round SIMULTANEOUSLY: one stacked
model with per-client parameter slices, batches stacked client-major, one
hipGraph replay per step for ALL clients.  This is the throughput backend
for client multiplexing on MI355X — sequential per-client training leaves
the 256-CU chip mostly idle on small models (measured 12% occupancy on
ResNet-18 deep layers), stacking multiplies every kernel's workgroups by
the client count.

Parameter flow per round:
  broadcast   : global flat [P] -> every stacked slice [G, P]
  train       : S graphed steps over stacked batches
  extract     : deltas [G, P] = slices - global ; per-client top-k
  (aggregation stays in the coordinator: weighted all-reduce over xGMI)

The stacked parameter order mirrors the template model's FlatParams
layout, so slice extraction is a segment-wise reshape.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..config import FederatedConfig
from ..models.stacked import build_stacked_model
from ..ops.fused_optim import make_flat_optimizer
from .client import FederatedClient


class StackedFlatParams:
    """Param-major stacked flat buffers: [param0: G x n0][param1: ...].

    Each stacked parameter [G, *shape] is a contiguous segment, so kernels
    see contiguous per-client weights; per-client flats are gathered
    segment-wise.
    """

    def __init__(self, model: torch.nn.Module, G: int,
                 device: torch.device) -> None:
        self.G = G
        self.device = device
        params = list(model.named_parameters())
        self.names = [n for n, _ in params]
        # template (per-client) shapes: drop the leading G dim
        self.t_shapes = [tuple(p.shape[1:]) for _, p in params]
        self.t_numels = [p.numel() // G for _, p in params]
        self.P = sum(self.t_numels)
        total = G * self.P
        self.flat_w = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_g = torch.zeros(total, dtype=torch.float32, device=device)
        offset = 0
        for (_, p), n in zip(params, self.t_numels):
            seg = self.flat_w[offset : offset + G * n].view(p.shape)
            seg.copy_(p.detach().to(device=device, dtype=torch.float32))
            p.data = seg
            p.grad = self.flat_g[offset : offset + G * n].view(p.shape)
            offset += G * n
        self._offsets = []
        off = 0
        for n in self.t_numels:
            self._offsets.append(off)
            off += G * n
        self.flat_bf16 = None  # allocated by attach_shadows

    def attach_shadows(self, model: torch.nn.Module) -> None:
        """Give every stacked conv a persistent bf16 view of its weights,
        plus a persistent PRE-TRANSPOSED dgrad-layout view.

        sync_shadow() is then ONE fused kernel per train step (inside the
        captured graph, after the optimizer) that casts the whole flat
        fp32 buffer to bf16 AND regenerates every conv's [G][CI][R][S][CO]
        dgrad weight in tiled-transpose passes — replacing both the
        per-layer autocast casts (~5% of step time, round 1) and the
        ~20 per-call permute+contiguous copies the dgrad path used to
        issue each step (~10% of step time — tools/gpu_debug/eager_attrib.py).
        """
        from ..models.stacked import SBatchNorm2d, SConv2d, SLinear
        from ..models.stacked_bert import SLayerNorm

        self.flat_bf16 = torch.empty_like(self.flat_w,
                                          dtype=torch.bfloat16)
        by_name = dict(zip(self.names,
                           zip(self._offsets, self.t_numels)))
        segA, segB = [], []
        t_off, blk = 0, 0
        conv_t = []  # (mod, t_off, shape)
        for mod_name, mod in model.named_modules():
            if isinstance(mod, SConv2d):
                off, n = by_name[f"{mod_name}.weight" if mod_name
                                 else "weight"]
                mod.wshadow = (self.flat_bf16[off : off + self.G * n]
                               .view(mod.weight.shape))
                # direct wgrad target: the param's flat-grad view
                mod.wgslot = mod.weight.grad
                G, CO, KH, KW, CI = mod.weight.shape
                if CO % 32 == 0 and CI % 32 == 0:
                    RS = KH * KW
                    tco, tci = CO // 32, CI // 32
                    nblk = G * tco * tci  # RS looped inside the block
                    segA.append([off, t_off, blk, blk + nblk])
                    segB.append([CO, CI, RS, tco, tci, G])
                    conv_t.append((mod, t_off, (G, CI, KH, KW, CO)))
                    t_off += G * n
                    blk += nblk
                # else (stem CI=3): dgrad is skipped anyway — the input
                # batch never needs a gradient (needs_input_grad gate)
            elif isinstance(mod, SBatchNorm2d):
                mod.gslot_w = mod.weight.grad.view(-1)
                mod.gslot_b = mod.bias.grad.view(-1)
            elif isinstance(mod, SLinear):
                off, n = by_name[f"{mod_name}.weight" if mod_name
                                 else "weight"]
                mod.wshadow = (self.flat_bf16[off : off + self.G * n]
                               .view(mod.weight.shape))
            elif isinstance(mod, SLayerNorm):
                mod.gslot_w = mod.weight.grad
                mod.gslot_b = mod.bias.grad
            else:
                from ..models.stacked_bert import SEmbedding

                if isinstance(mod, SEmbedding):
                    off, n = by_name[f"{mod_name}.weight" if mod_name
                                     else "weight"]
                    mod.wshadow = (
                        self.flat_bf16[off : off + self.G * n]
                        .view(mod.weight.shape))
        dev = self.device
        self.flat_bf16_t = torch.empty(max(t_off, 1),
                                       dtype=torch.bfloat16, device=dev)
        self._segA = torch.tensor(segA, dtype=torch.int64,
                                  device=dev).reshape(-1, 4)
        self._segB = torch.tensor(segB, dtype=torch.int32,
                                  device=dev).reshape(-1, 6)
        self._transpose_blocks = blk
        for mod, toff, shape in conv_t:
            n = 1
            for d in shape:
                n *= d
            mod.wshadow_t = self.flat_bf16_t[toff : toff + n].view(shape)
        self.sync_shadow()

    @torch.no_grad()
    def sync_shadow(self) -> None:
        if self.flat_bf16 is None:
            return
        from ..ops import native_enabled, require_ext

        if self.flat_w.is_cuda and native_enabled():
            require_ext().shadow_sync(
                self.flat_w, self.flat_bf16, self.flat_bf16_t,
                self._segA, self._segB, self._transpose_blocks,
            )
        else:
            self.flat_bf16.copy_(self.flat_w)

    # -- global <-> stacked ------------------------------------------------
    @torch.no_grad()
    def broadcast_global(self, global_flat: torch.Tensor) -> None:
        """Copy the per-client-layout global vector into every slice."""
        t_off = 0
        for off, n in zip(self._offsets, self.t_numels):
            seg = self.flat_w[off : off + self.G * n].view(self.G, n)
            seg.copy_(global_flat[t_off : t_off + n]
                      .unsqueeze(0).expand(self.G, n))
            t_off += n

    @torch.no_grad()
    def client_deltas(self, global_flat: torch.Tensor) -> torch.Tensor:
        """[G, P] per-client deltas in template layout."""
        out = torch.empty(self.G, self.P, dtype=torch.float32,
                          device=self.device)
        t_off = 0
        for off, n in zip(self._offsets, self.t_numels):
            seg = self.flat_w[off : off + self.G * n].view(self.G, n)
            out[:, t_off : t_off + n] = seg - global_flat[t_off : t_off + n]
            t_off += n
        return out


class StackedBuffers:
    """Float buffers (BN running stats) per client slice, flattened
    [G, Pb] in template order — aggregated like weights (FedAvg averages
    model state including BN statistics)."""

    def __init__(self, model: torch.nn.Module, G: int) -> None:
        self.G = G
        self.items = [
            (n, b) for n, b in model.named_buffers()
            if b.dtype.is_floating_point and b.shape[:1] == (G,)
        ]
        self.t_numels = [b.numel() // G for _, b in self.items]
        self.Pb = sum(self.t_numels)

    @torch.no_grad()
    def broadcast_global(self, global_buf: torch.Tensor) -> None:
        t_off = 0
        for (_, b), n in zip(self.items, self.t_numels):
            b.view(self.G, n).copy_(
                global_buf[t_off : t_off + n].unsqueeze(0).expand(self.G, n)
            )
            t_off += n

    @torch.no_grad()
    def gather(self) -> torch.Tensor:
        """[G, Pb] current per-client buffer states."""
        out = torch.empty(self.G, self.Pb,
                          device=self.items[0][1].device
                          if self.items else "cpu")
        t_off = 0
        for (_, b), n in zip(self.items, self.t_numels):
            out[:, t_off : t_off + n] = b.view(self.G, n).float()
            t_off += n
        return out


class StackedEngine:
    """Trains G clients simultaneously in G parameter slots.  The engine is
    slot-based: ``train_round`` takes the round's active clients (len == G),
    so a pool of engines keyed on active-count supports
    ``clients_per_round < num_clients`` without wasting compute on
    unsampled clients."""

    def __init__(
        self,
        cfg: FederatedConfig,
        G: int,
        device: torch.device,
    ) -> None:
        self.cfg = cfg
        self.G = G
        self.device = device
        self.model = build_stacked_model(
            cfg.model, cfg.data, self.G, dp_mode=cfg.dp.enabled
        )
        assert self.model is not None
        self.model = self.model.to(device)
        self.sflat = StackedFlatParams(self.model, self.G, device)
        if device.type == "cuda":
            self.sflat.attach_shadows(self.model)
        self.sbuf = StackedBuffers(self.model, self.G)
        # FedProx reference: pre-broadcast stacked copy (refreshed per round)
        self.flat_ref = (
            torch.zeros_like(self.sflat.flat_w)
            if cfg.train.fedprox_mu != 0.0 else None
        )
        self.optimizer = make_flat_optimizer(
            cfg.train.optimizer, self.sflat.flat_w, self.sflat.flat_g,
            cfg.train, flat_ref=self.flat_ref,
        )
        # stacked DP: per-sample grads for ALL clients' samples in one
        # backward (hook engine over the stacked modules), clip + per-
        # client mean in one grouped kernel sweep per parameter
        self.gs_engine = None
        if cfg.dp.enabled:
            from ..models.stacked import SConv2d, SLinear
            from ..ops.grad_sample import GradSampleEngine

            self.gs_engine = GradSampleEngine(self.model)
            # per-sample rules own the weight grads: autograd's mean
            # wgrad / fc-grad kernels would be overwritten — skip them
            for mod in self.model.modules():
                if isinstance(mod, (SConv2d, SLinear)):
                    mod.dp_mode = True
        self.runner = None
        if cfg.dp.enabled:
            # hipGraph capture of the WHOLE stacked DP step (hooks, clip,
            # grouped reduce, optimizer) — the eager path measured a ~20%
            # host-launch gap at 64 clients.  The DP models are
            # capture-safe: GroupNorm, no dropout, no sort/embedding.
            if device.type == "cuda" and cfg.train.use_graphs:
                bs = cfg.train.batch_size
                shape = (self.G * bs, cfg.model.in_channels,
                         cfg.model.image_size, cfg.model.image_size)
                self.runner = _DPStackedGraphRunner(
                    self, shape, device, cfg.train.dtype == "bfloat16",
                )
        elif cfg.model.name == "bert_base":
            # hipGraph replays with sort/embedding kernels intermittently
            # fault on ROCm 7.2 (see steprunner.py), and the stacked BERT
            # kernels are large enough that graphs buy ~nothing — eager.
            pass
        elif device.type == "cuda" and cfg.train.use_graphs:
            from .steprunner import GraphedStepRunner

            bs = cfg.train.batch_size
            shape = (self.G * bs, cfg.model.in_channels,
                     cfg.model.image_size, cfg.model.image_size)
            if cfg.model.name == "logreg":
                shape = (self.G * bs, cfg.model.in_channels,
                         cfg.model.image_size, cfg.model.image_size)
            self.runner = GraphedStepRunner(
                self.model, self.optimizer, self.sflat, shape,
                (self.G * bs,), torch.float32, torch.long, device,
                cfg.train.dtype == "bfloat16",
            )

    # ------------------------------------------------------------------
    def _dp_step(self, x, y, use_bf16: bool, graph_mode: bool = False):
        """One per-sample-clipped DP step for ALL G clients at once.

        Per-sample grads are weight-free quantities of (input, grad-out),
        so the hook rules run on the stacked modules unchanged with
        batch = G*B; the fused grouped reduction then clips each SAMPLE
        to C and means each CLIENT's rows in one launch per parameter
        (ops.dp.clip_and_reduce(groups=G))."""
        from ..ops import dp as dp_ops

        B = x.shape[0] // self.G
        self.optimizer.zero_grad()
        with self.gs_engine:
            with torch.autocast("cuda", torch.bfloat16,
                                enabled=use_bf16):
                loss_vec = self.model.loss_terms(x, y)
            loss_vec.sum().backward()
            per_sample = self.gs_engine.grad_samples()
            clipped = dp_ops.clip_and_reduce(
                per_sample, self.cfg.dp.clip_norm,
                pre_scale=float(B), groups=self.G,
            )
            trainable = [p for p in self.model.parameters()
                         if p.requires_grad]
            assert len(trainable) == len(clipped)
            for p, g in zip(trainable, clipped):
                p.grad.copy_(g.reshape(p.shape))
            self.gs_engine.clear()
        if graph_mode:
            self.optimizer.device_step()
        else:
            self.optimizer.step()
        self.sflat.sync_shadow()
        return loss_vec

    # ------------------------------------------------------------------
    def train_round(
        self, clients: List[FederatedClient],
        global_flat: torch.Tensor, global_buf: torch.Tensor,
        round_idx: int,
    ) -> Tuple[torch.Tensor, torch.Tensor, Dict[int, float], int]:
        """Train ``clients`` (len == G, one per slot).  Returns
        (deltas [G,P], buffers [G,Pb], losses per cid, samples/client)."""
        assert len(clients) == self.G
        self.clients = clients
        self.client_ids = [c.client_id for c in clients]
        cfg = self.cfg
        tc = cfg.train
        self.sflat.broadcast_global(global_flat)
        if global_buf.numel():
            self.sbuf.broadcast_global(global_buf.to(self.device))
        if self.flat_ref is not None:
            self.flat_ref.copy_(self.sflat.flat_w)
        self.sflat.sync_shadow()
        self.optimizer.reset()
        self.model.train()
        use_bf16 = tc.dtype == "bfloat16" and self.device.type == "cuda"

        gens = [c._gen(round_idx, "batches") for c in self.clients]
        loss_acc = torch.zeros(self.G, device=self.device)
        total_n = 0
        steps = 0
        budget_hit = False
        # one persistent generator per client across epochs, fresh batch
        # iterators each epoch — mirrors the sequential client.fit loop
        # (local_epochs was silently ignored here before).
        for _epoch in range(tc.local_epochs):
            if budget_hit:
                break
            iters = [
                c.shard.batches(tc.batch_size, generator=g, drop_last=False)
                for c, g in zip(self.clients, gens)
            ]
            while True:
                batches = []
                for it in iters:
                    batches.append(next(it, None))
                if any(b is None for b in batches):
                    break  # equal shard sizes => all exhaust together
                if (tc.max_local_steps is not None
                        and steps >= tc.max_local_steps):
                    budget_hit = True
                    break
                xs = torch.stack([b[0] for b in batches])  # [G, B, ...]
                ys = torch.stack([b[1] for b in batches])
                x = xs.reshape(-1, *xs.shape[2:]).to(self.device,
                                                     non_blocking=True)
                y = ys.reshape(-1, *ys.shape[2:]).to(self.device,
                                                     non_blocking=True)
                if self.runner is not None and self.runner.matches(x):
                    loss_vec = self.runner.step(x, y)
                elif self.gs_engine is not None:
                    loss_vec = self._dp_step(x, y, use_bf16)
                else:
                    self.optimizer.zero_grad()
                    with torch.autocast("cuda", torch.bfloat16,
                                        enabled=use_bf16):
                        loss_vec = self.model.loss_terms(x, y)
                    loss_vec.sum().backward()
                    self.optimizer.step()
                    self.sflat.sync_shadow()
                bs = batches[0][0].shape[0]
                loss_acc += loss_vec.detach() * bs
                total_n += bs
                steps += 1
        losses = {
            cid: float(v) / max(total_n, 1)
            for cid, v in zip(self.client_ids, loss_acc)
        }
        # control-plane parity with the sequential path: every client
        # reports its round result on the bus (CoLearn RESULT topic)
        from .bus import Topics

        for c in self.clients:
            if c.bus is not None:
                c.bus.publish(
                    Topics.RESULT.format(client_id=c.client_id),
                    {"round": round_idx, "n_samples": total_n,
                     "loss": losses[c.client_id]},
                )
        deltas = self.sflat.client_deltas(global_flat)
        bufs = self.sbuf.gather() if self.sbuf.Pb else torch.empty(self.G, 0)
        return deltas, bufs, losses, total_n


class StackedEnginePool:
    """Engines keyed on active-client count.  The full-pool engine (G =
    all local clients) is built eagerly — it is the steady-state path for
    every driver config; smaller engines are built lazily the first round
    a sampling/drop pattern needs them and cached (graph capture and
    stacked-weight allocation are one-time costs)."""

    # lazy engines kept beyond the full one: each holds G x weights +
    # optimizer state (+ DP graph pools with per-sample slabs), so an
    # unbounded cache under drop-injection (a new active count most
    # rounds) could exhaust HBM at 64-client scale
    MAX_LAZY = 2

    def __init__(self, cfg: FederatedConfig,
                 clients: List[FederatedClient],
                 device: torch.device) -> None:
        self.cfg = cfg
        self.all_clients = clients
        self.device = device
        self._engines: Dict[int, StackedEngine] = {
            len(clients): StackedEngine(cfg, len(clients), device)
        }
        self._lazy_lru: List[int] = []  # non-full engine keys, LRU order

    def get(self, n_active: int) -> StackedEngine:
        eng = self._engines.get(n_active)
        if eng is None:
            eng = StackedEngine(self.cfg, n_active, self.device)
            self._engines[n_active] = eng
        if n_active != len(self.all_clients):
            if n_active in self._lazy_lru:
                self._lazy_lru.remove(n_active)
            self._lazy_lru.append(n_active)
            while len(self._lazy_lru) > self.MAX_LAZY:
                evict = self._lazy_lru.pop(0)
                del self._engines[evict]
                if self.device.type == "cuda":
                    torch.cuda.empty_cache()
        return eng

    @property
    def full(self) -> StackedEngine:
        """The steady-state engine covering every local client."""
        return self._engines[len(self.all_clients)]

    @property
    def runner(self):
        return self.full.runner


class _DPStackedGraphRunner:
    """hipGraph capture of the full stacked-DP step (see StackedEngine).
    The per-sample hook Python runs once at capture; its kernels (and the
    ~90 GB of per-sample grad slabs they fill, held in the graph's
    private pool) replay each step."""

    def __init__(self, engine: "StackedEngine", shape, device,
                 use_bf16: bool, warmup_iters: int = 2) -> None:
        self.engine = engine
        self.enabled = device.type == "cuda"
        if not self.enabled:
            return
        self.static_x = torch.zeros(shape, dtype=torch.float32,
                                    device=device)
        self.static_y = torch.zeros((shape[0],), dtype=torch.long,
                                    device=device)
        self.static_x.normal_()
        self.static_y.random_(0, 2)
        engine.model.train()
        s = torch.cuda.Stream(device=device)
        s.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(s):
            for _ in range(warmup_iters):
                engine.optimizer.host_prepare()
                engine._dp_step(self.static_x, self.static_y, use_bf16,
                                graph_mode=True)
        torch.cuda.current_stream(device).wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        engine.optimizer.host_prepare()
        with torch.cuda.graph(self.graph):
            self.static_loss = engine._dp_step(
                self.static_x, self.static_y, use_bf16, graph_mode=True)
        engine.optimizer.t = 0
        torch.cuda.synchronize(device)

    def matches(self, x: torch.Tensor) -> bool:
        return self.enabled and tuple(x.shape) == tuple(self.static_x.shape)

    def step(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.engine.optimizer.host_prepare()
        self.graph.replay()
        return self.static_loss


def stackable(cfg: FederatedConfig, clients: List[FederatedClient]) -> bool:
    """Stacked execution applies when every local client has the same
    shard size (lock-step batches) and the model family is implemented."""
    if not cfg.train.use_stacked:
        return False
    if cfg.straggler_timeout_s is not None:
        # the per-client mid-fit deadline only exists on the sequential
        # path; in a lock-step stack no client can straggle relative to
        # its rank-mates, so honor the knob by falling back
        return False
    if cfg.model.name not in ("logreg", "resnet18", "resnet50",
                              "bert_base"):
        return False
    if cfg.dp.enabled and cfg.model.name not in ("resnet18", "resnet50"):
        return False  # stacked per-sample path exists for the conv models
    if len(clients) < 1:
        return False
    # G=1 is worth stacking too: the stacked engine carries the bf16
    # shadow weights and direct-to-flat grad writes, which the sequential
    # path does not (matters at 8 GPUs where each rank holds one client)
    sizes = {len(c.shard) for c in clients}
    return len(sizes) == 1
