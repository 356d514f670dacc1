"""hipGraph-captured training step (MI355X launch-overhead elimination).

The flagship workload multiplexes many small-model clients onto one GPU:
a ResNet-18 CIFAR step is ~100 GFLOP across ~200 kernel launches, so host
launch overhead dominates wall time (measured: 17 ms/step of gaps vs
~11 ms of kernel time before this).  Every client step has identical
shapes, so the whole (cast -> forward -> backward -> fused optimizer)
sequence is captured ONCE into a hipGraph and replayed per batch; per-step
host work collapses to two buffer copies + one graph launch.

State contract: the graph reads/writes only fixed-storage tensors — the
flat parameter/grad buffers (parallel.flat), optimizer state, BN running
stats, and the static input buffers.  Warmup + capture MUTATE that state;
the engine reloads every piece of it from the global model at the start
of each round (load_flat / buffer broadcast / optimizer.reset), so
construction needs no restore.  (A restore was tried and removed: ANY
default-stream copy_ issued between hipGraph capture and the first replay
intermittently faults the replay on ROCm 7.2 when the graph contains
sort/embedding kernels — the BERT path.  Keeping the post-capture window
clean avoids it.)  RNG-consuming ops (dropout) use PyTorch's graph-safe
philox machinery.
"""

from __future__ import annotations

from typing import Optional

import torch


class GraphedStepRunner:
    def __init__(
        self,
        model: torch.nn.Module,
        optimizer,
        flat,  # parallel.flat.FlatParams
        batch_shape,
        y_shape,
        x_dtype: torch.dtype,
        y_dtype: torch.dtype,
        device: torch.device,
        use_bf16: bool,
        warmup_iters: int = 3,
    ) -> None:
        self.device = device
        self.optimizer = optimizer
        self.enabled = device.type == "cuda"
        if not self.enabled:
            return
        self.static_x = torch.zeros(batch_shape, dtype=x_dtype,
                                    device=device)
        self.static_y = torch.zeros(y_shape, dtype=y_dtype, device=device)

        def one_step():
            flat.flat_g.zero_()
            loss_fn = getattr(model, "loss_terms", None)
            with torch.autocast("cuda", torch.bfloat16, enabled=use_bf16):
                loss = (loss_fn(self.static_x, self.static_y)
                        if loss_fn is not None
                        else model.loss(self.static_x, self.static_y))
            (loss.sum() if loss.dim() else loss).backward()
            optimizer.device_step()
            if hasattr(flat, "sync_shadow"):
                flat.sync_shadow()  # refresh bf16 shadow inside the graph
            return loss

        model.train()
        if self.static_x.dtype.is_floating_point:
            self.static_x.normal_()
        else:
            self.static_x.random_(0, 100)  # token ids (warmup only)
        if y_dtype == torch.long:
            self.static_y.random_(0, 2)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream(device=device)
        s.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(s):
            for _ in range(warmup_iters):
                optimizer.host_prepare()
                one_step()
        torch.cuda.current_stream(device).wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        optimizer.host_prepare()
        with torch.cuda.graph(self.graph):
            self.static_loss = one_step()
        optimizer.t = 0
        torch.cuda.synchronize(device)

    def matches(self, x: torch.Tensor) -> bool:
        return self.enabled and tuple(x.shape) == tuple(self.static_x.shape)

    def step(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """Run one captured train step; returns the (static) loss tensor."""
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.optimizer.host_prepare()
        self.graph.replay()
        return self.static_loss


def make_step_runner(model, optimizer, flat, cfg, device,
                     sample_shape) -> Optional[GraphedStepRunner]:
    """Build a runner for the config's full-batch shape, or None when
    graphs don't apply (CPU, DP per-sample path, or disabled)."""
    if device.type != "cuda" or not cfg.train.use_graphs:
        return None
    if cfg.dp.enabled:
        if cfg.data.dataset == "text_synth":
            return None
        bs = cfg.train.batch_size
        x_shape = (bs,) + tuple(sample_shape)
        try:
            return DPGraphedStepRunner(
                model, optimizer, flat, cfg.dp.clip_norm, x_shape, device,
                cfg.train.dtype == "bfloat16",
            )
        except NotImplementedError:
            return None  # no per-sample rule: eager vmap fallback
    if cfg.model.name == "bert_base":
        # hipGraph replays of the BERT step (sort/embedding kernels in the
        # graph) intermittently memory-fault on ROCm 7.2 once any other
        # work runs between capture and replay, and BERT's kernels are
        # large enough that graphs buy nothing (measured 1114 vs 1129
        # samples/s) — run it eager.
        return None
    bs = cfg.train.batch_size
    if cfg.data.dataset == "text_synth":
        x_shape = (bs, cfg.data.seq_len)
        y_shape = (bs, cfg.data.seq_len)
        x_dtype = y_dtype = torch.long
    else:
        x_shape = (bs,) + tuple(sample_shape)
        y_shape = (bs,)
        x_dtype, y_dtype = torch.float32, torch.long
    use_bf16 = cfg.train.dtype == "bfloat16"
    return GraphedStepRunner(
        model, optimizer, flat, x_shape, y_shape, x_dtype, y_dtype, device,
        use_bf16,
    )


class DPGraphedStepRunner:
    """hipGraph capture of the FULL DP-SGD step: forward, backward with
    per-sample-gradient hooks (ops.grad_sample), fused clip+reduce
    (ops.dp), clipped-grad copy-back, fused optimizer.  The hook Python
    runs once at capture; its kernels replay.  No RNG-consuming ops are in
    the DP models (GroupNorm, no dropout), and no sort/embedding kernels
    (the ROCm graph-replay hazard), so capture is safe here.
    """

    def __init__(self, model, optimizer, flat, clip_norm: float,
                 batch_shape, device: torch.device, use_bf16: bool,
                 warmup_iters: int = 2) -> None:
        from ..ops import dp as dp_ops
        from ..ops.grad_sample import GradSampleEngine

        self.device = device
        self.optimizer = optimizer
        self.enabled = device.type == "cuda"
        if not self.enabled:
            return
        bs = batch_shape[0]
        self.static_x = torch.zeros(batch_shape, dtype=torch.float32,
                                    device=device)
        self.static_y = torch.zeros((bs,), dtype=torch.long, device=device)
        eng = GradSampleEngine(model)

        def one_step():
            flat.flat_g.zero_()
            with eng:
                with torch.autocast("cuda", torch.bfloat16,
                                    enabled=use_bf16):
                    loss = model.loss(self.static_x, self.static_y)
                loss.backward()
                per_sample = eng.grad_samples()
                clipped = dp_ops.clip_and_reduce(per_sample, clip_norm,
                                                 pre_scale=float(bs))
                trainable = [p for _, p in model.named_parameters()
                             if p.requires_grad]
                assert len(trainable) == len(clipped)
                for p, g in zip(trainable, clipped):
                    p.grad.copy_(g.to(p.dtype))
                eng.clear()
            optimizer.device_step()
            return loss

        model.train()
        self.static_x.normal_()
        self.static_y.random_(0, 2)
        s = torch.cuda.Stream(device=device)
        s.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(s):
            for _ in range(warmup_iters):
                optimizer.host_prepare()
                one_step()
        torch.cuda.current_stream(device).wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        optimizer.host_prepare()
        with torch.cuda.graph(self.graph):
            self.static_loss = one_step()
        optimizer.t = 0
        torch.cuda.synchronize(device)

    def matches(self, x: torch.Tensor) -> bool:
        return self.enabled and tuple(x.shape) == tuple(self.static_x.shape)

    def step(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.optimizer.host_prepare()
        self.graph.replay()
        return self.static_loss
