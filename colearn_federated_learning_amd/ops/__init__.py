"""Native op layer: hand-written HIP/CDNA4 (gfx950) kernels + wrappers.

The reference has no native layer at all (pure-Python PySyft prototype,
SURVEY.md §2a negative inventory); the new build's native surface is the
BASELINE.json hot-path list: conv + GEMM fwd/bwd, BatchNorm/LayerNorm,
fused SGD/Adam, per-sample grad-clip + DP Gaussian noise, and top-k gradient
sparsification — all MFMA/LDS-tiled HIP for gfx950, no Triton, no
CUDA-compat shims.

Loading contract: the extension is built IN-TREE (``python setup.py
build_ext --inplace`` or ``__graft_entry__.build()``) so the ``.so`` ships
to GPU boxes with the repo snapshot.  On a machine WITH a GPU, every op
whose input lives on the GPU *requires* the extension and raises if it is
missing — there is no silent eager fallback on the GPU path.  On CPU-only
machines the pure-PyTorch reference implementations run instead (they are
also the numerics oracles for the kernel tests).
"""

from __future__ import annotations

import os
from typing import Optional

_EXT = None
_EXT_TRIED = False


def ext() -> Optional[object]:
    """Return the compiled HIP extension module, or None if not built."""
    global _EXT, _EXT_TRIED
    if not _EXT_TRIED:
        _EXT_TRIED = True
        try:
            from . import _colearn_hip  # type: ignore[attr-defined]

            _EXT = _colearn_hip
        except ImportError:
            _EXT = None
    return _EXT


def require_ext() -> object:
    """The GPU path must run native code: raise loudly if it can't."""
    m = ext()
    if m is None:
        raise RuntimeError(
            "colearn_federated_learning_amd: HIP extension _colearn_hip is "
            "not built but a GPU tensor reached a native op. Build it "
            "in-tree with `python setup.py build_ext --inplace` (or "
            "__graft_entry__.build()). The GPU hot path never falls back "
            "to eager PyTorch."
        )
    return m


def native_enabled() -> bool:
    """Native kernels are used for GPU tensors unless explicitly disabled
    (COLEARN_DISABLE_NATIVE=1 is for A/B benchmarking only)."""
    return os.environ.get("COLEARN_DISABLE_NATIVE", "0") != "1"


def is_vmapped(*tensors) -> bool:
    """True while any functorch transform (vmap/grad — the DP per-sample
    path) is active, or when a tensor is functorch-wrapped: native kernels
    can't see through the wrappers, so the composite fallbacks serve as
    the vmap rules."""
    import torch

    try:
        fc = torch._C._functorch
        if fc.peek_interpreter_stack() is not None:
            return True
        return any(fc.is_batchedtensor(t) for t in tensors)
    except Exception:
        return False
