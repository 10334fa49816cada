"""HIP/CDNA4 op library for MI355X (gfx950), with CPU reference fallbacks.

The compiled extension ``_dcr_hip.so`` lives in-tree next to this file
(built by ``dcr_amd.ops.build.build_extension()``, driven by
``__graft_entry__.build()``) so that it travels with repo snapshots.

Dispatch policy:
  * CPU tensors  -> plain PyTorch reference implementations (used by tests).
  * CUDA tensors -> the hand-written HIP kernels. If the extension is not
    importable on a machine that HAS a GPU this module raises loudly
    rather than silently falling back to eager PyTorch — a silent
    fallback would invalidate benchmark numbers. Set
    ``DCR_AMD_ALLOW_FALLBACK=1`` to override (debug only).

Reference capability map: SURVEY.md §2.4 enumerates the ops the upstream
repo (somepago/DCR) exercises through libraries; each kernel here is one
of those, fused for HBM3E-bound execution (8 TB/s, fuse elementwise into
producers, fp32 accumulation for norms/softmax).
"""
from __future__ import annotations

import importlib.util
import os
import sys
from pathlib import Path

import torch

_EXT = None
_EXT_ERR: str | None = None
_TRIED = False

_HERE = Path(__file__).resolve().parent


def _try_load():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    so = _HERE / "_dcr_hip.so"
    if not so.exists():
        _EXT_ERR = f"{so} not built (run __graft_entry__.build())"
        return None
    try:
        spec = importlib.util.spec_from_file_location("_dcr_hip", str(so))
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        sys.modules.setdefault("_dcr_hip", mod)
        _EXT = mod
    except Exception as e:  # pragma: no cover - load failure path
        _EXT_ERR = f"failed to load {so}: {e!r}"
        _EXT = None
    return _EXT


def ext() -> object | None:
    """Return the loaded HIP extension module, or None."""
    return _try_load()


def hip_ready() -> bool:
    return _try_load() is not None


def require_hip(opname: str) -> object:
    """Return the extension; raise if missing while a GPU is present."""
    m = _try_load()
    if m is None:
        if os.environ.get("DCR_AMD_ALLOW_FALLBACK") == "1":
            return None
        raise RuntimeError(
            f"dcr_amd op '{opname}' needs the HIP extension on GPU but it is "
            f"not available: {_EXT_ERR}. Build with __graft_entry__.build() "
            f"or set DCR_AMD_ALLOW_FALLBACK=1 (debug only)."
        )
    return m


def use_hip(*tensors: torch.Tensor) -> bool:
    """True when these tensors should run through HIP kernels."""
    if not tensors or not tensors[0].is_cuda:
        return False
    return True


# per-op HIP dispatch counters ("the native path is the one that runs"):
# incremented by the functional wrappers on every HIP-kernel dispatch.
from collections import Counter  # noqa: E402

dispatch_counts: "Counter[str]" = Counter()


def count_dispatch(name: str):
    dispatch_counts[name] += 1


from .functional import (  # noqa: E402,F401
    group_norm_silu,
    layer_norm,
    geglu,
    attention,
    add_noise,
    get_velocity,
    cfg_combine,
    lincomb,
)
from .adamw import FusedAdamW  # noqa: E402,F401
