"""Op dispatch layer.

Every hot op has two implementations:
  * `cpu_ref` — PyTorch fp64 reference (runs on CPU; used for tests and the
    host-only pipeline).
  * `hip` — hand-written HIP/CDNA4 kernels (gfx950) loaded from the in-tree
    extension `dpo_amd/ops/hip/libdpo_hip_ops.so`.

Dispatch rule: tensors on a CUDA (ROCm) device REQUIRE the HIP extension —
there is no silent eager fallback on GPU. CPU tensors use the torch
reference path.
"""
from __future__ import annotations

import torch

from . import cpu_ref

_hip_mod = None
_hip_err: Exception | None = None


def _load_hip():
    global _hip_mod, _hip_err
    if _hip_mod is not None or _hip_err is not None:
        return _hip_mod
    try:
        from . import hip_backend
        _hip_mod = hip_backend
    except Exception as e:  # noqa: BLE001
        _hip_err = e
        _hip_mod = None
    return _hip_mod


def hip_available() -> bool:
    return _load_hip() is not None


def backend_for(t: torch.Tensor):
    """Return the op backend module for a tensor's device. GPU tensors fail
    loudly if the HIP extension is missing (no eager fallback)."""
    if t.is_cuda:
        mod = _load_hip()
        if mod is None:
            raise RuntimeError(
                "dpo_amd HIP extension is required for GPU tensors but "
                f"failed to load: {_hip_err!r}. Build it with "
                "`python -m dpo_amd.ops.build` (or __graft_entry__.build()).")
        return mod
    return cpu_ref
