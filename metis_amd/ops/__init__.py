"""MI355X HIP op wrappers.

Dispatch policy:
* on GPU (ROCm) tensors the hand-written gfx950 kernels run — and if the
  in-tree extension is missing on a GPU machine, ops raise instead of
  silently falling back to eager PyTorch;
* on CPU tensors a plain PyTorch reference implementation runs, so the
  planner/runtime/test stack works in CPU-only containers (gloo).
"""

from __future__ import annotations

_EXT = None
_EXT_ERR: str = ""


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from metis_amd import _hip_ops  # built in-tree by setup.py

        _EXT = _hip_ops
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _EXT_ERR = str(e)
        _EXT = False
    return _EXT


def extension_available() -> bool:
    return bool(_load_extension())


def require_extension():
    ext = _load_extension()
    if not ext:
        raise RuntimeError(
            "metis_amd HIP extension is required on GPU but not built "
            f"(import error: {_EXT_ERR}). Run: PYTORCH_ROCM_ARCH=gfx950 "
            "python3 setup.py build_ext --inplace"
        )
    return ext


from metis_amd.ops.layernorm import LayerNorm, layer_norm  # noqa: E402
from metis_amd.ops.adamw import FusedAdamW  # noqa: E402

__all__ = [
    "LayerNorm",
    "layer_norm",
    "FusedAdamW",
    "extension_available",
    "require_extension",
]
