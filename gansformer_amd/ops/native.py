"""Loader for the in-tree HIP extension `gansformer_amd._C`.

The extension is built ahead-of-time for gfx950 (csrc/setup.py, driven by
`__graft_entry__.build()`); there is no import-time JIT (the reference
nvcc-JIT'd its two .cu files at import, ref src/dnnlib/tflib/custom_ops.py
[R] — we ship a prebuilt .so instead so gpurun snapshots carry it).

On a GPU machine, ops MUST run through the extension: `require_ext()`
raises if it is absent, so a missing native path fails loudly instead of
silently falling back to eager PyTorch.
"""

from __future__ import annotations

import torch

_C = None
_import_error: Exception | None = None

try:
    from gansformer_amd import _C as _C  # type: ignore  # built by csrc/setup.py
except ImportError as e:  # pragma: no cover - exercised only when .so missing
    _import_error = e


def have_ext() -> bool:
    return _C is not None


def require_ext():
    if _C is None:
        raise RuntimeError(
            "gansformer_amd._C HIP extension is not built, but an op was "
            "called on a GPU tensor. Build it in-tree with "
            "`python csrc/setup.py build_ext --inplace` "
            f"(import error: {_import_error!r})"
        )
    return _C


def use_native(*tensors: torch.Tensor) -> bool:
    """True if these tensors live on a HIP device (=> native kernels)."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if on_gpu:
        require_ext()
    return on_gpu
