"""Hot-op dispatch: hand-written CDNA4 HIP kernels on GPU, torch_ref on CPU.

The HIP extension (``raft_amd/ops/csrc``, built in-tree as
``raft_amd/ops/_hip_ops*.so``) is REQUIRED on a GPU box: if a CUDA/HIP device
is visible and the extension cannot be imported, ops fail loudly rather than
silently falling back to eager PyTorch (that would invalidate benchmarks).
Set ``RAFT_AMD_FORCE_TORCH=1`` to explicitly force the pure-PyTorch path.
"""
from __future__ import annotations

import os

import torch

_hip_ops = None
_hip_import_error: Exception | None = None

if os.environ.get("RAFT_AMD_FORCE_TORCH", "0") != "1":
    try:
        import importlib
        # NOTE: must be importlib (not `from raft_amd.ops import _hip_ops`)
        # — the `_hip_ops = None` sentinel above would shadow the submodule.
        _hip_ops = importlib.import_module("raft_amd.ops._hip_ops")
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _hip_import_error = e


def _try_bind():
    """(Re)try binding the extension — the .so may have been built after
    this package was first imported (e.g. __graft_entry__.build() in the
    same process)."""
    global _hip_ops, _hip_import_error
    if _hip_ops is not None:
        return _hip_ops
    if os.environ.get("RAFT_AMD_FORCE_TORCH", "0") == "1":
        return None
    try:
        import importlib
        importlib.invalidate_caches()
        _hip_ops = importlib.import_module("raft_amd.ops._hip_ops")
        _hip_import_error = None
    except ImportError as e:
        _hip_import_error = e
    return _hip_ops


def hip_available() -> bool:
    """True when the HIP extension is importable AND a GPU is present."""
    return _try_bind() is not None and torch.cuda.is_available()


def require_hip():
    """Return the HIP extension module, failing loudly if missing on GPU."""
    if _try_bind() is None:
        raise RuntimeError(
            "raft_amd HIP extension (_hip_ops) is not built but a GPU run "
            "requested it. Build in-tree with "
            "`python -m raft_amd.ops.build` (or __graft_entry__.build()). "
            f"Original import error: {_hip_import_error}")
    return _hip_ops


from raft_amd.ops import torch_ref  # noqa: E402,F401
from raft_amd.ops.functional import (  # noqa: E402,F401
    corr_volume, corr_pyramid, corr_lookup, gru_gates, convex_upsample,
    upflow8,
)
