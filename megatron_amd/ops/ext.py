"""Loader for the in-tree HIP ops extension (_C.so, gfx950).

The extension is built ahead of time by megatron_amd/ops/build.py (driven by
__graft_entry__.build()) and committed-adjacent in-tree so the .so travels with
the repo snapshot to GPU boxes. On a machine WITH a GPU the extension is
mandatory: a missing/broken _C.so raises instead of silently falling back to
eager PyTorch. On CPU-only machines (unit tests) callers use the torch
reference paths in megatron_amd/ops/functional.py.
"""

from __future__ import annotations

import importlib
import os
from types import ModuleType
from typing import Optional

import torch

_EXT: Optional[ModuleType] = None
_TRIED = False

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))


class OpsExtensionMissing(RuntimeError):
    pass


def available() -> bool:
    try:
        return load(required=False) is not None
    except Exception:
        return False


def load(required: Optional[bool] = None) -> Optional[ModuleType]:
    """Import megatron_amd/ops/_C.so. required defaults to
    torch.cuda.is_available(): on a GPU box the HIP path must run."""
    global _EXT, _TRIED
    if _EXT is not None:
        return _EXT
    if required is None:
        required = torch.cuda.is_available()
    if _TRIED and not required:
        return None
    _TRIED = True
    so_path = os.path.join(_OPS_DIR, "_C.so")
    if not os.path.exists(so_path):
        if required:
            raise OpsExtensionMissing(
                f"megatron_amd ops extension not found at {so_path}. "
                "Build it with: python -m megatron_amd.ops.build "
                "(or __graft_entry__.build())."
            )
        return None
    try:
        # module name must match TORCH_EXTENSION_NAME (PyInit__C)
        spec = importlib.util.spec_from_file_location("_C", so_path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
        return _EXT
    except Exception as e:
        if required:
            raise OpsExtensionMissing(
                f"megatron_amd ops extension failed to load from {so_path}: {e}"
            ) from e
        return None
