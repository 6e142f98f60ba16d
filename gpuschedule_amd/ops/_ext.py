"""Loader for the in-tree gfx950 extension.

Dispatch policy (loud-failure): on a GPU machine the HIP extension is
REQUIRED — a missing/unbuilt extension raises instead of silently
falling back to eager PyTorch.  The pure-torch reference paths are for
CPU (tests) only.
"""
from __future__ import annotations

import importlib.util
import os
from typing import Optional

import torch

from .build import BUILD_DIR, EXT_NAME

_ext = None
_tried = False


def _load_so() -> Optional[object]:
    so = os.path.join(BUILD_DIR, f"{EXT_NAME}.so")
    if not os.path.exists(so):
        return None
    spec = importlib.util.spec_from_file_location(EXT_NAME, so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def get_ext(required: bool = False):
    """Return the extension module, or None when absent on CPU."""
    global _ext, _tried
    if _ext is None and not _tried:
        _tried = True
        _ext = _load_so()
    if _ext is None and (required or torch.cuda.is_available()):
        raise RuntimeError(
            "gpuschedule_amd HIP extension not built: run "
            "`python -m gpuschedule_amd.ops.build` (gfx950 kernels are "
            "mandatory on GPU; no eager fallback)")
    return _ext


def have_ext() -> bool:
    try:
        return get_ext(required=False) is not None
    except RuntimeError:
        return False
