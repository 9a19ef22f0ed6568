"""Loader for the in-tree HIP extension.

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``python -m baton_amd.ops.build``) so the resulting ``.so`` travels with the
repo snapshot to GPU boxes. Policy:

  * no GPU visible (CPU CI) -> ``hip_ops() is None`` and the op wrappers use
    their torch fallbacks;
  * GPU visible but extension missing/unimportable -> ``require_hip()``
    raises RuntimeError. A silent eager fallback on a GPU box would
    invalidate every benchmark claim, so it is an error by design.
"""

from __future__ import annotations

import glob
import importlib
import os
from typing import Optional

import torch

_cached = None
_load_error: Optional[Exception] = None
_tried = False


def _try_load():
    global _cached, _load_error, _tried
    if _tried:
        return _cached
    _tried = True
    try:
        _cached = importlib.import_module("baton_amd.ops._hip_ops")
    except ImportError as e:
        _cached = None
        _load_error = e
    return _cached


def hip_available() -> bool:
    """True when a GPU is present AND the HIP extension loaded."""
    return torch.cuda.is_available() and _try_load() is not None


def hip_ops():
    """The extension module, or None when no GPU is present."""
    if not torch.cuda.is_available():
        return None
    return require_hip()


def require_hip():
    """The extension module; raises if a GPU is present but the extension
    is not importable (fail loudly — never a silent eager fallback)."""
    mod = _try_load()
    if mod is None:
        here = os.path.dirname(__file__)
        built = glob.glob(os.path.join(here, "_hip_ops*.so"))
        raise RuntimeError(
            "baton_amd HIP extension not importable on a GPU machine. "
            f"Found .so files: {built or 'none'}; import error: {_load_error!r}. "
            "Build it in-tree: python -m baton_amd.ops.build"
        )
    return mod
