"""Loader for the in-tree gfx950 HIP extension.

The extension is built IN-TREE (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) so the ``.so`` travels with the repo snapshot
to GPU boxes.  On a GPU box a missing/unloadable extension is a hard error:
silently running eager PyTorch would fake benchmark results.
"""

from __future__ import annotations

import glob
import importlib
import os
from types import ModuleType
from typing import Optional

_mod: Optional[ModuleType] = None
_err: Optional[Exception] = None
_tried = False


def _find_so() -> Optional[str]:
    pkg_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    hits = glob.glob(os.path.join(pkg_root, "_wap_hip*.so"))
    return hits[0] if hits else None


def load() -> Optional[ModuleType]:
    """Try to import the extension module; cache the result."""
    global _mod, _err, _tried
    if _tried:
        return _mod
    _tried = True
    try:
        import torch  # noqa: F401  (extension links against torch libs)
        _mod = importlib.import_module("wap_amd._wap_hip")
    except Exception as e:  # ImportError or ABI mismatch
        _err = e
        _mod = None
    return _mod


def available() -> bool:
    return load() is not None


def require() -> ModuleType:
    m = load()
    if m is None:
        so = _find_so()
        raise RuntimeError(
            "wap_amd HIP extension is required on GPU but could not be "
            f"loaded (so={'found ' + so if so else 'NOT FOUND in-tree'}). "
            "Build it with `python setup.py build_ext --inplace` "
            f"(underlying error: {_err!r}). Set WAP_KERNELS=eager only for "
            "explicit eager A/B runs — never for benchmarks."
        )
    return m
