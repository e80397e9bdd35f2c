"""Hot-op dispatch: hand-written CDNA4 HIP kernels with an eager oracle.

Every op here has two implementations with identical math:

- ``eager``: plain PyTorch ops — the numerical reference oracle (SURVEY.md
  §5: numerics tests compare the HIP kernel against this in fp32), and the
  CPU path.
- ``hip``: the gfx950 extension in ``wap_amd/csrc`` (built in-tree by
  ``setup.py`` / ``__graft_entry__.build``), used on ROCm devices.

Backend selection: ``hip`` is the default whenever the tensors live on a
GPU.  If the extension is missing on a GPU box this package FAILS LOUDLY
(RuntimeError) rather than silently falling back — a silent eager fallback
would invalidate every GPU benchmark.  Set ``WAP_KERNELS=eager`` (or call
``set_backend("eager")``) to A/B the eager path explicitly.
"""

from __future__ import annotations

import os

_BACKEND = os.environ.get("WAP_KERNELS", "auto")
_VALID = ("auto", "eager", "hip")


def set_backend(name: str) -> None:
    global _BACKEND
    if name not in _VALID:
        raise ValueError(f"backend must be one of {_VALID}, got {name!r}")
    _BACKEND = name


def get_backend() -> str:
    return _BACKEND


_EAGER_OPS = set(
    s for s in os.environ.get("WAP_EAGER_OPS", "").split(",") if s
)


def op_eager(name: str) -> bool:
    """Per-op eager override (debug/bisect): WAP_EAGER_OPS=msc,attn,..."""
    return name in _EAGER_OPS


def use_hip(tensor) -> bool:
    """True if this op call should run the HIP kernel."""
    if _BACKEND == "eager":
        return False
    if not tensor.is_cuda:
        return False
    if _BACKEND == "hip":
        from . import ext
        ext.require()  # raises if unavailable
        return True
    # auto: on GPU the HIP path is mandatory (fail loudly if missing).
    from . import ext
    ext.require()
    return True


from .functional import (  # noqa: E402
    attention_score,
    coverage_attention,
    fused_gru_cell,
    gru_cell,
    gru_gates,
    masked_softmax_context,
    unfold_1ch,
)

__all__ = [
    "set_backend", "get_backend", "use_hip",
    "attention_score", "coverage_attention", "masked_softmax_context",
    "gru_cell", "gru_gates", "fused_gru_cell", "unfold_1ch",
]
