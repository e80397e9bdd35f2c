"""Checkpoint / resume (reference layout contract).

Layout: ``<dir>/latest.pt`` (every save), ``<dir>/best.pt`` (keyed by valid
WER), with ``dictionary.txt`` copied alongside — a user of the reference
finds the same keep-best + latest files.  The payload includes model,
optimizer, epoch/step, best WER, and RNG states for exact resume
(tests/test_checkpoint.py asserts bit-equal resume).
"""

from __future__ import annotations

import os
import random
import shutil
from typing import Any, Dict, Optional

import numpy as np
import torch


def save_checkpoint(
    ckpt_dir: str,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer],
    epoch: int,
    step: int,
    best_wer: float,
    is_best: bool,
    vocab_path: Optional[str] = None,
    extra: Optional[Dict[str, Any]] = None,
) -> str:
    os.makedirs(ckpt_dir, exist_ok=True)
    state = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "epoch": epoch,
        "step": step,
        "best_wer": best_wer,
        "rng": {
            "torch": torch.get_rng_state(),
            "cuda": torch.cuda.get_rng_state_all() if torch.cuda.is_available() else None,
            "numpy": np.random.get_state(),
            "python": random.getstate(),
        },
        "extra": extra or {},
    }
    latest = os.path.join(ckpt_dir, "latest.pt")
    tmp = latest + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, latest)  # atomic: never leave a torn latest.pt
    if is_best:
        shutil.copyfile(latest, os.path.join(ckpt_dir, "best.pt"))
    if vocab_path and os.path.exists(vocab_path):
        dst = os.path.join(ckpt_dir, "dictionary.txt")
        if os.path.abspath(vocab_path) != os.path.abspath(dst):
            shutil.copyfile(vocab_path, dst)
    return latest


def load_checkpoint(
    path: str,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    restore_rng: bool = True,
    map_location: str = "cpu",
) -> Dict[str, Any]:
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer") is not None:
        optimizer.load_state_dict(state["optimizer"])
    if restore_rng and "rng" in state:
        rng = state["rng"]
        torch.set_rng_state(rng["torch"].cpu() if hasattr(rng["torch"], "cpu") else rng["torch"])
        if rng.get("cuda") is not None and torch.cuda.is_available():
            try:
                torch.cuda.set_rng_state_all(rng["cuda"])
            except RuntimeError:
                pass  # device-count mismatch on resume: keep fresh cuda rng
        np.random.set_state(rng["numpy"])
        random.setstate(rng["python"])
    return state
