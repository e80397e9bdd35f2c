"""Synthetic CROHME-shape data generator.

There is no network in this environment (no real CROHME download), so
training/benchmarking runs on synthetic data of the same shape distribution
as CROHME (~60-300 px tall, up to ~2000 px wide grayscale
images; captions = whitespace-tokenized LaTeX up to 200 tokens).  The
generator draws random stroke-like line segments so images are not all-zero
(zero-filled inputs inflate throughput via DVFS — cdna_hip_programming.md
§5.4 rule 25) and emits the exact on-disk formats of the reference family
(features pickle + caption txt), so the whole pipeline downstream of
``gen_pkl.py`` is exercised unchanged.
"""

from __future__ import annotations

import pickle
from typing import Dict, List, Tuple

import numpy as np

from .vocab import Vocab


def _random_strokes(rng: np.random.Generator, h: int, w: int) -> np.ndarray:
    """Cheap handwriting-ish image: random polylines on a uint8 canvas."""
    img = np.zeros((h, w), dtype=np.uint8)
    n_strokes = int(rng.integers(4, 12))
    for _ in range(n_strokes):
        x = float(rng.uniform(0, w - 1))
        y = float(rng.uniform(0, h - 1))
        steps = int(rng.integers(10, 60))
        dx = float(rng.uniform(-3, 3))
        dy = float(rng.uniform(-2, 2))
        for _ in range(steps):
            dx += float(rng.uniform(-1, 1))
            dy += float(rng.uniform(-1, 1))
            x = min(max(x + dx, 0.0), w - 1.0)
            y = min(max(y + dy, 0.0), h - 1.0)
            xi, yi = int(x), int(y)
            img[max(yi - 1, 0): yi + 2, max(xi - 1, 0): xi + 2] = rng.integers(160, 255)
    return img


def synth_samples(
    n: int,
    vocab: Vocab,
    seed: int = 0,
    min_h: int = 48,
    max_h: int = 192,
    min_w: int = 128,
    max_w: int = 768,
    min_len: int = 4,
    max_len: int = 60,
) -> Tuple[Dict[str, np.ndarray], Dict[str, str]]:
    """Return ({uid: uint8[H,W]}, {uid: caption}) of n synthetic samples."""
    rng = np.random.default_rng(seed)
    feats: Dict[str, np.ndarray] = {}
    caps: Dict[str, str] = {}
    # token ids 1..K-1 (0 is <eos>, never inside a caption)
    ids = np.arange(1, len(vocab))
    for i in range(n):
        uid = f"synth_{seed}_{i:06d}"
        h = int(rng.integers(min_h, max_h + 1))
        w = int(rng.integers(min_w, max_w + 1))
        feats[uid] = _random_strokes(rng, h, w)
        length = int(rng.integers(min_len, max_len + 1))
        toks: List[str] = [vocab.id_to_token[int(t)] for t in rng.choice(ids, size=length)]
        caps[uid] = " ".join(toks)
    return feats, caps


def write_synth_dataset(
    out_pkl: str,
    out_captions: str,
    n: int,
    vocab: Vocab,
    seed: int = 0,
    **kw,
) -> None:
    """Materialize the reference's on-disk formats."""
    feats, caps = synth_samples(n, vocab, seed=seed, **kw)
    with open(out_pkl, "wb") as f:
        pickle.dump(feats, f, protocol=pickle.HIGHEST_PROTOCOL)
    with open(out_captions, "w", encoding="utf-8") as f:
        for uid, cap in caps.items():
            f.write(f"{uid}\t{cap}\n")


def fixed_shape_batch(
    batch: int,
    height: int,
    width: int,
    seq_len: int,
    n_tokens: int,
    seed: int = 0,
):
    """Uniform-shape synthetic batch for benchmarking (bench.py): returns
    numpy (x[B,1,H,W] float32 in [0,1], y[B,T] int64, full masks)."""
    rng = np.random.default_rng(seed)
    x = np.stack([_random_strokes(rng, height, width) for _ in range(batch)])
    x = (x.astype(np.float32) / 255.0)[:, None]
    y = rng.integers(1, n_tokens, size=(batch, seq_len)).astype(np.int64)
    y[:, -1] = 0  # eos
    return x, y
