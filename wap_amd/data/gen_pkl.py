"""Offline data preparation: image directory -> features pickle.

Capability parity with the reference's ``gen_pkl.py``:
walk a directory of grayscale expression images (png/bmp/pgm), store
``{uid: uint8[H, W]}`` into a pickle.  PIL/cv2 are not guaranteed in this
image, so we read PNG/PGM with a minimal numpy path and fall back to PIL if
importable.

Usage:  python -m wap_amd.data.gen_pkl IMAGE_DIR OUT_PKL
"""

from __future__ import annotations

import os
import pickle
import sys
import zlib
from typing import Dict

import numpy as np


def _read_pgm(path: str) -> np.ndarray:
    with open(path, "rb") as f:
        data = f.read()
    if not data.startswith(b"P5"):
        raise ValueError("only binary PGM (P5) supported")
    # header: P5 <w> <h> <maxval> then raster
    fields, idx = [], 2
    while len(fields) < 3:
        while idx < len(data) and data[idx : idx + 1].isspace():
            idx += 1
        if data[idx : idx + 1] == b"#":
            while data[idx : idx + 1] not in (b"\n", b""):
                idx += 1
            continue
        start = idx
        while idx < len(data) and not data[idx : idx + 1].isspace():
            idx += 1
        fields.append(int(data[start:idx]))
    idx += 1
    w, h, maxval = fields
    img = np.frombuffer(data, dtype=np.uint8, count=w * h, offset=idx)
    return img.reshape(h, w).copy()


def read_gray(path: str) -> np.ndarray:
    ext = os.path.splitext(path)[1].lower()
    if ext == ".pgm":
        return _read_pgm(path)
    try:  # pragma: no cover - PIL optional
        from PIL import Image  # type: ignore

        return np.asarray(Image.open(path).convert("L"), dtype=np.uint8)
    except ImportError as e:
        raise RuntimeError(
            f"cannot read {path}: PIL unavailable and format not PGM"
        ) from e


def gen_pkl(image_dir: str, out_pkl: str) -> int:
    feats: Dict[str, np.ndarray] = {}
    for name in sorted(os.listdir(image_dir)):
        base, ext = os.path.splitext(name)
        if ext.lower() not in (".png", ".bmp", ".pgm", ".jpg", ".jpeg"):
            continue
        feats[base] = read_gray(os.path.join(image_dir, name))
    with open(out_pkl, "wb") as f:
        pickle.dump(feats, f, protocol=pickle.HIGHEST_PROTOCOL)
    return len(feats)


if __name__ == "__main__":  # pragma: no cover
    if len(sys.argv) != 3:
        print(__doc__)
        sys.exit(2)
    n = gen_pkl(sys.argv[1], sys.argv[2])
    print(f"wrote {n} images -> {sys.argv[2]}")
