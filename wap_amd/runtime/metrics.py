"""Structured metrics: stdout lines + JSONL file."""

from __future__ import annotations

import json
import os
import sys
import time
from typing import Any, Dict, Optional, TextIO


class MetricsLogger:
    def __init__(self, jsonl_path: Optional[str] = None, rank: int = 0,
                 stream: TextIO = sys.stdout):
        self.rank = rank
        self.stream = stream
        self.f: Optional[TextIO] = None
        if jsonl_path and rank == 0:
            os.makedirs(os.path.dirname(os.path.abspath(jsonl_path)), exist_ok=True)
            self.f = open(jsonl_path, "a", encoding="utf-8")

    def log(self, kind: str, **fields: Any) -> None:
        if self.rank != 0:
            return
        rec: Dict[str, Any] = {"ts": time.time(), "kind": kind, **fields}
        parts = " ".join(
            f"{k}={v:.4g}" if isinstance(v, float) else f"{k}={v}"
            for k, v in fields.items()
        )
        print(f"[{kind}] {parts}", file=self.stream, flush=True)
        if self.f:
            self.f.write(json.dumps(rec) + "\n")
            self.f.flush()

    def close(self) -> None:
        if self.f:
            self.f.close()
            self.f = None
