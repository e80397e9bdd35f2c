"""Run configuration (SURVEY.md §A.4 family defaults).

The reference hardcodes a hyperparameter dict in its main file; here a
single dataclass + argparse overrides serves train.py / decode.py / bench.py.
"""

from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class WAPConfig:
    # model
    n_tokens: int = 112
    encoder: str = "densenet"          # densenet | vgg
    embed_dim: int = 256               # m
    hidden_dim: int = 256              # n
    attention_dim: int = 512
    coverage_ch: int = 512
    coverage_kernel: int = 11
    growth: int = 24                   # DenseNet growth k
    # data / bucketing
    dictionary: str = "dictionary.txt"
    train_pkl: str = ""
    train_captions: str = ""
    valid_pkl: str = ""
    valid_captions: str = ""
    batch_size: int = 16
    batch_imagesize: int = 500000
    maxlen: int = 200
    max_imagesize: int = 500000
    synthetic: int = 0                 # >0: generate N synthetic samples instead of files
    # optimization
    optimizer: str = "adadelta"        # adadelta | adam
    lr: float = 1.0
    adadelta_rho: float = 0.95
    adadelta_eps: float = 1e-8
    weight_decay: float = 1e-4
    grad_clip: float = 100.0
    epochs: int = 10
    seed: int = 1234
    patience: int = 15                 # early stop on valid WER plateau
    # runtime
    device: str = "auto"               # auto | cpu | cuda
    dtype: str = "bf16"                # bf16 | fp32 (compute dtype on GPU)
    kernels: str = "auto"              # auto | eager | hip
    # hipGraph-captured train steps (multi-shape trainer).  Requires the
    # first backward of the process to run on a non-default stream (done by
    # the trainer; default-stream AccumulateGrad binding breaks captures on
    # ROCm 7.x) — see PERF.md for the full list of platform hazards.
    graph: str = "auto"
    graph_pad_hw: int = 64             # bucket shape padding grid for graphs
    graph_pad_t: int = 16
    log_every: int = 20
    valid_every_epochs: int = 1
    ckpt_dir: str = "checkpoints"
    metrics_jsonl: str = "metrics.jsonl"
    resume: str = ""                   # path to latest.pt
    # decode
    beam: int = 10
    decode_maxlen: int = 200
    # distributed (filled from env by train.py)
    bucket_mb: int = 8                 # DP gradient bucket size


def add_config_args(p: argparse.ArgumentParser, cfg: Optional[WAPConfig] = None) -> None:
    cfg = cfg or WAPConfig()
    for f in dataclasses.fields(WAPConfig):
        default = getattr(cfg, f.name)
        arg = "--" + f.name.replace("_", "-")
        if f.type == "bool" or isinstance(default, bool):
            p.add_argument(arg, type=int, default=int(default))
        else:
            p.add_argument(arg, type=type(default), default=default)


def config_from_args(args: argparse.Namespace) -> WAPConfig:
    kw = {}
    for f in dataclasses.fields(WAPConfig):
        if hasattr(args, f.name):
            kw[f.name] = getattr(args, f.name)
    return WAPConfig(**kw)
