from .checkpoint import load_checkpoint, save_checkpoint
from .decode import BeamState, beam_decode, greedy_decode
from .metrics import MetricsLogger

__all__ = [
    "load_checkpoint", "save_checkpoint",
    "BeamState", "beam_decode", "greedy_decode",
    "MetricsLogger",
]
