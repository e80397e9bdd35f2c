from .attention import CoverageAttention
from .decoder import GRUCellWAP, Parser
from .encoder import DenseNetEncoder, VGGEncoder, build_encoder, downsample_mask
from .wap import WAPModel, masked_ce

__all__ = [
    "CoverageAttention", "GRUCellWAP", "Parser",
    "DenseNetEncoder", "VGGEncoder", "build_encoder", "downsample_mask",
    "WAPModel", "masked_ce",
]
