"""Watcher: fully-convolutional encoders (DenseNet-B and VGG variants).

Behavioral spec per SURVEY.md §A.1 (paper-fixed; reference class
``Watcher_train`` — file:line unverifiable, mount empty):

- DenseNet-B: conv7x7 s2 (48ch) -> pool2 -> [3 dense blocks x 16 bottleneck
  layers (BN-ReLU-1x1conv(4k) -> BN-ReLU-3x3conv(k)), growth k=24] with
  transitions (BN-ReLU-1x1 conv theta=0.5 + avgpool2) between -> D=684
  channels at stride 16.  Channel walk: 48+16*24=432 -> 216 -> 216+384=600
  -> 300 -> 300+384=684.
- VGG-FCN: stacked 3x3 conv+ReLU blocks with 4 max-pools, stride 16 (toy /
  BASELINE config-1 encoder).

Both return (annotations a[B, D, H', W'], mask[B, H', W']) where the mask is
the input pixel mask max-pooled to the feature stride (a feature cell is
valid if any contributing pixel is).
"""

from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor, nn
import torch.nn.functional as F


def conv_1ch(x: Tensor, weight: Tensor, bias=None, stride: int = 1,
             padding: int = 0) -> Tensor:
    """Conv2d for 1-input-channel tensors via unfold + rocBLAS GEMM.

    MIOpen (ROCm 7.2) falls back to its NAIVE nchw kernels for 1-channel
    bf16 convs — measured 74% of the whole WAP train step (fwd+dgrad+wrw;
    profiles/r01_baseline_kernel_stats.md).  unfold+matmul routes the same
    math through hipBLASLt instead.  x[B,1,H,W], weight[C,1,kh,kw].
    """
    from .. import ops
    b = x.shape[0]
    c, _, kh, kw = weight.shape
    ho = (x.shape[2] + 2 * padding - kh) // stride + 1
    wo = (x.shape[3] + 2 * padding - kw) // stride + 1
    cols = ops.unfold_1ch(x, kh, kw, stride, padding)         # [B,kh*kw,L]
    out = weight.view(c, kh * kw) @ cols                      # [B,C,L]
    if bias is not None:
        out = out + bias.view(1, c, 1)
    return out.view(b, c, ho, wo)


def downsample_mask(x_mask: Tensor, stride: int) -> Tensor:
    """[B,H,W] pixel mask -> [B,H',W'] feature-grid mask (max-pool)."""
    m = x_mask.unsqueeze(1)
    m = F.max_pool2d(m, kernel_size=stride, stride=stride, ceil_mode=True)
    return m.squeeze(1)


class _Bottleneck(nn.Module):
    """BN-ReLU-1x1(4k) -> BN-ReLU-3x3(k), concatenated onto the input."""

    def __init__(self, in_ch: int, growth: int, dropout: float = 0.2):
        super().__init__()
        inter = 4 * growth
        self.bn1 = nn.BatchNorm2d(in_ch)
        self.conv1 = nn.Conv2d(in_ch, inter, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(inter)
        self.conv2 = nn.Conv2d(inter, growth, 3, padding=1, bias=False)
        self.dropout = dropout

    def forward(self, x: Tensor) -> Tensor:
        from ..ops import fused_conv
        if fused_conv.can_fuse(x, self.conv1.weight):
            # hand-written MFMA implicit-GEMM conv with BN+ReLU fused into
            # the load stage (csrc/conv_mfma.hip)
            out = fused_conv.bnrelu_conv(x, self.conv1.weight, self.bn1)
            out = fused_conv.bnrelu_conv(out, self.conv2.weight, self.bn2)
        elif fused_conv.can_fuse_bnrelu(x):
            # BN+ReLU as cast-free hand kernels, convs stay MIOpen (the
            # full fused-conv path is opt-in; this elementwise fusion
            # wins on the default NCHW path)
            out = self.conv1(fused_conv.bnrelu(x, self.bn1))
            out = self.conv2(fused_conv.bnrelu(out, self.bn2))
        else:
            out = self.conv1(F.relu(self.bn1(x)))
            out = self.conv2(F.relu(self.bn2(out)))
        if self.dropout > 0 and self.training:
            out = F.dropout(out, self.dropout)
        return torch.cat([x, out], dim=1)


class _Transition(nn.Module):
    """BN-ReLU-1x1 (theta=0.5 compression) + 2x2 avgpool."""

    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        self.bn = nn.BatchNorm2d(in_ch)
        self.conv = nn.Conv2d(in_ch, out_ch, 1, bias=False)

    def forward(self, x: Tensor) -> Tensor:
        from ..ops import fused_conv
        if fused_conv.can_fuse_transition(x, self.conv.weight):
            # hand stats/mask kernels + rocBLAS GEMMs (K > 96; aten's
            # channels_last wrw here picked a ~5 ms CK kernel)
            out = fused_conv.bnrelu_conv1x1_big(x, self.conv.weight, self.bn)
        elif fused_conv.can_fuse_bnrelu(x):
            out = self.conv(fused_conv.bnrelu(x, self.bn))
        else:
            out = self.conv(F.relu(self.bn(x)))
        return F.avg_pool2d(out, 2, ceil_mode=True)


class DenseNetEncoder(nn.Module):
    stride = 16

    def __init__(self, growth: int = 24, layers_per_block: int = 16,
                 n_blocks: int = 3, init_ch: int = 48, theta: float = 0.5,
                 dropout: float = 0.2):
        super().__init__()
        self.stem = nn.Conv2d(1, init_ch, 7, stride=2, padding=3, bias=False)
        self.stem_bn = nn.BatchNorm2d(init_ch)
        ch = init_ch
        blocks = []
        for b in range(n_blocks):
            layers = []
            for _ in range(layers_per_block):
                layers.append(_Bottleneck(ch, growth, dropout))
                ch += growth
            blocks.append(nn.Sequential(*layers))
            if b != n_blocks - 1:
                out_ch = int(ch * theta)
                blocks.append(_Transition(ch, out_ch))
                ch = out_ch
        self.blocks = nn.Sequential(*blocks)
        self.out_channels = ch  # 684 with defaults

    def forward(self, x: Tensor, x_mask: Tensor) -> Tuple[Tensor, Tensor]:
        h = conv_1ch(x, self.stem.weight, stride=2, padding=3)
        if x.is_contiguous(memory_format=torch.channels_last):
            h = h.to(memory_format=torch.channels_last)
        from ..ops import fused_conv as _fc
        if _fc.can_fuse_bnrelu(h):
            h = _fc.bnrelu(h, self.stem_bn)
        else:
            h = F.relu(self.stem_bn(h))
        h = F.max_pool2d(h, 2, ceil_mode=True)   # stride 4 after stem+pool
        a = self.blocks(h)
        mask = downsample_mask(x_mask, self.stride)
        # guard: ceil_mode pooling can differ by a cell from mask pooling
        mask = mask[:, : a.shape[2], : a.shape[3]]
        return a, mask


class VGGEncoder(nn.Module):
    """Small stride-16 FCN for the toy/CPU config (BASELINE config 1)."""

    stride = 16

    def __init__(self, channels=(32, 64, 64, 128)):
        super().__init__()
        layers = []
        in_ch = 1
        for ch in channels:
            layers += [
                nn.Conv2d(in_ch, ch, 3, padding=1), nn.ReLU(inplace=True),
                nn.Conv2d(ch, ch, 3, padding=1), nn.ReLU(inplace=True),
                nn.MaxPool2d(2, ceil_mode=True),
            ]
            in_ch = ch
        self.features = nn.Sequential(*layers)
        self.out_channels = channels[-1]

    def forward(self, x: Tensor, x_mask: Tensor) -> Tuple[Tensor, Tensor]:
        a = self.features(x)
        mask = downsample_mask(x_mask, self.stride)
        mask = mask[:, : a.shape[2], : a.shape[3]]
        return a, mask


def build_encoder(name: str, **kw) -> nn.Module:
    if name == "densenet":
        return DenseNetEncoder(**kw)
    if name == "vgg":
        return VGGEncoder(**kw)
    raise ValueError(f"unknown encoder {name!r}")
