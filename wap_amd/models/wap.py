"""The full WAP model: Watcher (encoder) + Attender/Parser (decoder) + loss.

Loss is the reference's masked token-level cross-entropy (SURVEY.md §3.1
#7):  sum_{t,b} y_mask * CE(logits_tb, y_tb) / sum y_mask.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
from torch import Tensor, nn
import torch.nn.functional as F

from .decoder import Parser
from .encoder import build_encoder


class WAPModel(nn.Module):
    def __init__(
        self,
        n_tokens: int = 112,
        encoder: str = "densenet",
        embed_dim: int = 256,
        hidden_dim: int = 256,
        attention_dim: int = 512,
        coverage_ch: int = 512,
        coverage_kernel: int = 11,
        encoder_kw: Optional[dict] = None,
    ):
        super().__init__()
        self.encoder_name = encoder
        self.watcher = build_encoder(encoder, **(encoder_kw or {}))
        self.parser = Parser(
            n_tokens=n_tokens,
            annotation_dim=self.watcher.out_channels,
            embed_dim=embed_dim,
            hidden_dim=hidden_dim,
            attention_dim=attention_dim,
            coverage_ch=coverage_ch,
            coverage_kernel=coverage_kernel,
        )
        self.n_tokens = n_tokens

    def use_channels_last(self) -> "WAPModel":
        """Run the encoder in NHWC (channels_last): MIOpen's bf16 conv fast
        paths are NHWC igemm kernels and otherwise transpose around every
        conv (measured batched_transpose_* kernels); NHWC also makes the
        attention's [B,L,D] annotation view copy-free."""
        self._channels_last = True
        self.watcher.to(memory_format=torch.channels_last)
        return self

    def encode(self, x: Tensor, x_mask: Tensor) -> Tuple[Tensor, Tensor]:
        if getattr(self, "_channels_last", False):
            x = x.to(memory_format=torch.channels_last)
        return self.watcher(x, x_mask)

    def forward(
        self, x: Tensor, x_mask: Tensor, y: Tensor, y_mask: Tensor
    ) -> Dict[str, Tensor]:
        a, mask = self.encode(x, x_mask)
        logits = self.parser(a, mask, y)       # [B,T,K]
        loss, n_tok = masked_ce(logits, y, y_mask)
        return {"loss": loss, "logits": logits, "n_tokens": n_tok}


def masked_ce(logits: Tensor, y: Tensor, y_mask: Tensor) -> Tuple[Tensor, Tensor]:
    """Masked cross entropy, normalized by the token count."""
    b, t, k = logits.shape
    ce = F.cross_entropy(
        logits.reshape(b * t, k).float(), y.reshape(b * t), reduction="none"
    ).view(b, t)
    denom = y_mask.sum().clamp_min(1.0)
    return (ce * y_mask).sum() / denom, denom
