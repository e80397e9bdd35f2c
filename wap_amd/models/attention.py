"""Attender: coverage-based additive attention.

Per decode step t over spatial positions i in 1..L (L = H'*W'):

    beta_t = sum_{l<t} alpha_l                     # coverage map
    F      = Q * beta_t                            # 11x11 conv, q=512 ch
    e_ti   = nu . tanh(W_a s_hat + U_a a_i + U_f F_i)
    alpha  = masked_softmax_i(e)
    c_t    = sum_i alpha_i a_i

U_a a is constant across steps for one image, so it is precomputed once per
sequence (``precompute``).  The score + masked softmax + context reduction
run through ``wap_amd.ops`` (HIP-fused on gfx950).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor, nn

from .. import ops


class CoverageAttention(nn.Module):
    def __init__(self, annotation_dim: int = 684, state_dim: int = 256,
                 attention_dim: int = 512, coverage_ch: int = 512,
                 coverage_kernel: int = 11):
        super().__init__()
        self.W_a = nn.Linear(state_dim, attention_dim, bias=False)
        self.U_a = nn.Linear(annotation_dim, attention_dim, bias=True)
        self.conv_Q = nn.Conv2d(1, coverage_ch, coverage_kernel,
                                padding=coverage_kernel // 2, bias=False)
        self.U_f = nn.Linear(coverage_ch, attention_dim, bias=True)
        self.nu = nn.Parameter(torch.zeros(attention_dim))
        nn.init.normal_(self.nu, std=attention_dim ** -0.5)
        self.annotation_dim = annotation_dim
        self.attention_dim = attention_dim

    def precompute(self, a: Tensor) -> Tuple[Tensor, Tensor]:
        """a[B,D,H,W] -> (a_flat[B,L,D], ua[B,L,A]) done once per sequence."""
        b, d, h, w = a.shape
        a_flat = a.flatten(2).transpose(1, 2).contiguous()  # [B,L,D]
        ua = self.U_a(a_flat)                               # [B,L,A]
        return a_flat, ua

    def merged_coverage_weight(self) -> Tensor:
        """U_f o conv_Q collapses to ONE 11x11 conv: conv_Q has 1 input
        channel, so U_f(conv_Q(beta)) == conv2d(beta, U_f.weight @ Q).
        Tiny [A,q]@[q,KK] GEMM per step; removes the [B*L,q]x[q,A]
        projection GEMM and the F tensor from the per-step path."""
        q_ch, _, kh, kw = self.conv_Q.weight.shape
        w = self.U_f.weight @ self.conv_Q.weight.view(q_ch, kh * kw)
        return w.view(self.attention_dim, 1, kh, kw)

    def attend(
        self,
        q: Tensor,          # [B, A] precomputed W_a s_hat
        a_flat: Tensor,     # [B, L, D]
        ua: Tensor,         # [B, L, A] precomputed U_a a
        coverage: Tensor,   # [B, H', W'] running sum of alphas
        mask: Tensor,       # [B, H', W'] feature-grid mask
        w_cov: Optional[Tensor] = None,  # merged_coverage_weight(), hoisted
        w_pack=None,        # ops.pack_coverage_weight(w_cov), hoisted
    ) -> Tuple[Tensor, Tensor]:
        """Attention with the query GEMM and coverage-weight merge done by
        the caller (the training unroll batches/hoists them across steps).
        Returns (context[B,D], alpha[B,L])."""
        b, hp, wp = coverage.shape
        w = self.merged_coverage_weight() if w_cov is None else w_cov
        kh = w.shape[2]
        from ..ops import functional as opf
        if opf.can_fuse_cov_score(coverage, w, ua):
            # single-launch fused path (csrc/fattn.hip): conv-over-coverage
            # + bias + additive score in one kernel, fwd and bwd
            alpha, ctx = opf.fused_cov_attention(
                q, ua, coverage, w, w_pack, self.U_f.bias, self.nu, a_flat,
                mask.flatten(1)
            )
            return ctx, alpha
        # 1-channel conv via unfold+GEMM (MIOpen naive-kernel bypass, see
        # encoder.conv_1ch) — this runs once per decode timestep.
        from .encoder import conv_1ch
        uf = conv_1ch(coverage.unsqueeze(1), w, padding=kh // 2)  # [B,A,H',W']
        uf = uf.flatten(2).transpose(1, 2) + self.U_f.bias        # [B,L,A]
        alpha, ctx = ops.coverage_attention(
            q, ua, uf, self.nu, a_flat, mask.flatten(1)
        )
        return ctx, alpha

    def forward(
        self,
        s_hat: Tensor,      # [B, n] pre-attention decoder state
        a_flat: Tensor,     # [B, L, D]
        ua: Tensor,         # [B, L, A] precomputed U_a a
        coverage: Tensor,   # [B, H', W'] running sum of alphas
        mask: Tensor,       # [B, H', W'] feature-grid mask
        w_cov: Optional[Tensor] = None,
        w_pack=None,
    ) -> Tuple[Tensor, Tensor]:
        """Returns (context[B,D], alpha[B,L])."""
        q = self.W_a(s_hat)                                 # [B,A]
        return self.attend(q, a_flat, ua, coverage, mask, w_cov, w_pack)
