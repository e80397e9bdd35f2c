"""Parser: conditioned two-cell GRU decoder + maxout readout.

Per step t:

    s_hat_t = GRU1(E y_{t-1}, s_{t-1})       # m=256 embedding, n=256 hidden
    c_t     = Attend(s_hat_t, a, beta_t)     # coverage attention
    s_t     = GRU2(c_t, s_hat_t)
    o_t     = W_o maxout_2(W_s s_t + W_c c_t + W_y E y_{t-1})  # 256->128->K
    s_0     = tanh(W_init . masked_mean_i(a_i))
    y_0     = zero embedding (start "token")

GRU steps run through wap_amd.ops (fused gate kernel in training, fully
fused single-launch cell in decode).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor, nn

from .. import ops
from .attention import CoverageAttention


class GRUCellWAP(nn.Module):
    """GRU cell with explicit packed weights so the HIP fused kernel maps
    1:1 (gate order r|z|n, PyTorch convention)."""

    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.w_x = nn.Parameter(torch.empty(3 * hidden_dim, input_dim))
        self.w_h = nn.Parameter(torch.empty(3 * hidden_dim, hidden_dim))
        self.b_x = nn.Parameter(torch.zeros(3 * hidden_dim))
        self.b_h = nn.Parameter(torch.zeros(3 * hidden_dim))
        nn.init.xavier_uniform_(self.w_x)
        nn.init.orthogonal_(self.w_h[:hidden_dim])
        nn.init.orthogonal_(self.w_h[hidden_dim:2 * hidden_dim])
        nn.init.orthogonal_(self.w_h[2 * hidden_dim:])

    def forward(self, x: Tensor, h: Tensor) -> Tensor:
        return ops.gru_cell(x, h, self.w_x, self.w_h, self.b_x, self.b_h)

    @torch.no_grad()
    def step_fused(self, x: Tensor, h: Tensor) -> Tensor:
        return ops.fused_gru_cell(x, h, self.w_x, self.w_h, self.b_x, self.b_h)


class Parser(nn.Module):
    def __init__(
        self,
        n_tokens: int = 112,
        annotation_dim: int = 684,
        embed_dim: int = 256,
        hidden_dim: int = 256,
        attention_dim: int = 512,
        coverage_ch: int = 512,
        coverage_kernel: int = 11,
        maxout_pieces: int = 2,
        readout_dim: int = 128,
    ):
        super().__init__()
        self.embedding = nn.Embedding(n_tokens, embed_dim)
        self.gru1 = GRUCellWAP(embed_dim, hidden_dim)
        self.gru2 = GRUCellWAP(annotation_dim, hidden_dim)
        self.attention = CoverageAttention(
            annotation_dim, hidden_dim, attention_dim, coverage_ch, coverage_kernel
        )
        self.W_init = nn.Linear(annotation_dim, hidden_dim)
        # readout: maxout over `maxout_pieces` of readout_dim
        self.W_s = nn.Linear(hidden_dim, maxout_pieces * readout_dim)
        self.W_c = nn.Linear(annotation_dim, maxout_pieces * readout_dim, bias=False)
        self.W_y = nn.Linear(embed_dim, maxout_pieces * readout_dim, bias=False)
        self.W_o = nn.Linear(readout_dim, n_tokens)
        self.maxout_pieces = maxout_pieces
        self.readout_dim = readout_dim
        self.hidden_dim = hidden_dim
        self.n_tokens = n_tokens

    def init_state(self, a_flat: Tensor, mask_flat: Tensor) -> Tensor:
        """s_0 = tanh(W_init . masked_mean(a))."""
        denom = mask_flat.sum(dim=1, keepdim=True).clamp_min(1.0)
        mean_a = (a_flat * mask_flat.unsqueeze(-1)).sum(dim=1) / denom
        return torch.tanh(self.W_init(mean_a))

    def readout(self, s: Tensor, c: Tensor, ey: Tensor,
                w_cat: Optional[Tensor] = None) -> Tensor:
        """Maxout readout -> logits[..., K] (any leading dims).

        W_s s + W_c c + W_y ey == [s|c|ey] @ [W_s|W_c|W_y]^T: one GEMM
        instead of three + two adds.  The training unroll calls this ONCE
        over the stacked [B,T,...] states (the readout does not feed back
        into the recurrence), so its forward+backward GEMMs and the weight
        concatenations happen once per sequence, not once per step.  The
        beam loop passes a hoisted ``w_cat`` (BeamState buffer)."""
        if w_cat is None:
            w_cat = torch.cat(
                [self.W_s.weight, self.W_c.weight, self.W_y.weight], dim=1)
        x_cat = torch.cat([s, c.to(s.dtype), ey.to(s.dtype)], dim=-1)
        pre = torch.nn.functional.linear(x_cat, w_cat, self.W_s.bias)
        pre = pre.view(*pre.shape[:-1], self.readout_dim, self.maxout_pieces)
        return self.W_o(pre.max(dim=-1).values)

    def step(
        self,
        y_prev_emb: Tensor,   # [B, m]
        s_prev: Tensor,       # [B, n]
        a_flat: Tensor,       # [B, L, D]
        ua: Tensor,           # [B, L, A]
        coverage: Tensor,     # [B, H', W']
        mask: Tensor,         # [B, H', W']
        fused: bool = False,
        w_cov: Optional[Tensor] = None,
        w_pack=None,
        w_cat: Optional[Tensor] = None,
    ) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
        """One decode step -> (logits, s_t, c_t, alpha)."""
        gru1 = self.gru1.step_fused if fused else self.gru1
        gru2 = self.gru2.step_fused if fused else self.gru2
        s_hat = gru1(y_prev_emb, s_prev)
        c_t, alpha = self.attention(s_hat, a_flat, ua, coverage, mask,
                                    w_cov, w_pack)
        s_t = gru2(c_t, s_hat)
        logits = self.readout(s_t, c_t, y_prev_emb, w_cat)
        return logits, s_t, c_t, alpha

    def forward(
        self,
        a: Tensor,          # [B, D, H', W'] annotations
        mask: Tensor,       # [B, H', W']
        y: Tensor,          # [B, T] target tokens (teacher forcing)
    ) -> Tensor:
        """Teacher-forced unroll -> logits[B, T, K].

        Launch-count-aware restructuring of the per-step math (the unroll is
        the dispatch-bound part of the train step; see PERF.md):

        - the GRU1 input gates depend only on the (known) teacher-forced
          tokens: ONE [B*T] GEMM up front instead of one per step;
        - the attention query W_a s_hat and GRU2's h-side gates both consume
          s_hat: merged into one GEMM per step;
        - the merged 11x11 coverage-conv weight is built once per sequence;
        - the maxout readout does not feed the recurrence: it runs ONCE over
          the stacked states after the loop (one fwd GEMM + one bwd GEMM
          pair instead of T of each).

        Per-step math is identical to :meth:`step` (tested)."""
        b, _, hp, wp = a.shape
        a_flat, ua = self.attention.precompute(a)
        mask_flat = mask.flatten(1)
        s = self.init_state(a_flat, mask_flat)
        # f32 from step 0 (alpha is f32, so step>=1 coverage is f32 anyway);
        # decode's BeamState uses the same dtype so the two paths match
        coverage = torch.zeros(b, hp, wp, device=a.device, dtype=torch.float32)
        T = y.shape[1]
        att = self.attention
        # teacher-forced embeddings for steps 0..T-1 (step 0 = zero "token")
        emb = self.embedding(y[:, : T - 1])
        emb_all = torch.cat([emb.new_zeros(b, 1, emb.shape[-1]), emb], dim=1)
        gx1_all = torch.nn.functional.linear(emb_all, self.gru1.w_x,
                                             self.gru1.b_x)       # [B,T,3n]
        A = att.attention_dim
        w_qh = torch.cat([att.W_a.weight, self.gru2.w_h], dim=0)  # [A+3n, n]
        b_qh = torch.cat([self.gru2.b_h.new_zeros(A), self.gru2.b_h])
        w_cov = att.merged_coverage_weight()
        import os
        from ..ops import functional as opf
        if os.environ.get("WAP_MANUAL_BPTT", "1") != "0":
            # hand-written BPTT over the unroll (ops/unroll.py): same
            # per-step kernels, but the backward batches every recurrent
            # weight grad into one GEMM and keeps carried grads in
            # preallocated buffers (the autograd tape's per-step grad
            # GEMMs + accumulation adds dominate the dispatch count)
            from ..ops.unroll import decoder_unroll
            dt = a.dtype
            s_all, c_all = decoder_unroll(
                gx1_all.to(dt), s.to(dt), a_flat, ua, mask_flat,
                w_qh.to(dt), b_qh.to(dt), self.gru2.w_x.to(dt),
                self.gru2.b_x.to(dt), self.gru1.w_h.to(dt),
                self.gru1.b_h.to(dt), w_cov, self.attention.U_f.bias,
                self.attention.nu, hp, wp)
            return self.readout(s_all, c_all, emb_all)
        w_pack = (opf.pack_coverage_weight(w_cov)
                  if opf.can_fuse_cov_score(coverage, w_cov, ua) else None)
        ss, cs = [], []
        for t in range(T):
            gh1 = torch.nn.functional.linear(s, self.gru1.w_h, self.gru1.b_h)
            s_hat = ops.gru_gates(gx1_all[:, t], gh1, s)
            qh = torch.nn.functional.linear(s_hat, w_qh, b_qh)
            c_t, alpha = att.attend(qh[:, :A], a_flat, ua, coverage, mask,
                                    w_cov, w_pack)
            gx2 = torch.nn.functional.linear(c_t, self.gru2.w_x, self.gru2.b_x)
            s = ops.gru_gates(gx2, qh[:, A:], s_hat)
            coverage = coverage + alpha.view(b, hp, wp)
            ss.append(s)
            cs.append(c_t)
        s_all = torch.stack(ss, dim=1)                            # [B,T,n]
        c_all = torch.stack(cs, dim=1)                            # [B,T,D]
        return self.readout(s_all, c_all, emb_all)
