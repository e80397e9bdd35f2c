"""Greedy and beam-search decode.

Reference semantics: encode once; loop t < maxlen running one Parser step
for the k live hypotheses (batched), top-k over k*K expansions, hypotheses
emitting <eos>=0 die; final ranking by length-normalized score (score/len);
greedy = the k=1 path.

MI355X-first design: the whole beam step is tensorized with STATIC shapes
(always B*k rows; dead hypotheses are padded, never dropped) so the per-token
step is hipGraph-capturable (`CapturedBeamStep`): on ROCm the graph replay
removes the per-launch host overhead that dominates this latency-bound loop.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
from torch import Tensor

from ..data.vocab import EOS_ID

NEG_INF = -1e30


@torch.no_grad()
def _expand_for_beams(t: Tensor, k: int) -> Tensor:
    """[B, ...] -> [B*k, ...] repeating each row k times."""
    return t.repeat_interleave(k, dim=0)


class BeamState:
    """Static-shape beam state: all tensors keep [B*k, ...] layout."""

    def __init__(self, model, x: Tensor, x_mask: Tensor, beam: int):
        self.model = model
        self.k = beam
        a, mask = model.encode(x, x_mask)
        b, d, hp, wp = a.shape
        self.b, self.hp, self.wp = b, hp, wp
        a_flat, ua = model.parser.attention.precompute(a)
        self.a_flat = _expand_for_beams(a_flat, beam)
        self.ua = _expand_for_beams(ua, beam)
        self.mask = _expand_for_beams(mask, beam)
        mask_flat = self.mask.flatten(1)
        self.s = model.parser.init_state(self.a_flat, mask_flat)
        self.coverage = torch.zeros(b * beam, hp, wp, device=a.device, dtype=a.dtype)
        self.emb_prev = torch.zeros(b * beam, model.parser.embedding.embedding_dim,
                                    device=a.device, dtype=a.dtype)
        self.scores = torch.full((b, beam), NEG_INF, device=a.device)
        self.scores[:, 0] = 0.0
        self.alive = torch.ones(b, beam, dtype=torch.bool, device=a.device)
        self.lengths = torch.zeros(b, beam, dtype=torch.long, device=a.device)
        self.tokens: List[Tensor] = []     # list of [B, k] per step
        self.parents: List[Tensor] = []    # list of [B, k] back-pointers

    def step(self, fused: bool = True) -> bool:
        """One expansion step.  Returns True when every hypothesis is dead."""
        m = self.model.parser
        b, k = self.b, self.k
        logits, s_new, _, alpha = m.step(
            self.emb_prev, self.s, self.a_flat, self.ua, self.coverage,
            self.mask, fused=fused,
        )
        logprob = torch.log_softmax(logits.float(), dim=-1).view(b, k, -1)
        K = logprob.shape[-1]
        # Dead hypotheses may only emit <eos> at zero cost (score frozen).
        dead = ~self.alive
        if dead.any():
            frozen = torch.full_like(logprob, NEG_INF)
            frozen[:, :, EOS_ID] = 0.0
            logprob = torch.where(dead.unsqueeze(-1), frozen, logprob)
        total = self.scores.unsqueeze(-1) + logprob          # [B,k,K]
        top_scores, flat_idx = total.view(b, k * K).topk(k, dim=1)
        parent = flat_idx // K                               # [B,k]
        token = flat_idx % K                                 # [B,k]

        # Gather/reorder states by parent beam.
        gather_rows = (parent + torch.arange(b, device=parent.device)
                       .unsqueeze(1) * k).view(-1)           # [B*k]
        self.s = s_new[gather_rows]
        new_cov = self.coverage + alpha.view(b * k, self.hp, self.wp)
        self.coverage = new_cov[gather_rows]
        parent_alive = torch.gather(self.alive, 1, parent)
        self.lengths = torch.gather(self.lengths, 1, parent) + parent_alive.long()
        self.alive = parent_alive & (token != EOS_ID)
        self.scores = top_scores
        self.tokens.append(token)
        self.parents.append(parent)
        emb = self.model.parser.embedding(token.view(-1))
        # Dead rows keep the zero embedding (value irrelevant: frozen).
        self.emb_prev = emb.to(self.emb_prev.dtype)
        return bool((~self.alive).all().item())

    def extract(self) -> List[List[int]]:
        """Backtrace the best (length-normalized) hypothesis per image."""
        b, k = self.b, self.k
        norm = self.scores / self.lengths.clamp_min(1).float()
        best = norm.argmax(dim=1)                            # [B]
        out: List[List[int]] = []
        tokens = [t.cpu() for t in self.tokens]
        parents = [p.cpu() for p in self.parents]
        best = best.cpu()
        for i in range(b):
            beam_idx = int(best[i])
            seq: List[int] = []
            for t in range(len(tokens) - 1, -1, -1):
                seq.append(int(tokens[t][i, beam_idx]))
                beam_idx = int(parents[t][i, beam_idx])
            seq.reverse()
            if EOS_ID in seq:
                seq = seq[: seq.index(EOS_ID)]
            out.append(seq)
        return out


@torch.no_grad()
def beam_decode(
    model,
    x: Tensor,
    x_mask: Tensor,
    beam: int = 10,
    maxlen: int = 200,
    fused: bool = True,
) -> List[List[int]]:
    """Beam-search decode a batch -> token id lists (no eos)."""
    model.eval()
    state = BeamState(model, x, x_mask, beam)
    for _ in range(maxlen):
        if state.step(fused=fused):
            break
    return state.extract()


@torch.no_grad()
def greedy_decode(
    model, x: Tensor, x_mask: Tensor, maxlen: int = 200, fused: bool = True
) -> List[List[int]]:
    return beam_decode(model, x, x_mask, beam=1, maxlen=maxlen, fused=fused)
