"""Greedy and beam-search decode.

Reference semantics: encode once; loop t < maxlen running one Parser step
for the k live hypotheses (batched), top-k over k*K expansions, hypotheses
emitting <eos>=0 die; final ranking by length-normalized score (score/len);
greedy = the k=1 path.

MI355X-first design: the per-token step is a single sync-free tensor
function over STATIC [B*k, ...] state buffers (dead hypotheses are padded,
never dropped), updated in place — so the whole step (two fused GRU cells,
coverage conv, fused attention, readout, top-k, state reorder) is
hipGraph-captured once and replayed per token (``use_graph=True``): replay
removes the per-launch host overhead that dominates this latency-bound
loop (MI355X_MICROARCH.md row graph-replay-floor).  The host only syncs on
the liveness flag every ``sync_every`` tokens.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch


def _capture_mode() -> str:
    """hipGraph capture mode: thread_local allows other host threads (the
    pinned-memory prefetcher) to touch HIP during capture; override with
    WAP_CAPTURE_MODE=global for A/B."""
    import os
    return os.environ.get("WAP_CAPTURE_MODE", "thread_local")

from torch import Tensor

from ..data.vocab import EOS_ID

NEG_INF = -1e30


@torch.no_grad()
def _expand_for_beams(t: Tensor, k: int) -> Tensor:
    """[B, ...] -> [B*k, ...] repeating each row k times."""
    return t.repeat_interleave(k, dim=0)


@torch.no_grad()
def _beam_step_(model, S: Dict[str, Tensor], b: int, k: int, fused: bool) -> None:
    """One beam expansion; updates every state buffer of S IN PLACE with no
    host synchronization and static shapes (hipGraph-capturable)."""
    m = model.parser
    hp, wp = S["coverage"].shape[1], S["coverage"].shape[2]
    logits, s_new, _, alpha = m.step(
        S["emb"], S["s"], S["a_flat"], S["ua"], S["coverage"], S["mask"],
        fused=fused, w_cov=S["w_cov"], w_pack=(S["wpad"], S["wT"]),
        w_cat=S["w_cat"],
    )
    logprob = torch.log_softmax(logits.float(), dim=-1).view(b, k, -1)
    # Dead hypotheses may only emit <eos> at zero cost (score frozen);
    # the frozen-row template is a static per-problem buffer.
    K = logprob.shape[-1]
    dead = ~S["alive"]
    logprob = torch.where(dead.unsqueeze(-1), S["frozen"], logprob)
    total = S["scores"].unsqueeze(-1) + logprob               # [B,k,K]
    top_scores, flat_idx = total.view(b, k * K).topk(k, dim=1)
    parent = flat_idx // K                                    # [B,k]
    token = flat_idx % K                                      # [B,k]

    rows = (parent + S["brow"]).view(-1)                      # [B*k]
    S["s"].copy_(s_new[rows])
    new_cov = S["coverage"] + alpha.view(b * k, hp, wp).to(S["coverage"].dtype)
    S["coverage"].copy_(new_cov[rows])
    parent_alive = torch.gather(S["alive"], 1, parent)
    S["lengths"].copy_(torch.gather(S["lengths"], 1, parent) + parent_alive.long())
    # Bank finished hypotheses: a hyp that emits <eos> here could later be
    # evicted from the k-slot beam by a temporarily-better live candidate;
    # recording the death event lets extract() still rank it.
    S["died"].copy_(parent_alive & (token == EOS_ID))
    S["alive"].copy_(parent_alive & (token != EOS_ID))
    S["scores"].copy_(top_scores)
    S["token"].copy_(token)
    S["parent"].copy_(parent)
    S["emb"].copy_(model.parser.embedding(token.view(-1)).to(S["emb"].dtype))


class BeamState:
    """Static-shape beam state: all tensors keep [B*k, ...] layout."""

    def __init__(self, model, x: Tensor, x_mask: Tensor, beam: int):
        self.model = model
        self.k = beam
        a, mask = model.encode(x, x_mask)
        b, d, hp, wp = a.shape
        self.b, self.hp, self.wp = b, hp, wp
        a_flat, ua = model.parser.attention.precompute(a)
        dev = a.device
        mask_bk = _expand_for_beams(mask, beam)
        a_flat_bk = _expand_for_beams(a_flat, beam).contiguous()
        s0 = model.parser.init_state(a_flat_bk, mask_bk.flatten(1))
        # merged coverage-conv weight (+ its packed form for the fused
        # kernel), built once per decode problem — derived from live
        # weights; reset_from refreshes them so cached graphs never replay
        # a stale copy after training steps
        w_cov = (model.parser.attention.merged_coverage_weight()
                 .detach().contiguous())
        from ..ops.functional import pack_coverage_weight
        wpad, wT = pack_coverage_weight(w_cov)
        pm = model.parser
        w_cat = torch.cat([pm.W_s.weight, pm.W_c.weight, pm.W_y.weight],
                          dim=1).detach().contiguous()
        frozen = torch.full((b, beam, pm.n_tokens), NEG_INF, device=dev)
        frozen[:, :, EOS_ID] = 0.0
        self.S: Dict[str, Tensor] = {
            "w_cov": w_cov,
            "wpad": wpad,
            "wT": wT,
            "w_cat": w_cat,
            "frozen": frozen,
            "a_flat": a_flat_bk,
            "ua": _expand_for_beams(ua, beam).contiguous(),
            "mask": mask_bk.contiguous(),
            "s": s0.contiguous(),
            # coverage accumulates in f32, matching the training path (the
            # teacher-forced unroll promotes coverage to f32 at step 1)
            "coverage": torch.zeros(b * beam, hp, wp, device=dev,
                                    dtype=torch.float32),
            "emb": torch.zeros(b * beam, model.parser.embedding.embedding_dim,
                               device=dev, dtype=s0.dtype),
            "scores": torch.full((b, beam), NEG_INF, device=dev),
            "alive": torch.ones(b, beam, dtype=torch.bool, device=dev),
            "died": torch.zeros(b, beam, dtype=torch.bool, device=dev),
            "lengths": torch.zeros(b, beam, dtype=torch.long, device=dev),
            "token": torch.zeros(b, beam, dtype=torch.long, device=dev),
            "parent": torch.zeros(b, beam, dtype=torch.long, device=dev),
            "brow": (torch.arange(b, device=dev) * beam).unsqueeze(1),
        }
        self.S["scores"][:, 0] = 0.0
        # per-step records: list mode (append clones) until
        # ensure_record(maxlen) switches to preallocated [T,B,k] buffers —
        # no allocation and one batched D2H at extract() instead of 4
        # clones + 4 transfers per decoded token
        self.tokens = []     # [B, k] per step (or [T,B,k] buffer)
        self.parents = []    # [B, k] back-pointers
        self.dieds = []      # [B, k] finished-here flags
        self.dscores = []    # [B, k] scores at that step
        self.n_rec = 0
        self._rec_cap = 0
        self.graph = None

    def reset_from(self, other: "BeamState") -> None:
        """Reuse this state's (captured) buffers for a new decode problem of
        the same shape: copy the per-problem tensors in, reset the mutable
        state.  The hipGraph stays valid (same storage)."""
        S, O = self.S, other.S
        for n in ("a_flat", "ua", "mask", "s", "coverage", "emb", "w_cov",
                  "wpad", "wT", "w_cat"):
            S[n].copy_(O[n])
        S["scores"].fill_(NEG_INF)
        S["scores"][:, 0] = 0.0
        S["alive"].fill_(True)
        S["died"].fill_(False)
        S["lengths"].zero_()
        self.n_rec = 0
        if self._rec_cap == 0:
            self.tokens, self.parents = [], []
            self.dieds, self.dscores = [], []

    def step(self, fused: bool = True) -> None:
        _beam_step_(self.model, self.S, self.b, self.k, fused)
        self.record()

    def ensure_record(self, maxlen: int) -> None:
        """Switch per-step records to preallocated [maxlen,B,k] buffers."""
        if self._rec_cap >= maxlen:
            self.n_rec = 0
            return
        dev = self.S["token"].device
        b, k = self.b, self.k
        self.tokens = torch.empty(maxlen, b, k, dtype=torch.long, device=dev)
        self.parents = torch.empty_like(self.tokens)
        self.dieds = torch.empty(maxlen, b, k, dtype=torch.bool, device=dev)
        self.dscores = torch.empty(maxlen, b, k, device=dev)
        self._rec_cap = maxlen
        self.n_rec = 0

    def record(self) -> None:
        if self._rec_cap:
            t = self.n_rec
            self.tokens[t].copy_(self.S["token"], non_blocking=True)
            self.parents[t].copy_(self.S["parent"], non_blocking=True)
            self.dieds[t].copy_(self.S["died"], non_blocking=True)
            self.dscores[t].copy_(self.S["scores"], non_blocking=True)
        else:
            self.tokens.append(self.S["token"].clone())
            self.parents.append(self.S["parent"].clone())
            self.dieds.append(self.S["died"].clone())
            self.dscores.append(self.S["scores"].clone())
        self.n_rec += 1

    # ---- hipGraph capture -------------------------------------------------
    def capture(self, fused: bool = True, autocast_bf16: bool = False) -> None:
        """Capture one beam step into a hipGraph (CUDAGraph on PyTorch-ROCm);
        subsequent ``replay()`` calls re-run it on the static buffers."""
        from .graph_train import safe_stream_sync
        safe_stream_sync()  # NEVER hipDeviceSynchronize after graph replays
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):  # warm up allocator/MIOpen off-graph
            for _ in range(2):
                _beam_step_(self.model, self.S, self.b, self.k, fused)
        torch.cuda.current_stream().wait_stream(side)
        safe_stream_sync()
        g = torch.cuda.CUDAGraph()
        ctx = torch.autocast("cuda", dtype=torch.bfloat16, enabled=autocast_bf16)
        with ctx, torch.cuda.graph(g, capture_error_mode=_capture_mode()):
            _beam_step_(self.model, self.S, self.b, self.k, fused)
        self.graph = g

    def replay(self) -> None:
        self.graph.replay()
        self.record()

    def all_dead(self) -> bool:
        return bool((~self.S["alive"]).all().item())

    def extract(self) -> List[List[int]]:
        """Best (length-normalized) hypothesis per image.

        Candidates are the union of (a) banked death events (score frozen at
        the step the hyp emitted <eos>, length = that step + 1) — so a
        finished hyp evicted from the beam later is still considered — and
        (b) the live hypotheses surviving at the last step; ranked together
        by length-normalized score.  (Surviving dead slots are a subset of
        their own death events.)"""
        b = self.b
        T = self.n_rec
        if self._rec_cap:
            tokens = self.tokens[:T].cpu()
            parents = self.parents[:T].cpu()
            dieds = self.dieds[:T].cpu()
            dscores = self.dscores[:T].cpu()
        else:
            tokens = [t.cpu() for t in self.tokens]
            parents = [p.cpu() for p in self.parents]
            dieds = [d.cpu() for d in self.dieds]
            dscores = [s.cpu() for s in self.dscores]
        live_norm = (self.S["scores"]
                     / self.S["lengths"].clamp_min(1).float()).cpu()
        alive = self.S["alive"].cpu()
        out: List[List[int]] = []
        for i in range(b):
            best = None  # (norm_score, step, slot)
            for t in range(T):
                d = dieds[t][i]
                if d.any():
                    for slot in d.nonzero().flatten().tolist():
                        sc = float(dscores[t][i, slot]) / (t + 1)
                        if best is None or sc > best[0]:
                            best = (sc, t, slot)
            for slot in range(self.k):
                if alive[i, slot] or best is None:
                    sc = float(live_norm[i, slot])
                    if best is None or sc > best[0]:
                        best = (sc, T - 1, slot)
            _, t0, beam_idx = best
            seq: List[int] = []
            for t in range(t0, -1, -1):
                seq.append(int(tokens[t][i, beam_idx]))
                beam_idx = int(parents[t][i, beam_idx])
            seq.reverse()
            if EOS_ID in seq:
                seq = seq[: seq.index(EOS_ID)]
            out.append(seq)
        return out


# Captured-graph cache: capture costs ~100 ms, so graphs + their static
# buffers persist across beam_decode calls, keyed by problem shape.  Replay
# reads weights through the live parameter storage, so training in between
# does not invalidate a graph.  Bounded LRU.
_GRAPH_CACHE: Dict[tuple, BeamState] = {}
_GRAPH_CACHE_CAP = 16
_MODEL_KEYS = 0


def _model_key(model) -> int:
    """Stable per-model cache id (id(model) can be recycled after GC while
    cached graphs still hold the OLD model's weights)."""
    global _MODEL_KEYS
    k = getattr(model, "_wap_decode_key", None)
    if k is None:
        _MODEL_KEYS += 1
        k = _MODEL_KEYS
        model._wap_decode_key = k
    return k


def _graph_key(model, state: BeamState, fused: bool, autocast_on: bool):
    return (_model_key(model), state.b, state.k, state.hp, state.wp,
            state.S["a_flat"].shape[-1], str(state.S["a_flat"].dtype),
            fused, autocast_on)


@torch.no_grad()
def beam_decode(
    model,
    x: Tensor,
    x_mask: Tensor,
    beam: int = 10,
    maxlen: int = 200,
    fused: bool = False,
    use_graph: Optional[bool] = None,
    sync_every: int = 8,
) -> List[List[int]]:
    """Beam-search decode a batch -> token id lists (no eos).

    ``fused=True`` routes the GRU cells through the single-launch VALU
    GEMV kernel — an r01 launch-count optimization that hipGraph replay
    made obsolete: under graphs the rocBLAS gate-GEMM path (default)
    measured 18.9k vs 7.9k tok/s (PERF.md r02)."""
    model.eval()
    state = BeamState(model, x, x_mask, beam)
    if use_graph is None:
        use_graph = x.is_cuda
    state.ensure_record(maxlen)
    if use_graph and x.is_cuda:
        autocast_on = torch.is_autocast_enabled("cuda")
        key = _graph_key(model, state, fused, autocast_on)
        cached = _GRAPH_CACHE.get(key)
        if cached is None:
            state.capture(fused=fused, autocast_bf16=autocast_on)
            # capture mutated the state: re-init from a fresh encode
            state.reset_from(BeamState(model, x, x_mask, beam))
            if len(_GRAPH_CACHE) >= _GRAPH_CACHE_CAP:
                _GRAPH_CACHE.pop(next(iter(_GRAPH_CACHE)))
            _GRAPH_CACHE[key] = state
        else:
            cached.reset_from(state)
            state = cached
            state.ensure_record(maxlen)
        for t in range(maxlen):
            state.replay()
            if (t + 1) % sync_every == 0 and state.all_dead():
                break
    else:
        for t in range(maxlen):
            state.step(fused=fused)
            if state.all_dead():
                break
    return state.extract()


@torch.no_grad()
def greedy_decode(
    model, x: Tensor, x_mask: Tensor, maxlen: int = 200, fused: bool = True,
    use_graph: Optional[bool] = None,
) -> List[List[int]]:
    return beam_decode(model, x, x_mask, beam=1, maxlen=maxlen, fused=fused,
                       use_graph=use_graph)
