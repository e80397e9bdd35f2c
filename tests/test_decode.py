"""Beam/greedy decode semantics."""

import torch

from wap_amd.data.vocab import EOS_ID
from wap_amd.models import WAPModel
from wap_amd.runtime import beam_decode, greedy_decode

torch.manual_seed(0)


def _toy_model():
    return WAPModel(n_tokens=12, encoder="vgg", embed_dim=16, hidden_dim=16,
                    attention_dim=16, coverage_ch=8, coverage_kernel=3)


def test_greedy_terminates_and_no_eos_inside():
    m = _toy_model().eval()
    x = torch.rand(3, 1, 32, 48)
    xm = torch.ones(3, 32, 48)
    hyps = greedy_decode(m, x, xm, maxlen=25)
    assert len(hyps) == 3
    for h in hyps:
        assert len(h) <= 25
        assert EOS_ID not in h


def test_greedy_is_argmax_rollout():
    """beam=1 must equal a manual argmax rollout of parser.step."""
    m = _toy_model().eval()
    x = torch.rand(1, 1, 32, 32)
    xm = torch.ones(1, 32, 32)
    hyp = greedy_decode(m, x, xm, maxlen=10)[0]

    with torch.no_grad():
        a, mask = m.encode(x, xm)
        a_flat, ua = m.parser.attention.precompute(a)
        s = m.parser.init_state(a_flat, mask.flatten(1))
        cov = torch.zeros(1, a.shape[2], a.shape[3])
        emb = torch.zeros(1, 16)
        manual = []
        for _ in range(10):
            logits, s, _, alpha = m.parser.step(emb, s, a_flat, ua, cov, mask)
            tok = int(logits.argmax(-1))
            if tok == EOS_ID:
                break
            manual.append(tok)
            cov = cov + alpha.view_as(cov)
            emb = m.parser.embedding(torch.tensor([tok]))
    assert hyp == manual


def test_beam_score_not_worse_than_greedy():
    """The beam-k best length-normalized logprob >= the greedy path's."""
    m = _toy_model().eval()
    x = torch.rand(2, 1, 32, 48)
    xm = torch.ones(2, 32, 48)

    def seq_score(hyp):
        # re-score a hypothesis (incl. its eos) under the model
        with torch.no_grad():
            a, mask = m.encode(x[:1], xm[:1])
            a_flat, ua = m.parser.attention.precompute(a)
            s = m.parser.init_state(a_flat, mask.flatten(1))
            cov = torch.zeros(1, a.shape[2], a.shape[3])
            emb = torch.zeros(1, 16)
            total = 0.0
            for tok in hyp + [EOS_ID]:
                logits, s, _, alpha = m.parser.step(emb, s, a_flat, ua, cov, mask)
                total += torch.log_softmax(logits.float(), -1)[0, tok].item()
                cov = cov + alpha.view_as(cov)
                emb = m.parser.embedding(torch.tensor([tok]))
            return total / max(len(hyp) + 1, 1)

    g = greedy_decode(m, x[:1], xm[:1], maxlen=12)[0]
    b = beam_decode(m, x[:1], xm[:1], beam=4, maxlen=12)[0]
    assert seq_score(b) >= seq_score(g) - 1e-5


def test_beam_batch_matches_single():
    """Batched beam search must equal per-image beam search."""
    m = _toy_model().eval()
    x = torch.rand(3, 1, 32, 32)
    xm = torch.ones(3, 32, 32)
    batched = beam_decode(m, x, xm, beam=3, maxlen=12)
    singles = [beam_decode(m, x[i:i+1], xm[i:i+1], beam=3, maxlen=12)[0]
               for i in range(3)]
    assert batched == singles


def test_decode_all_masked_image_is_safe():
    """A fully-padded image must decode without NaN/crash (alpha=0, ctx=0)."""
    m = _toy_model().eval()
    x = torch.zeros(2, 1, 32, 32)
    xm = torch.ones(2, 32, 32)
    xm[1] = 0  # image 1 entirely padding
    hyps = beam_decode(m, x, xm, beam=3, maxlen=8)
    assert len(hyps) == 2
    assert all(len(h) <= 8 for h in hyps)


def test_config_args_roundtrip():
    import argparse
    from wap_amd.config import WAPConfig, add_config_args, config_from_args
    p = argparse.ArgumentParser()
    add_config_args(p)
    cfg = config_from_args(p.parse_args([
        "--batch-size", "7", "--encoder", "vgg", "--graph", "off",
        "--beam", "3",
    ]))
    assert cfg.batch_size == 7 and cfg.encoder == "vgg"
    assert cfg.graph == "off" and cfg.beam == 3
    assert cfg.maxlen == WAPConfig().maxlen  # defaults preserved


def test_beamstate_reset_from_reproduces_decode():
    """reset_from must fully re-initialize a state: decoding problem B in
    buffers previously used for problem A equals a fresh decode of B (the
    decode graph cache relies on this)."""
    from wap_amd.runtime.decode import BeamState
    m = _toy_model().eval()
    xa = torch.rand(2, 1, 32, 32)
    xb = torch.rand(2, 1, 32, 32)
    xm = torch.ones(2, 32, 32)
    with torch.no_grad():
        fresh = beam_decode(m, xb, xm, beam=3, maxlen=10, use_graph=False)
        sa = BeamState(m, xa, xm, 3)
        for _ in range(4):
            sa.step()
        sb = BeamState(m, xb, xm, 3)
        sa.reset_from(sb)
        for _ in range(10):
            sa.step()
            if sa.all_dead():
                break
        reused = sa.extract()
    assert reused == fresh


def test_banked_finished_hyp_survives_eviction():
    """A hypothesis that emits <eos> and is later pushed out of the k-slot
    beam must still be rankable at extract() time (death-event banking)."""
    from wap_amd.runtime.decode import BeamState
    st = BeamState.__new__(BeamState)
    st.b, st.k = 1, 2
    st._rec_cap = 0
    T = 3
    st.n_rec = T
    # slot 0 dies at t=1 with a GOOD normalized score (-1.0/2 = -0.5); by the
    # end both surviving slots are live continuations with worse norm scores.
    st.tokens = [torch.tensor([[4, 5]]), torch.tensor([[EOS_ID, 6]]),
                 torch.tensor([[7, 8]])]
    st.parents = [torch.tensor([[0, 0]]), torch.tensor([[0, 1]]),
                  torch.tensor([[1, 1]])]
    st.dieds = [torch.zeros(1, 2, dtype=torch.bool) for _ in range(T)]
    st.dieds[1][0, 0] = True
    st.dscores = [torch.zeros(1, 2) for _ in range(T)]
    st.dscores[1][0, 0] = -1.0
    st.S = {
        "scores": torch.tensor([[-9.0, -9.5]]),
        "lengths": torch.tensor([[3, 3]]),
        "alive": torch.tensor([[True, True]]),
    }
    out = st.extract()
    # banked: backtrace from (t=1, slot 0): t1 slot0 tok=eos, parent 0;
    # t0 slot0 tok=4 -> [4, eos] -> strip eos -> [4]
    assert out == [[4]]


def test_live_hyp_beats_worse_banked():
    """Banking must not override a better live hypothesis."""
    from wap_amd.runtime.decode import BeamState
    st = BeamState.__new__(BeamState)
    st.b, st.k = 1, 2
    st._rec_cap = 0
    st.n_rec = 2
    st.tokens = [torch.tensor([[4, 5]]), torch.tensor([[EOS_ID, 6]])]
    st.parents = [torch.tensor([[0, 0]]), torch.tensor([[0, 1]])]
    st.dieds = [torch.zeros(1, 2, dtype=torch.bool) for _ in range(2)]
    st.dieds[1][0, 0] = True
    st.dscores = [torch.zeros(1, 2) for _ in range(2)]
    st.dscores[1][0, 0] = -8.0          # norm -4.0
    st.S = {
        "scores": torch.tensor([[-8.0, -1.0]]),   # slot 1 live, norm -0.5
        "lengths": torch.tensor([[2, 2]]),
        "alive": torch.tensor([[False, True]]),
    }
    out = st.extract()
    assert out == [[5, 6]]
