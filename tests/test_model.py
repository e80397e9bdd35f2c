"""Model layer: encoder shape contracts, loss math, tiny overfit."""

import pytest
import torch

from wap_amd.models import DenseNetEncoder, VGGEncoder, WAPModel, masked_ce

torch.manual_seed(0)


def test_densenet_channel_walk():
    enc = DenseNetEncoder()
    assert enc.out_channels == 684  # 48+384=432 ->216 ->600 ->300 ->684


def test_encoder_shapes_and_mask():
    for enc in (VGGEncoder(), DenseNetEncoder(growth=8, layers_per_block=2)):
        enc.eval()
        x = torch.randn(2, 1, 49, 130).abs()
        xm = torch.ones(2, 49, 130)
        xm[1, 30:, :] = 0  # image 1 is shorter
        a, m = enc(x, xm)
        assert a.shape[0] == 2 and a.shape[1] == enc.out_channels
        # stride-16 contract (ceil division)
        assert a.shape[2] == (49 + 15) // 16 and a.shape[3] == (130 + 15) // 16
        assert m.shape == a.shape[2:] or m.shape == (2, a.shape[2], a.shape[3])
        assert m[0].all()
        assert (m[1, 2:, :] == 0).all()  # masked rows beyond 30/16


def test_masked_ce():
    B, T, K = 2, 3, 5
    logits = torch.zeros(B, T, K)
    y = torch.zeros(B, T, dtype=torch.long)
    ym = torch.ones(B, T)
    loss, n = masked_ce(logits, y, ym)
    assert torch.allclose(loss, torch.tensor(5.0).log())  # uniform CE = ln K
    ym2 = ym.clone()
    ym2[:, 2] = 0
    loss2, n2 = masked_ce(logits, y, ym2)
    assert torch.allclose(loss, loss2) and n2 == 4


def test_forward_backward_all_params():
    m = WAPModel(n_tokens=20, encoder="vgg")
    x = torch.rand(2, 1, 32, 48)
    xm = torch.ones(2, 32, 48)
    y = torch.randint(1, 20, (2, 5))
    ym = torch.ones(2, 5)
    out = m(x, xm, y, ym)
    out["loss"].backward()
    missing = [n for n, p in m.named_parameters() if p.grad is None]
    assert not missing, f"params with no grad: {missing}"


def test_tiny_overfit_loss_decreases():
    """BASELINE config 1 integration test: loss must decrease on a fixed
    toy batch (VGG encoder, 32x32, CPU, world 1)."""
    torch.manual_seed(7)
    m = WAPModel(n_tokens=15, encoder="vgg", embed_dim=32, hidden_dim=32,
                 attention_dim=32, coverage_ch=16, coverage_kernel=5)
    x = torch.rand(2, 1, 32, 32)
    xm = torch.ones(2, 32, 32)
    y = torch.tensor([[1, 2, 3, 0], [4, 5, 6, 0]])
    ym = torch.ones(2, 4)
    opt = torch.optim.Adam(m.parameters(), lr=3e-3)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        out = m(x, xm, y, ym)
        out["loss"].backward()
        opt.step()
        losses.append(out["loss"].item())
    assert losses[-1] < losses[0] * 0.5, losses[::10]


def test_attention_normalized_and_masked():
    m = WAPModel(n_tokens=15, encoder="vgg", embed_dim=16, hidden_dim=16,
                 attention_dim=16, coverage_ch=8, coverage_kernel=3)
    m.eval()
    x = torch.rand(2, 1, 32, 64)
    xm = torch.ones(2, 32, 64)
    xm[0, :, 32:] = 0  # right half of image 0 is padding
    a, mask = m.encode(x, xm)
    a_flat, ua = m.parser.attention.precompute(a)
    s = m.parser.init_state(a_flat, mask.flatten(1))
    cov = torch.zeros(2, a.shape[2], a.shape[3])
    emb = torch.zeros(2, 16)
    _, _, _, alpha = m.parser.step(emb, s, a_flat, ua, cov, mask)
    assert torch.allclose(alpha.sum(1), torch.ones(2), atol=1e-5)
    am = alpha.view(2, a.shape[2], a.shape[3])
    assert (am[0, :, 2:] == 0).all()  # no attention on padding


def test_forward_matches_stepwise_rollout():
    """The batched teacher-forced unroll (forward) must produce the same
    logits as composing parser.step() per timestep — the decode path and the
    training path share per-step math."""
    import torch
    from wap_amd.models import WAPModel
    torch.manual_seed(11)
    m = WAPModel(n_tokens=13, encoder="vgg", embed_dim=12, hidden_dim=12,
                 attention_dim=16, coverage_ch=8, coverage_kernel=3).eval()
    b, T = 3, 7
    x = torch.rand(b, 1, 32, 48)
    xm = torch.ones(b, 32, 48)
    xm[1, :, 30:] = 0
    y = torch.randint(1, 13, (b, T))
    with torch.no_grad():
        a, mask = m.encode(x, xm)
        batched = m.parser(a, mask, y)
        # manual per-step rollout
        a_flat, ua = m.parser.attention.precompute(a)
        s = m.parser.init_state(a_flat, mask.flatten(1))
        hp, wp = a.shape[2], a.shape[3]
        cov = torch.zeros(b, hp, wp)
        emb = torch.zeros(b, 12)
        outs = []
        for t in range(T):
            logits, s, _, alpha = m.parser.step(emb, s, a_flat, ua, cov, mask)
            outs.append(logits)
            cov = cov + alpha.view(b, hp, wp)
            emb = m.parser.embedding(y[:, t])
        stepwise = torch.stack(outs, dim=1)
    assert torch.allclose(batched, stepwise, atol=1e-5), \
        (batched - stepwise).abs().max().item()


def test_memorization_drives_wer_to_zero():
    """End-to-end learning-signal proof: overfitting 8
    fixed synthetic samples must drive greedy-decode WER to exactly 0 —
    the only whole-pipeline check of the gradient path possible without
    real CROHME data."""
    import torch
    from wap_amd.models import WAPModel
    from wap_amd.data import Vocab, synth_samples, Sample, Bucket, prepare_batch
    from wap_amd.runtime import greedy_decode
    from wap_amd.eval import score_pairs

    torch.manual_seed(7)
    v = Vocab.default()
    feats, caps = synth_samples(8, v, seed=11, min_h=24, max_h=32, min_w=32,
                                max_w=48, min_len=2, max_len=4)
    samples = [Sample(u, feats[u], v.encode(caps[u])) for u in feats]
    batch = prepare_batch(Bucket(samples))
    m = WAPModel(n_tokens=112, encoder="vgg", embed_dim=32, hidden_dim=32,
                 attention_dim=32, coverage_ch=8, coverage_kernel=3)
    opt = torch.optim.Adadelta(m.parameters(), lr=1.0, rho=0.95, eps=1e-8)
    wer = 1.0
    for it in range(600):
        m.train()
        opt.zero_grad()
        out = m(batch["x"], batch["x_mask"], batch["y"], batch["y_mask"])
        out["loss"].backward()
        torch.nn.utils.clip_grad_norm_(m.parameters(), 100.0)
        opt.step()
        if (it + 1) % 50 == 0:
            m.eval()
            hyps = greedy_decode(m, batch["x"], batch["x_mask"], maxlen=8)
            pairs = [([v.id_to_token[j] for j in s.label],
                      [v.id_to_token[j] for j in hyps[i]])
                     for i, s in enumerate(samples)]
            wer = score_pairs(pairs).wer
            if wer == 0.0:
                break
    assert wer == 0.0, f"model failed to memorize 8 samples (wer={wer})"
