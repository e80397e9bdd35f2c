"""Checkpoint save -> restore -> bit-equal resume."""

import torch

from wap_amd.models import WAPModel
from wap_amd.runtime import load_checkpoint, save_checkpoint


def _mk():
    torch.manual_seed(11)
    m = WAPModel(n_tokens=10, encoder="vgg", embed_dim=8, hidden_dim=8,
                 attention_dim=8, coverage_ch=4, coverage_kernel=3)
    opt = torch.optim.Adadelta(m.parameters(), lr=1.0, weight_decay=1e-4)
    return m, opt


def _batch(seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(2, 1, 32, 32, generator=g)
    xm = torch.ones(2, 32, 32)
    y = torch.randint(1, 10, (2, 4), generator=g)
    ym = torch.ones(2, 4)
    return x, xm, y, ym


def _step(m, opt, seed):
    m.train()
    opt.zero_grad()
    x, xm, y, ym = _batch(seed)
    out = m(x, xm, y, ym)
    out["loss"].backward()
    opt.step()
    return out["loss"].item()


def test_exact_resume(tmp_path):
    ckpt = str(tmp_path / "ck")
    # run A: 2 steps, save, 2 more steps
    m1, o1 = _mk()
    _step(m1, o1, 1), _step(m1, o1, 2)
    save_checkpoint(ckpt, m1, o1, epoch=0, step=2, best_wer=1.0, is_best=True)
    la = [_step(m1, o1, 3), _step(m1, o1, 4)]

    # run B: fresh model, restore, same 2 steps -> identical losses + params
    m2, o2 = _mk()
    _step(m2, o2, 99)  # desync on purpose before restore
    state = load_checkpoint(str(tmp_path / "ck/latest.pt"), m2, o2)
    assert state["step"] == 2
    lb = [_step(m2, o2, 3), _step(m2, o2, 4)]
    assert la == lb
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert n1 == n2 and torch.equal(p1, p2), n1


def test_best_and_latest_files(tmp_path):
    ckpt = str(tmp_path / "ck")
    m, o = _mk()
    save_checkpoint(ckpt, m, o, 0, 1, 0.9, is_best=True,
                    vocab_path="dictionary.txt")
    save_checkpoint(ckpt, m, o, 1, 2, 0.9, is_best=False)
    import os
    assert os.path.exists(os.path.join(ckpt, "latest.pt"))
    assert os.path.exists(os.path.join(ckpt, "best.pt"))
    assert os.path.exists(os.path.join(ckpt, "dictionary.txt"))
    best = torch.load(os.path.join(ckpt, "best.pt"), weights_only=False)
    latest = torch.load(os.path.join(ckpt, "latest.pt"), weights_only=False)
    assert best["step"] == 1 and latest["step"] == 2
