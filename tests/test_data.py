"""Data layer: vocab, bucketing sampler properties, padding/masks, synth."""

import os
import pickle

import numpy as np
import pytest
import torch

from wap_amd.data import (
    Bucket, BucketLoader, Sample, Vocab, load_dict, load_pairs, make_buckets,
    prepare_batch, shard_buckets, synth_samples, write_synth_dataset,
)
from wap_amd.data.vocab import EOS_ID


def test_vocab_contract(vocab):
    assert len(vocab) == 112
    assert vocab.token_to_id["<eos>"] == 0
    ids = vocab.encode(r"\frac 1 2")
    assert len(ids) == 3 and all(i > 0 for i in ids)
    assert vocab.decode(ids + [EOS_ID, 5]) == r"\frac 1 2"


def test_load_dict_rejects_bad_eos(tmp_path):
    p = tmp_path / "d.txt"
    p.write_text("a 0\n<eos> 1\n")
    with pytest.raises(ValueError):
        load_dict(str(p))


def _mk_samples(n=40, seed=0, min_h=16, max_h=80, min_w=16, max_w=160):
    v = Vocab.default()
    feats, caps = synth_samples(n, v, seed=seed, min_h=min_h, max_h=max_h,
                                min_w=min_w, max_w=max_w, min_len=2, max_len=12)
    return [Sample(u, feats[u], v.encode(caps[u])) for u in feats]


def test_bucketing_properties():
    samples = _mk_samples(60)
    budget = 40000
    buckets, dropped = make_buckets(samples, batch_size=8, batch_imagesize=budget,
                                    maxlen=200, max_imagesize=500000)
    assert len(dropped) == 0
    # every sample appears exactly once
    uids = [s.uid for b in buckets for s in b.samples]
    assert sorted(uids) == sorted(s.uid for s in samples)
    for b in buckets:
        assert len(b) <= 8
        hmax = max(s.image.shape[0] for s in b.samples)
        wmax = max(s.image.shape[1] for s in b.samples)
        if len(b) > 1:  # padded pixel cost respects the budget
            assert hmax * wmax * len(b) <= budget
    # deterministic
    buckets2, _ = make_buckets(samples, batch_size=8, batch_imagesize=budget)
    assert [[s.uid for s in b.samples] for b in buckets] == \
           [[s.uid for s in b.samples] for b in buckets2]


def test_bucketing_drops_oversize():
    samples = _mk_samples(10)
    samples[0] = Sample("big", np.zeros((900, 900), np.uint8), [1, 2])
    samples[1] = Sample("long", samples[1].image, list(range(1, 110)) * 3)
    buckets, dropped = make_buckets(samples, maxlen=200, max_imagesize=500000)
    assert len(dropped) == 2
    uids = {s.uid for b in buckets for s in b.samples}
    assert "big" not in uids and "long" not in uids


def test_shard_balance_and_equal_steps():
    samples = _mk_samples(80)
    buckets, _ = make_buckets(samples, batch_size=4, batch_imagesize=30000)
    world = 4
    shards = [shard_buckets(buckets, r, world) for r in range(world)]
    steps = [len(s) for s in shards]
    assert len(set(steps)) == 1 and steps[0] > 0  # equal step count (collectives)
    loads = [sum(b.pixels for b in s) for s in shards]
    assert max(loads) <= 1.5 * max(min(loads), 1)  # near-balanced pixels
    # every bucket lands on some rank (nothing dropped); at most the
    # remainder (< world) buckets are duplicated as padding
    ids = [id(b) for s in shards for b in s]
    assert set(ids) == {id(b) for b in buckets}
    assert len(ids) - len(set(ids)) < world


def test_prepare_batch_padding_and_masks():
    samples = _mk_samples(6)
    batch = prepare_batch(Bucket(samples))
    x, xm, y, ym = batch["x"], batch["x_mask"], batch["y"], batch["y_mask"]
    assert x.shape[0] == 6 and x.shape[1] == 1
    assert x.min() >= 0 and x.max() <= 1
    for i, s in enumerate(samples):
        h, w = s.image.shape
        assert xm[i, :h, :w].all() and xm[i].sum() == h * w
        assert (x[i, 0] * (1 - xm[i])).abs().sum() == 0  # zero-padded
        t = len(s.label)
        assert y[i, :t].tolist() == s.label
        assert y[i, t] == EOS_ID
        assert ym[i].sum() == t + 1


def test_loader_epoch_shuffle_deterministic():
    samples = _mk_samples(30)
    buckets, _ = make_buckets(samples, batch_size=4, batch_imagesize=30000)
    l1 = BucketLoader(buckets, seed=3)
    l2 = BucketLoader(buckets, seed=3)
    l1.set_epoch(1), l2.set_epoch(1)
    o1 = [b["uids"] for b in l1]
    o2 = [b["uids"] for b in l2]
    assert o1 == o2
    l2.set_epoch(2)
    o3 = [b["uids"] for b in l2]
    assert o1 != o3  # reshuffled across epochs


def test_on_disk_roundtrip(tmp_path, vocab):
    pkl, cap = str(tmp_path / "f.pkl"), str(tmp_path / "c.txt")
    write_synth_dataset(pkl, cap, 12, vocab, seed=5)
    with open(pkl, "rb") as f:
        feats = pickle.load(f)
    assert all(isinstance(v, np.ndarray) and v.dtype == np.uint8 for v in feats.values())
    samples = load_pairs(pkl, cap, vocab)
    assert len(samples) == 12
    assert all(len(s.label) > 0 for s in samples)
    assert all(s.image.max() > 0 for s in samples)  # not zero-filled (DVFS rule)


def test_pad_to_keeps_loss_exact():
    """pad_to rounds shapes up + pads the batch with all-masked rows; with a
    BN-free encoder (VGG) the masked loss must be bit-comparable."""
    import torch
    from wap_amd.models import WAPModel
    torch.manual_seed(31)
    m = WAPModel(n_tokens=15, encoder="vgg", embed_dim=16, hidden_dim=16,
                 attention_dim=16, coverage_ch=8, coverage_kernel=3).eval()
    samples = _mk_samples(3, seed=9, min_h=20, max_h=40, min_w=30, max_w=50)
    for s in samples:  # clamp ids into the toy model's 15-token vocab
        s.label = [1 + (i % 13) for i in s.label]
    from wap_amd.data import Bucket, prepare_batch
    plain = prepare_batch(Bucket(samples))
    padded = prepare_batch(Bucket(samples), pad_to=(32, 32, 8, 6))
    assert padded["x"].shape[0] == 6
    assert padded["x"].shape[2] % 32 == 0 and padded["x"].shape[3] % 32 == 0
    assert padded["y"].shape[1] % 8 == 0
    with torch.no_grad():
        l1 = m(plain["x"], plain["x_mask"], plain["y"], plain["y_mask"])["loss"]
        l2 = m(padded["x"], padded["x_mask"], padded["y"], padded["y_mask"])["loss"]
    assert torch.allclose(l1, l2, atol=1e-5), (l1.item(), l2.item())


def test_shard_more_ranks_than_buckets():
    samples = _mk_samples(4)
    from wap_amd.data import make_buckets, shard_buckets
    buckets, _ = make_buckets(samples, batch_size=2, batch_imagesize=500000)
    world = len(buckets) + 3
    shards = [shard_buckets(buckets, r, world) for r in range(world)]
    steps = {len(s) for s in shards}
    # every rank runs the same (non-zero) step count: short ranks are padded
    # with duplicate buckets, none sits out an epoch (no deadlock)
    assert steps == {1}
    covered = {id(b) for s in shards for b in s}
    assert covered == {id(b) for b in buckets}


def test_vocab_oov_raises(vocab):
    import pytest as _pytest
    with _pytest.raises(KeyError):
        vocab.encode("definitely_not_a_token")


def test_gen_pkl_cli_roundtrip(tmp_path):
    """gen_pkl CLI: image dir (binary PGM) -> {uid: uint8[H,W]} pickle."""
    import subprocess
    import sys
    d = tmp_path / "imgs"
    d.mkdir()
    rng = np.random.default_rng(3)
    imgs = {}
    for uid in ("a_1", "b_2"):
        h, w = int(rng.integers(10, 30)), int(rng.integers(10, 40))
        arr = rng.integers(0, 256, (h, w)).astype(np.uint8)
        imgs[uid] = arr
        header = f"P5\n{w} {h}\n255\n".encode()
        (d / f"{uid}.pgm").write_bytes(header + arr.tobytes())
    out = tmp_path / "feat.pkl"
    r = subprocess.run(
        [sys.executable, "-m", "wap_amd.data.gen_pkl", str(d), str(out)],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    with open(out, "rb") as f:
        feats = pickle.load(f)
    assert set(feats) == set(imgs)
    for uid in imgs:
        assert feats[uid].dtype == np.uint8
        assert (feats[uid] == imgs[uid]).all()
