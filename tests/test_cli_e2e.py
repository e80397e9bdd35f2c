"""Full reference-UX pipeline on CPU: train.py -> decode.py -> compute-wer.py
(BASELINE config 1: VGG encoder, toy images, world 1)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(900)
def test_train_decode_wer_pipeline(tmp_path):
    env = dict(os.environ, WAP_KERNELS="eager")
    ckpt = str(tmp_path / "ck")
    metrics = str(tmp_path / "m.jsonl")
    r = subprocess.run(
        [sys.executable, "train.py", "--synthetic", "24", "--encoder", "vgg",
         "--epochs", "1", "--batch-size", "4", "--batch-imagesize", "60000",
         "--device", "cpu", "--log-every", "3", "--decode-maxlen", "10",
         "--ckpt-dir", ckpt, "--metrics-jsonl", metrics,
         "--dictionary", "dictionary.txt"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(os.path.join(ckpt, "latest.pt"))
    assert os.path.exists(os.path.join(ckpt, "best.pt"))
    assert os.path.exists(os.path.join(ckpt, "dictionary.txt"))
    recs = [json.loads(l) for l in open(metrics)]
    kinds = {r["kind"] for r in recs}
    assert {"setup", "train", "valid"} <= kinds

    out = str(tmp_path / "pred.txt")
    r = subprocess.run(
        [sys.executable, "decode.py", "--ckpt", os.path.join(ckpt, "best.pt"),
         "--encoder", "vgg", "--synthetic", "6", "--beam", "3",
         "--maxlen", "8", "--device", "cpu", "--kernels", "eager",
         "--out", out, "--batch-imagesize", "60000"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert "WER" in r.stdout
    lines = open(out).read().strip().splitlines()
    assert len(lines) == 6
    assert all("\t" in l or len(l.split()) >= 1 for l in lines)

    # compute-wer.py CLI on the produced caption file vs itself -> WER 0
    r = subprocess.run(
        [sys.executable, "compute-wer.py", out, out],
        cwd=REPO, capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr
    assert "WER 0.0000" in r.stdout and "ExpRate 1.0000" in r.stdout
