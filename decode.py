#!/usr/bin/env python3
"""Decode a test set with a trained WAP checkpoint -> caption file (+WER).

Reference-equivalent entrypoint (the Translate/test path of the TF
reference):

    python decode.py --ckpt checkpoints/best.pt \
        --test-pkl offline-test.pkl --test-captions test_captions.txt \
        --out predictions.txt --beam 10

Output file uses the reference caption format ("uid<TAB>tok tok ...");
when --test-captions is given the WER/ExpRate are printed (same scoring as
compute-wer.py).  --synthetic N decodes N synthetic samples instead.
"""

from __future__ import annotations

import argparse
import time

import torch

from wap_amd import ops
from wap_amd.data import (
    BucketLoader, Sample, Vocab, load_pairs, make_buckets, prepare_batch,
    synth_samples,
)
from wap_amd.eval import score_pairs
from wap_amd.models import WAPModel
from wap_amd.runtime import beam_decode, load_checkpoint


def main() -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--ckpt", type=str, default="")
    p.add_argument("--dictionary", type=str, default="dictionary.txt")
    p.add_argument("--test-pkl", type=str, default="")
    p.add_argument("--test-captions", type=str, default="")
    p.add_argument("--synthetic", type=int, default=0)
    p.add_argument("--out", type=str, default="predictions.txt")
    p.add_argument("--beam", type=int, default=10)
    p.add_argument("--maxlen", type=int, default=200)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--batch-imagesize", type=int, default=500000)
    p.add_argument("--encoder", type=str, default="densenet")
    p.add_argument("--device", type=str, default="auto")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--kernels", type=str, default="auto")
    args = p.parse_args()

    ops.set_backend(args.kernels)
    device = torch.device(
        "cuda" if (args.device == "auto" and torch.cuda.is_available())
        else (args.device if args.device != "auto" else "cpu")
    )
    vocab = Vocab.from_file(args.dictionary)

    if args.synthetic > 0:
        feats, caps = synth_samples(args.synthetic, vocab, seed=7)
        samples = [Sample(u, feats[u], vocab.encode(caps[u])) for u in feats]
    else:
        samples = load_pairs(args.test_pkl, args.test_captions, vocab)
    buckets, dropped = make_buckets(samples, args.batch_size,
                                    args.batch_imagesize)

    model = WAPModel(n_tokens=len(vocab), encoder=args.encoder).to(device)
    if args.ckpt:
        load_checkpoint(args.ckpt, model, restore_rng=False, map_location=str(device))
    model.eval()

    use_bf16 = device.type == "cuda" and args.dtype == "bf16"
    pairs = []
    n_tok = 0
    t0 = time.perf_counter()
    with open(args.out, "w", encoding="utf-8") as f:
        for bucket in buckets:
            batch = prepare_batch(bucket, device)
            with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
                hyps = beam_decode(model, batch["x"], batch["x_mask"],
                                   beam=args.beam, maxlen=args.maxlen)
            for i, hyp in enumerate(hyps):
                text = " ".join(vocab.id_to_token[j] for j in hyp)
                f.write(f"{batch['uids'][i]}\t{text}\n")
                n_tok += len(hyp)
                t = int(batch["y_mask"][i].sum().item()) - 1
                ref_ids = batch["y"][i, :t].tolist()
                pairs.append((
                    [vocab.id_to_token[j] for j in ref_ids],
                    text.split(),
                ))
        # Over-size/over-length samples are not decodable under the image
        # budget, but they must still appear in the output file and count as
        # full deletions in the printed WER (so this score agrees with
        # compute-wer.py run against the complete caption file).
        for s in dropped:
            f.write(f"{s.uid}\t\n")
            pairs.append(([vocab.id_to_token[j] for j in s.label], []))
    if dropped:
        print(f"note: {len(dropped)} samples exceeded maxlen/max_imagesize; "
              "emitted empty hypotheses for them")
    dt = time.perf_counter() - t0
    print(f"decoded {len(pairs)} expressions, {n_tok} tokens "
          f"in {dt:.2f}s ({n_tok / max(dt, 1e-9):.1f} tok/s) -> {args.out}")
    if pairs and (args.test_captions or args.synthetic):
        r = score_pairs(pairs)
        print(f"WER {r.wer:.4f}  ExpRate {r.exprate:.4f}  "
              f"<=1 {r.exprate_le1:.4f}  <=2 {r.exprate_le2:.4f}")


if __name__ == "__main__":
    main()
