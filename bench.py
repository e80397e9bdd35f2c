#!/usr/bin/env python3
"""Flagship benchmark: WAP-DenseNet training step throughput (+ decode tok/s).

Contract (driver-run):
    python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches via torch.distributed.run with one rank per GPU
over RCCL; this script reads RANK/LOCAL_RANK/WORLD_SIZE from the env.

Measures the BASELINE.json headline: train images/sec on the WAP-DenseNet
model, batch 16 per GPU of synthetic CROHME-shape (128x512) images, bf16
autocast, Adadelta + grad clip (the full training step including optimizer),
weak scaling (per-GPU work fixed).  After the timed region, rank 0 also
measures beam-10 decode tok/s and reports it under "extra".

Rank 0 prints exactly ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from wap_amd import ops
from wap_amd.data.synth import fixed_shape_batch
from wap_amd.models import WAPModel
from wap_amd.parallel import BucketedDDP, init_distributed
from wap_amd.runtime import beam_decode


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--height", type=int, default=128)
    p.add_argument("--width", type=int, default=512)
    p.add_argument("--seq-len", type=int, default=32)
    p.add_argument("--encoder", type=str, default="densenet")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--kernels", type=str, default="auto")
    p.add_argument("--beam", type=int, default=10)
    p.add_argument("--decode-steps", type=int, default=0,
                   help="decode measurement rounds (default 1 on GPU, 0 on CPU)")
    p.add_argument("--no-decode", action="store_true")
    p.add_argument("--channels-last", type=str, default="off",
                   choices=["on", "off"])
    p.add_argument("--graph", type=str, default="auto",
                   choices=["auto", "on", "off"],
                   help="hipGraph-captured train step")
    args = p.parse_args()

    ops.set_backend(args.kernels)
    info = init_distributed("auto")
    rank, world = info["rank"], info["world_size"]
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        dev_idx = info["local_rank"] % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)
    else:
        device = torch.device("cpu")
    torch.manual_seed(1234 + rank)

    use_bf16 = on_gpu and args.dtype == "bf16"

    def autocast():
        return torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16)

    model = WAPModel(n_tokens=112, encoder=args.encoder).to(device)
    if on_gpu and args.channels_last == "on":
        model.use_channels_last()
    ddp = BucketedDDP(model)
    opt = torch.optim.Adadelta(model.parameters(), lr=1.0, rho=0.95,
                               eps=1e-8, weight_decay=1e-4)

    x_np, y_np = fixed_shape_batch(args.batch, args.height, args.width,
                                   args.seq_len, 112, seed=17 + rank)
    x = torch.from_numpy(x_np).to(device)
    x_mask = torch.ones(args.batch, args.height, args.width, device=device)
    y = torch.from_numpy(y_np).to(device)
    y_mask = torch.ones(args.batch, args.seq_len, device=device)

    batch = {"x": x, "x_mask": x_mask, "y": y, "y_mask": y_mask}
    gstep = None
    if on_gpu and args.graph != "off":
        try:
            from wap_amd.runtime.graph_train import GraphedTrainStep
            gstep = GraphedTrainStep(model, ddp, opt, clip=100.0,
                                     autocast_bf16=use_bf16)
            gstep.initialize(batch)
        except Exception as e:  # pragma: no cover - graph capture issues
            if args.graph == "on":
                raise
            import sys
            print(f"graph capture unavailable, eager fallback: {e!r}",
                  file=sys.stderr)
            gstep = None

    def train_step():
        if gstep is not None:
            loss = gstep.step()
            if os.environ.get("WAP_GRAD_TRACE"):
                import sys
                gn = sum(float(b.buffer.float().norm() ** 2)
                         for b in ddp.buckets) ** 0.5
                print(f"grad_norm={gn:.4g}", file=sys.stderr)
            return loss
        ddp.zero_grad()
        with autocast():
            out = model(x, x_mask, y, y_mask)
        out["loss"].backward()
        ddp.finish_gradient_sync()
        ddp.clip_grad_norm_(100.0)
        opt.step()
        return out["loss"]

    def barrier_sync():
        # Sync via a D2H scalar read (hipMemcpyWithStream), NEVER
        # hipStreamSynchronize / hipDeviceSynchronize: on ROCm 7.x both of
        # those corrupt the next hipGraph replay's backward (measured:
        # grads explode exactly one step after every such sync; the
        # memcpy-style sync .item() performs is safe).  Rank alignment
        # uses a dummy all-reduce on the current stream instead of
        # dist.barrier() for the same reason: the barrier's internal wait
        # may device-sync, while a stream collective + the D2H drain is
        # exactly the (proven-safe) pattern the graphed trainer runs every
        # step.  The read drains the current stream, which carries all our
        # work, so timing is exact.
        if world > 1:
            t = torch.ones(1, device=device if on_gpu else "cpu")
            torch.distributed.all_reduce(t)
        if on_gpu and not os.environ.get("WAP_NO_BARRIER"):
            from wap_amd.runtime.graph_train import safe_stream_sync
            safe_stream_sync()

    import sys
    for i in range(args.warmup):
        loss = train_step()
        if rank == 0 and os.environ.get("WAP_LOSS_TRACE"):
            print(f"warmup[{i}] loss={float(loss.item()):.4f}", file=sys.stderr)
    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = train_step()
        if rank == 0 and os.environ.get("WAP_LOSS_TRACE"):
            print(f"step[{i}] loss={float(loss.item()):.4f}", file=sys.stderr)
    # read the loss BEFORE the closing barrier: .item() is itself the safe
    # stream drain, and doing it first keeps the reported final_loss from
    # any rank-alignment side effects
    final_loss = float(loss.item())
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks of the elapsed time (slowest rank defines throughput).
    if world > 1:
        t = torch.tensor([elapsed], device=device if on_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    value = world * args.batch * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    extra = {"final_loss": final_loss}
    decode_rounds = args.decode_steps if args.decode_steps else (1 if on_gpu else 0)
    if rank == 0 and decode_rounds > 0 and not args.no_decode:
        # Fixed-seed fresh model: hypothesis lengths (the tok/s numerator)
        # must not depend on how far the timed training happened to get.
        torch.manual_seed(4242)
        dmodel = WAPModel(n_tokens=112, encoder=args.encoder).to(device)
        dmodel.eval()
        model = dmodel
        db = min(args.batch, 8)
        with autocast():
            beam_decode(model, x[:db], x_mask[:db], beam=args.beam, maxlen=32)
        if on_gpu:
            from wap_amd.runtime.graph_train import safe_stream_sync
            safe_stream_sync()
        tok = 0
        td0 = time.perf_counter()
        for _ in range(decode_rounds):
            with autocast():
                hyps = beam_decode(model, x[:db], x_mask[:db],
                                   beam=args.beam, maxlen=64)
            tok += sum(max(len(h), 1) for h in hyps)
        if on_gpu:
            from wap_amd.runtime.graph_train import safe_stream_sync
            safe_stream_sync()
        td = time.perf_counter() - td0
        extra["decode_tok_s"] = tok / td
        extra["decode_beam"] = args.beam
        extra["decode_batch"] = db
        # contract-length decode (reference maxlen=200):
        # a random-init model rarely emits eos, so this measures the
        # full-length beam loop, not early termination
        with autocast():
            beam_decode(model, x[:db], x_mask[:db], beam=args.beam,
                        maxlen=200)  # warm the maxlen-200 graph shapes
        if on_gpu:
            safe_stream_sync()
        tok2 = 0
        td0 = time.perf_counter()
        for _ in range(decode_rounds):
            with autocast():
                hyps = beam_decode(model, x[:db], x_mask[:db],
                                   beam=args.beam, maxlen=200)
            tok2 += sum(max(len(h), 1) for h in hyps)
        if on_gpu:
            safe_stream_sync()
        extra["decode_tok_s_maxlen200"] = tok2 / (time.perf_counter() - td0)

    if rank == 0:
        print(json.dumps({
            "metric": "train images/sec, WAP-DenseNet CROHME-shape",
            "value": value,
            "unit": "images/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_bf16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"wap-{args.encoder}",
                "global_batch": world * args.batch,
                "seq_len": args.seq_len,
                "image": f"{args.height}x{args.width}",
                "parallelism": f"dp{world}",
                "kernels": ops.get_backend(),
                "graph": gstep is not None,
                "channels_last": on_gpu and args.channels_last == "on",
            },
            "extra": extra,
        }))


if __name__ == "__main__":
    main()
