#!/usr/bin/env python3
"""Train WAP on CROHME-format (or synthetic) data.

Reference-equivalent entrypoint (`python Train.py` in the TF reference,
SURVEY.md §4.1), extended with DP over RCCL/xGMI:

    # single device (CPU or 1 GPU)
    python train.py --synthetic 512 --encoder vgg --epochs 2

    # 8-GPU data parallel on one MI355X node
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --synthetic 8192

Real data: --train-pkl/--train-captions/--valid-pkl/--valid-captions with
the reference's pickle + caption-file formats.
"""

from __future__ import annotations

import argparse
import math
import os
import time

import torch

from wap_amd import ops
from wap_amd.config import WAPConfig, add_config_args, config_from_args
from wap_amd.data import (
    BucketLoader, Sample, Vocab, load_pairs, make_buckets, synth_samples,
)
from wap_amd.eval import score_pairs
from wap_amd.models import WAPModel
from wap_amd.parallel import BucketedDDP, init_distributed
from wap_amd.runtime import (
    MetricsLogger, greedy_decode, load_checkpoint, save_checkpoint,
)


def pick_device(cfg: WAPConfig, local_rank: int) -> torch.device:
    if cfg.device == "cpu" or (cfg.device == "auto" and not torch.cuda.is_available()):
        return torch.device("cpu")
    idx = local_rank % torch.cuda.device_count()
    torch.cuda.set_device(idx)
    return torch.device("cuda", idx)


def build_optimizer(cfg: WAPConfig, model: torch.nn.Module) -> torch.optim.Optimizer:
    if cfg.optimizer == "adadelta":
        return torch.optim.Adadelta(
            model.parameters(), lr=cfg.lr, rho=cfg.adadelta_rho,
            eps=cfg.adadelta_eps, weight_decay=cfg.weight_decay,
        )
    if cfg.optimizer == "adam":
        return torch.optim.Adam(
            model.parameters(), lr=cfg.lr, weight_decay=cfg.weight_decay
        )
    raise ValueError(f"unknown optimizer {cfg.optimizer!r}")


def load_split(cfg: WAPConfig, vocab: Vocab, split: str, seed: int):
    if cfg.synthetic > 0:
        n = cfg.synthetic if split == "train" else max(cfg.synthetic // 8, 8)
        feats, caps = synth_samples(n, vocab, seed=seed)
        samples = [Sample(u, feats[u], vocab.encode(caps[u])) for u in feats]
    else:
        pkl = cfg.train_pkl if split == "train" else cfg.valid_pkl
        cap = cfg.train_captions if split == "train" else cfg.valid_captions
        if not pkl:
            return None
        samples = load_pairs(pkl, cap, vocab)
    buckets, dropped = make_buckets(
        samples, cfg.batch_size, cfg.batch_imagesize, cfg.maxlen, cfg.max_imagesize
    )
    return buckets, dropped, samples


def validate(model, buckets, vocab, cfg, device, autocast_ctx,
             use_graph: bool = True) -> float:
    """Greedy-decode the validation buckets -> WER (rank 0 only).

    ``use_graph``: capture/replay the decode step per shape.  Callers
    pass False while a graphed TRAINER holds captured train-step graphs —
    capturing decode graphs on top of them produced a GPU memory-access
    fault on ROCm 7.x (r02, reproduced: epoch trains clean, fault appears
    exactly at the first validation capture; decode graphs alone and
    train graphs alone are both stable)."""
    pairs = []
    on_gpu = torch.device(device).type == "cuda"
    # pad validation buckets onto the SAME shape grid training uses,
    # including the batch dim, so the hipGraph-captured decode step is
    # reused across buckets (a handful of captures per run instead of
    # per-shape eager loops)
    pad_to = ((cfg.graph_pad_hw, cfg.graph_pad_hw, cfg.graph_pad_t,
               cfg.batch_size) if on_gpu else None)
    loader = BucketLoader(buckets, device=device, pad_to=pad_to)
    with torch.no_grad():
        for batch in loader:
            with autocast_ctx():
                hyps = greedy_decode(
                    model, batch["x"], batch["x_mask"],
                    maxlen=cfg.decode_maxlen, fused=False,
                    use_graph=on_gpu and use_graph,
                )
            refs = batch["y"]
            masks = batch["y_mask"]
            for i in range(len(batch["uids"])):   # skip batch-padded rows
                hyp = hyps[i]
                t = int(masks[i].sum().item()) - 1  # exclude eos slot
                ref_ids = refs[i, :t].tolist()
                pairs.append((
                    [vocab.id_to_token[j] for j in ref_ids],
                    [vocab.id_to_token[j] for j in hyp],
                ))
    return score_pairs(pairs)


def main() -> None:
    p = argparse.ArgumentParser(description=__doc__)
    add_config_args(p)
    cfg = config_from_args(p.parse_args())

    ops.set_backend(cfg.kernels)
    info = init_distributed(cfg.device)
    rank, world, local_rank = info["rank"], info["world_size"], info["local_rank"]
    device = pick_device(cfg, local_rank)
    torch.manual_seed(cfg.seed + rank)

    vocab = Vocab.from_file(cfg.dictionary)
    train_buckets, dropped, _ = load_split(cfg, vocab, "train", cfg.seed)
    valid = load_split(cfg, vocab, "valid", cfg.seed + 999)

    log = MetricsLogger(cfg.metrics_jsonl, rank=rank)
    log.log("setup", world_size=world, device=str(device),
            n_buckets=len(train_buckets), dropped=len(dropped),
            encoder=cfg.encoder, kernels=ops.get_backend())

    model = WAPModel(
        n_tokens=len(vocab), encoder=cfg.encoder, embed_dim=cfg.embed_dim,
        hidden_dim=cfg.hidden_dim, attention_dim=cfg.attention_dim,
        coverage_ch=cfg.coverage_ch, coverage_kernel=cfg.coverage_kernel,
        encoder_kw=({"growth": cfg.growth} if cfg.encoder == "densenet"
                    else None),
    ).to(device)
    optimizer = build_optimizer(cfg, model)

    start_epoch, gstep, best_wer = 0, 0, math.inf
    if cfg.resume and os.path.exists(cfg.resume):
        state = load_checkpoint(cfg.resume, model, optimizer)
        start_epoch = state["epoch"] + 1
        gstep = state["step"]
        best_wer = state["best_wer"]
        log.log("resume", path=cfg.resume, epoch=start_epoch, best_wer=best_wer)

    ddp = BucketedDDP(model, bucket_bytes=cfg.bucket_mb << 20)
    use_bf16 = device.type == "cuda" and cfg.dtype == "bf16"
    use_graph = device.type == "cuda" and cfg.graph != "off"
    trainer = None
    pad_to = None
    if device.type == "cuda":
        # Collapse bucket shapes onto a grid: (a) lets captured graphs be
        # reused, (b) works around a MIOpen BatchNorm segfault after a
        # handful of distinct input shapes (ROCm 7.x, reproduced with all
        # WAP kernels disabled; see PERF.md).
        pad_to = (cfg.graph_pad_hw, cfg.graph_pad_hw, cfg.graph_pad_t,
                  cfg.batch_size)
    if use_graph:
        from wap_amd.runtime.graph_train import MultiShapeGraphedTrainer
        trainer = MultiShapeGraphedTrainer(
            model, ddp, optimizer, clip=cfg.grad_clip, autocast_bf16=use_bf16
        )
    if trainer is not None:
        # The pinned-memory prefetch thread issues H2D copies while the
        # trainer may be mid-hipGraph-capture on the main thread; a
        # variable-shape training run produced a GPU memory-access fault
        # with the two interleaved (r02).  Graph capture dominates the
        # steady state here anyway, so the loader stays synchronous when
        # graphs are on.
        os.environ.setdefault("WAP_NO_PREFETCH", "1")
    loader = BucketLoader(
        train_buckets, rank=rank, world_size=world, seed=cfg.seed,
        device=device, pad_to=pad_to,
    )

    def autocast_ctx():
        return torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16)

    bad_epochs = 0
    for epoch in range(start_epoch, cfg.epochs):
        model.train()
        loader.set_epoch(epoch)
        t0 = time.perf_counter()
        seen_imgs = 0
        running = 0.0
        for i, batch in enumerate(loader):
            if trainer is not None:
                loss_t = trainer.step(batch)
            else:
                ddp.zero_grad()
                with autocast_ctx():
                    out = model(batch["x"], batch["x_mask"],
                                batch["y"], batch["y_mask"])
                out["loss"].backward()
                ddp.finish_gradient_sync()
                if cfg.grad_clip > 0:
                    ddp.clip_grad_norm_(cfg.grad_clip)
                optimizer.step()
                loss_t = out["loss"]
            gstep += 1
            seen_imgs += batch["x"].shape[0]
            running += loss_t.item()
            if (i + 1) % cfg.log_every == 0:
                dt = time.perf_counter() - t0
                log.log("train", epoch=epoch, step=gstep,
                        loss=running / cfg.log_every,
                        imgs_per_s=seen_imgs * world / max(dt, 1e-9))
                running = 0.0

        stop = torch.zeros(1, device=device if device.type == "cuda" else "cpu")
        if (epoch + 1) % cfg.valid_every_epochs == 0 and valid is not None:
            wer = math.nan
            if rank == 0:
                model.eval()
                res = validate(model, valid[0], vocab, cfg, device,
                               autocast_ctx, use_graph=(trainer is None))
                wer = res.wer
                is_best = wer < best_wer
                best_wer = min(best_wer, wer)
                save_checkpoint(cfg.ckpt_dir, model, optimizer, epoch, gstep,
                                best_wer, is_best, vocab_path=cfg.dictionary)
                log.log("valid", epoch=epoch, wer=wer, exprate=res.exprate,
                        best_wer=best_wer, is_best=int(is_best))
                bad_epochs = 0 if is_best else bad_epochs + 1
                if bad_epochs >= cfg.patience:
                    stop += 1
        if world > 1:
            import torch.distributed as dist
            dist.broadcast(stop, src=0)
        if stop.item() > 0:
            log.log("early_stop", epoch=epoch)
            break

    log.close()


if __name__ == "__main__":
    main()
