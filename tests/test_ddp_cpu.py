"""BucketedDDP on gloo (CPU, world 2): reduced grads == mean of per-rank
grads; params stay identical across ranks after a step."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from wap_amd.models import WAPModel


def _mk_model():
    torch.manual_seed(21)
    return WAPModel(n_tokens=10, encoder="vgg", embed_dim=8, hidden_dim=8,
                    attention_dim=8, coverage_ch=4, coverage_kernel=3)


def _rank_batch(rank):
    g = torch.Generator().manual_seed(100 + rank)
    x = torch.rand(2, 1, 32, 32, generator=g)
    xm = torch.ones(2, 32, 32)
    y = torch.randint(1, 10, (2, 4), generator=g)
    ym = torch.ones(2, 4)
    return x, xm, y, ym


def _single_process_reference():
    """Mean of the two per-rank grads, computed without distributed."""
    grads = {}
    for rank in range(2):
        m = _mk_model()
        x, xm, y, ym = _rank_batch(rank)
        m(x, xm, y, ym)["loss"].backward()
        for n, p in m.named_parameters():
            grads.setdefault(n, []).append(p.grad.clone())
    return {n: torch.stack(g).mean(0) for n, g in grads.items()}


def _worker(rank, world, port, out_dir):
    from wap_amd import ops
    ops.set_backend("eager")
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    from wap_amd.parallel import BucketedDDP

    m = _mk_model()
    ddp = BucketedDDP(m, bucket_bytes=1 << 18)  # small buckets: multiple fire
    assert len(ddp.buckets) > 1
    ddp.zero_grad()
    x, xm, y, ym = _rank_batch(rank)
    out = m(x, xm, y, ym)
    out["loss"].backward()
    ddp.finish_gradient_sync()
    torch.save({n: p.grad.clone() for n, p in m.named_parameters()},
               os.path.join(out_dir, f"grads_{rank}.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bucketed_ddp_grads_match_reference(tmp_path):
    port = 29731
    mp.spawn(_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)
    ref = _single_process_reference()
    g0 = torch.load(tmp_path / "grads_0.pt", weights_only=False)
    g1 = torch.load(tmp_path / "grads_1.pt", weights_only=False)
    for n, rg in ref.items():
        assert torch.allclose(g0[n], rg, atol=1e-6), n
        assert torch.equal(g0[n], g1[n]), n  # identical across ranks


def test_world1_is_noop():
    from wap_amd.parallel import BucketedDDP
    m = _mk_model()
    ddp = BucketedDDP(m)
    ddp.zero_grad()
    x, xm, y, ym = _rank_batch(0)
    m(x, xm, y, ym)["loss"].backward()
    ddp.finish_gradient_sync()  # must not raise without a process group
    assert all(p.grad is not None for p in m.parameters())


def _frozen_worker(rank, world, port, out_dir):
    from wap_amd import ops
    ops.set_backend("eager")
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    from wap_amd.parallel import BucketedDDP

    m = _mk_model()
    # freeze the embedding: its bucket's hook never fires; the
    # finish_gradient_sync fallback must still reduce every bucket
    m.parser.W_o.weight.requires_grad_(False)
    ddp = BucketedDDP(m, bucket_bytes=1 << 16)
    ddp.zero_grad()
    x, xm, y, ym = _rank_batch(rank)
    m(x, xm, y, ym)["loss"].backward()
    ddp.finish_gradient_sync()  # must not deadlock or raise
    torch.save({n: p.grad.clone() for n, p in m.named_parameters()
                if p.requires_grad},
               os.path.join(out_dir, f"fg_{rank}.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_frozen_param_buckets_still_sync(tmp_path):
    port = 29733
    mp.spawn(_frozen_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)
    g0 = torch.load(tmp_path / "fg_0.pt", weights_only=False)
    g1 = torch.load(tmp_path / "fg_1.pt", weights_only=False)
    for n in g0:
        assert torch.equal(g0[n], g1[n]), n


def _trainer_worker(rank, world, port, out_dir):
    """World-4 collective-alignment test of MultiShapeGraphedTrainer's
    step() control flow (graphs_enabled=False: eager stand-ins execute,
    the cache/capture/fallback branching and the collective placement are
    the real code).  Ranks deliberately DIVERGE in caching behavior —
    rank 0 never caches (max_graphs=0, always the eager-fallback branch),
    others cache per shape — while batches cycle through several shapes.
    Params must stay identical across ranks after every step."""
    from wap_amd import ops
    ops.set_backend("eager")
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world
    )
    from wap_amd.parallel import BucketedDDP
    from wap_amd.runtime.graph_train import MultiShapeGraphedTrainer

    m = _mk_model()
    ddp = BucketedDDP(m, bucket_bytes=1 << 18)
    opt = torch.optim.Adadelta(m.parameters(), lr=1.0)
    trainer = MultiShapeGraphedTrainer(
        m, ddp, opt, clip=1.0, autocast_bf16=False,
        max_graphs=0 if rank == 0 else 8, graphs_enabled=False,
    )
    shapes = [(32, 32, 4), (32, 48, 5), (48, 32, 4), (32, 32, 6)]
    g = torch.Generator().manual_seed(500 + rank)
    for step in range(6):
        h, w, t = shapes[step % len(shapes)]
        batch = {
            "x": torch.rand(2, 1, h, w, generator=g),
            "x_mask": torch.ones(2, h, w),
            "y": torch.randint(1, 10, (2, t), generator=g),
            "y_mask": torch.ones(2, t),
        }
        trainer.step(batch)
        # params identical across ranks after every step
        for n, p in m.named_parameters():
            ref = p.data.clone()
            dist.broadcast(ref, src=0)
            assert torch.allclose(p.data, ref, atol=1e-6), (step, n, rank)
    if rank == 0:
        assert trainer.n_eager_fallback > 0      # rank 0 took the fallback
    else:
        assert len(trainer.cache) > 0            # others took the cache
    dist.barrier()
    dist.destroy_process_group()


def test_multishape_trainer_world4_alignment(tmp_path):
    port = 29737
    mp.spawn(_trainer_worker, args=(4, port, str(tmp_path)), nprocs=4,
             join=True)


def test_bucket_clip_matches_torch_cpu():
    """BucketedDDP.clip_grad_norm_ over flat buffers == torch's clip
    (single process, CPU — the GPU twin lives in test_kernels_gpu)."""
    import torch.nn as nn
    from wap_amd.parallel import BucketedDDP
    torch.manual_seed(51)
    m1 = nn.Sequential(nn.Linear(32, 64), nn.Linear(64, 16))
    m2 = nn.Sequential(nn.Linear(32, 64), nn.Linear(64, 16))
    m2.load_state_dict(m1.state_dict())
    x = torch.randn(4, 32)
    ddp = BucketedDDP(m2, bucket_bytes=4096)  # multiple buckets
    m1(x).square().sum().backward()
    m2(x).square().sum().backward()
    n1 = torch.nn.utils.clip_grad_norm_(m1.parameters(), 0.25)
    n2 = ddp.clip_grad_norm_(0.25)
    assert torch.allclose(n1, n2, rtol=1e-6)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.grad, p2.grad, rtol=1e-6, atol=1e-8)
