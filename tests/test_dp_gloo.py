"""Distributed DP engine tests without a GPU: the bucketed
all-reduce engine over the gloo backend, world_size=2, on 127.0.0.1."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from melgan_amd.models.generator import Generator


def _make_model(seed=0):
    torch.manual_seed(seed)
    return Generator(mel_channels=16, out_channels=2, base_channels=16,
                     ratios=[4, 2], n_residual_layers=2)


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from melgan_amd.parallel.dp import BucketAllReduceDP

        model = _make_model(seed=123 + rank)  # different init per rank:
        # broadcast at engine construction must equalise them
        engine = BucketAllReduceDP(model, bucket_cap_mb=0.004)  # 4 KB cap
        # on a 15 KB model -> genuinely multiple buckets (the earlier
        # 0.05 MB cap silently left ONE bucket), exercising the
        # launch-per-bucket path

        torch.manual_seed(1000 + rank)  # different DATA per rank
        mel = torch.randn(2, 16, 8)
        y = model(mel)
        y.pow(2).mean().backward()
        engine.finish()

        grads = torch.cat([p.grad.flatten() for p in model.parameters()])
        params = torch.cat([p.data.flatten() for p in model.parameters()])
        results[rank] = (grads, params, dict(engine.last_stats))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_bucketed_allreduce_matches_large_batch():
    world = 2
    port = 29531
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker, args=(world, port, results), nprocs=world, join=True)

    g0, p0, s0 = results[0]
    g1, p1, _ = results[1]
    # both ranks ended with identical (averaged) gradients and identical params
    assert torch.allclose(g0, g1, atol=1e-6)
    assert torch.equal(p0, p1)

    # finish() recorded the overlap artifact bench.py ships at world>1
    assert s0["buckets"] >= 2, "0.05 MB cap should split into many buckets"
    assert s0["bytes"] == g0.numel() * 4  # fp32 grads, every param bucketed
    assert 0 <= s0["pre_completed"] <= s0["buckets"]
    assert s0["finish_wait_ms"] >= 0
    assert len(s0["bucket_ms"]) == s0["buckets"]
    assert all(ms >= 0 for ms in s0["bucket_ms"])

    # single-process reference: same model (rank0 init), batch = both ranks'
    model = _make_model(seed=123)  # engine broadcast rank0 init
    grads_accum = None
    for rank in range(world):
        torch.manual_seed(1000 + rank)
        mel = torch.randn(2, 16, 8)
        model.zero_grad()
        y = model(mel)
        y.pow(2).mean().backward()
        g = torch.cat([p.grad.flatten() for p in model.parameters()])
        grads_accum = g if grads_accum is None else grads_accum + g
    ref = grads_accum / world
    assert torch.allclose(g0, ref, atol=1e-5), \
        (g0 - ref).abs().max().item()


@pytest.mark.timeout(60)
def test_world1_noop():
    from melgan_amd.parallel.dp import BucketAllReduceDP

    model = _make_model()
    engine = BucketAllReduceDP(model)
    assert not engine.enabled
    y = model(torch.randn(1, 16, 8))
    y.mean().backward()
    engine.finish()  # no-op, must not raise
    assert engine.last_stats == {}  # no overlap artifact at world 1


def _rebind_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from melgan_amd.parallel.dp import BucketAllReduceDP

        model = _make_model(seed=123)
        engine = BucketAllReduceDP(model, bucket_cap_mb=0.004)
        # sever the bucket binding the way an external zero_grad would:
        # autograd then allocates FRESH grad storage and the hook must
        # fold it into the bucket view and rebind (the _on_grad fallback)
        for p in model.parameters():
            p.grad = None
        torch.manual_seed(1000 + rank)
        mel = torch.randn(2, 16, 8)
        model(mel).pow(2).mean().backward()
        engine.finish()
        rebound = all(
            p.grad.data_ptr() == b.views[i].data_ptr()
            for b in engine.buckets
            for i, p in enumerate(b.params))
        grads = torch.cat([p.grad.flatten() for p in model.parameters()])
        results[rank] = (grads, rebound)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_grad_rebind_fallback_world2():
    """A backward after p.grad was externally set to None still lands in
    the buckets (fold + rebind) and all-reduces correctly."""
    world = 2
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_rebind_worker, args=(world, 29541, results), nprocs=world,
             join=True)
    g0, rebound0 = results[0]
    g1, rebound1 = results[1]
    assert rebound0 and rebound1, "p.grad not rebound to the bucket views"
    assert torch.allclose(g0, g1, atol=1e-6)

    # reference: average of the two ranks' single-process grads
    model = _make_model(seed=123)
    acc = None
    for rank in range(world):
        torch.manual_seed(1000 + rank)
        mel = torch.randn(2, 16, 8)
        model.zero_grad()
        model(mel).pow(2).mean().backward()
        g = torch.cat([p.grad.flatten() for p in model.parameters()])
        acc = g if acc is None else acc + g
    assert torch.allclose(g0, acc / world, atol=1e-5)


def _gan_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from melgan_amd.config import HParam, merge_default
        from melgan_amd.engine.trainer import Trainer

        raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 2, num_workers: 0, discriminator_train_start_steps: 0}
audio: {segment_length: 4096, hop_length: 256}
model: {out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2]}
log: {chkpt_dir: /tmp/dpg_ck, log_dir: /tmp/dpg_lg}
backend: {amp: none, fused_adam: false}
"""
        hp = merge_default(HParam(raw=raw))
        torch.manual_seed(100 + rank)  # different init; broadcast equalises
        tr = Trainer(hp, f"dpgan{rank}", rank=rank, world_size=world,
                     device=torch.device("cpu"), enable_tb=False)
        torch.manual_seed(500 + rank)  # different data per rank
        mel = torch.randn(2, 80, 16)
        audio = torch.rand(2, 1, 4096) * 1.6 - 0.8
        for _ in range(2):
            # full GAN step incl. discriminator phase: the g-phase
            # adversarial backward must NOT leak into the disc DP
            # buckets (engine raises if it does)
            g_loss, parts, d_loss = tr.train_step(mel, audio)
            assert d_loss is not None and torch.isfinite(d_loss)
        gp = torch.cat([p.data.flatten()
                        for p in tr.generator.parameters()])
        dp = torch.cat([p.data.flatten()
                        for p in tr.discriminator.parameters()])
        results[rank] = (gp, dp)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_full_gan_step_world2():
    """Both ranks stay bit-identical through full adversarial steps
    (regression for the cross-model DP-hook leak: the g-phase backward
    used to fill the discriminator's buckets with adversarial grads,
    making the d-step all-reduce stale — engine now raises on it)."""
    world = 2
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_gan_worker, args=(world, 29537, results), nprocs=world,
             join=True)
    g0, d0 = results[0]
    g1, d1 = results[1]
    assert torch.equal(g0, g1), "generator params diverged across ranks"
    assert torch.equal(d0, d1), "discriminator params diverged across ranks"


def _resume_worker(rank, world, port, ck_path, log_dir, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from melgan_amd.config import HParam, merge_default
        from melgan_amd.engine.trainer import Trainer

        raw = f"""
data: {{train_dir: null, val_dir: null}}
train: {{batch_size: 2, num_workers: 0, discriminator_train_start_steps: 0}}
audio: {{segment_length: 4096, hop_length: 256}}
model: {{out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2]}}
log: {{chkpt_dir: {log_dir}/ck2, log_dir: {log_dir}/lg{rank}}}
backend: {{amp: none, fused_adam: false}}
"""
        hp = merge_default(HParam(raw=raw))
        torch.manual_seed(500 + rank)  # rank-varying init: load must win
        tr = Trainer(hp, f"resume{rank}", rank=rank, world_size=world,
                     device=torch.device("cpu"), enable_tb=False)
        tr.load(ck_path)
        step0 = tr.step
        torch.manual_seed(2000 + rank)  # rank-specific data
        mel = torch.randn(2, 80, 16)
        audio = torch.rand(2, 1, 4096) * 1.6 - 0.8
        tr.train_step(mel, audio)
        params = torch.cat([p.data.flatten()
                            for p in tr.generator.parameters()])
        results[rank] = (step0, params)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_resume_scale_out_world2(tmp_path):
    """Production scale-out: train single-process, checkpoint, resume on
    TWO ranks — both ranks restore the same state (broadcast at engine
    construction must not clobber the loaded weights) and stay in
    lockstep after a DP step on different data."""
    from melgan_amd.config import HParam, merge_default
    from melgan_amd.engine.trainer import Trainer

    raw = f"""
data: {{train_dir: null, val_dir: null}}
train: {{batch_size: 2, num_workers: 0, discriminator_train_start_steps: 0}}
audio: {{segment_length: 4096, hop_length: 256}}
model: {{out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2]}}
log: {{chkpt_dir: {tmp_path}/ck, log_dir: {tmp_path}/lg}}
backend: {{amp: none, fused_adam: false}}
"""
    hp = merge_default(HParam(raw=raw))
    torch.manual_seed(0)
    tr = Trainer(hp, "seed_run", device=torch.device("cpu"),
                 enable_tb=False)
    mel = torch.randn(2, 80, 16)
    audio = torch.rand(2, 1, 4096) * 1.6 - 0.8
    tr.train_step(mel, audio)
    tr.train_step(mel, audio)
    ck = tr.save()
    assert ck

    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_resume_worker, args=(2, 29533, ck, str(tmp_path), results),
             nprocs=2, join=True)
    s0, p0 = results[0]
    s1, p1 = results[1]
    assert s0 == 2 and s1 == 2  # step counter resumed on both ranks
    assert torch.equal(p0, p1)  # ranks in lockstep after the DP step
