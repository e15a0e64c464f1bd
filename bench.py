#!/usr/bin/env python3
"""Benchmark entry point (driver contract).

    python bench.py --gpus N --steps K --warmup W

Runs the flagship training step — multi-band (4-sub-band PQMF) MelGAN
22.05 kHz bf16 training (BASELINE.json config 2; config 3 at N>1 via
torchrun, one rank per GPU over RCCL) — on synthetic mels with
random-init weights, and prints ONE JSON line on rank 0:

    {"metric": "train_mel_frames_per_sec", "value": <aggregate over all
     GPUs>, ...}

Timing: W untimed warmup steps, then EXACTLY K steps bracketed by
barrier + torch.cuda.synchronize() on both sides; elapsed is the MAX
over ranks.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
from melgan_amd.parallel.dp import init_distributed


def _reexec_torchrun(nproc: int):
    """Replace this process with `torch.distributed.run --nproc-per-node N
    bench.py <same args>` (rendezvous on 127.0.0.1 — container hostnames
    may not resolve)."""
    import socket
    import sys

    with socket.socket() as s:  # pick a free rendezvous port
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    os.execvp(sys.executable, [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
        "--master-port", str(port),
        os.path.abspath(__file__), *sys.argv[1:]])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--config", default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), "configs",
        "mb_22k_4band.yaml"))
    ap.add_argument("--batch-size", type=int, default=None)
    ap.add_argument("--adversarial", action="store_true", default=True,
                    help="include discriminator phase (full GAN step)")
    ap.add_argument("--no-adversarial", dest="adversarial",
                    action="store_false")
    ap.add_argument("--no-rtf", dest="rtf", action="store_false",
                    default=True,
                    help="skip the inference-RTF measurement")
    ap.add_argument("--inference-rtf", dest="rtf", action="store_true",
                    help="(default) also measure inference RTF")
    args = ap.parse_args()
    if args.steps < 1 or args.warmup < 0:
        raise SystemExit("bench.py: --steps must be >= 1 and --warmup >= 0")

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.gpus > 1 and world == 1:
        # invoked directly with --gpus N: re-exec under torchrun so the
        # bench REALLY runs N ranks (one per GPU) instead of silently
        # measuring one
        _reexec_torchrun(args.gpus)
        return  # unreachable (execvp replaces the process)
    if world > 1:
        init_distributed()
        got = torch.distributed.get_world_size()
        if args.gpus > 1 and got != args.gpus:
            raise SystemExit(
                f"bench.py: asked for --gpus {args.gpus} but world_size is "
                f"{got} — launcher/env mismatch, refusing to report a bogus "
                "n_gpus")
        world = got
    have_gpu = torch.cuda.is_available()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0"))) \
        if have_gpu else torch.device("cpu")

    hp = merge_default(HParam(args.config))
    if args.batch_size:
        hp.train.batch_size = args.batch_size
    if not have_gpu:
        hp.backend.amp = "none"
        hp.backend.fused_adam = False
    # bench measures the steady adversarial regime (full GAN step) unless
    # --no-adversarial: warmup-phase-only steps would skip discriminator
    # work inside the timed region
    hp.train.discriminator_train_start_steps = 0 if args.adversarial \
        else 10 ** 9
    hp.log.log_dir = os.path.join("gpurun_out", "bench_logs") \
        if os.path.isdir("gpurun_out") else "/tmp/melgan_bench_logs"

    torch.manual_seed(1234 + rank)
    trainer = Trainer(hp, f"bench_r{rank}", rank=rank, world_size=world,
                      device=device, enable_tb=False)

    B = hp.train.batch_size
    seg = hp.audio.segment_length
    frames = seg // hp.audio.hop_length
    # pre-generate a small rotation of synthetic batches on device
    gen = torch.Generator().manual_seed(4321 + rank)
    batches = []
    for _ in range(4):
        mel = (torch.randn(B, hp.audio.n_mel_channels, frames,
                           generator=gen) * 2.0 - 6.0).to(device)
        audio = (torch.rand(B, 1, seg, generator=gen) * 1.6 - 0.8).to(device)
        batches.append((mel, audio))

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        trainer.train_step(*batches[i % len(batches)])
    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        trainer.train_step(*batches[i % len(batches)])
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if have_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t)

    peak_hbm_gb = round(torch.cuda.max_memory_allocated() / 2 ** 30, 2) \
        if have_gpu else None

    # DP overlap artifact: how much of the gradient
    # all-reduce was already complete when backward ended, per model
    comm = None
    if world > 1 and trainer.dp_g.last_stats:
        comm = {"g": trainer.dp_g.last_stats,
                "d": trainer.dp_d.last_stats or None}

    rtf = None
    if args.rtf and rank == 0:
        rtf = measure_rtf(hp, device)

    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000.0
        frames_per_step = B * frames * world
        value = frames_per_step / (elapsed / args.steps)
        out = {
            "metric": "train_mel_frames_per_sec",
            "value": round(value, 1),
            "unit": "mel-frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"mb-melgan-{hp.model.out_channels}band-"
                         f"{hp.audio.sampling_rate // 1000}k",
                "global_batch": B * world,
                "seq_len": frames,
                "parallelism": f"dp{world}",
                "segment_samples": seg,
                "adversarial": args.adversarial,
                "peak_hbm_gb": peak_hbm_gb,
            },
        }
        if rtf is not None:
            out["config"][
                f"inference_rtf_{hp.audio.sampling_rate // 1000}k"] = rtf
        if comm is not None:
            out["config"]["dp_overlap"] = comm
        print(json.dumps(out))

    # Success path ends HERE with an explicit clean exit: interpreter/HIP
    # teardown after a full GPU training run can segfault (observed rc 139
    # AFTER all output, ROCm 7.2 — see NOTES.md "exit-teardown segfault"),
    # and under torchrun a rank dying at teardown turns a finished bench
    # into a reported failure.  Flush, tear down the process group, and
    # exit without running the hazardous destructor chain.
    import sys

    if world > 1 and torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    sys.stdout.flush()
    sys.stderr.flush()
    os._exit(0)


def measure_rtf(hp, device, n_frames: int = 860, reps: int = 20,
                dtype=torch.bfloat16):
    """Inference RTF: ~10 s of audio per run on the fused MFMA fwd path
    (bf16 weights/activations; PQMF synthesis fp32)."""
    from melgan_amd.models import build_generator
    from melgan_amd.ops.pqmf import PQMF

    if device.type != "cuda":  # CPU smoke runs: keep it to seconds
        n_frames, reps = 128, 2

    g = build_generator(hp).to(device).eval()
    g.remove_weight_norm()
    if device.type == "cuda" and dtype is not None:
        g = g.to(dtype)
    pqmf = PQMF(hp.model.out_channels).to(device) \
        if hp.model.out_channels > 1 else None
    mel = (torch.randn(1, hp.audio.n_mel_channels, n_frames,
                       device=device) * 2.0 - 6.0)
    if device.type == "cuda" and dtype is not None:
        mel = mel.to(dtype)
    sr = hp.audio.sampling_rate
    with torch.no_grad():
        for _ in range(3):
            sub = g(mel)
            audio = pqmf.synthesis(sub.float()) if pqmf else sub
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            sub = g(mel)
            audio = pqmf.synthesis(sub.float()) if pqmf else sub
        if device.type == "cuda":
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
    audio_s = audio.size(-1) / sr
    return round(dt / audio_s, 6)


if __name__ == "__main__":
    main()
