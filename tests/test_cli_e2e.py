"""End-to-end CLI round trip on CPU (reference workflow parity):
preprocess.py (wav -> .mel) -> train.py (few steps, ckpt) ->
inference.py (.mel/.npy -> .wav, RTF report)."""

import json
import os
import subprocess
import sys
import wave

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY_CFG = """
data: {{train_dir: {d}, val_dir: {d}}}
train:
  batch_size: 2
  num_workers: 0
  discriminator_train_start_steps: 0
audio:
  n_mel_channels: 80
  segment_length: 4096
  pad_short: 2000
  hop_length: 256
  filter_length: 1024
  win_length: 1024
  sampling_rate: 22050
model: {{out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2]}}
log: {{summary_interval: 2, validation_interval: 2, save_interval: 3,
      chkpt_dir: {ck}, log_dir: {lg}}}
backend: {{amp: none, fused_adam: false}}
"""


def _write_wav(path, seconds=1.0, sr=22050):
    t = np.arange(int(seconds * sr))
    sig = (0.4 * np.sin(2 * np.pi * 220 * t / sr) +
           0.1 * np.random.RandomState(0).randn(len(t)))
    pcm = np.clip(sig * 32767, -32768, 32767).astype(np.int16)
    with wave.open(path, "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(pcm.tobytes())


def _run(args, **kw):
    return subprocess.run([sys.executable] + args, cwd=REPO,
                          capture_output=True, text=True, timeout=600, **kw)


@pytest.mark.timeout(900)
def test_preprocess_train_infer_roundtrip(tmp_path):
    data = tmp_path / "wavs"
    data.mkdir()
    for i in range(3):
        _write_wav(str(data / f"utt{i}.wav"), seconds=0.5)
    cfg = tmp_path / "cfg.yaml"
    cfg.write_text(TINY_CFG.format(d=str(data), ck=str(tmp_path / "ck"),
                                   lg=str(tmp_path / "lg")))

    # preprocess: wav -> .mel
    r = _run(["preprocess.py", "-c", str(cfg), "-d", str(data)])
    assert r.returncode == 0, r.stderr[-2000:]
    mels = list(data.glob("*.mel"))
    assert len(mels) == 3

    # train 3 steps on the real (tiny) dataset; checkpoint at step 3.
    # validation_interval=2 makes validate() run mid-train on the REAL
    # val set (full-length uncropped clips — regression for the
    # synthesised-length overshoot crash)
    r = _run(["train.py", "-c", str(cfg), "-n", "e2e", "--max-steps", "3"])
    assert r.returncode == 0, r.stderr[-2000:]
    cks = sorted((tmp_path / "ck").glob("e2e_*.pt"))
    assert cks, "no checkpoint written"

    # family resume contract: train.py -p continues from the checkpoint
    r = _run(["train.py", "-c", str(cfg), "-n", "e2e_resume",
              "-p", str(cks[-1]), "--max-steps", "6"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert sorted((tmp_path / "ck").glob("e2e_resume_0006.pt")), \
        "resume did not continue to step 6"

    # inference: mel .npy -> wav with RTF report
    meldir = tmp_path / "mels"
    meldir.mkdir()
    mel = torch.load(mels[0], weights_only=True).numpy()
    np.save(str(meldir / "utt0.npy"), mel)
    out = tmp_path / "wav_out"
    r = _run(["inference.py", "-c", str(cfg), "-p", str(cks[-1]),
              "-i", str(meldir), "-o", str(out)])
    assert r.returncode == 0, r.stderr[-2000:]
    wavs = list(out.glob("*.wav"))
    assert len(wavs) == 1
    assert "TOTAL RTF" in r.stdout
    with wave.open(str(wavs[0]), "rb") as w:
        assert w.getframerate() == 22050
        nframes_full = w.getnframes()
        assert nframes_full > 10000

    # chunked streaming inference from the CLI produces the same-length wav
    out2 = tmp_path / "wav_out_chunked"
    r = _run(["inference.py", "-c", str(cfg), "-p", str(cks[-1]),
              "-i", str(meldir), "-o", str(out2), "--chunk-frames", "8"])
    assert r.returncode == 0, r.stderr[-2000:]
    with wave.open(str(next(out2.glob("*.wav"))), "rb") as w:
        # inference() pads+trims, chunked streams the raw mel — lengths
        # agree up to the pad_frames trim margin
        assert abs(w.getnframes() - nframes_full) <= 2 * 10 * 256

    # .mel input files (preprocess output) work directly too
    meldir2 = tmp_path / "mels_raw"
    meldir2.mkdir()
    (meldir2 / "utt0.mel").write_bytes(mels[0].read_bytes())
    out3 = tmp_path / "wav_out_mel"
    r = _run(["inference.py", "-c", str(cfg), "-p", str(cks[-1]),
              "-i", str(meldir2), "-o", str(out3)])
    assert r.returncode == 0, r.stderr[-2000:]
    with wave.open(str(next(out3.glob("*.wav"))), "rb") as w:
        assert w.getnframes() == nframes_full  # same mel -> same length


@pytest.mark.timeout(300)
def test_bench_cpu_contract(tmp_path):
    """bench.py emits the driver-contract JSON line (CPU, tiny)."""
    cfg = tmp_path / "cfg.yaml"
    cfg.write_text(TINY_CFG.format(d="null", ck=str(tmp_path / "ck"),
                                   lg=str(tmp_path / "lg")))
    r = _run(["bench.py", "--steps", "2", "--warmup", "1",
              "--config", str(cfg), "--batch-size", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    rec = json.loads(line)
    assert rec["metric"] == "train_mel_frames_per_sec"
    assert rec["n_gpus"] == 1 and rec["steps"] == 2
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp1"
    assert rec["config"]["inference_rtf_22k"] > 0  # RTF on by default


@pytest.mark.timeout(600)
def test_bench_gpus2_reexecs_torchrun(tmp_path):
    """`bench.py --gpus 2` really runs 2 ranks (gloo on CPU) and ships
    the DP overlap artifact — the driver's SCALE invocation contract
    (VERDICT r1 missing #1/#3)."""
    cfg = tmp_path / "cfg.yaml"
    cfg.write_text(TINY_CFG.format(d="null", ck=str(tmp_path / "ck"),
                                   lg=str(tmp_path / "lg")))
    # NO --no-rtf: the driver's SCALE invocation runs with RTF on, so the
    # rank-0-only RTF measurement at world>1 must be covered too
    r = _run(["bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
              "--config", str(cfg), "--batch-size", "2"])
    assert r.returncode == 0, r.stderr[-2000:]
    recs = []
    for line in r.stdout.splitlines():  # torchrun interleaves rank output
        try:
            rec = json.loads(line)
        except ValueError:
            continue
        if isinstance(rec, dict) and rec.get("metric"):
            recs.append(rec)
    assert len(recs) == 1, "exactly one JSON line (rank 0)"
    rec = recs[0]
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 4  # 2 per rank x 2 ranks
    ov = rec["config"]["dp_overlap"]
    assert ov["g"]["buckets"] >= 1 and ov["g"]["bytes"] > 0
    assert len(ov["g"]["bucket_ms"]) == ov["g"]["buckets"]
    assert rec["config"]["inference_rtf_22k"] > 0  # rank-0 RTF at world>1


@pytest.mark.timeout(600)
def test_train_cli_world2_gloo(tmp_path):
    """train.py under torchrun with 2 ranks (gloo on CPU): rendezvous on
    127.0.0.1, per-rank sharding, rank-0-only checkpoint/metrics — the
    same path the driver's multi-GPU run takes over RCCL."""
    import socket

    cfg = tmp_path / "cfg.yaml"
    cfg.write_text(TINY_CFG.format(d="null", ck=str(tmp_path / "ck"),
                                   lg=str(tmp_path / "lg")))
    with socket.socket() as s:  # free rendezvous port (xdist-safe)
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    r = _run(["-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", str(port), "train.py", "-c", str(cfg),
              "-n", "dist2", "--synthetic", "--max-steps", "4"])
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    cks = sorted((tmp_path / "ck").glob("dist2_*.pt"))
    assert cks, "rank 0 wrote no checkpoint"
    # exactly ONE metrics stream (rank 0); rank 1 must not write
    metrics = list((tmp_path / "lg").glob("dist2_metrics.jsonl"))
    assert len(metrics) == 1
