#!/usr/bin/env python3
"""Inference entry point (reference-compatible CLI).

    python inference.py -c config.yaml -p checkpoint.pt -i mel_folder -o out

Loads the generator only, removes weight norm, iterates mel files
(.npy / .mel), synthesises audio (PQMF synthesis for multi-band) and
writes 16-bit WAVs.  Reports RTF (generation time / audio duration) —
the BASELINE.json inference metric.
"""

from __future__ import annotations

import argparse
import glob
import os
import time
import wave

import numpy as np
import torch

from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.checkpoint import load_checkpoint, load_flexible
from melgan_amd.models import build_generator
from melgan_amd.ops.pqmf import PQMF

MAX_WAV_VALUE = 32768.0


def write_wav(path: str, audio: np.ndarray, sr: int):
    pcm = np.clip(audio * MAX_WAV_VALUE, -MAX_WAV_VALUE,
                  MAX_WAV_VALUE - 1).astype(np.int16)
    with wave.open(path, "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(pcm.tobytes())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-c", "--config", required=True)
    ap.add_argument("-p", "--checkpoint_path", required=True)
    ap.add_argument("-i", "--input_folder", required=True,
                    help="folder of mel .npy/.mel files")
    ap.add_argument("-o", "--output_folder", default="generated")
    ap.add_argument("-d", "--device", default=None)
    ap.add_argument("--fp32", action="store_true",
                    help="run the generator in fp32 (default bf16 on GPU)")
    ap.add_argument("--chunk-frames", type=int, default=0,
                    help="stream long mels in chunks of this many frames "
                         "(overlap-discard, O(chunk) memory — SURVEY §5.7; "
                         "0 = whole utterance at once)")
    args = ap.parse_args()

    hp = merge_default(HParam(args.config))
    device = torch.device(args.device) if args.device else (
        torch.device("cuda") if torch.cuda.is_available()
        else torch.device("cpu"))

    generator = build_generator(hp).to(device)
    ck = load_checkpoint(args.checkpoint_path, map_location=device)
    state = ck["model_g"] if "model_g" in ck else ck
    load_flexible(generator, state)
    generator.eval()
    generator.remove_weight_norm()
    use_bf16 = device.type == "cuda" and not args.fp32
    if use_bf16:
        generator = generator.to(torch.bfloat16)
    multiband = hp.model.out_channels > 1
    pqmf = PQMF(hp.model.out_channels,
                cutoff_ratio=hp.model.get("pqmf_cutoff")).to(device) \
        if multiband else None

    os.makedirs(args.output_folder, exist_ok=True)
    mel_files = sorted(glob.glob(os.path.join(args.input_folder, "*.npy")) +
                       glob.glob(os.path.join(args.input_folder, "*.mel")))
    if not mel_files:
        raise FileNotFoundError(f"no mel files in {args.input_folder}")

    sr = hp.audio.sampling_rate
    total_gen_s = 0.0
    total_audio_s = 0.0
    with torch.no_grad():
        for path in mel_files:
            if path.endswith(".npy"):
                mel = torch.from_numpy(np.load(path)).float()
            else:
                mel = torch.load(path, map_location="cpu",
                                 weights_only=True).float()
            if mel.dim() == 2:
                mel = mel.unsqueeze(0)
            mel = mel.to(device)
            if use_bf16:
                mel = mel.to(torch.bfloat16)
            if device.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            if args.chunk_frames:
                audio_sub = generator.inference_chunked(
                    mel, chunk_frames=args.chunk_frames).float()
            else:
                audio_sub = generator.inference(mel).float()
            audio = pqmf.synthesis(audio_sub) if multiband else audio_sub
            if device.type == "cuda":
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            audio = audio[0, 0].float().cpu().numpy()
            total_gen_s += dt
            total_audio_s += len(audio) / sr
            out = os.path.join(
                args.output_folder,
                os.path.splitext(os.path.basename(path))[0] + ".wav")
            write_wav(out, audio, sr)
            print(f"{out}: {len(audio)/sr:.2f}s audio in {dt*1000:.1f} ms "
                  f"(RTF {dt/(len(audio)/sr):.4f})")
    print(f"TOTAL RTF: {total_gen_s/total_audio_s:.5f} "
          f"({total_audio_s/total_gen_s:.1f}x realtime) @ {sr} Hz")

    # explicit clean exit on success (HIP teardown segfault after GPU
    # work — NOTES.md "exit-teardown segfault")
    import sys

    sys.stdout.flush()
    sys.stderr.flush()
    os._exit(0)


if __name__ == "__main__":
    main()
