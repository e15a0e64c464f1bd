#!/usr/bin/env python3
"""Perf-regression sweep: runs bench.py across the config
matrix / batch sizes and appends one JSON line per point to stdout.

    python benchmarks/sweep.py                 # all configs, default B
    python benchmarks/sweep.py --batches 64 128 256
"""

import argparse
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CONFIGS = [
    "configs/mb_22k_4band.yaml",
    "configs/mb_24k_vctk.yaml",
    "configs/fb_44k_8band.yaml",
]


def run_point(config, batch, steps, warmup, rtf):
    cmd = [sys.executable, "bench.py", "--config", config,
           "--steps", str(steps), "--warmup", str(warmup)]
    if batch:
        cmd += ["--batch-size", str(batch)]
    if rtf:
        cmd += ["--inference-rtf"]
    r = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                       timeout=1800)
    if r.returncode != 0:
        return {"config": config, "batch": batch, "error":
                r.stderr.strip().splitlines()[-1] if r.stderr else "?"}
    return json.loads(r.stdout.strip().splitlines()[-1])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--configs", nargs="*", default=CONFIGS)
    ap.add_argument("--batches", nargs="*", type=int, default=[None])
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--rtf", action="store_true")
    args = ap.parse_args()
    for cfg in args.configs:
        for b in args.batches:
            print(json.dumps(run_point(cfg, b, args.steps, args.warmup,
                                       args.rtf)), flush=True)


if __name__ == "__main__":
    main()
