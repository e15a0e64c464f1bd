#!/usr/bin/env python3
"""Microbenchmark for the k41 tower kernels (xstage tier vs the g16
tier, toggled by MELGAN_AMD_NO_XSTAGE=1): per-direction achieved
TFLOP/s at the four tower layer shapes, B=128, scale 1."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

# (tag, Cin, Cout, groups, T)  -- all K=41, s=4, pad=20, zero pad
TOWERS = [
    ("L1", 16, 64, 4, 16384),
    ("L2", 64, 256, 16, 4096),
    ("L3", 256, 1024, 64, 1024),
    ("L4", 1024, 1024, 256, 256),
]


def bench(fn, reps=20, warmup=4):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    import melgan_amd._C as C

    B, K, s, p = 128, 41, 4, 20
    mode = "g16" if os.environ.get("MELGAN_AMD_NO_XSTAGE") == "1" \
        else "xstage"
    for tag, Cin, Cout, g, T in TOWERS:
        To = (T + 2 * p - K) // s + 1
        x = (torch.randn(B, Cin, T, device="cuda") * 0.3).to(torch.bfloat16)
        w = (torch.randn(Cout, Cin // g, K, device="cuda") * 0.1) \
            .to(torch.bfloat16)
        gz = (torch.randn(B, Cout, To, device="cuda") * 0.3) \
            .to(torch.bfloat16)
        flops = 2.0 * B * To * Cout * (Cin // g) * K

        t_f = bench(lambda: C.conv1d_fwd(x, w, None, s, 1, p, False, g,
                                         1, 0.2, False))
        t_bd = bench(lambda: C.conv1d_bwd_data(gz, w, s, 1, p, False, g, T))
        t_bw = bench(lambda: C.conv1d_bwd_weight(gz, x, s, 1, p, False, g,
                                                 K, False, 0.2))
        print(json.dumps({
            "mode": mode, "shape": tag,
            "fwd_us": round(t_f * 1e6, 1), "fwd_tf": round(flops / t_f / 1e12, 1),
            "bwdd_us": round(t_bd * 1e6, 1), "bwdd_tf": round(flops / t_bd / 1e12, 1),
            "bwdw_us": round(t_bw * 1e6, 1), "bwdw_tf": round(flops / t_bw / 1e12, 1),
        }))


if __name__ == "__main__":
    main()
