#!/usr/bin/env python3
"""Per-kernel microbenchmark: times the fused conv ops on the driver
configs' hot shapes and prints achieved TFLOP/s (JSON lines)."""

import json
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


SHAPES = [
    # (tag, B, Cin, Cout, K, stride, dil, groups, pad, T)
    ("disc_k5", 32, 1024, 1024, 5, 1, 1, 1, 2, 64),
    ("res256_k3", 32, 256, 256, 3, 1, 1, 1, 3, 512),
    ("res128_k3d3", 32, 128, 128, 3, 1, 3, 1, 9, 2048),
    ("res64_k3d9", 32, 64, 64, 3, 1, 9, 1, 27, 4096),
    ("mel_k7", 32, 80, 512, 7, 1, 1, 1, 3, 64),
    ("disc_k41_L1", 32, 16, 64, 41, 4, 1, 4, 20, 16384),
    ("disc_k41_L3", 32, 256, 1024, 41, 4, 1, 64, 20, 1024),
]


def bench_op(fn, reps=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    import melgan_amd._C as C

    assert torch.cuda.is_available(), "GPU micro-bench: run on an MI355X box (gpurun)"
    for tag, B, Cin, Cout, K, s, d, g, p, T in SHAPES:
        x = torch.randn(B, Cin, T, device="cuda").to(torch.bfloat16)
        w = (torch.randn(Cout, Cin // g, K, device="cuda") * 0.1) \
            .to(torch.bfloat16)
        To = (T + 2 * p - ((K - 1) * d + 1)) // s + 1
        gz = torch.randn(B, Cout, To, device="cuda").to(torch.bfloat16)
        flops = 2.0 * B * To * Cout * (Cin // g) * K

        t_f = bench_op(lambda: C.conv1d_fwd(x, w, None, s, d, p, False, g,
                                            0, 0.0, False))
        t_d = bench_op(lambda: C.conv1d_bwd_data(gz, w, s, d, p, False, g,
                                                 T))
        t_w = bench_op(lambda: C.conv1d_bwd_weight(gz, x, s, d, p, False,
                                                   g, K, False, 0.0))
        print(json.dumps({
            "shape": tag, "useful_gflop": round(flops / 1e9, 2),
            "fwd_us": round(t_f * 1e6, 1),
            "fwd_tflops": round(flops / t_f / 1e12, 1),
            "bwd_data_us": round(t_d * 1e6, 1),
            "bwd_data_tflops": round(flops / t_d / 1e12, 1),
            "bwd_weight_us": round(t_w * 1e6, 1),
            "bwd_weight_tflops": round(flops / t_w / 1e12, 1),
        }), flush=True)


if __name__ == "__main__":
    main()
