import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import melgan_amd._C as C

torch.manual_seed(0)
# correctness: asymmetric operands, several shapes incl. edges
for (M, N, Kp) in [(128, 128, 32), (256, 512, 64), (1024, 2048, 5120),
                   (100, 300, 64), (64, 2048, 768)]:
    A = (torch.randn(M, Kp, device="cuda") * 0.5).to(torch.bfloat16)
    Bt = (torch.randn(N, Kp, device="cuda") * 0.5).to(torch.bfloat16)
    Cc = C.gemm_bt(A, Bt)
    ref = (A.float() @ Bt.float().t())
    rel = (Cc.float() - ref).abs().max() / ref.abs().max()
    print(f"gemm {M}x{N}x{Kp}: rel {float(rel):.5f}")

# perf: disc-k5-like + big square
for (M, N, Kp, reps) in [(1024, 2048, 5120, 50), (4096, 4096, 4096, 20)]:
    A = (torch.randn(M, Kp, device="cuda") * 0.5).to(torch.bfloat16)
    Bt = (torch.randn(N, Kp, device="cuda") * 0.5).to(torch.bfloat16)
    for _ in range(5):
        Cc = C.gemm_bt(A, Bt)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        Cc = C.gemm_bt(A, Bt)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    tf = 2.0 * M * N * Kp / dt / 1e12
    print(f"gemm {M}x{N}x{Kp}: {dt*1e6:.1f} us = {tf:.0f} TF")
