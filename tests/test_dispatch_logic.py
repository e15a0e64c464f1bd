"""CPU-testable units of the GPU dispatch layer: eligibility predicates
and table builders (pure logic, no kernels)."""

import torch

from melgan_amd.ops import fused
from melgan_amd.utils.profiling import PhaseTimer


def test_gemm_fwd_eligibility():
    bf = torch.zeros(1, dtype=torch.bfloat16)
    f32 = torch.zeros(1)
    # disc k5 / res stacks: eligible
    assert fused._gemm_fwd_eligible(bf, 1024, 1024, 5, 1, 1)
    assert fused._gemm_fwd_eligible(bf, 256, 256, 3, 1, 1)
    # strided / grouped / small / fp32: not
    assert not fused._gemm_fwd_eligible(bf, 1024, 1024, 5, 4, 1)
    assert not fused._gemm_fwd_eligible(bf, 1024, 1024, 5, 1, 4)
    assert not fused._gemm_fwd_eligible(bf, 64, 64, 3, 1, 1)   # 192 < 384
    assert not fused._gemm_fwd_eligible(bf, 512, 4, 7, 1, 1)   # Cout < 64
    assert not fused._gemm_fwd_eligible(f32, 1024, 1024, 5, 1, 1)


def test_gemm_bwd_eligibility():
    assert fused._gemm_bwd_eligible(torch.bfloat16, 128, 128, 3)
    assert not fused._gemm_bwd_eligible(torch.float32, 128, 128, 3)
    assert not fused._gemm_bwd_eligible(torch.bfloat16, 32, 128, 3)


def test_gemm_env_killswitch(monkeypatch):
    bf = torch.zeros(1, dtype=torch.bfloat16)
    monkeypatch.setenv("MELGAN_AMD_NO_GEMM", "1")
    assert not fused._gemm_fwd_eligible(bf, 1024, 1024, 5, 1, 1)


def test_l1_table_layout():
    f = [torch.randn(4, 8), torch.randn(2, 100000)]
    r = [torch.randn_like(t) for t in f]
    g = [torch.empty_like(t) for t in f]
    tbl, ch = fused._l1_table(f, r, g)
    nt = 2
    assert tbl.numel() == 3 * nt + nt  # f|r|g pointers + numels
    assert tbl[3 * nt + 0] == 32 and tbl[3 * nt + 1] == 200000
    # chunking: 32 fits one chunk; 200000 -> ceil(200000/65536)=4
    assert ch.numel() == 1 + 4
    assert (ch[0] >> 48) == 0 and (ch[1] >> 48) == 1


def test_phase_timer_disabled_cpu():
    tm = PhaseTimer(enabled=True)  # no CUDA here -> disabled
    assert not tm.enabled
    tm.mark("x")
    tm.mark(None)
    assert tm.collect() == {}
