"""CPU-testable units of the GPU dispatch layer: eligibility predicates
and table builders (pure logic, no kernels)."""

import pytest
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


# ---------------------------------------------------------------------------
# ConvT phase-decomposed GEMM route (fused._aphase_cached et al.): verify
# the A-fold identity ON CPU by emulating the GEMM engine with F.conv1d.
# ---------------------------------------------------------------------------
def _convt_mod(cin, cout, r):
    from melgan_amd.ops.wnconv import WNConvTranspose1d

    torch.manual_seed(0)
    return WNConvTranspose1d(cin, cout, r)


@pytest.mark.parametrize("r", [2, 3, 4, 5, 8])
def test_convt_aphase_fold_matches_composite(r):
    import torch.nn.functional as F

    mod = _convt_mod(12, 6, r)
    w_c = mod.current_weight().detach()
    aph = fused._aphase_cached(mod, w_c)
    bias_ph = fused._biasph_cached(mod)
    cin, cout, K = w_c.shape
    B, T = 2, 17
    x = torch.randn(B, cin, T)
    # emulate conv1d_fwd_gemm: K=3 pad=1 conv, act_before leaky prologue
    xp = F.leaky_relu(x, mod.act_slope) if mod.act_before else x
    Aw = aph[:, :3 * cin].reshape(r * cout, cin, 3)
    yph = F.conv1d(xp, Aw, bias_ph, padding=1)
    y = yph.view(B, r, cout, T).permute(0, 2, 3, 1).reshape(B, cout, T * r)
    ref = mod._composite(x)
    assert (y - ref).abs().max() < 1e-5


@pytest.mark.parametrize("r", [2, 4, 8])
def test_convt_a2_fold_matches_bwd_data(r):
    import torch.nn.functional as F

    mod = _convt_mod(10, 5, r)
    mod.act_before = False
    w_c = mod.current_weight().detach()
    a2 = fused._a2_cached(mod, w_c)
    cin, cout, K = w_c.shape
    B, T = 2, 13
    x = torch.randn(B, cin, T, requires_grad=True)
    y = F.conv_transpose1d(x, w_c, None, stride=r, padding=mod.padding,
                           output_padding=mod.output_padding)
    gy = torch.randn_like(y)
    y.backward(gy)
    # emulate conv1d_fwd_gemm(gy, a2, K=2r, stride=r, pad=p)
    Aw = a2[:, :cout * K].reshape(cin, cout, K)
    gx = F.conv1d(gy, Aw, None, stride=r, padding=mod.padding)
    assert (gx - x.grad).abs().max() < 1e-5


@pytest.mark.parametrize("r", [2, 4, 8])
def test_convt_weightgrad_roleswap_identity(r):
    """Third leg of the convT GEMM route: the weight gradient equals the
    ROLE-SWAPPED conv1d weight-grad — treat gy as the conv input and the
    (pre-activated) x as the grad_output, same stride/pad as bwd-data.
    This is what conv1d_bwd_weight_gemm computes on the split-K engine."""
    import torch.nn.functional as F

    cin, cout, K = 10, 5, 2 * r
    p = r // 2
    torch.manual_seed(0)
    x = torch.randn(2, cin, 13)
    w = torch.randn(cin, cout, K, requires_grad=True)
    y = F.conv_transpose1d(x, w, None, stride=r, padding=p)
    gy = torch.randn_like(y)
    y.backward(gy)

    gw = torch.nn.grad.conv1d_weight(gy, (cin, cout, K), x,
                                     stride=(r,), padding=(p,))
    assert (gw - w.grad).abs().max() < 1e-4, \
        (gw - w.grad).abs().max().item()


def test_convt_gemm_eligibility():
    mod8 = _convt_mod(512, 256, 8)
    mod_pqmf = _convt_mod(4, 1, 4)  # PQMF synthesis shape: stays off GEMM
    bf = torch.zeros(1, 512, 8, dtype=torch.bfloat16)
    f32 = torch.zeros(1, 512, 8)
    assert fused._convt_gemm_ok(bf, mod8)
    assert not fused._convt_gemm_ok(f32, mod8)
    bf_p = torch.zeros(1, 4, 8, dtype=torch.bfloat16)
    assert not fused._convt_gemm_ok(bf_p, mod_pqmf)
