// MFMA implicit-GEMM tier for stride-1 conv1d on gfx950 (CDNA4).
//
// Covers the dominant kernels of the multi-band MelGAN step (SURVEY.md
// §2.2 K1): the generator's dilated residual convs (k=3 d∈{1,3,9,27},
// k=1, k=7) and the discriminator's k=5/k=3 convs — profiled at 92% of
// step time on the v1 VALU tier.
//
// Mapping (guide §3/§5): mfma_f32_16x16x32_bf16 per-wave tiles; a
// workgroup of 4 waves (2x2) computes a 64(co) x 64(t) output tile;
// the Cin*K contraction is decomposed as ci-chunks of 32 (MFMA K) with
// the k-taps looped around the MFMA — the input tile is staged in LDS
// ONCE per ci-chunk ([t][ci] transposed image, 40-ci padded rows for
// conflict-free ds_read_b128) and re-read shifted for each tap.
// fp32 accumulate in AGPRs; bias/activation fused in the epilogue;
// reflection padding folded into the staging address math.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace melgan {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 64;   // co tile
constexpr int BN = 64;   // t tile
constexpr int CI_CH = 32;  // MFMA contraction chunk
constexpr int LDW = 40;  // padded ci stride (80 B rows: 16B-aligned,
                         // bank-conflict-free for 16-lane b128 groups)

DEV_INLINE unsigned short load_pad_bf16(const unsigned short* row, int i,
                                        int len, bool reflect) {
  if (reflect) return row[reflect_idx(i, len)];
  if (i < 0 || i >= len) return 0;
  return row[i];
}

DEV_INLINE unsigned short leaky_bf16(unsigned short u, float slope) {
  union { unsigned short s; __bf16 b; } in{u};
  float f = (float)in.b;
  if (f < 0.f) f *= slope;
  union { __bf16 b; unsigned short s; } out{(__bf16)f};
  return out.s;
}

// ---------------------------------------------------------------------------
// forward: y[b][co][t] = act(bias + sum_{ci,k} w[co][ci][k] *
//                            pre(x[b][ci][t + k*d - p]))
// grid: (ceil(To/64), ceil(Cout/64), B); block 256 (4 waves, 2x2)
// dynamic LDS: lds_x[span][LDW] ++ lds_w[K][64][LDW]   (ushort)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
conv1d_fwd_mfma_kernel(const unsigned short* __restrict__ x,
                       const unsigned short* __restrict__ w,
                       const float* __restrict__ bias,
                       unsigned short* __restrict__ y, int B, int Cin,
                       int Tin, int Cout, int To, int K, int dil, int pad,
                       bool reflect, int act, float slope, bool act_before) {
  extern __shared__ __attribute__((aligned(16))) unsigned short smem[];
  const int span = BN + (K - 1) * dil;
  unsigned short* lds_x = smem;              // [span][LDW]
  unsigned short* lds_w = smem + span * LDW;  // [K][BM][LDW]

  const int t0 = blockIdx.x * BN;
  const int co0 = blockIdx.y * BM;
  const int b = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;  // wave grid 2x2
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2][2] = {};
  const int gt0 = t0 - pad;

  for (int ci0 = 0; ci0 < Cin; ci0 += CI_CH) {
    // ---- stage x chunk: [t][ci] transposed image ----
    for (int idx = tid; idx < span * CI_CH; idx += 256) {
      const int ci = idx / span;       // consecutive tid -> consecutive t
      const int t = idx - ci * span;   // (coalesced global reads)
      unsigned short v = 0;
      if (ci0 + ci < Cin) {
        const unsigned short* xr = x + ((int64_t)b * Cin + ci0 + ci) * Tin;
        v = load_pad_bf16(xr, gt0 + t, Tin, reflect);
        if (act_before) v = leaky_bf16(v, slope);
      }
      lds_x[t * LDW + ci] = v;
    }
    // ---- stage w chunk: [k][co][ci] ----
    for (int idx = tid; idx < K * BM * CI_CH; idx += 256) {
      const int k = idx / (BM * CI_CH);
      const int r = idx - k * BM * CI_CH;
      const int co = r / CI_CH;
      const int ci = r - co * CI_CH;
      unsigned short v = 0;
      if (co0 + co < Cout && ci0 + ci < Cin)
        v = w[((int64_t)(co0 + co) * Cin + ci0 + ci) * K + k];
      lds_w[(k * BM + co) * LDW + ci] = v;
    }
    __syncthreads();

    for (int k = 0; k < K; ++k) {
      const unsigned short* wk = lds_w + k * BM * LDW;
      const int tb = wc * 32 + row + k * dil;  // row==col for B operand
      bf16x8 a0 = *(const bf16x8*)(wk + (wr * 32 + row) * LDW + kk0);
      bf16x8 a1 = *(const bf16x8*)(wk + (wr * 32 + 16 + row) * LDW + kk0);
      bf16x8 b0 = *(const bf16x8*)(lds_x + tb * LDW + kk0);
      bf16x8 b1 = *(const bf16x8*)(lds_x + (tb + 16) * LDW + kk0);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: bias + act, bf16 store ----
  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int m = 0; m < 2; ++m)
    for (int n = 0; n < 2; ++n) {
      const int t = t0 + wc * 32 + n * 16 + col;
      if (t >= To) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + wr * 32 + m * 16 + row4 + r;
        if (co >= Cout) continue;
        float o = acc[m][n][r] + (bias ? bias[co] : 0.f);
        if (!act_before) o = apply_act(o, act, slope);
        union { __bf16 b; unsigned short s; } cv{(__bf16)o};
        y[((int64_t)b * Cout + co) * To + t] = cv.s;
      }
    }
}

// ---------------------------------------------------------------------------
// backward-data (stride 1): gxpad[b][ci][q] = sum_{co,k} w[co][ci][k] *
//                                             gz[b][co][q - k*d]
// grid: (ceil(Tpad/64), ceil(Cin/64), B); LDS: gzT[span][LDW] ++ wT[K][64][LDW]
// The reflect/zero un-padding fold runs as a separate elementwise kernel.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
conv1d_bwd_data_mfma_kernel(const unsigned short* __restrict__ gz,
                            const unsigned short* __restrict__ w,
                            unsigned short* __restrict__ gxpad, int B,
                            int Cin, int Tpad, int Cout, int To, int K,
                            int dil) {
  extern __shared__ __attribute__((aligned(16))) unsigned short smem[];
  const int span = BN + (K - 1) * dil;
  unsigned short* lds_gz = smem;              // [t][co]
  unsigned short* lds_w = smem + span * LDW;  // [k][ci][co]

  const int q0 = blockIdx.x * BN;
  const int ci0 = blockIdx.y * BM;
  const int b = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2][2] = {};
  const int gt0 = q0 - (K - 1) * dil;  // gz staging origin

  for (int co0 = 0; co0 < Cout; co0 += CI_CH) {
    for (int idx = tid; idx < span * CI_CH; idx += 256) {
      const int co = idx / span;
      const int t = idx - co * span;
      unsigned short v = 0;
      const int gt = gt0 + t;
      if (co0 + co < Cout && gt >= 0 && gt < To)
        v = gz[((int64_t)b * Cout + co0 + co) * To + gt];
      lds_gz[t * LDW + co] = v;
    }
    for (int idx = tid; idx < K * BM * CI_CH; idx += 256) {
      const int k = idx / (BM * CI_CH);
      const int r = idx - k * BM * CI_CH;
      const int ci = r / CI_CH;
      const int co = r - ci * CI_CH;
      unsigned short v = 0;
      if (ci0 + ci < Cin && co0 + co < Cout)
        v = w[((int64_t)(co0 + co) * Cin + ci0 + ci) * K + k];
      lds_w[(k * BM + ci) * LDW + co] = v;
    }
    __syncthreads();

    for (int k = 0; k < K; ++k) {
      const unsigned short* wk = lds_w + k * BM * LDW;
      // q - k*d  ->  staged t = q_local + (K-1-k)*d
      const int tb = wc * 32 + row + (K - 1 - k) * dil;
      bf16x8 a0 = *(const bf16x8*)(wk + (wr * 32 + row) * LDW + kk0);
      bf16x8 a1 = *(const bf16x8*)(wk + (wr * 32 + 16 + row) * LDW + kk0);
      bf16x8 b0 = *(const bf16x8*)(lds_gz + tb * LDW + kk0);
      bf16x8 b1 = *(const bf16x8*)(lds_gz + (tb + 16) * LDW + kk0);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int m = 0; m < 2; ++m)
    for (int n = 0; n < 2; ++n) {
      const int q = q0 + wc * 32 + n * 16 + col;
      if (q >= Tpad) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int ci = ci0 + wr * 32 + m * 16 + row4 + r;
        if (ci >= Cin) continue;
        union { __bf16 bb; unsigned short s; } cv{(__bf16)acc[m][n][r]};
        gxpad[((int64_t)b * Cin + ci) * Tpad + q] = cv.s;
      }
    }
}

// un-pad fold: gx[j] = gxpad[p+j] (+ mirrored edge terms for reflect)
// src_len >= Tin + 2*pad (the zero-stuffed bwd-data path over-computes)
template <typename T>
__global__ void unpad_fold_kernel(const T* __restrict__ gxpad,
                                  T* __restrict__ gx, int BC, int Tin,
                                  int pad, bool reflect, int src_len) {
  const int64_t n = (int64_t)BC * Tin;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  const int Tpad = Tin + 2 * pad;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n) return;
  // carried (bc, j) split: one divide per thread, not per element
  int64_t bc = idx / Tin;
  int j = (int)(idx - bc * Tin);
  for (; idx < n; idx += step) {
    const T* src = gxpad + bc * src_len;
    float v = static_cast<float>(src[pad + j]);
    if (reflect && pad > 0) {
      if (j > 0 && j <= pad) v += static_cast<float>(src[pad - j]);
      const int jm = 2 * Tin - 2 - j;
      if (jm >= Tin && (pad + jm) < Tpad)
        v += static_cast<float>(src[pad + jm]);
    }
    gx[idx] = static_cast<T>(v);
    j += (int)(step % Tin);
    bc += step / Tin;
    if (j >= Tin) {
      j -= Tin;
      ++bc;
    }
  }
}

// ---------------------------------------------------------------------------
// generic grouped forward (any stride / dilation / groups / K):
//   y[b][co][t] = act(bias + sum_{cig,k} w[co][cig][k] *
//                     pre(x[b][g*Cpg+cig][t*s + k*d - p]))
// The contraction runs over FUSED (cig,k) columns in chunks of 32; the
// shifted input rows are expanded into LDS per chunk.  Covers the
// discriminator's grouped strided k=41 towers, k=15
// (Cin=1), and — via zero-stuffing + weight flip on the host — strided
// conv backward-data and conv_transpose1d forward (K2).
// grid: (ceil(To/64), G * ceil(Copg/64), B)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
conv1d_fwd_mfma_g_kernel(const unsigned short* __restrict__ x,
                         const unsigned short* __restrict__ w,
                         const float* __restrict__ bias,
                         unsigned short* __restrict__ y, int B, int Cin,
                         int Tin, int Cout, int To, int K, int stride,
                         int dil, int pad, bool reflect, int groups, int act,
                         float slope, bool act_before) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[2 * 64 * LDW];
  unsigned short* lds_w = smem;              // [co][kk]
  unsigned short* lds_xs = smem + 64 * LDW;  // [col=t][kk]

  const int Copg = Cout / groups;
  const int Cpg = Cin / groups;
  const int mtiles = (Copg + BM - 1) / BM;
  const int g = blockIdx.y / mtiles;
  const int mt = blockIdx.y - g * mtiles;
  const int co_base = g * Copg + mt * BM;
  const int co_end = (g + 1) * Copg;
  const int t0 = blockIdx.x * BN;
  const int b = blockIdx.z;
  const int Npg = Cpg * K;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2][2] = {};

  for (int nn0 = 0; nn0 < Npg; nn0 += CI_CH) {
    for (int idx = tid; idx < BM * CI_CH; idx += 256) {
      const int co = idx / CI_CH;
      const int kk = idx - co * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (co_base + co < co_end && nn < Npg)
        v = w[((int64_t)(co_base + co) * Cpg + nn / K) * K + nn % K];
      lds_w[co * LDW + kk] = v;
    }
    for (int idx = tid; idx < BN * CI_CH; idx += 256) {
      const int col = idx / CI_CH;
      const int kk = idx - col * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (nn < Npg && t0 + col < To) {
        const int cig = nn / K;
        const int k = nn - cig * K;
        const unsigned short* xr =
            x + ((int64_t)b * Cin + g * Cpg + cig) * Tin;
        v = load_pad_bf16(xr, (t0 + col) * stride + k * dil - pad, Tin,
                          reflect);
        if (act_before) v = leaky_bf16(v, slope);
      }
      lds_xs[col * LDW + kk] = v;
    }
    __syncthreads();

    {
      bf16x8 a0 = *(const bf16x8*)(lds_w + (wr * 32 + row) * LDW + kk0);
      bf16x8 a1 = *(const bf16x8*)(lds_w + (wr * 32 + 16 + row) * LDW + kk0);
      bf16x8 b0 = *(const bf16x8*)(lds_xs + (wc * 32 + row) * LDW + kk0);
      bf16x8 b1 =
          *(const bf16x8*)(lds_xs + (wc * 32 + 16 + row) * LDW + kk0);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int m = 0; m < 2; ++m)
    for (int n = 0; n < 2; ++n) {
      const int t = t0 + wc * 32 + n * 16 + col;
      if (t >= To) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co_base + wr * 32 + m * 16 + row4 + r;
        if (co >= co_end || co >= Cout) continue;
        float o = acc[m][n][r] + (bias ? bias[co] : 0.f);
        if (!act_before) o = apply_act(o, act, slope);
        union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
        y[((int64_t)b * Cout + co) * To + t] = cv.s;
      }
    }
}

// 16-row variant for small groups (disc towers: Copg in {4,16}):
// tile 16(co) x 128(t), 4 waves each own 32 t-columns. Cuts the M-tile
// masking waste of the 64-row kernel by 4-16x on grouped k=41 convs.
// grid: (ceil(To/128), G * ceil(Copg/16), B)
__global__ void __launch_bounds__(256)
conv1d_fwd_mfma_g16_kernel(const unsigned short* __restrict__ x,
                           const unsigned short* __restrict__ w,
                           const float* __restrict__ bias,
                           unsigned short* __restrict__ y, int B, int Cin,
                           int Tin, int Cout, int To, int K, int stride,
                           int dil, int pad, bool reflect, int groups,
                           int act, float slope, bool act_before) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[
      16 * LDW + 128 * LDW];
  unsigned short* lds_w = smem;               // [16co][kk]
  unsigned short* lds_xs = smem + 16 * LDW;   // [col=t(128)][kk]

  const int Copg = Cout / groups;
  const int Cpg = Cin / groups;
  const int mtiles = (Copg + 15) / 16;
  const int g = blockIdx.y / mtiles;
  const int mt = blockIdx.y - g * mtiles;
  const int co_base = g * Copg + mt * 16;
  const int co_end = (g + 1) * Copg;
  const int t0 = blockIdx.x * 128;
  const int b = blockIdx.z;
  const int Npg = Cpg * K;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;       // wave owns cols [wid*32, wid*32+32)
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2] = {};

  for (int nn0 = 0; nn0 < Npg; nn0 += CI_CH) {
    for (int idx = tid; idx < 16 * CI_CH; idx += 256) {
      const int co = idx / CI_CH;
      const int kk = idx - co * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (co_base + co < co_end && nn < Npg)
        v = w[((int64_t)(co_base + co) * Cpg + nn / K) * K + nn % K];
      lds_w[co * LDW + kk] = v;
    }
    for (int idx = tid; idx < 128 * CI_CH; idx += 256) {
      const int col = idx / CI_CH;
      const int kk = idx - col * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (nn < Npg && t0 + col < To) {
        const int cig = nn / K;
        const int k = nn - cig * K;
        const unsigned short* xr =
            x + ((int64_t)b * Cin + g * Cpg + cig) * Tin;
        v = load_pad_bf16(xr, (t0 + col) * stride + k * dil - pad, Tin,
                          reflect);
        if (act_before) v = leaky_bf16(v, slope);
      }
      lds_xs[col * LDW + kk] = v;
    }
    __syncthreads();
    {
      bf16x8 a0 = *(const bf16x8*)(lds_w + row * LDW + kk0);
      bf16x8 b0 = *(const bf16x8*)(lds_xs + (wid * 32 + row) * LDW + kk0);
      bf16x8 b1 =
          *(const bf16x8*)(lds_xs + (wid * 32 + 16 + row) * LDW + kk0);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int n = 0; n < 2; ++n) {
    const int t = t0 + wid * 32 + n * 16 + col;
    if (t >= To) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int co = co_base + row4 + r;
      if (co >= co_end || co >= Cout) continue;
      float o = acc[n][r] + (bias ? bias[co] : 0.f);
      if (!act_before) o = apply_act(o, act, slope);
      union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
      y[((int64_t)b * Cout + co) * To + t] = cv.s;
    }
  }
}

// ---------------------------------------------------------------------------
// phase-decomposed strided backward-data (zero-pad geometry only).
// KEY IDENTITY: for output j = u*s + phi - p the contributing gz index
// t = (j + p - k)/s with k = phi + i*s is  t = u - i  — INDEPENDENT of
// the phase phi.  So the M rows of one MFMA tile are (cig, phi) pairs
// sharing one B operand: for the k=41 discriminator towers
// (Cpg=4, s=4) the 16-row tile is EXACTLY filled — zero masked waste.
//   gx[b][g*Cpg+cig][u*s+phi-p] = sum_{co,i} w[g*Copg+co][cig][phi+i*s]
//                                 * gz[b][g*Copg+co][u-i]
// grid: (ceil(Umax/128), G * ceil(Cpg*s/16), B)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
convbwd_phased16_kernel(const unsigned short* __restrict__ gz,
                        const unsigned short* __restrict__ w,
                        unsigned short* __restrict__ gx, int B, int Cin,
                        int Tin, int Cout, int To, int K, int stride,
                        int pad, int groups, int utiles) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[
      16 * LDW + 128 * LDW];
  unsigned short* lds_w = smem;              // [16 (cig,phi)][kk]
  unsigned short* lds_gs = smem + 16 * LDW;  // [col=u(128)][kk]

  const int Copg = Cout / groups;
  const int Cpg = Cin / groups;
  const int rows_pg = Cpg * stride;          // (cig, phi) rows per group
  const int mtiles = (rows_pg + 15) / 16;
  const int g = blockIdx.y / mtiles;
  const int mt = blockIdx.y - g * mtiles;
  const int row_base = mt * 16;
  const int u0 = blockIdx.x * 128;
  const int b = blockIdx.z;
  const int nkmax = (K + stride - 1) / stride;
  const int Npg = Copg * nkmax;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2] = {};

  for (int nn0 = 0; nn0 < Npg; nn0 += CI_CH) {
    for (int idx = tid; idx < 16 * CI_CH; idx += 256) {
      const int r = idx / CI_CH;
      const int kk = idx - r * CI_CH;
      const int nn = nn0 + kk;
      const int rg = row_base + r;           // (cig, phi): phi fastest
      unsigned short v = 0;
      if (nn < Npg && rg < rows_pg) {
        const int cig = rg / stride;
        const int phi = rg - cig * stride;
        const int co = nn / nkmax;
        const int i = nn - co * nkmax;
        const int k = phi + i * stride;
        if (k < K)
          v = w[((int64_t)(g * Copg + co) * Cpg + cig) * K + k];
      }
      lds_w[r * LDW + kk] = v;
    }
    for (int idx = tid; idx < 128 * CI_CH; idx += 256) {
      const int kk = idx / 128;        // consecutive tid -> consecutive u
      const int col = idx - kk * 128;  // (coalesced gz reads per row)
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (nn < Npg) {
        const int co = nn / nkmax;
        const int i = nn - co * nkmax;
        const int t = u0 + col - i;
        if (t >= 0 && t < To)
          v = gz[((int64_t)b * Cout + g * Copg + co) * To + t];
      }
      lds_gs[col * LDW + kk] = v;
    }
    __syncthreads();
    {
      bf16x8 a0 = *(const bf16x8*)(lds_w + row * LDW + kk0);
      bf16x8 b0 = *(const bf16x8*)(lds_gs + (wid * 32 + row) * LDW + kk0);
      bf16x8 b1 =
          *(const bf16x8*)(lds_gs + (wid * 32 + 16 + row) * LDW + kk0);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int n = 0; n < 2; ++n) {
    const int u = u0 + wid * 32 + n * 16 + col;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int rg = row_base + row4 + r;
      if (rg >= rows_pg) continue;
      const int cig = rg / stride;
      const int phi = rg - cig * stride;
      const int j = u * stride + phi - pad;
      if (j < 0 || j >= Tin) continue;
      const int ci = g * Cpg + cig;
      union { __bf16 bb; unsigned short s; } cv{(__bf16)acc[n][r]};
      gx[((int64_t)b * Cin + ci) * Tin + j] = cv.s;
    }
  }
}

// ---------------------------------------------------------------------------
// deep-contraction / small-T forward (disc k=5 etc.):
// batch folded into the column dimension (N = B*To), 256-column tiles,
// fused (ci,k) contraction — 4x more output per staging round than the
// 64-column kernels, which are staging/barrier-bound at these shapes.
// stride 1, groups 1.  grid: (ceil(B*To/256), ceil(Cout/64), 1)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
conv1d_fwd_mfma_deep_kernel(const unsigned short* __restrict__ x,
                            const unsigned short* __restrict__ w,
                            const float* __restrict__ bias,
                            unsigned short* __restrict__ y,
                            float* __restrict__ yws, int B, int Cin,
                            int Tin, int Cout, int To, int K, int dil,
                            int pad, bool reflect, int act, float slope,
                            bool act_before, int l2cin, int l2to) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[
      (64 + 256) * LDW];
  unsigned short* lds_w = smem;               // [64co][kk]
  unsigned short* lds_xs = smem + 64 * LDW;   // [col][kk]

  const int cg0 = blockIdx.x * 256;           // global (b,t) column origin
  const int co0 = blockIdx.y * BM;
  const int NK = Cin * K;
  const int64_t NCOL = (int64_t)B * To;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;      // wc: 128-col half
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4 acc[2][8] = {};

  // split-K over gridDim.z: this block covers chunk indices
  // [z*per, min((z+1)*per, nchunks))
  const int nchunks = (NK + CI_CH - 1) / CI_CH;
  const int per = (nchunks + gridDim.z - 1) / gridDim.z;
  const int c_lo = blockIdx.z * per;
  const int c_hi = min(c_lo + per, nchunks);

  for (int cix = c_lo; cix < c_hi; ++cix) {
    const int nn0 = cix * CI_CH;
    for (int idx = tid; idx < BM * CI_CH; idx += 256) {
      const int co = idx / CI_CH;
      const int kk = idx - co * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      if (co0 + co < Cout && nn < NK) {
        int ci, k;
        if (l2cin >= 0) { ci = nn & (Cin - 1); k = nn >> l2cin; }
        else { ci = nn % Cin; k = nn / Cin; }
        v = w[((int64_t)(co0 + co) * Cin + ci) * K + k];
      }
      lds_w[co * LDW + kk] = v;
    }
    for (int idx = tid; idx < 256 * CI_CH; idx += 256) {
      const int col = idx / CI_CH;
      const int kk = idx - col * CI_CH;
      const int nn = nn0 + kk;
      unsigned short v = 0;
      const int64_t cg = cg0 + col;
      if (nn < NK && cg < NCOL) {
        int b, t;
        if (l2to >= 0) { b = (int)(cg >> l2to); t = (int)cg & (To - 1); }
        else { b = (int)(cg / To); t = (int)(cg - (int64_t)b * To); }
        int ci, k;
        if (l2cin >= 0) { ci = nn & (Cin - 1); k = nn >> l2cin; }
        else { ci = nn % Cin; k = nn / Cin; }
        const unsigned short* xr = x + ((int64_t)b * Cin + ci) * Tin;
        v = load_pad_bf16(xr, t + k * dil - pad, Tin, reflect);
        if (act_before) v = leaky_bf16(v, slope);
      }
      lds_xs[col * LDW + kk] = v;
    }
    __syncthreads();

    {
      bf16x8 a0 = *(const bf16x8*)(lds_w + (wr * 32 + row) * LDW + kk0);
      bf16x8 a1 = *(const bf16x8*)(lds_w + (wr * 32 + 16 + row) * LDW + kk0);
#pragma unroll
      for (int n = 0; n < 8; ++n) {
        bf16x8 bn = *(const bf16x8*)(lds_xs +
                                     (wc * 128 + n * 16 + row) * LDW + kk0);
        acc[0][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bn, acc[0][n], 0, 0, 0);
        acc[1][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bn, acc[1][n], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int m = 0; m < 2; ++m)
    for (int n = 0; n < 8; ++n) {
      const int64_t cg = cg0 + wc * 128 + n * 16 + col;
      if (cg >= NCOL) continue;
      int b, t;
      if (l2to >= 0) { b = (int)(cg >> l2to); t = (int)cg & (To - 1); }
      else { b = (int)(cg / To); t = (int)(cg - (int64_t)b * To); }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + wr * 32 + m * 16 + row4 + r;
        if (co >= Cout) continue;
        const int64_t oi = ((int64_t)b * Cout + co) * To + t;
        if (gridDim.z == 1) {
          float o = acc[m][n][r] + (bias ? bias[co] : 0.f);
          if (!act_before) o = apply_act(o, act, slope);
          union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
          y[oi] = cv.s;
        } else {
          atomicAdd(&yws[oi], acc[m][n][r]);
        }
      }
    }
}

// finalize after split-K: y = act(ws + bias)
__global__ void deep_finalize_kernel(const float* __restrict__ yws,
                                     const float* __restrict__ bias,
                                     unsigned short* __restrict__ y, int B,
                                     int Cout, int To, int act, float slope,
                                     bool act_before) {
  const int64_t nall = (int64_t)B * Cout * To;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nall;
       i += step) {
    const int co = (int)((i / To) % Cout);
    float o = yws[i] + (bias ? bias[co] : 0.f);
    if (!act_before) o = apply_act(o, act, slope);
    union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
    y[i] = cv.s;
  }
}

// dot-form forward for tiny Cout (disc score conv 1024->1 k3): one block
// per (b, t-tile); 256 threads reduce over Cin*K via LDS tree.
__global__ void __launch_bounds__(256)
conv1d_fwd_dot_kernel(const unsigned short* __restrict__ x,
                      const unsigned short* __restrict__ w,
                      const float* __restrict__ bias,
                      unsigned short* __restrict__ y, int B, int Cin,
                      int Tin, int Cout, int To, int K, int dil, int pad,
                      bool reflect, int act, float slope, bool act_before) {
  __shared__ float red[256];
  const int t = blockIdx.x;
  const int co = blockIdx.y;
  const int b = blockIdx.z;
  const int NK = Cin * K;
  float acc = 0.f;
  for (int nn = threadIdx.x; nn < NK; nn += 256) {
    const int ci = nn / K;
    const int k = nn - ci * K;
    const unsigned short* xr = x + ((int64_t)b * Cin + ci) * Tin;
    unsigned short v = load_pad_bf16(xr, t + k * dil - pad, Tin, reflect);
    if (act_before) v = leaky_bf16(v, slope);
    union { unsigned short s; __bf16 bb; } xv{v};
    union { unsigned short s; __bf16 bb; } wv{
        w[((int64_t)co * Cin + ci) * K + k]};
    acc = fmaf((float)wv.bb, (float)xv.bb, acc);
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    float o = red[0] + (bias ? bias[co] : 0.f);
    if (!act_before) o = apply_act(o, act, slope);
    union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
    y[((int64_t)b * Cout + co) * To + t] = cv.s;
  }
}

// zero-stuff along t: y[b][c][t*s] = x[b][c][t] (rest 0)
__global__ void zero_stuff_kernel(const unsigned short* __restrict__ x,
                                  unsigned short* __restrict__ y,
                                  int64_t BC, int Tin, int stride,
                                  int Lout) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t n = BC * Lout;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = i; idx < n; idx += step) {
    const int64_t bc = idx / Lout;
    const int t = (int)(idx - bc * Lout);
    unsigned short v = 0;
    if (t % stride == 0 && t / stride < Tin) v = x[bc * Tin + t / stride];
    y[idx] = v;
  }
}

// ---------------------------------------------------------------------------
// generalized backward-weight: GEMM  M=Cout (grouped), N=Cpg*K, Kred=B*To
//   gw[co][cig][k] += sum_{b,t} A[b][co][t] * pre(Bx[b][g*Cpg+cig][t*s+k*d-p])
// act_a: LeakyReLU on the A operand (conv_transpose weight grad, where
//        A is the pre-activation input x); act_b: LeakyReLU on B (conv
//        with fused input activation).  Shared by conv1d (A=gz, B=x) and
//        conv_transpose1d (A=x, B=gy, roles swapped by the wrapper).
// grid: (ceil(Cpg*K/64), G*ceil(Copg/64), NSPLIT); fp32 atomicAdd epilogue.
// ---------------------------------------------------------------------------
constexpr int TCH = 64;   // t per chunk
constexpr int LDT = 72;   // padded t stride (144 B rows)

// 16-row weight-grad variant for small groups (disc k=41 towers,
// Copg in {4,16}): cuts the 64-row tile's masked-MFMA waste 4-16x.
// waves: 4, each owns 16 of the 64 (cig,k) columns; M tile = 16 co.
// grid: (ceil(Cpg*K/64), G*ceil(Copg/16), NSPLIT)
__global__ void __launch_bounds__(256)
conv1d_bwd_weight_mfma16_kernel(const unsigned short* __restrict__ gz,
                                const unsigned short* __restrict__ x,
                                float* __restrict__ gw, int B, int Cin,
                                int Tin, int Cout, int To, int K,
                                int stride, int dil, int pad, bool reflect,
                                int groups, float slope, bool act_a,
                                bool act_b, int nsplit) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[
      (16 + 64) * LDT];
  unsigned short* lds_gz = smem;             // [16co][t]
  unsigned short* lds_xs = smem + 16 * LDT;  // [col=(cig,k)][t]

  const int Copg = Cout / groups;
  const int Cpg = Cin / groups;
  const int mtiles = (Copg + 15) / 16;
  const int g = blockIdx.y / mtiles;
  const int mt = blockIdx.y - g * mtiles;
  const int co_base = g * Copg + mt * 16;
  const int co_end = (g + 1) * Copg;
  const int n0 = blockIdx.x * 64;
  const int N = Cpg * K;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;           // wave owns cols [wid*16, wid*16+16)
  const int row = lane & 15;

  const int tchunks = (To + TCH - 1) / TCH;
  const int64_t total = (int64_t)B * tchunks;

  f32x4 acc = {};

  for (int64_t c = blockIdx.z; c < total; c += nsplit) {
    const int b = (int)(c / tchunks);
    const int t0 = (int)(c - (int64_t)b * tchunks) * TCH;
    for (int idx = tid; idx < 16 * TCH; idx += 256) {
      const int co = idx / TCH;
      const int t = idx - co * TCH;
      unsigned short v = 0;
      if (co_base + co < co_end && t0 + t < To) {
        v = gz[((int64_t)b * Cout + co_base + co) * To + t0 + t];
        if (act_a) v = leaky_bf16(v, slope);
      }
      lds_gz[co * LDT + t] = v;
    }
    for (int idx = tid; idx < 64 * (TCH / 4); idx += 256) {
      const int cc = idx / (TCH / 4);
      const int t4 = (idx - cc * (TCH / 4)) * 4;
      const int nn = n0 + cc;
      ushort4 v = {0, 0, 0, 0};
      if (nn < N) {
        const int ci = g * Cpg + nn / K;
        const int k = nn % K;
        const unsigned short* xr = x + ((int64_t)b * Cin + ci) * Tin;
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          unsigned short e = 0;
          if (t0 + t4 + u < To) {
            e = load_pad_bf16(xr, (t0 + t4 + u) * stride + k * dil - pad,
                              Tin, reflect);
            if (act_b) e = leaky_bf16(e, slope);
          }
          (&v.x)[u] = e;
        }
      }
      *(ushort4*)(lds_xs + cc * LDT + t4) = v;
    }
    __syncthreads();

#pragma unroll
    for (int step = 0; step < 2; ++step) {
      const int tt = step * 32 + (lane >> 4) * 8;
      bf16x8 a0 = *(const bf16x8*)(lds_gz + row * LDT + tt);
      bf16x8 b0 = *(const bf16x8*)(lds_xs + (wid * 16 + row) * LDT + tt);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  {
    const int nn = n0 + wid * 16 + col;
    if (nn < N) {
      const int cig = nn / K;
      const int k = nn - cig * K;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co_base + row4 + r;
        if (co < co_end && co < Cout)
          atomicAdd(&gw[((int64_t)co * Cpg + cig) * K + k], acc[r]);
      }
    }
  }
}


__global__ void __launch_bounds__(256)
conv1d_bwd_weight_mfma_kernel(const unsigned short* __restrict__ gz,
                              const unsigned short* __restrict__ x,
                              float* __restrict__ gw, int B, int Cin,
                              int Tin, int Cout, int To, int K, int stride,
                              int dil, int pad, bool reflect, int groups,
                              float slope, bool act_a, bool act_b,
                              int nsplit) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[2 * 64 * LDT];
  unsigned short* lds_gz = smem;            // [co][t]
  unsigned short* lds_xs = smem + 64 * LDT;  // [col=(cig,k)][t]

  const int Copg = Cout / groups;
  const int Cpg = Cin / groups;
  const int mtiles = (Copg + BM - 1) / BM;
  const int g = blockIdx.y / mtiles;
  const int mt = blockIdx.y - g * mtiles;
  const int co_base = g * Copg + mt * BM;
  const int co_end = (g + 1) * Copg;
  const int n0 = blockIdx.x * 64;   // (cig,k) tile origin within the group
  const int N = Cpg * K;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int row = lane & 15;

  const int tchunks = (To + TCH - 1) / TCH;
  const int64_t total = (int64_t)B * tchunks;

  f32x4 acc[2][2] = {};

  for (int64_t c = blockIdx.z; c < total; c += nsplit) {
    const int b = (int)(c / tchunks);
    const int t0 = (int)(c - (int64_t)b * tchunks) * TCH;
    // A stage (gz rows, always t-contiguous): ushort4 when the whole
    // 64-t run is interior, scalar tail otherwise
    const bool a_vec = (t0 + TCH <= To) && !act_a;
    if (a_vec) {
      for (int idx = tid; idx < 64 * (TCH / 4); idx += 256) {
        const int co = idx / (TCH / 4);
        const int t4 = (idx - co * (TCH / 4)) * 4;
        ushort4 v = {0, 0, 0, 0};
        if (co_base + co < co_end)
          v = *(const ushort4*)(gz +
                ((int64_t)b * Cout + co_base + co) * To + t0 + t4);
        *(ushort4*)(lds_gz + co * LDT + t4) = v;
      }
    } else {
      for (int idx = tid; idx < 64 * TCH; idx += 256) {
        const int co = idx / TCH;
        const int t = idx - co * TCH;
        unsigned short v = 0;
        if (co_base + co < co_end && t0 + t < To) {
          v = gz[((int64_t)b * Cout + co_base + co) * To + t0 + t];
          if (act_a) v = leaky_bf16(v, slope);
        }
        lds_gz[co * LDT + t] = v;
      }
    }
    // B stage (shifted x rows): vectorize per-column when the run is
    // interior, 8-byte aligned and stride-1
    for (int idx = tid; idx < 64 * (TCH / 4); idx += 256) {
      const int cc = idx / (TCH / 4);
      const int t4 = (idx - cc * (TCH / 4)) * 4;
      const int nn = n0 + cc;
      ushort4 v = {0, 0, 0, 0};
      if (nn < N) {
        const int ci = g * Cpg + nn / K;
        const int k = nn % K;
        const unsigned short* xr = x + ((int64_t)b * Cin + ci) * Tin;
        const int base = (t0 + t4) * stride + k * dil - pad;
        if (stride == 1 && !reflect && (base & 3) == 0 && base >= 0 &&
            base + 4 <= Tin && t0 + t4 + 4 <= To) {
          v = *(const ushort4*)(xr + base);
          if (act_b) {
            v.x = leaky_bf16(v.x, slope);
            v.y = leaky_bf16(v.y, slope);
            v.z = leaky_bf16(v.z, slope);
            v.w = leaky_bf16(v.w, slope);
          }
        } else {
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            unsigned short e = 0;
            if (t0 + t4 + u < To) {
              e = load_pad_bf16(xr, (t0 + t4 + u) * stride + k * dil - pad,
                                Tin, reflect);
              if (act_b) e = leaky_bf16(e, slope);
            }
            (&v.x)[u] = e;
          }
        }
      }
      *(ushort4*)(lds_xs + cc * LDT + t4) = v;
    }
    __syncthreads();

#pragma unroll
    for (int step = 0; step < 2; ++step) {
      const int tt = step * 32 + (lane >> 4) * 8;
      bf16x8 a0 = *(const bf16x8*)(lds_gz + (wr * 32 + row) * LDT + tt);
      bf16x8 a1 = *(const bf16x8*)(lds_gz + (wr * 32 + 16 + row) * LDT + tt);
      bf16x8 b0 = *(const bf16x8*)(lds_xs + (wc * 32 + row) * LDT + tt);
      bf16x8 b1 = *(const bf16x8*)(lds_xs + (wc * 32 + 16 + row) * LDT + tt);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
  for (int m = 0; m < 2; ++m)
    for (int n = 0; n < 2; ++n) {
      const int nn = n0 + wc * 32 + n * 16 + col;
      if (nn >= N) continue;
      const int cig = nn / K;
      const int k = nn - cig * K;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co_base + wr * 32 + m * 16 + row4 + r;
        if (co >= co_end || co >= Cout) continue;
        atomicAdd(&gw[((int64_t)co * Cpg + cig) * K + k], acc[m][n][r]);
      }
    }
}

}  // namespace melgan

// ===========================================================================
// host-side dispatch (called from the wrappers in kernels.hip)
// ===========================================================================
using namespace melgan;

// xstage tier for the grouped strided k=41 towers (xstage.hip)
bool xstage_fwd_eligible(int64_t Cin, int64_t Cout, int64_t K,
                         int64_t stride, int64_t dil, int64_t groups,
                         bool reflect);
torch::Tensor xstage_fwd(torch::Tensor x, torch::Tensor w,
                         const float* bias_p, int64_t stride, int64_t pad,
                         int64_t groups, int64_t act, double slope,
                         bool act_before, int64_t To);
bool xstage_bwd_data_eligible(int64_t Cin, int64_t Cout, int64_t K,
                              int64_t stride, int64_t dil, int64_t groups,
                              bool reflect);
torch::Tensor xstage_bwd_data(torch::Tensor gz, torch::Tensor w,
                              int64_t stride, int64_t pad, int64_t groups,
                              int64_t Tin);
bool xstage_bwd_weight_eligible(int64_t Cin, int64_t Cout, int64_t K,
                                int64_t stride, int64_t dil,
                                int64_t groups, bool reflect);
torch::Tensor xstage_bwd_weight(torch::Tensor gz, torch::Tensor x,
                                int64_t stride, int64_t pad,
                                int64_t groups, int64_t K, bool act_b,
                                double slope);

static inline int cdiv_i(int64_t a, int64_t b) {
  return (int)((a + b - 1) / b);
}

bool mfma_enabled() {
  static const bool disabled = []() {
    const char* e = getenv("MELGAN_AMD_NO_MFMA");
    return e && e[0] == '1';
  }();
  return !disabled;
}

// specialized stride-1 groups-1 kernel (x span staged once per ci-chunk)
bool conv1d_mfma_eligible(const torch::Tensor& x, int64_t Cin, int64_t K,
                          int64_t stride, int64_t dil, int64_t groups) {
  if (x.scalar_type() != torch::kBFloat16) return false;
  if (stride != 1 || groups != 1) return false;
  if (Cin < 16 || K > 8) return false;
  const int span = 64 + (int)((K - 1) * dil);
  const size_t lds = (size_t)(span + K * 64) * LDW * 2;
  if (lds > 64 * 1024) return false;
  return mfma_enabled();
}

// generic grouped kernel: any stride/dil/groups/K for bf16
bool conv1d_mfma_eligible_any(const torch::Tensor& x) {
  return x.scalar_type() == torch::kBFloat16 && mfma_enabled();
}

torch::Tensor conv1d_fwd_mfma(torch::Tensor x, torch::Tensor w,
                              const float* bias_p, int64_t dil, int64_t pad,
                              bool reflect, int64_t act, double slope,
                              bool act_before, int64_t To) {
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int Cout = w.size(0), K = w.size(2);
  auto y = torch::empty({B, Cout, (int)To}, x.options());
  auto stream0 = at::hip::getCurrentHIPStream();
  if (Cout <= 4 && Cin >= 256) {
    // tiny-M: dot-form (one block per output sample)
    dim3 grid((int)To, Cout, B);
    hipLaunchKernelGGL(conv1d_fwd_dot_kernel, grid, dim3(256), 0, stream0,
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(), bias_p,
                       (unsigned short*)y.data_ptr(), B, Cin, Tin, Cout,
                       (int)To, K, (int)dil, (int)pad, reflect, (int)act,
                       (float)slope, act_before);
    return y;
  }
  static const bool use_deep = []() {
    const char* e = getenv("MELGAN_AMD_DEEP");
    return e && e[0] == '1';
  }();
  // NOTE: measured SLOWER than the specialized kernel on the disc k5
  // shapes (51.6k vs 54.5k mel-frames/s end to end) — kept behind
  // MELGAN_AMD_DEEP=1 for further tuning; see profiles/README.md.
  if (use_deep && (int64_t)Cin * K >= 2048 && To <= 256 && Cout >= 32) {
    // deep contraction, small T: fold batch into 256-column tiles,
    // split-K to fill the chip when (cols x M) tiles are few
    const int ct = cdiv_i((int64_t)B * To, 256);
    const int mt = cdiv_i(Cout, BM);
    const int nchunks = cdiv_i((int64_t)Cin * K, CI_CH);
    int nsplit = (int)std::min<int64_t>(
        nchunks, std::max<int64_t>(1, 512 / std::max(1, ct * mt)));
    dim3 grid(ct, mt, nsplit);
    torch::Tensor yws;
    float* yws_p = nullptr;
    if (nsplit > 1) {
      yws = torch::zeros({B, Cout, (int)To},
                         x.options().dtype(torch::kFloat));
      yws_p = yws.data_ptr<float>();
    }
    auto log2i = [](int v) {
      return (v & (v - 1)) == 0 ? __builtin_ctz(v) : -1;
    };
    hipLaunchKernelGGL(conv1d_fwd_mfma_deep_kernel, grid, dim3(256), 0,
                       stream0, (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(), bias_p,
                       (unsigned short*)y.data_ptr(), yws_p, B, Cin, Tin,
                       Cout, (int)To, K, (int)dil, (int)pad, reflect,
                       (int)act, (float)slope, act_before, log2i(Cin),
                       log2i((int)To));
    if (nsplit > 1) {
      const int64_t nall = (int64_t)B * Cout * To;
      hipLaunchKernelGGL(deep_finalize_kernel,
                         dim3((int)std::min<int64_t>(2048,
                                                     (nall + 255) / 256)),
                         dim3(256), 0, stream0, yws_p, bias_p,
                         (unsigned short*)y.data_ptr(), B, Cout, (int)To,
                         (int)act, (float)slope, act_before);
    }
    return y;
  }
  const int span = 64 + (K - 1) * (int)dil;
  const size_t lds = (size_t)(span + K * 64) * LDW * sizeof(unsigned short);
  dim3 grid(cdiv_i(To, BN), cdiv_i(Cout, BM), B);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv1d_fwd_mfma_kernel, grid, dim3(256), lds, stream,
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(), bias_p,
                     (unsigned short*)y.data_ptr(), B, Cin, Tin, Cout,
                     (int)To, K, (int)dil, (int)pad, reflect, (int)act,
                     (float)slope, act_before);
  return y;
}

torch::Tensor conv1d_fwd_mfma_generic(torch::Tensor x, torch::Tensor w,
                                      const float* bias_p, int64_t stride,
                                      int64_t dil, int64_t pad, bool reflect,
                                      int64_t groups, int64_t act,
                                      double slope, bool act_before,
                                      int64_t To) {
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int Cout = w.size(0), K = w.size(2);
  const int Copg = Cout / (int)groups;
  if (xstage_fwd_eligible(Cin, Cout, K, stride, dil, groups, reflect))
    return xstage_fwd(x, w, bias_p, stride, pad, groups, act, slope,
                      act_before, To);
  auto y = torch::empty({B, Cout, (int)To}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  if (Copg <= 16) {
    dim3 grid(cdiv_i(To, 128), (int)groups * cdiv_i(Copg, 16), B);
    hipLaunchKernelGGL(conv1d_fwd_mfma_g16_kernel, grid, dim3(256), 0,
                       stream, (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(), bias_p,
                       (unsigned short*)y.data_ptr(), B, Cin, Tin, Cout,
                       (int)To, K, (int)stride, (int)dil, (int)pad, reflect,
                       (int)groups, (int)act, (float)slope, act_before);
    return y;
  }
  dim3 grid(cdiv_i(To, BN), (int)groups * cdiv_i(Copg, BM), B);
  hipLaunchKernelGGL(conv1d_fwd_mfma_g_kernel, grid, dim3(256), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(), bias_p,
                     (unsigned short*)y.data_ptr(), B, Cin, Tin, Cout,
                     (int)To, K, (int)stride, (int)dil, (int)pad, reflect,
                     (int)groups, (int)act, (float)slope, act_before);
  return y;
}

void run_unpad_fold(torch::Tensor gxpad, torch::Tensor gx, int BC,
                           int Tin, int pad, bool reflect, int src_len) {
  const int64_t n = (int64_t)BC * Tin;
  const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gx.scalar_type(), "unpad_fold", [&] {
        hipLaunchKernelGGL((unpad_fold_kernel<scalar_t>), dim3(blocks),
                           dim3(256), 0, stream,
                           gxpad.data_ptr<scalar_t>(),
                           gx.data_ptr<scalar_t>(), BC, Tin, pad, reflect,
                           src_len);
      });
}

torch::Tensor conv1d_bwd_data_mfma(torch::Tensor gz, torch::Tensor w,
                                   int64_t dil, int64_t pad, bool reflect,
                                   int64_t Tin) {
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cin = w.size(1), K = w.size(2);
  const int Tpad = (int)Tin + 2 * (int)pad;
  auto gxpad = torch::empty({B, Cin, Tpad}, gz.options());
  const int span = 64 + (K - 1) * (int)dil;
  const size_t lds = (size_t)(span + K * 64) * LDW * sizeof(unsigned short);
  dim3 grid(cdiv_i(Tpad, BN), cdiv_i(Cin, BM), B);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv1d_bwd_data_mfma_kernel, grid, dim3(256), lds,
                     stream, (const unsigned short*)gz.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)gxpad.data_ptr(), B, Cin, Tpad, Cout,
                     To, K, (int)dil);
  if (pad == 0) return gxpad;
  auto gx = torch::empty({B, Cin, (int)Tin}, gz.options());
  run_unpad_fold(gxpad, gx, B * Cin, (int)Tin, (int)pad, reflect, Tpad);
  return gx;
}

// generic bwd-data: zero-stuff gz (stride>1), flip/permute w, run the
// grouped generic forward on padded coordinates, fold the padding.
torch::Tensor conv1d_bwd_data_mfma_generic(torch::Tensor gz,
                                           torch::Tensor w, int64_t stride,
                                           int64_t dil, int64_t pad,
                                           bool reflect, int64_t groups,
                                           int64_t Tin) {
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cpg = w.size(1), K = w.size(2);
  const int Cin = Cpg * (int)groups;
  const int Copg = Cout / (int)groups;
  auto stream = at::hip::getCurrentHIPStream();

  // w~ [Cin][Copg][K] = w[g*Copg+cog][cig][K-1-k]
  auto wf = w.view({(int)groups, Copg, Cpg, K})
                .permute({0, 2, 1, 3})
                .flip(-1)
                .reshape({Cin, Copg, K})
                .contiguous();

  const int pad2 = (K - 1) * (int)dil;
  const int Tpad = (int)Tin + 2 * (int)pad;
  const int CpgX = Cpg;  // (per-group input channels == gx channels/group)
  if (xstage_bwd_data_eligible(Cin, Cout, K, stride, dil, groups, reflect))
    return xstage_bwd_data(gz, w, stride, pad, groups, Tin);
  if (!reflect && stride > 1 && dil == 1 &&
      CpgX * stride <= 64) {  // (cig,phi) rows per group fit few m-tiles
    auto gx = torch::empty({B, Cin, (int)Tin}, gz.options());
    // Umax covers all j in [0, Tin): u <= (Tin - 1 + pad)/stride
    const int Umax = ((int)Tin - 1 + (int)pad) / (int)stride + 1;
    const int utiles = cdiv_i(Umax, 128);
    dim3 grid(utiles,
              (int)groups * cdiv_i(CpgX * (int)stride, 16), B);
    auto stream2 = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(convbwd_phased16_kernel, grid, dim3(256), 0, stream2,
                       (const unsigned short*)gz.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (unsigned short*)gx.data_ptr(), B, Cin, (int)Tin,
                       Cout, To, K, (int)stride, (int)pad, (int)groups,
                       utiles);
    return gx;
  }
  torch::Tensor gz_in = gz;
  int Lin = To;
  if (stride > 1) {
    // cover Tpad: trailing zeros beyond the last real sample are harmless
    Lin = std::max((To - 1) * (int)stride + 1, Tpad - pad2);
    gz_in = torch::empty({B, Cout, Lin}, gz.options());
    const int64_t n = (int64_t)B * Cout * Lin;
    const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
    hipLaunchKernelGGL(zero_stuff_kernel, dim3(blocks), dim3(256), 0,
                       stream, (const unsigned short*)gz.data_ptr(),
                       (unsigned short*)gz_in.data_ptr(),
                       (int64_t)B * Cout, To, (int)stride, Lin);
  }
  // gxpad' = conv(gz_in, w~, stride 1, dil d, zero-pad (K-1)d)
  const int Lout = Lin + pad2;  // Lin + 2*pad2 - (K-1)*dil
  TORCH_CHECK(Lout >= Tpad, "bwd_data geometry: Lout < Tpad");
  auto gxpad = conv1d_fwd_mfma_generic(gz_in, wf, nullptr, 1, dil, pad2,
                                       false, groups, 0, 0.0, false, Lout);
  auto gx = torch::empty({B, Cin, (int)Tin}, gz.options());
  run_unpad_fold(gxpad, gx, B * Cin, (int)Tin, (int)pad, reflect, Lout);
  return gx;
}

torch::Tensor conv1d_bwd_weight_mfma(torch::Tensor gz, torch::Tensor x,
                                     int64_t stride, int64_t dil,
                                     int64_t pad, bool reflect,
                                     int64_t groups, int64_t K, bool act_a,
                                     bool act_b, double slope) {
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cin = x.size(1), Tin = x.size(2);
  const int Cpg = Cin / (int)groups;
  const int Copg = Cout / (int)groups;
  const int N = Cpg * (int)K;
  if (!act_a &&
      xstage_bwd_weight_eligible(Cin, Cout, K, stride, dil, groups,
                                 reflect))
    return xstage_bwd_weight(gz, x, stride, pad, groups, K, act_b, slope);
  auto gw = torch::zeros({Cout, Cpg, (int)K},
                         x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (Copg <= 16) {
    const int gx16 = cdiv_i(N, 64);
    const int gy16 = (int)groups * cdiv_i(Copg, 16);
    const int tchunks16 = cdiv_i(To, TCH);
    const int64_t total16 = (int64_t)B * tchunks16;
    int ns16 = (int)std::min<int64_t>(
        total16, std::max<int64_t>(1, 2048 / std::max(1, gx16 * gy16)));
    dim3 grid16(gx16, gy16, ns16);
    hipLaunchKernelGGL(conv1d_bwd_weight_mfma16_kernel, grid16, dim3(256),
                       0, stream, (const unsigned short*)gz.data_ptr(),
                       (const unsigned short*)x.data_ptr(),
                       gw.data_ptr<float>(), B, Cin, Tin, Cout, To, (int)K,
                       (int)stride, (int)dil, (int)pad, reflect,
                       (int)groups, (float)slope, act_a, act_b, ns16);
    return gw;
  }
  const int gx_ = cdiv_i(N, 64);
  const int gy_ = (int)groups * cdiv_i(Copg, 64);
  const int tchunks = cdiv_i(To, TCH);
  const int64_t total = (int64_t)B * tchunks;
  int nsplit = (int)std::min<int64_t>(
      total, std::max<int64_t>(1, 2048 / std::max(1, gx_ * gy_)));
  dim3 grid(gx_, gy_, nsplit);
  hipLaunchKernelGGL(conv1d_bwd_weight_mfma_kernel, grid, dim3(256), 0,
                     stream, (const unsigned short*)gz.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     gw.data_ptr<float>(), B, Cin, Tin, Cout, To, (int)K,
                     (int)stride, (int)dil, (int)pad, reflect, (int)groups,
                     (float)slope, act_a, act_b, nsplit);
  return gw;
}

// conv_transpose forward: zero-stuff x, flip/permute w, generic conv.
torch::Tensor conv_transpose1d_fwd_mfma(torch::Tensor x, torch::Tensor w,
                                        const float* bias_p, int64_t stride,
                                        int64_t pad, int64_t out_pad,
                                        int64_t act, double slope,
                                        bool act_before, int64_t To) {
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int K = w.size(2);
  auto stream = at::hip::getCurrentHIPStream();
  // w' [Cout][Cin][k'] = w[ci][co][K-1-k']
  auto wf = w.permute({1, 0, 2}).flip(-1).contiguous();
  const int Lin = (Tin - 1) * (int)stride + 1 + (int)out_pad;
  auto x_up = torch::empty({B, Cin, Lin}, x.options());
  {
    const int64_t n = (int64_t)B * Cin * Lin;
    const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
    hipLaunchKernelGGL(zero_stuff_kernel, dim3(blocks), dim3(256), 0,
                       stream, (const unsigned short*)x.data_ptr(),
                       (unsigned short*)x_up.data_ptr(), (int64_t)B * Cin,
                       Tin, (int)stride, Lin);
  }
  const int pad2 = K - 1 - (int)pad;
  TORCH_CHECK(pad2 >= 0, "conv_transpose geometry: pad > K-1");
  return conv1d_fwd_mfma_generic(x_up, wf, bias_p, 1, 1, pad2, false, 1,
                                 act, slope, act_before, To);
}

// python-visible un-pad fold (GEMM bwd-data path): gxpad (B, C, Tpad)
// -> gx (B, C, Tin) with reflect edge folding
torch::Tensor unpad_fold_py(torch::Tensor gxpad, int64_t Tin, int64_t pad,
                            bool reflect) {
  const int B = gxpad.size(0), C = gxpad.size(1);
  const int src_len = gxpad.size(2);
  TORCH_CHECK(src_len >= Tin + 2 * pad);
  auto gx = torch::empty({B, C, (int)Tin}, gxpad.options());
  run_unpad_fold(gxpad, gx, B * C, (int)Tin, (int)pad, reflect, src_len);
  return gx;
}
