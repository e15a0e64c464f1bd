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
template <typename T>
__global__ void unpad_fold_kernel(const T* __restrict__ gxpad,
                                  T* __restrict__ gx, int BC, int Tin,
                                  int pad, bool reflect) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t n = (int64_t)BC * Tin;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  const int Tpad = Tin + 2 * pad;
  for (int64_t idx = i; idx < n; idx += step) {
    const int64_t bc = idx / Tin;
    const int j = (int)(idx - bc * Tin);
    const T* src = gxpad + bc * Tpad;
    float v = static_cast<float>(src[pad + j]);
    if (reflect && pad > 0) {
      if (j > 0 && j <= pad) v += static_cast<float>(src[pad - j]);
      const int jm = 2 * Tin - 2 - j;
      if (jm >= Tin && (pad + jm) < Tpad)
        v += static_cast<float>(src[pad + jm]);
    }
    gx[idx] = static_cast<T>(v);
  }
}

// ---------------------------------------------------------------------------
// backward-weight (stride 1): GEMM  M=Cout, N=Cin*K, Kred=B*To
//   gw[co][ci*K+k] += sum_{b,t} gz[b][co][t] * pre(x[b][ci][t + k*d - p])
// grid: (ceil(N/64), ceil(Cout/64), NSPLIT); fp32 atomicAdd epilogue.
// LDS: lds_gz[64co][72t] ++ lds_xs[64col][72t]
// ---------------------------------------------------------------------------
constexpr int TCH = 64;   // t per chunk
constexpr int LDT = 72;   // padded t stride (144 B rows)

__global__ void __launch_bounds__(256)
conv1d_bwd_weight_mfma_kernel(const unsigned short* __restrict__ gz,
                              const unsigned short* __restrict__ x,
                              float* __restrict__ gw, int B, int Cin,
                              int Tin, int Cout, int To, int K, int dil,
                              int pad, bool reflect, float slope,
                              bool act_before, int nsplit) {
  __shared__ __attribute__((aligned(16))) unsigned short smem[2 * 64 * LDT];
  unsigned short* lds_gz = smem;            // [co][t]
  unsigned short* lds_xs = smem + 64 * LDT;  // [col=(ci,k)][t]

  const int n0 = blockIdx.x * 64;   // (ci,k) tile origin
  const int co0 = blockIdx.y * 64;  // co tile origin
  const int N = Cin * K;
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
    for (int idx = tid; idx < 64 * TCH; idx += 256) {
      const int co = idx / TCH;
      const int t = idx - co * TCH;
      unsigned short v = 0;
      if (co0 + co < Cout && t0 + t < To)
        v = gz[((int64_t)b * Cout + co0 + co) * To + t0 + t];
      lds_gz[co * LDT + t] = v;
    }
    for (int idx = tid; idx < 64 * TCH; idx += 256) {
      const int cc = idx / TCH;
      const int t = idx - cc * TCH;
      unsigned short v = 0;
      const int nn = n0 + cc;
      if (nn < N && t0 + t < To) {
        const int ci = nn / K;
        const int k = nn - ci * K;
        const unsigned short* xr = x + ((int64_t)b * Cin + ci) * Tin;
        v = load_pad_bf16(xr, (t0 + t) + k * dil - pad, Tin, reflect);
        if (act_before) v = leaky_bf16(v, slope);
      }
      lds_xs[cc * LDT + t] = v;
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
      const int ci = nn / K;
      const int k = nn - ci * K;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int co = co0 + wr * 32 + m * 16 + row4 + r;
        if (co >= Cout) continue;
        atomicAdd(&gw[((int64_t)co * Cin + ci) * K + k], acc[m][n][r]);
      }
    }
}

}  // namespace melgan

// ===========================================================================
// host-side dispatch (called from the wrappers in kernels.hip)
// ===========================================================================
using namespace melgan;

static inline int cdiv_i(int64_t a, int64_t b) {
  return (int)((a + b - 1) / b);
}

bool conv1d_mfma_eligible(const torch::Tensor& x, int64_t Cin, int64_t K,
                          int64_t stride, int64_t dil, int64_t groups) {
  if (x.scalar_type() != torch::kBFloat16) return false;
  if (stride != 1 || groups != 1) return false;
  if (Cin < 16 || K > 8) return false;
  const int span = 64 + (int)((K - 1) * dil);
  const size_t lds = (size_t)(span + K * 64) * LDW * 2;
  if (lds > 64 * 1024) return false;
  static const bool disabled = []() {
    const char* e = getenv("MELGAN_AMD_NO_MFMA");
    return e && e[0] == '1';
  }();
  return !disabled;
}

torch::Tensor conv1d_fwd_mfma(torch::Tensor x, torch::Tensor w,
                              const float* bias_p, int64_t dil, int64_t pad,
                              bool reflect, int64_t act, double slope,
                              bool act_before, int64_t To) {
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int Cout = w.size(0), K = w.size(2);
  auto y = torch::empty({B, Cout, (int)To}, x.options());
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
  const int64_t n = (int64_t)B * Cin * Tin;
  const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gz.scalar_type(), "unpad_fold", [&] {
        hipLaunchKernelGGL((unpad_fold_kernel<scalar_t>), dim3(blocks),
                           dim3(256), 0, stream,
                           gxpad.data_ptr<scalar_t>(),
                           gx.data_ptr<scalar_t>(), B * Cin, (int)Tin,
                           (int)pad, reflect);
      });
  return gx;
}

torch::Tensor conv1d_bwd_weight_mfma(torch::Tensor gz, torch::Tensor x,
                                     int64_t dil, int64_t pad, bool reflect,
                                     int64_t K, bool act_before,
                                     double slope) {
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cin = x.size(1), Tin = x.size(2);
  const int N = Cin * (int)K;
  auto gw = torch::zeros({Cout, Cin, (int)K},
                         x.options().dtype(torch::kFloat));
  const int gx_ = cdiv_i(N, 64), gy_ = cdiv_i(Cout, 64);
  const int tchunks = cdiv_i(To, TCH);
  const int64_t total = (int64_t)B * tchunks;
  int nsplit = (int)std::min<int64_t>(
      total, std::max<int64_t>(1, 2048 / std::max(1, gx_ * gy_)));
  dim3 grid(gx_, gy_, nsplit);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(conv1d_bwd_weight_mfma_kernel, grid, dim3(256), 0,
                     stream, (const unsigned short*)gz.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     gw.data_ptr<float>(), B, Cin, Tin, Cout, To, (int)K,
                     (int)dil, (int)pad, reflect, (float)slope, act_before,
                     nsplit);
  return gw;
}
