// Dedicated kernels for single-input-channel convs (the discriminator
// towers' k=15 input conv, Cin=1 -> Cout=16).  On the MFMA tiers these
// shapes are 94-98% masked (M or contraction rows of 1 in 16/64-row
// tiles); the work is tiny (15 MACs/output) and VALU-shaped: stage the
// single padded input row in LDS once per tile and loop co over it.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace melgan {

constexpr int C1_TL = 1024;  // outputs per block (t)

// y[b][co][t] = act(bias + sum_k w[co][0][k] * pre(x[b][0][t*s+k*d-p]))
template <typename T>
__global__ void __launch_bounds__(256)
cin1_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                const float* __restrict__ bias, T* __restrict__ y, int B,
                int Tin, int Cout, int To, int K, int stride, int dil,
                int pad, bool reflect, int act, float slope,
                bool act_before) {
  extern __shared__ __attribute__((aligned(16))) float sm1[];
  float* sw = sm1;            // [Cout*K]
  float* sx = sm1 + Cout * K;  // [span]
  const int t0 = blockIdx.x * C1_TL;
  const int b = blockIdx.y;
  const int tid = threadIdx.x;
  const int span = (C1_TL - 1) * stride + (K - 1) * dil + 1;
  const int g0 = t0 * stride - pad;

  for (int i = tid; i < Cout * K; i += 256)
    sw[i] = static_cast<float>(w[i]);
  const T* xr = x + (int64_t)b * Tin;
  for (int i = tid; i < span; i += 256) {
    float v = read_pad(xr, g0 + i, Tin, reflect);
    sx[i] = leaky_pre(v, slope, act_before);
  }
  __syncthreads();

  for (int co = 0; co < Cout; ++co) {
    const float* wr = sw + co * K;
    const float bv = bias ? bias[co] : 0.f;
    for (int tl = tid; tl < C1_TL; tl += 256) {
      const int t = t0 + tl;
      if (t >= To) break;
      float acc = bv;
      const float* xs = sx + tl * stride;
#pragma unroll 5
      for (int k = 0; k < K; ++k) acc = fmaf(wr[k], xs[k * dil], acc);
      if (!act_before) acc = apply_act(acc, act, slope);
      y[((int64_t)b * Cout + co) * To + t] = static_cast<T>(acc);
    }
  }
}

// gxpad[b][0][q] = sum_{co,k} w[co][0][k] * gz[b][co][q - k*d]
// (stride 1 only; host un-pads/edge-folds afterwards)
template <typename T>
__global__ void __launch_bounds__(256)
cin1_bwd_data_kernel(const T* __restrict__ gz, const T* __restrict__ w,
                     T* __restrict__ gxpad, int B, int Tpad, int Cout,
                     int To, int K, int dil) {
  extern __shared__ __attribute__((aligned(16))) float sm2[];
  float* sw = sm2;             // [Cout*K]
  // gz span staged as scalar_t to halve LDS: [co][span]
  T* sz = (T*)(sm2 + Cout * K);
  const int t0 = blockIdx.x * C1_TL;
  const int b = blockIdx.y;
  const int tid = threadIdx.x;
  const int ext = (K - 1) * dil;
  const int span = C1_TL + ext;

  for (int i = tid; i < Cout * K; i += 256)
    sw[i] = static_cast<float>(w[i]);
  for (int i = tid; i < Cout * span; i += 256) {
    const int co = i / span;
    const int v = i - co * span;
    const int t = t0 - ext + v;
    sz[co * span + v] =
        (t >= 0 && t < To) ? gz[((int64_t)b * Cout + co) * To + t]
                           : (T)0;
  }
  __syncthreads();

  for (int ql = tid; ql < C1_TL; ql += 256) {
    const int q = t0 + ql;
    if (q >= Tpad) break;
    float acc = 0.f;
    for (int co = 0; co < Cout; ++co) {
      const float* wr = sw + co * K;
      const T* zr = sz + co * span + ql + ext;
#pragma unroll 5
      for (int k = 0; k < K; ++k)
        acc = fmaf(wr[k], static_cast<float>(zr[-k * dil]), acc);
    }
    gxpad[(int64_t)b * Tpad + q] = static_cast<T>(acc);
  }
}

// (a dedicated cin1 weight-grad kernel lived here; it measured 870us
// per call vs ~100us for the mfma16 path on the k15 shape — the 16-row
// MFMA tile wins the (tiny N, huge B*T) reduction — so it was removed.
// See the dispatch note in conv1d_bwd_weight, kernels.hip.)

}  // namespace melgan

using namespace melgan;

static inline int c1cdiv(int64_t a, int64_t b) {
  return (int)((a + b - 1) / b);
}

bool cin1_eligible(int64_t Cin, int64_t Cout, int64_t K, int64_t stride,
                   int64_t groups) {
  return Cin == 1 && groups == 1 && Cout <= 32 && K <= 63 && stride == 1;
}

torch::Tensor cin1_fwd(torch::Tensor x, torch::Tensor w,
                       const float* bias_p, int64_t dil, int64_t pad,
                       bool reflect, int64_t act, double slope,
                       bool act_before, int64_t To) {
  const int B = x.size(0), Tin = x.size(2);
  const int Cout = w.size(0), K = w.size(2);
  auto y = torch::empty({B, Cout, (int)To}, x.options());
  const int span = (C1_TL - 1) + (K - 1) * (int)dil + 1;
  const size_t lds = (size_t)(Cout * K + span) * 4;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(c1cdiv(To, C1_TL), B);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "cin1_fwd", [&] {
        hipLaunchKernelGGL((cin1_fwd_kernel<scalar_t>), grid, dim3(256),
                           lds, stream, x.data_ptr<scalar_t>(),
                           w.data_ptr<scalar_t>(), bias_p,
                           y.data_ptr<scalar_t>(), B, Tin, Cout, (int)To,
                           K, 1, (int)dil, (int)pad, reflect, (int)act,
                           (float)slope, act_before);
      });
  return y;
}

// returns gxpad (B, 1, Tin + 2*pad); caller un-pads / edge-folds
torch::Tensor cin1_bwd_data_pad(torch::Tensor gz, torch::Tensor w,
                                int64_t dil, int64_t pad, int64_t Tin) {
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int K = w.size(2);
  const int Tpad = (int)Tin + 2 * (int)pad;
  auto gxpad = torch::empty({B, 1, Tpad}, gz.options());
  const int span = C1_TL + (K - 1) * (int)dil;
  const size_t lds =
      (size_t)(Cout * K) * 4 + (size_t)(Cout * span) * gz.element_size();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(c1cdiv(Tpad, C1_TL), B);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gz.scalar_type(), "cin1_bwd_data", [&] {
        hipLaunchKernelGGL((cin1_bwd_data_kernel<scalar_t>), grid,
                           dim3(256), lds, stream,
                           gz.data_ptr<scalar_t>(),
                           w.data_ptr<scalar_t>(),
                           gxpad.data_ptr<scalar_t>(), B, Tpad, Cout, To,
                           K, (int)dil);
      });
  return gxpad;
}

