// melgan_amd gfx950 (CDNA4 / MI355X) kernels — v1 correctness tier.
//
// Implements the fused ops of SURVEY.md §2.2 (K1-K12 replacements):
//   conv1d_fwd / conv1d_bwd_data / conv1d_bwd_weight   (K1, K3, K9-analysis)
//   conv_transpose1d_fwd / _bwd_data / _bwd_weight     (K2, K9-synthesis)
//   act_bwd, leaky_bwd_inplace, conv1d_bias_grad       (K6, K7 epilogues)
//   multi_tensor_adam_tbl                              (K12)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//   - wave64: blocks are multiples of 64; the t (time) axis maps to lanes
//     so global loads coalesce; the co axis maps to waves so weight reads
//     are wave-uniform (scalar-cache path).
//   - weight-norm is folded by the wn_fold kernel (w = g*v/||v||, one
//     launch, cached per optimizer step); conv kernels see plain
//     weights (K5 eliminated from the hot path).
//   - reflection/zero padding is folded into the address calculation
//     (K8 eliminated); the reflect *backward* edge-fold is computed
//     in-place by evaluating the mirrored padded positions.
//   - LeakyReLU prologue (act_before) / LeakyReLU-Tanh epilogue fused.
//   - fp32 accumulation always; bf16/fp16/fp32 storage via AT_DISPATCH.
//
// v2 (MFMA implicit-GEMM tiles for the Cin*K contraction) layers on top —
// see conv1d_mfma.hip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV_INLINE __device__ __forceinline__

namespace melgan {

constexpr int ACT_NONE = 0;
constexpr int ACT_LEAKY = 1;
constexpr int ACT_TANH = 2;

DEV_INLINE float apply_act(float v, int act, float slope) {
  if (act == ACT_LEAKY) return v > 0.f ? v : v * slope;
  if (act == ACT_TANH) return tanhf(v);
  return v;
}

DEV_INLINE float leaky_pre(float v, float slope, bool on) {
  return (on && v < 0.f) ? v * slope : v;
}

// reflect index into [0, T): -1 -> 1, T -> T-2 (torch 'reflect' mode)
DEV_INLINE int reflect_idx(int i, int T) {
  if (i < 0) i = -i;
  if (i >= T) i = 2 * T - 2 - i;
  return i;
}

// padded read with zero or reflect semantics
template <typename T>
DEV_INLINE float read_pad(const T* __restrict__ row, int i, int len,
                          bool reflect) {
  if (reflect) return static_cast<float>(row[reflect_idx(i, len)]);
  if (i < 0 || i >= len) return 0.f;
  return static_cast<float>(row[i]);
}

// ---------------------------------------------------------------------------
// conv1d forward:  y[b][co][t] = act(bias + sum_{cig,k} w[co][cig][k] *
//                                pre(x[b][g*Cpg+cig][t*s + k*d - p]))
// grid: (ceil(To/BT), ceil(Cout/BC), B);  block: BT*BC threads (t fastest)
// ---------------------------------------------------------------------------
template <typename T, int BT, int BC>
__global__ void conv1d_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ bias, T* __restrict__ y,
    int B, int Cin, int Tin, int Cout, int To, int K, int stride, int dil,
    int pad, bool reflect, int groups, int act, float slope,
    bool act_before) {
  const int t = blockIdx.x * BT + (threadIdx.x % BT);
  const int co = blockIdx.y * BC + (threadIdx.x / BT);
  const int b = blockIdx.z;
  if (t >= To || co >= Cout) return;
  const int Cpg = Cin / groups;
  const int Copg = Cout / groups;
  const int g = co / Copg;
  const T* xb = x + ((int64_t)b * Cin + (int64_t)g * Cpg) * Tin;
  const T* wc = w + (int64_t)co * Cpg * K;
  float acc = bias ? bias[co] : 0.f;
  const int base = t * stride - pad;
  for (int ci = 0; ci < Cpg; ++ci) {
    const T* xr = xb + (int64_t)ci * Tin;
    const T* wr = wc + ci * K;
#pragma unroll 4
    for (int k = 0; k < K; ++k) {
      float xv = read_pad(xr, base + k * dil, Tin, reflect);
      xv = leaky_pre(xv, slope, act_before);
      acc = fmaf(static_cast<float>(wr[k]), xv, acc);
    }
  }
  if (!act_before) acc = apply_act(acc, act, slope);
  y[((int64_t)b * Cout + co) * To + t] = static_cast<T>(acc);
}

// ---------------------------------------------------------------------------
// conv1d backward-data with reflect edge folding.
//   gxpad[q] = sum_{k, t: t*s + k*d == q} sum_co w[co][ci][k] * gz[co][t]
//   gx[j] = gxpad[p+j]  (+ gxpad[p-j] and gxpad[2T-2+p-j] mirrors, reflect)
// grid: (ceil(Tin/BT), ceil(Cin/BC), B)
// ---------------------------------------------------------------------------
template <typename T, int BT, int BC>
__global__ void conv1d_bwd_data_kernel(
    const T* __restrict__ gz, const T* __restrict__ w, T* __restrict__ gx,
    int B, int Cin, int Tin, int Cout, int To, int K, int stride, int dil,
    int pad, bool reflect, int groups) {
  const int j = blockIdx.x * BT + (threadIdx.x % BT);
  const int ci = blockIdx.y * BC + (threadIdx.x / BT);
  const int b = blockIdx.z;
  if (j >= Tin || ci >= Cin) return;
  const int Cpg = Cin / groups;
  const int Copg = Cout / groups;
  const int g = ci / Cpg;
  const int cig = ci - g * Cpg;
  const T* gzb = gz + ((int64_t)b * Cout + (int64_t)g * Copg) * To;
  const T* wg = w + (((int64_t)g * Copg) * Cpg + cig) * K;

  auto gxpad_at = [&](int q) -> float {
    float s = 0.f;
    for (int k = 0; k < K; ++k) {
      int r = q - k * dil;
      if (r < 0) break;  // k*dil grows; r only shrinks
      if (r % stride) continue;
      int t = r / stride;
      if (t >= To) continue;
      const T* wp = wg + k;
      const T* gzp = gzb + t;
      float a = 0.f;
      for (int co = 0; co < Copg; ++co)
        a = fmaf(static_cast<float>(wp[(int64_t)co * Cpg * K]),
                 static_cast<float>(gzp[(int64_t)co * To]), a);
      s += a;
    }
    return s;
  };

  float acc = gxpad_at(pad + j);
  if (reflect && pad > 0) {
    if (j > 0 && j <= pad) acc += gxpad_at(pad - j);
    const int jm = 2 * Tin - 2 - j;  // source of mirrored tail position
    // tail mirrors: q in [T+pad-? ...]; q = pad + 2T-2-j in range
    if (j < Tin - 1 && (2 * Tin - 2 - j) >= Tin &&
        (pad + jm) < (Tin + 2 * pad))
      acc += gxpad_at(pad + jm);
  }
  gx[((int64_t)b * Cin + ci) * Tin + j] = static_cast<T>(acc);
}

// ---------------------------------------------------------------------------
// conv1d backward-weight (fp32 out, atomic t-chunked reduction):
//   gw[co][cig][k] += sum_{b,t in chunk} gz[b][co][t] *
//                     pre(x[b][g*Cpg+cig][t*s + k*d - p])
// grid: (ceil(Cpg*K/256), Cout, TCH) — TCH chunks over (b,t)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void conv1d_bwd_weight_kernel(
    const T* __restrict__ gz, const T* __restrict__ x,
    float* __restrict__ gw, int B, int Cin, int Tin, int Cout, int To,
    int K, int stride, int dil, int pad, bool reflect, int groups,
    float slope, bool act_before, int nchunks) {
  const int cik = blockIdx.x * blockDim.x + threadIdx.x;
  const int co = blockIdx.y;
  const int Cpg = Cin / groups;
  if (cik >= Cpg * K) return;
  const int cig = cik / K;
  const int k = cik - cig * K;
  const int Copg = Cout / groups;
  const int g = co / Copg;
  const int64_t bt_total = (int64_t)B * To;
  const int64_t per = (bt_total + nchunks - 1) / nchunks;
  const int64_t lo = (int64_t)blockIdx.z * per;
  const int64_t hi = min(lo + per, bt_total);
  float acc = 0.f;
  for (int64_t bt = lo; bt < hi; ++bt) {
    const int b = (int)(bt / To);
    const int t = (int)(bt - (int64_t)b * To);
    float gzv = static_cast<float>(gz[((int64_t)b * Cout + co) * To + t]);
    const T* xr = x + ((int64_t)b * Cin + (int64_t)g * Cpg + cig) * Tin;
    float xv = read_pad(xr, t * stride + k * dil - pad, Tin, reflect);
    xv = leaky_pre(xv, slope, act_before);
    acc = fmaf(gzv, xv, acc);
  }
  if (nchunks == 1)
    gw[((int64_t)co * Cpg + cig) * K + k] = acc;
  else
    atomicAdd(&gw[((int64_t)co * Cpg + cig) * K + k], acc);
}

// ---------------------------------------------------------------------------
// conv_transpose1d forward (gather form, no atomics — SURVEY.md K2):
//   y[b][co][t] = act(bias + sum_ci sum_{k≡(t+p) mod s} w[ci][co][k] *
//                 pre(x[b][ci][(t+p-k)/s]))
// grid: (ceil(To/BT), ceil(Cout/BC), B)
// ---------------------------------------------------------------------------
template <typename T, int BT, int BC>
__global__ void convt1d_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ w,
    const float* __restrict__ bias, T* __restrict__ y,
    int B, int Cin, int Tin, int Cout, int To, int K, int stride, int pad,
    float slope, bool act_before) {
  const int t = blockIdx.x * BT + (threadIdx.x % BT);
  const int co = blockIdx.y * BC + (threadIdx.x / BT);
  const int b = blockIdx.z;
  if (t >= To || co >= Cout) return;
  const T* xb = x + (int64_t)b * Cin * Tin;
  float acc = bias ? bias[co] : 0.f;
  const int tp = t + pad;
  const int k0 = tp % stride;
  for (int ci = 0; ci < Cin; ++ci) {
    const T* xr = xb + (int64_t)ci * Tin;
    const T* wr = w + ((int64_t)ci * Cout + co) * K;
    for (int k = k0; k < K; k += stride) {
      const int m = (tp - k) / stride;
      if (m < 0) break;  // k grows => m shrinks; m<0 stays <0
      if (m >= Tin) continue;
      float xv = leaky_pre(static_cast<float>(xr[m]), slope, act_before);
      acc = fmaf(static_cast<float>(wr[k]), xv, acc);
    }
  }
  y[((int64_t)b * Cout + co) * To + t] = static_cast<T>(acc);
}

// conv_transpose1d backward-weight:
//   gw[ci][co][k] = sum_{b,m} pre(x[b][ci][m]) * gy[b][co][m*s - p + k]
// grid: (ceil(Cout*K/256), Cin, TCH)
template <typename T>
__global__ void convt1d_bwd_weight_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    float* __restrict__ gw, int B, int Cin, int Tin, int Cout, int To,
    int K, int stride, int pad, float slope, bool act_before, int nchunks) {
  const int cok = blockIdx.x * blockDim.x + threadIdx.x;
  const int ci = blockIdx.y;
  if (cok >= Cout * K) return;
  const int co = cok / K;
  const int k = cok - co * K;
  const int64_t bm_total = (int64_t)B * Tin;
  const int64_t per = (bm_total + nchunks - 1) / nchunks;
  const int64_t lo = (int64_t)blockIdx.z * per;
  const int64_t hi = min(lo + per, bm_total);
  float acc = 0.f;
  for (int64_t bm = lo; bm < hi; ++bm) {
    const int b = (int)(bm / Tin);
    const int m = (int)(bm - (int64_t)b * Tin);
    const int t = m * stride - pad + k;
    if (t < 0 || t >= To) continue;
    float xv = leaky_pre(
        static_cast<float>(x[((int64_t)b * Cin + ci) * Tin + m]), slope,
        act_before);
    float gv = static_cast<float>(gy[((int64_t)b * Cout + co) * To + t]);
    acc = fmaf(gv, xv, acc);
  }
  if (nchunks == 1)
    gw[((int64_t)ci * Cout + co) * K + k] = acc;
  else
    atomicAdd(&gw[((int64_t)ci * Cout + co) * K + k], acc);
}

// ---------------------------------------------------------------------------
// elementwise epilogue/prologue backward + bias grad
// ---------------------------------------------------------------------------
template <typename T>
__global__ void act_bwd_kernel(const T* __restrict__ gy,
                               const T* __restrict__ y, T* __restrict__ gz,
                               int64_t n, int act, float slope) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = i; idx < n; idx += step) {
    float g = static_cast<float>(gy[idx]);
    float yv = static_cast<float>(y[idx]);
    float r;
    if (act == ACT_LEAKY) r = yv > 0.f ? g : g * slope;
    else if (act == ACT_TANH) r = g * (1.f - yv * yv);
    else r = g;
    gz[idx] = static_cast<T>(r);
  }
}

template <typename T>
__global__ void leaky_bwd_inplace_kernel(T* __restrict__ gx,
                                         const T* __restrict__ x, int64_t n,
                                         float slope) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = i; idx < n; idx += step) {
    if (static_cast<float>(x[idx]) <= 0.f)
      gx[idx] = static_cast<T>(static_cast<float>(gx[idx]) * slope);
  }
}

// fused activation-backward + bias-grad: gz = act'(y)*gy written once,
// per-channel sum of gz accumulated in the same pass (saves the full
// re-read of gz that a separate bias_grad kernel costs — VERDICT r1
// weak #6 / next #8).
template <typename T>
__global__ void act_bwd_biasgrad_kernel(const T* __restrict__ gy,
                                        const T* __restrict__ y,
                                        T* __restrict__ gz,
                                        float* __restrict__ gb, int B,
                                        int C, int Tlen, int act,
                                        float slope, int nchunks) {
  __shared__ float red[256];
  const int c = blockIdx.x;
  const int64_t bt_total = (int64_t)B * Tlen;
  const int64_t per = (bt_total + nchunks - 1) / nchunks;
  const int64_t lo = (int64_t)blockIdx.y * per;
  const int64_t hi = min(lo + per, bt_total);
  float acc = 0.f;
  // carried (b, t) split: one divide per thread, not per element
  int64_t bt = lo + threadIdx.x;
  int b = (int)(bt / Tlen);
  int t = (int)(bt - (int64_t)b * Tlen);
  for (; bt < hi; bt += blockDim.x) {
    const int64_t idx = ((int64_t)b * C + c) * Tlen + t;
    float g = static_cast<float>(gy[idx]);
    float r;
    if (act == ACT_LEAKY) {
      r = static_cast<float>(y[idx]) > 0.f ? g : g * slope;
    } else if (act == ACT_TANH) {
      const float yv = static_cast<float>(y[idx]);
      r = g * (1.f - yv * yv);
    } else {
      r = g;
    }
    gz[idx] = static_cast<T>(r);
    acc += r;
    t += blockDim.x;
    while (t >= Tlen) {
      t -= Tlen;
      ++b;
    }
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    if (nchunks == 1) gb[c] = red[0];
    else atomicAdd(&gb[c], red[0]);
  }
}

// gbias[c] = sum_{b,t} gz[b][c][t];  grid: (C, chunks), atomic combine
template <typename T>
__global__ void bias_grad_kernel(const T* __restrict__ gz,
                                 float* __restrict__ gb, int B, int C,
                                 int Tlen, int nchunks) {
  __shared__ float red[256];
  const int c = blockIdx.x;
  const int64_t bt_total = (int64_t)B * Tlen;
  const int64_t per = (bt_total + nchunks - 1) / nchunks;
  const int64_t lo = (int64_t)blockIdx.y * per;
  const int64_t hi = min(lo + per, bt_total);
  float acc = 0.f;
  int64_t bt = lo + threadIdx.x;
  int b = (int)(bt / Tlen);
  int t = (int)(bt - (int64_t)b * Tlen);
  for (; bt < hi; bt += blockDim.x) {
    acc += static_cast<float>(gz[((int64_t)b * C + c) * Tlen + t]);
    t += blockDim.x;
    while (t >= Tlen) {
      t -= Tlen;
      ++b;
    }
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    if (nchunks == 1) gb[c] = red[0];
    else atomicAdd(&gb[c], red[0]);
  }
}

// ---------------------------------------------------------------------------
// fused weight-norm fold / backward (K5 elimination):
//   wn_fold:  n[c] = max(||v[c]||, 1e-12); w[c] = v[c] * g[c] / n[c]
//   wn_bwd:   dot = <gw[c], v[c]>; gg[c] = dot/n;
//             gv[c] = (g/n) gw[c] - (g dot / n^3) v[c]
// One kernel launch each, replacing the ~6-kernel torch chains per conv.
// grid: (C0); block 256.
// ---------------------------------------------------------------------------
template <typename TO>
__global__ void wn_fold_kernel(const float* __restrict__ v,
                               const float* __restrict__ g,
                               TO* __restrict__ w, float* __restrict__ n_out,
                               int C0, int M) {
  __shared__ float red[256];
  const int c = blockIdx.x;
  const float* vr = v + (int64_t)c * M;
  float acc = 0.f;
  for (int i = threadIdx.x; i < M; i += blockDim.x) {
    const float vv = vr[i];
    acc += vv * vv;
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  const float n = fmaxf(sqrtf(red[0]), 1e-12f);
  const float scale = g[c] / n;
  if (threadIdx.x == 0) n_out[c] = n;
  TO* wr = w + (int64_t)c * M;
  for (int i = threadIdx.x; i < M; i += blockDim.x)
    wr[i] = static_cast<TO>(vr[i] * scale);
}

__global__ void wn_bwd_kernel(const float* __restrict__ gw,
                              const float* __restrict__ v,
                              const float* __restrict__ g,
                              const float* __restrict__ n,
                              float* __restrict__ gv,
                              float* __restrict__ gg, int C0, int M) {
  __shared__ float red[256];
  const int c = blockIdx.x;
  const float* gwr = gw + (int64_t)c * M;
  const float* vr = v + (int64_t)c * M;
  float acc = 0.f;
  for (int i = threadIdx.x; i < M; i += blockDim.x)
    acc += gwr[i] * vr[i];
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  const float dot = red[0];
  const float nc = n[c];
  const float a = g[c] / nc;
  const float bcoef = g[c] * dot / (nc * nc * nc);
  if (threadIdx.x == 0) gg[c] = dot / nc;
  float* gvr = gv + (int64_t)c * M;
  for (int i = threadIdx.x; i < M; i += blockDim.x)
    gvr[i] = a * gwr[i] - bcoef * vr[i];
}

// ---------------------------------------------------------------------------
// multi-tensor fused Adam (K12): one launch for ALL tensors.
// chunk table: chunk_tensor[i] = tensor index, chunk_off[i] = elem offset.
// ---------------------------------------------------------------------------
constexpr int ADAM_CHUNK = 1 << 13;  // ~900 blocks for a 7M-param model

__global__ void multi_tensor_adam_kernel(
    float** __restrict__ p_ptrs, float** __restrict__ g_ptrs,
    float** __restrict__ m_ptrs, float** __restrict__ v_ptrs,
    const int64_t* __restrict__ numels,
    const int64_t* __restrict__ chunk_map,  // (tensor_idx<<48) | offset
    float lr, float beta1, float beta2, float eps, float wd, float bc1,
    float bc2) {
  const int64_t packed = chunk_map[blockIdx.x];
  const int ti = (int)(packed >> 48);
  const int64_t off = packed & ((1LL << 48) - 1);
  const int64_t n = numels[ti];
  float* p = p_ptrs[ti] + off;
  float* g = g_ptrs[ti] + off;
  float* m = m_ptrs[ti] + off;
  float* v = v_ptrs[ti] + off;
  const int64_t len = min((int64_t)ADAM_CHUNK, n - off);
  const float step_size = lr / bc1;
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x) {
    float gi = g[i];
    float mi = beta1 * m[i] + (1.f - beta1) * gi;
    float vi = beta2 * v[i] + (1.f - beta2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    float denom = sqrtf(vi / bc2) + eps;
    float pi = p[i];
    if (wd != 0.f) pi *= (1.f - lr * wd);
    p[i] = pi - step_size * mi / denom;
  }
}

}  // namespace melgan

// ===========================================================================
// host wrappers
// ===========================================================================
using namespace melgan;

#define CHECK_IN(x) \
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be CUDA contiguous")

static inline int cdiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

// MFMA tier (conv1d_mfma.hip) — bf16 implicit GEMM
bool conv1d_mfma_eligible(const torch::Tensor& x, int64_t Cin, int64_t K,
                          int64_t stride, int64_t dil, int64_t groups);
bool conv1d_mfma_eligible_any(const torch::Tensor& x);
void run_unpad_fold(torch::Tensor gxpad, torch::Tensor gx, int BC, int Tin,
                    int pad, bool reflect, int src_len);
torch::Tensor conv1d_fwd_mfma(torch::Tensor x, torch::Tensor w,
                              const float* bias_p, int64_t dil, int64_t pad,
                              bool reflect, int64_t act, double slope,
                              bool act_before, int64_t To);
torch::Tensor conv1d_fwd_mfma_generic(torch::Tensor x, torch::Tensor w,
                                      const float* bias_p, int64_t stride,
                                      int64_t dil, int64_t pad, bool reflect,
                                      int64_t groups, int64_t act,
                                      double slope, bool act_before,
                                      int64_t To);
torch::Tensor conv1d_bwd_data_mfma(torch::Tensor gz, torch::Tensor w,
                                   int64_t dil, int64_t pad, bool reflect,
                                   int64_t Tin);
torch::Tensor conv1d_bwd_data_mfma_generic(torch::Tensor gz,
                                           torch::Tensor w, int64_t stride,
                                           int64_t dil, int64_t pad,
                                           bool reflect, int64_t groups,
                                           int64_t Tin);
torch::Tensor conv1d_bwd_weight_mfma(torch::Tensor gz, torch::Tensor x,
                                     int64_t stride, int64_t dil,
                                     int64_t pad, bool reflect,
                                     int64_t groups, int64_t K, bool act_a,
                                     bool act_b, double slope);
torch::Tensor conv_transpose1d_fwd_mfma(torch::Tensor x, torch::Tensor w,
                                        const float* bias_p, int64_t stride,
                                        int64_t pad, int64_t out_pad,
                                        int64_t act, double slope,
                                        bool act_before, int64_t To);

// STFT pipeline (stft.hip)
torch::Tensor stft_frame(torch::Tensor x, int64_t fft, int64_t hop,
                         int64_t win, torch::Tensor window);
torch::Tensor stft_frame_bwd(torch::Tensor gframes, torch::Tensor window,
                             int64_t fft, int64_t hop, int64_t win,
                             int64_t S);
torch::Tensor complex_mag_clamp(torch::Tensor spec_ri, double mn);
torch::Tensor mag_bwd(torch::Tensor gmag, torch::Tensor spec_ri, double mn);
torch::Tensor avg_pool1d_421(torch::Tensor x);
torch::Tensor avg_pool1d_421_bwd(torch::Tensor gy, int64_t Tin);
torch::Tensor stft_loss_reduce(torch::Tensor mx, torch::Tensor my);
torch::Tensor stft_loss_bwd(torch::Tensor mx, torch::Tensor my,
                            torch::Tensor sums, torch::Tensor gouts);
torch::Tensor multi_l1_fwd(torch::Tensor tbl, torch::Tensor chunks,
                           int64_t nt, torch::ScalarType dt);
void multi_l1_bwd(torch::Tensor tbl, torch::Tensor chunks, int64_t nt,
                  torch::Tensor gout, torch::ScalarType dt);

// Cin==1 VALU kernels (cin1.hip): the k=15 disc-input conv shapes
bool cin1_eligible(int64_t Cin, int64_t Cout, int64_t K, int64_t stride,
                   int64_t groups);
torch::Tensor cin1_fwd(torch::Tensor x, torch::Tensor w,
                       const float* bias_p, int64_t dil, int64_t pad,
                       bool reflect, int64_t act, double slope,
                       bool act_before, int64_t To);
torch::Tensor cin1_bwd_data_pad(torch::Tensor gz, torch::Tensor w,
                                int64_t dil, int64_t pad, int64_t Tin);
void run_unpad_fold(torch::Tensor gxpad, torch::Tensor gx, int BC,
                    int Tin, int pad, bool reflect, int src_len);

// dedicated PQMF polyphase kernels (pqmf.hip)
torch::Tensor pqmf_gather(torch::Tensor x, torch::Tensor fpad, int64_t KT,
                          int64_t N, int64_t pad, double scale);
torch::Tensor pqmf_scatter(torch::Tensor x, torch::Tensor fpad,
                           int64_t KT, int64_t N, int64_t pad,
                           double scale);

// glds GEMM engine (gemm.hip)
torch::Tensor conv1d_fwd_gemm(torch::Tensor x, torch::Tensor A_pad,
                              c10::optional<torch::Tensor> bias,
                              int64_t K, int64_t stride, int64_t dil,
                              int64_t pad, bool reflect, int64_t act,
                              double slope, bool act_before, int64_t To,
                              int64_t ilv);
torch::Tensor gemm_bt(torch::Tensor A, torch::Tensor Bt);
torch::Tensor unpad_fold_py(torch::Tensor gxpad, int64_t Tin, int64_t pad,
                            bool reflect);
torch::Tensor conv1d_bwd_weight_gemm(torch::Tensor gz, torch::Tensor x,
                                     int64_t K, int64_t stride, int64_t dil,
                                     int64_t pad, bool reflect,
                                     bool act_before, double slope);

torch::Tensor conv1d_fwd(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias, int64_t stride,
                         int64_t dil, int64_t pad, bool reflect,
                         int64_t groups, int64_t act, double slope,
                         bool act_before) {
  CHECK_IN(x);
  CHECK_IN(w);
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int Cout = w.size(0), K = w.size(2);
  const int To = (Tin + 2 * (int)pad - ((K - 1) * (int)dil + 1)) /
                     (int)stride + 1;
  TORCH_CHECK(To > 0, "conv1d_fwd: empty output");
  auto y = torch::empty({B, Cout, To}, x.options());
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  if (cin1_eligible(Cin, Cout, K, stride, groups))
    return cin1_fwd(x, w, bias_p, dil, pad, reflect, act, slope,
                    act_before, To);
  if (conv1d_mfma_eligible(x, Cin, K, stride, dil, groups))
    return conv1d_fwd_mfma(x, w, bias_p, dil, pad, reflect, act, slope,
                           act_before, To);
  if (conv1d_mfma_eligible_any(x))
    return conv1d_fwd_mfma_generic(x, w, bias_p, stride, dil, pad, reflect,
                                   groups, act, slope, act_before, To);
  constexpr int BT = 64, BC = 4;
  dim3 grid(cdiv(To, BT), cdiv(Cout, BC), B);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "conv1d_fwd", [&] {
        hipLaunchKernelGGL((conv1d_fwd_kernel<scalar_t, BT, BC>), grid,
                           dim3(BT * BC), 0, stream,
                           x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                           bias_p, y.data_ptr<scalar_t>(), B, Cin, Tin, Cout,
                           To, K, (int)stride, (int)dil, (int)pad, reflect,
                           (int)groups, (int)act, (float)slope, act_before);
      });
  return y;
}

torch::Tensor conv1d_bwd_data(torch::Tensor gz, torch::Tensor w,
                              int64_t stride, int64_t dil, int64_t pad,
                              bool reflect, int64_t groups, int64_t Tin) {
  CHECK_IN(gz);
  CHECK_IN(w);
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cin = w.size(1) * groups, K = w.size(2);
  static const bool use_deep_bwd = []() {
    const char* e = getenv("MELGAN_AMD_DEEP");
    return e && e[0] == '1';
  }();
  if (use_deep_bwd && conv1d_mfma_eligible_any(gz) && stride == 1 &&
      groups == 1 && (int64_t)Cout * K >= 2048 && Tin + 2 * pad <= 256 &&
      Cin >= 32) {
    // deep-contraction small-T (disc k5, mel conv): flipped-weight direct
    // conv on the batch-folded 256-column kernel, then un-pad fold
    auto wf = w.permute({1, 0, 2}).flip(-1).contiguous();
    const int Tpad = (int)Tin + 2 * (int)pad;
    auto gxpad = conv1d_fwd_mfma(gz, wf, nullptr, dil, (K - 1) * (int)dil,
                                 false, 0, 0.0, false, Tpad);
    if (pad == 0) return gxpad;
    auto gx = torch::empty({B, (int)Cin, (int)Tin}, gz.options());
    run_unpad_fold(gxpad, gx, B * (int)Cin, (int)Tin, (int)pad, reflect,
                   Tpad);
    return gx;
  }
  if (cin1_eligible(Cin, Cout, K, stride, groups)) {
    auto gxpad = cin1_bwd_data_pad(gz, w, dil, pad, Tin);
    if (pad == 0) return gxpad;
    auto gx = torch::empty({B, 1, (int)Tin}, gz.options());
    run_unpad_fold(gxpad, gx, B, (int)Tin, (int)pad, reflect,
                   (int)Tin + 2 * (int)pad);
    return gx;
  }
  if (conv1d_mfma_eligible(gz, Cout, K, stride, dil, groups))
    return conv1d_bwd_data_mfma(gz, w, dil, pad, reflect, Tin);
  if (conv1d_mfma_eligible_any(gz))
    return conv1d_bwd_data_mfma_generic(gz, w, stride, dil, pad, reflect,
                                        groups, Tin);
  auto gx = torch::empty({B, (int)Cin, (int)Tin}, gz.options());
  constexpr int BT = 64, BC = 4;
  dim3 grid(cdiv(Tin, BT), cdiv(Cin, BC), B);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gz.scalar_type(), "conv1d_bwd_data", [&] {
        hipLaunchKernelGGL((conv1d_bwd_data_kernel<scalar_t, BT, BC>), grid,
                           dim3(BT * BC), 0, stream,
                           gz.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                           gx.data_ptr<scalar_t>(), B, Cin, (int)Tin, Cout,
                           To, K, (int)stride, (int)dil, (int)pad, reflect,
                           (int)groups);
      });
  return gx;
}

torch::Tensor conv1d_bwd_weight(torch::Tensor gz, torch::Tensor x,
                                int64_t stride, int64_t dil, int64_t pad,
                                bool reflect, int64_t groups, int64_t K,
                                bool act_before, double slope) {
  CHECK_IN(gz);
  CHECK_IN(x);
  const int B = gz.size(0), Cout = gz.size(1), To = gz.size(2);
  const int Cin = x.size(1), Tin = x.size(2);
  const int Cpg = Cin / (int)groups;
  // NOTE: a dedicated cin1 weight-grad kernel (cin1.hip) measured 870us
  // per call vs the mfma16 path's ~100us on the k15 shape — the 16-row
  // MFMA tile wins on the (tiny N, huge B*T) reduction; keep mfma16.
  if (conv1d_mfma_eligible_any(x))
    return conv1d_bwd_weight_mfma(gz, x, stride, dil, pad, reflect, groups,
                                  K, /*act_a=*/false,
                                  /*act_b=*/act_before, slope);
  auto gw = torch::empty({Cout, Cpg, (int)K},
                         x.options().dtype(torch::kFloat));
  // chunk (b,t) so small weights still fill the chip
  int64_t bt = (int64_t)B * To;
  int nchunks = (int)std::min<int64_t>(
      std::max<int64_t>(1, bt / 8192),
      std::max<int64_t>(1, 4096 / std::max(1, cdiv(Cpg * K, 256) *
                                                  (int)Cout)));
  if (nchunks > 1) gw.zero_();
  dim3 grid(cdiv(Cpg * K, 256), Cout, nchunks);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "conv1d_bwd_weight", [&] {
        hipLaunchKernelGGL((conv1d_bwd_weight_kernel<scalar_t>), grid,
                           dim3(256), 0, stream, gz.data_ptr<scalar_t>(),
                           x.data_ptr<scalar_t>(), gw.data_ptr<float>(), B,
                           Cin, Tin, Cout, To, (int)K, (int)stride, (int)dil,
                           (int)pad, reflect, (int)groups, (float)slope,
                           act_before, nchunks);
      });
  return gw;
}

torch::Tensor conv_transpose1d_fwd(torch::Tensor x, torch::Tensor w,
                                   c10::optional<torch::Tensor> bias,
                                   int64_t stride, int64_t pad,
                                   int64_t out_pad, double slope,
                                   bool act_before) {
  CHECK_IN(x);
  CHECK_IN(w);
  const int B = x.size(0), Cin = x.size(1), Tin = x.size(2);
  const int Cout = w.size(1), K = w.size(2);
  const int To = (Tin - 1) * (int)stride - 2 * (int)pad + K + (int)out_pad;
  torch::Tensor bias_f;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  if (conv1d_mfma_eligible_any(x))
    return conv_transpose1d_fwd_mfma(
        x, w, bias_p, stride, pad, out_pad, ACT_NONE, slope, act_before, To);
  auto y = torch::empty({B, Cout, To}, x.options());
  constexpr int BT = 64, BC = 4;
  dim3 grid(cdiv(To, BT), cdiv(Cout, BC), B);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "convt1d_fwd", [&] {
        hipLaunchKernelGGL((convt1d_fwd_kernel<scalar_t, BT, BC>), grid,
                           dim3(BT * BC), 0, stream,
                           x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                           bias_p, y.data_ptr<scalar_t>(), B, Cin, Tin, Cout,
                           To, K, (int)stride, (int)pad, (float)slope,
                           act_before);
      });
  return y;
}

torch::Tensor conv_transpose1d_bwd_data(torch::Tensor gy, torch::Tensor w,
                                        int64_t stride, int64_t pad,
                                        int64_t out_pad, int64_t Tin) {
  // gx[ci][m] = sum_{co,k} w[ci][co][k] * gy[co][m*s - p + k]
  // == direct conv of gy with weight(Cin, Cout, K), stride s, pad p.
  CHECK_IN(gy);
  CHECK_IN(w);
  const int B = gy.size(0), Cout = gy.size(1), To = gy.size(2);
  const int Cin = w.size(0), K = w.size(2);
  // gx[ci][m] = direct conv of gy with weight viewed (Cin, Cout, K):
  // reuse the conv1d_fwd dispatcher (MFMA generic path for bf16).
  if (conv1d_mfma_eligible_any(gy))
    return conv1d_fwd_mfma_generic(gy, w, nullptr, stride, 1, pad, false, 1,
                                   ACT_NONE, 0.0, false, Tin);
  auto gx = torch::empty({B, Cin, (int)Tin}, gy.options());
  constexpr int BT = 64, BC = 4;
  dim3 grid(cdiv(Tin, BT), cdiv(Cin, BC), B);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gy.scalar_type(), "convt1d_bwd_data", [&] {
        hipLaunchKernelGGL((conv1d_fwd_kernel<scalar_t, BT, BC>), grid,
                           dim3(BT * BC), 0, stream,
                           gy.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                           nullptr, gx.data_ptr<scalar_t>(), B, Cout, To,
                           Cin, (int)Tin, K, (int)stride, 1, (int)pad, false,
                           1, ACT_NONE, 0.f, false);
      });
  return gx;
}

torch::Tensor conv_transpose1d_bwd_weight(torch::Tensor gy, torch::Tensor x,
                                          int64_t stride, int64_t pad,
                                          int64_t out_pad, int64_t K,
                                          bool act_before, double slope) {
  CHECK_IN(gy);
  CHECK_IN(x);
  const int B = gy.size(0), Cout = gy.size(1), To = gy.size(2);
  const int Cin = x.size(1), Tin = x.size(2);
  // gw[ci][co][k] = sum_{b,m} pre(x[ci][m]) * gy[co][m*s - p + k]:
  // the generalized MFMA weight-grad kernel with roles swapped
  // (A = x [act_a], B = gy), dil=1.
  if (conv1d_mfma_eligible_any(x))
    return conv1d_bwd_weight_mfma(x, gy, stride, 1, pad, false, 1, K,
                                  /*act_a=*/act_before, /*act_b=*/false,
                                  slope);
  auto gw = torch::empty({Cin, Cout, (int)K},
                         x.options().dtype(torch::kFloat));
  int64_t bm = (int64_t)B * Tin;
  int nchunks = (int)std::min<int64_t>(
      std::max<int64_t>(1, bm / 8192),
      std::max<int64_t>(1, 4096 / std::max(1, cdiv(Cout * K, 256) * Cin)));
  if (nchunks > 1) gw.zero_();
  dim3 grid(cdiv(Cout * K, 256), Cin, nchunks);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "convt1d_bwd_weight", [&] {
        hipLaunchKernelGGL((convt1d_bwd_weight_kernel<scalar_t>), grid,
                           dim3(256), 0, stream, gy.data_ptr<scalar_t>(),
                           x.data_ptr<scalar_t>(), gw.data_ptr<float>(), B,
                           Cin, Tin, Cout, To, (int)K, (int)stride,
                           (int)pad, (float)slope, act_before, nchunks);
      });
  return gw;
}

torch::Tensor act_bwd(torch::Tensor gy, torch::Tensor y, int64_t act,
                      double slope) {
  CHECK_IN(gy);
  CHECK_IN(y);
  auto gz = torch::empty_like(gy);
  const int64_t n = gy.numel();
  const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gy.scalar_type(), "act_bwd", [&] {
        hipLaunchKernelGGL((act_bwd_kernel<scalar_t>), dim3(blocks),
                           dim3(256), 0, stream, gy.data_ptr<scalar_t>(),
                           y.data_ptr<scalar_t>(), gz.data_ptr<scalar_t>(),
                           n, (int)act, (float)slope);
      });
  return gz;
}

std::vector<torch::Tensor> act_bwd_biasgrad(torch::Tensor gy,
                                            torch::Tensor y, int64_t act,
                                            double slope) {
  CHECK_IN(gy);
  CHECK_IN(y);
  const int B = gy.size(0), C = gy.size(1), Tlen = gy.size(2);
  auto gz = torch::empty_like(gy);
  const int nchunks = (int)std::min<int64_t>(
      std::max<int64_t>(1, (int64_t)B * Tlen / 16384),
      std::max<int64_t>(1, 2048 / C));
  auto gb = nchunks == 1
                ? torch::empty({C}, gy.options().dtype(torch::kFloat))
                : torch::zeros({C}, gy.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gy.scalar_type(), "act_bwd_biasgrad", [&] {
        hipLaunchKernelGGL((act_bwd_biasgrad_kernel<scalar_t>),
                           dim3(C, nchunks), dim3(256), 0, stream,
                           gy.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           gz.data_ptr<scalar_t>(), gb.data_ptr<float>(),
                           B, C, Tlen, (int)act, (float)slope, nchunks);
      });
  return {gz, gb};
}

void leaky_bwd_inplace(torch::Tensor gx, torch::Tensor x, double slope) {
  CHECK_IN(gx);
  CHECK_IN(x);
  const int64_t n = gx.numel();
  const int blocks = (int)std::min<int64_t>(2048, (n + 255) / 256);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gx.scalar_type(), "leaky_bwd_inplace", [&] {
        hipLaunchKernelGGL((leaky_bwd_inplace_kernel<scalar_t>),
                           dim3(blocks), dim3(256), 0, stream,
                           gx.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                           n, (float)slope);
      });
}

torch::Tensor conv1d_bias_grad(torch::Tensor gz) {
  CHECK_IN(gz);
  const int B = gz.size(0), C = gz.size(1), Tlen = gz.size(2);
  const int nchunks = (int)std::min<int64_t>(
      std::max<int64_t>(1, (int64_t)B * Tlen / 16384),
      std::max<int64_t>(1, 2048 / C));
  auto gb = nchunks == 1
                ? torch::empty({C}, gz.options().dtype(torch::kFloat))
                : torch::zeros({C}, gz.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gz.scalar_type(), "bias_grad", [&] {
        hipLaunchKernelGGL((bias_grad_kernel<scalar_t>), dim3(C, nchunks),
                           dim3(256), 0, stream, gz.data_ptr<scalar_t>(),
                           gb.data_ptr<float>(), B, C, Tlen, nchunks);
      });
  return gb;
}

std::vector<torch::Tensor> wn_fold(torch::Tensor v, torch::Tensor g,
                                   torch::ScalarType out_dtype) {
  CHECK_IN(v);
  TORCH_CHECK(v.scalar_type() == torch::kFloat, "wn_fold: fp32 v");
  const int C0 = v.size(0);
  const int M = (int)(v.numel() / C0);
  auto gc = g.to(torch::kFloat).flatten().contiguous();
  auto w = torch::empty(v.sizes(), v.options().dtype(out_dtype));
  auto n = torch::empty({C0}, v.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, out_dtype, "wn_fold", [&] {
        hipLaunchKernelGGL((wn_fold_kernel<scalar_t>), dim3(C0), dim3(256),
                           0, stream, v.data_ptr<float>(),
                           gc.data_ptr<float>(), w.data_ptr<scalar_t>(),
                           n.data_ptr<float>(), C0, M);
      });
  return {w, n};
}

std::vector<torch::Tensor> wn_bwd(torch::Tensor gw, torch::Tensor v,
                                  torch::Tensor g, torch::Tensor n) {
  CHECK_IN(gw);
  CHECK_IN(v);
  const int C0 = v.size(0);
  const int M = (int)(v.numel() / C0);
  auto gc = g.to(torch::kFloat).flatten().contiguous();
  auto gv = torch::empty_like(v);
  auto gg = torch::empty({C0}, v.options());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(wn_bwd_kernel, dim3(C0), dim3(256), 0, stream,
                     gw.data_ptr<float>(), v.data_ptr<float>(),
                     gc.data_ptr<float>(), n.data_ptr<float>(),
                     gv.data_ptr<float>(), gg.data_ptr<float>(), C0, M);
  return {gv, gg};
}

// (a per-step-H2D multi_tensor_adam variant lived here; superseded by
// multi_tensor_adam_tbl below, which caches the device tables across
// steps — zero per-step H2D, hipGraph-capture safe.  See ops/adam.py.)

void multi_tensor_adam_tbl(torch::Tensor tbl, torch::Tensor chunks,
                           int64_t nt, double lr, double beta1, double beta2,
                           double eps, double wd, int64_t step) {
  TORCH_CHECK(tbl.is_cuda() && chunks.is_cuda());
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();
  int64_t* tp = tbl.data_ptr<int64_t>();
  hipLaunchKernelGGL(multi_tensor_adam_kernel,
                     dim3((int)chunks.numel()), dim3(256), 0, stream,
                     (float**)tp, (float**)(tp + nt), (float**)(tp + 2 * nt),
                     (float**)(tp + 3 * nt), tp + 4 * nt,
                     chunks.data_ptr<int64_t>(), (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)wd, bc1, bc2);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv1d_fwd", &conv1d_fwd);
  m.def("conv1d_bwd_data", &conv1d_bwd_data);
  m.def("conv1d_bwd_weight", &conv1d_bwd_weight);
  m.def("conv_transpose1d_fwd", &conv_transpose1d_fwd);
  m.def("conv_transpose1d_bwd_data", &conv_transpose1d_bwd_data);
  m.def("conv_transpose1d_bwd_weight", &conv_transpose1d_bwd_weight);
  m.def("act_bwd", &act_bwd);
  m.def("act_bwd_biasgrad", &act_bwd_biasgrad);
  m.def("leaky_bwd_inplace", &leaky_bwd_inplace);
  m.def("conv1d_bias_grad", &conv1d_bias_grad);
  m.def("wn_fold", &wn_fold);
  m.def("wn_bwd", &wn_bwd);
  m.def("stft_frame", &stft_frame);
  m.def("stft_frame_bwd", &stft_frame_bwd);
  m.def("complex_mag_clamp", &complex_mag_clamp);
  m.def("mag_bwd", &mag_bwd);
  m.def("avg_pool1d_421", &avg_pool1d_421);
  m.def("avg_pool1d_421_bwd", &avg_pool1d_421_bwd);
  m.def("stft_loss_reduce", &stft_loss_reduce);
  m.def("stft_loss_bwd", &stft_loss_bwd);
  m.def("multi_l1_fwd", &multi_l1_fwd);
  m.def("multi_l1_bwd", &multi_l1_bwd);
  m.def("conv1d_fwd_gemm", &conv1d_fwd_gemm);
  m.def("gemm_bt", &gemm_bt);
  m.def("unpad_fold", &unpad_fold_py);
  m.def("conv1d_bwd_weight_gemm", &conv1d_bwd_weight_gemm);
  m.def("pqmf_gather", &pqmf_gather);
  m.def("pqmf_scatter", &pqmf_scatter);
  m.def("multi_tensor_adam_tbl", &multi_tensor_adam_tbl);
}
