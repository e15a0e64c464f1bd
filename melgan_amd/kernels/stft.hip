// Fused STFT-loss pipeline pieces on gfx950.
//
// torch.stft on ROCm lowers to ~6 kernels per call (reflect-pad
// materialisation, as_strided copy, window mul, rocFFT, abs, clamp).
// Here the framing prologue (reflect pad + window, no materialised pad)
// and the |.|+clamp epilogue are ONE kernel each; the batched R2C FFT
// itself stays on rocFFT via torch.fft.rfft between them (its autograd
// is handled by torch).  All fp32 (GAN loss numerics).
//
//   stft_frame(x, fft, hop, win, window) -> (B, F, fft) frames
//   stft_frame_bwd(gframes, window, ...) -> gx  (overlap-add gather +
//                                           reflect un-pad fold)
//   complex_mag_clamp(spec_ri, min)      -> |S| clamped
//   mag_bwd(gmag, spec_ri, mag, min)     -> gspec_ri

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace melgan {

// frames[b][f][j] = wpad[j] * xpad[b][f*hop + j]
// xpad = reflect-pad(x, fft/2);  wpad = window centered in fft
__global__ void stft_frame_kernel(const float* __restrict__ x,
                                  const float* __restrict__ window,
                                  float* __restrict__ frames, int B, int S,
                                  int F, int fft, int hop, int win) {
  const int off = (fft - win) / 2;
  const int pad = fft / 2;
  const int64_t n = (int64_t)B * F * fft;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const int j = (int)(i % fft);
    const int64_t bf = i / fft;
    const int f = (int)(bf % F);
    const int b = (int)(bf / F);
    float v = 0.f;
    const int jw = j - off;
    if (jw >= 0 && jw < win) {
      const int q = f * hop + j - pad;  // un-padded coordinate
      v = window[jw] * x[(int64_t)b * S + reflect_idx(q, S)];
    }
    frames[i] = v;
  }
}

// gxpad[b][q] = sum_{f: 0<=q-f*hop<fft} gframes[b][f][q-f*hop] * wpad[...]
__global__ void stft_frame_bwd_kernel(const float* __restrict__ gframes,
                                      const float* __restrict__ window,
                                      float* __restrict__ gxpad, int B,
                                      int Spad, int F, int fft, int hop,
                                      int win) {
  const int off = (fft - win) / 2;
  const int64_t n = (int64_t)B * Spad;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const int q = (int)(i % Spad);
    const int b = (int)(i / Spad);
    // contributing frames: f*hop <= q < f*hop + fft
    const int f_lo = (q >= fft) ? ((q - fft) / hop + 1) : 0;
    float acc = 0.f;
    for (int f = f_lo; f <= q / hop && f < F; ++f) {
      const int j = q - f * hop;
      const int jw = j - off;
      if (jw >= 0 && jw < win)
        acc += gframes[((int64_t)b * F + f) * fft + j] * window[jw];
    }
    gxpad[i] = acc;
  }
}

__global__ void mag_clamp_kernel(const float* __restrict__ spec_ri,
                                 float* __restrict__ mag, int64_t n,
                                 float mn) {
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const float re = spec_ri[2 * i];
    const float im = spec_ri[2 * i + 1];
    mag[i] = fmaxf(sqrtf(re * re + im * im), mn);
  }
}

// d|S|/dS = S/|S|; zero where the clamp was active (|S| < mn)
__global__ void mag_bwd_kernel(const float* __restrict__ gmag,
                               const float* __restrict__ spec_ri,
                               float* __restrict__ gspec_ri, int64_t n,
                               float mn) {
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const float re = spec_ri[2 * i];
    const float im = spec_ri[2 * i + 1];
    const float r = sqrtf(re * re + im * im);
    float s = 0.f;
    if (r >= mn && r > 0.f) s = gmag[i] / r;
    gspec_ri[2 * i] = re * s;
    gspec_ri[2 * i + 1] = im * s;
  }
}

// declared in conv1d_mfma.hip
template <typename T>
__global__ void unpad_fold_stft_kernel(const T* __restrict__ gxpad,
                                       T* __restrict__ gx, int BC, int Tin,
                                       int pad, bool reflect, int src_len) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t n = (int64_t)BC * Tin;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  const int Tpad = Tin + 2 * pad;
  for (int64_t idx = i; idx < n; idx += step) {
    const int64_t bc = idx / Tin;
    const int j = (int)(idx - bc * Tin);
    const T* src = gxpad + bc * src_len;
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
// K4: discriminator input down-scaling  avg_pool1d(k=4, s=2, pad=1,
// count_include_pad=False)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void avg_pool1d_kernel(const T* __restrict__ x,
                                  T* __restrict__ y, int64_t BC, int Tin,
                                  int To) {
  const int64_t n = BC * To;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const int64_t bc = i / To;
    const int t = (int)(i - bc * To);
    const T* row = x + bc * Tin;
    float acc = 0.f;
    int cnt = 0;
    const int base = t * 2 - 1;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int q = base + j;
      if (q >= 0 && q < Tin) {
        acc += static_cast<float>(row[q]);
        ++cnt;
      }
    }
    y[i] = static_cast<T>(acc / cnt);
  }
}

template <typename T>
__global__ void avg_pool1d_bwd_kernel(const T* __restrict__ gy,
                                      T* __restrict__ gx, int64_t BC,
                                      int Tin, int To) {
  const int64_t n = BC * Tin;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const int64_t bc = i / Tin;
    const int q = (int)(i - bc * Tin);
    const T* gr = gy + bc * To;
    float acc = 0.f;
    // outputs t whose window [2t-1, 2t+2] contains q
    for (int t = max(0, (q - 2 + 1) / 2); t <= (q + 1) / 2 && t < To; ++t) {
      const int base = t * 2 - 1;
      if (q < base || q > base + 3) continue;
      int cnt = 0;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int qq = base + j;
        if (qq >= 0 && qq < Tin) ++cnt;
      }
      acc += static_cast<float>(gr[t]) / cnt;
    }
    gx[i] = static_cast<T>(acc);
  }
}

// ---------------------------------------------------------------------------
// K11: fused STFT-loss reductions over a pair of magnitude tensors:
//   out[0] = sum((my - mx)^2)   (spectral-convergence numerator^2)
//   out[1] = sum(my^2)          (denominator^2)
//   out[2] = sum(|log mx - log my|)
// ---------------------------------------------------------------------------
__global__ void stft_loss_reduce_kernel(const float* __restrict__ mx,
                                        const float* __restrict__ my,
                                        float* __restrict__ out, int64_t n) {
  __shared__ float red[3][256];
  float a = 0.f, b = 0.f, c = 0.f;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const float x = mx[i], y = my[i];
    const float d = y - x;
    a += d * d;
    b += y * y;
    c += fabsf(logf(x) - logf(y));
  }
  red[0][threadIdx.x] = a;
  red[1][threadIdx.x] = b;
  red[2][threadIdx.x] = c;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off)
      for (int r = 0; r < 3; ++r)
        red[r][threadIdx.x] += red[r][threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicAdd(&out[0], red[0][0]);
    atomicAdd(&out[1], red[1][0]);
    atomicAdd(&out[2], red[2][0]);
  }
}

// backward wrt mx only (my is the no-grad target):
//   d sc/d mx = (mx - my) / (sqrt(num)*sqrt(den))
//   d l1/d mx = sign(log mx - log my) / (n * mx)
// gx = gsc * dsc + gl1 * dl1
__global__ void stft_loss_bwd_kernel(const float* __restrict__ mx,
                                     const float* __restrict__ my,
                                     const float* __restrict__ sums,
                                     const float* __restrict__ gouts,
                                     float* __restrict__ gx, int64_t n) {
  const float gsc = gouts[0], gl1 = gouts[1];
  const float num = sqrtf(sums[0]);
  const float den = fmaxf(sqrtf(sums[1]), 1e-7f);
  const float inv_scden = 1.f / fmaxf(num * den, 1e-12f);
  const float inv_n = 1.f / (float)n;
  const int64_t step = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += step) {
    const float x = mx[i], y = my[i];
    float g = gsc * (x - y) * inv_scden;
    const float ld = logf(x) - logf(y);
    g += gl1 * ((ld > 0.f) ? 1.f : (ld < 0.f ? -1.f : 0.f)) * inv_n / x;
    gx[i] = g;
  }
}

// ---------------------------------------------------------------------------
// K11b: multi-tensor feature-matching L1 over ~20 (fake, real) feature
// pairs: ONE forward reduction launch, ONE backward launch.
// chunk_map packs (tensor_idx << 48) | elem_offset (adam-style table).
// ---------------------------------------------------------------------------
template <typename T>
__global__ void multi_l1_fwd_kernel(T** __restrict__ f_ptrs,
                                    T** __restrict__ r_ptrs,
                                    const int64_t* __restrict__ numels,
                                    const int64_t* __restrict__ chunk_map,
                                    float* __restrict__ partial, int nt) {
  __shared__ float red[256];
  const int64_t packed = chunk_map[blockIdx.x];
  const int ti = (int)(packed >> 48);
  const int64_t off = packed & ((1LL << 48) - 1);
  const int64_t n = numels[ti];
  const T* f = f_ptrs[ti] + off;
  const T* r = r_ptrs[ti] + off;
  const int64_t len = min((int64_t)(1 << 16), n - off);
  // per-tensor mean then mean over tensors: weight 1/(nt * n)
  const float wgt = 1.f / ((float)nt * (float)n);
  float acc = 0.f;
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x)
    acc += fabsf(static_cast<float>(f[i]) - static_cast<float>(r[i]));
  red[threadIdx.x] = acc * wgt;
  __syncthreads();
  for (int o = 128; o > 0; o >>= 1) {
    if (threadIdx.x < o) red[threadIdx.x] += red[threadIdx.x + o];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(partial, red[0]);
}

template <typename T>
__global__ void multi_l1_bwd_kernel(T** __restrict__ f_ptrs,
                                    T** __restrict__ r_ptrs,
                                    T** __restrict__ g_ptrs,
                                    const int64_t* __restrict__ numels,
                                    const int64_t* __restrict__ chunk_map,
                                    const float* __restrict__ gout, int nt) {
  const int64_t packed = chunk_map[blockIdx.x];
  const int ti = (int)(packed >> 48);
  const int64_t off = packed & ((1LL << 48) - 1);
  const int64_t n = numels[ti];
  const T* f = f_ptrs[ti] + off;
  const T* r = r_ptrs[ti] + off;
  T* g = g_ptrs[ti] + off;
  const int64_t len = min((int64_t)(1 << 16), n - off);
  const float scale = gout[0] / ((float)nt * (float)n);
  for (int64_t i = threadIdx.x; i < len; i += blockDim.x) {
    const float d = static_cast<float>(f[i]) - static_cast<float>(r[i]);
    g[i] = static_cast<T>(d > 0.f ? scale : (d < 0.f ? -scale : 0.f));
  }
}

}  // namespace melgan

using namespace melgan;

static inline int gblocks(int64_t n) {
  return (int)std::min<int64_t>(2048, (n + 255) / 256);
}

torch::Tensor stft_frame(torch::Tensor x, int64_t fft, int64_t hop,
                         int64_t win, torch::Tensor window) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kFloat);
  const int B = x.size(0), S = x.size(1);
  const int F = S / (int)hop + 1;  // center=True frame count
  auto frames = torch::empty({B, F, (int)fft}, x.options());
  auto wc = window.contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  const int64_t n = (int64_t)B * F * fft;
  hipLaunchKernelGGL(stft_frame_kernel, dim3(gblocks(n)), dim3(256), 0,
                     stream, x.data_ptr<float>(), wc.data_ptr<float>(),
                     frames.data_ptr<float>(), B, S, F, (int)fft, (int)hop,
                     (int)win);
  return frames;
}

torch::Tensor stft_frame_bwd(torch::Tensor gframes, torch::Tensor window,
                             int64_t fft, int64_t hop, int64_t win,
                             int64_t S) {
  TORCH_CHECK(gframes.is_cuda() && gframes.is_contiguous());
  const int B = gframes.size(0), F = gframes.size(1);
  const int pad = (int)fft / 2;
  const int Spad = (int)S + 2 * pad;
  auto gxpad = torch::empty({B, Spad},
                            gframes.options().dtype(torch::kFloat));
  auto wc = window.contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(stft_frame_bwd_kernel,
                     dim3(gblocks((int64_t)B * Spad)), dim3(256), 0, stream,
                     gframes.data_ptr<float>(), wc.data_ptr<float>(),
                     gxpad.data_ptr<float>(), B, Spad, F, (int)fft,
                     (int)hop, (int)win);
  auto gx = torch::empty({B, (int)S}, gxpad.options());
  hipLaunchKernelGGL(unpad_fold_stft_kernel<float>,
                     dim3(gblocks((int64_t)B * S)), dim3(256), 0, stream,
                     gxpad.data_ptr<float>(), gx.data_ptr<float>(), B,
                     (int)S, pad, true, Spad);
  return gx;
}

torch::Tensor complex_mag_clamp(torch::Tensor spec_ri, double mn) {
  TORCH_CHECK(spec_ri.is_cuda() && spec_ri.is_contiguous());
  auto sizes = spec_ri.sizes().vec();
  TORCH_CHECK(sizes.back() == 2);
  sizes.pop_back();
  auto mag = torch::empty(sizes, spec_ri.options());
  const int64_t n = mag.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mag_clamp_kernel, dim3(gblocks(n)), dim3(256), 0,
                     stream, spec_ri.data_ptr<float>(),
                     mag.data_ptr<float>(), n, (float)mn);
  return mag;
}

torch::Tensor mag_bwd(torch::Tensor gmag, torch::Tensor spec_ri, double mn) {
  TORCH_CHECK(gmag.is_cuda() && gmag.is_contiguous());
  auto gspec = torch::empty_like(spec_ri);
  const int64_t n = gmag.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mag_bwd_kernel, dim3(gblocks(n)), dim3(256), 0, stream,
                     gmag.data_ptr<float>(), spec_ri.data_ptr<float>(),
                     gspec.data_ptr<float>(), n, (float)mn);
  return gspec;
}

torch::Tensor avg_pool1d_421(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int64_t BC = x.size(0) * x.size(1);
  const int Tin = x.size(2);
  const int To = (Tin + 2 * 1 - 4) / 2 + 1;
  auto y = torch::empty({x.size(0), x.size(1), To}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "avgpool", [&] {
        hipLaunchKernelGGL((avg_pool1d_kernel<scalar_t>),
                           dim3(gblocks(BC * To)), dim3(256), 0, stream,
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           BC, Tin, To);
      });
  return y;
}

torch::Tensor avg_pool1d_421_bwd(torch::Tensor gy, int64_t Tin) {
  TORCH_CHECK(gy.is_cuda() && gy.is_contiguous());
  const int64_t BC = gy.size(0) * gy.size(1);
  const int To = gy.size(2);
  auto gx = torch::empty({gy.size(0), gy.size(1), (int)Tin}, gy.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gy.scalar_type(), "avgpool_bwd", [&] {
        hipLaunchKernelGGL((avg_pool1d_bwd_kernel<scalar_t>),
                           dim3(gblocks(BC * Tin)), dim3(256), 0, stream,
                           gy.data_ptr<scalar_t>(), gx.data_ptr<scalar_t>(),
                           BC, (int)Tin, To);
      });
  return gx;
}

torch::Tensor stft_loss_reduce(torch::Tensor mx, torch::Tensor my) {
  TORCH_CHECK(mx.is_cuda() && mx.is_contiguous() && my.is_contiguous());
  auto out = torch::zeros({3}, mx.options());
  const int64_t n = mx.numel();
  auto stream = at::hip::getCurrentHIPStream();
  const int rb = (int)std::min<int64_t>(512, (n + 255) / 256);
  hipLaunchKernelGGL(stft_loss_reduce_kernel, dim3(rb), dim3(256),
                     0, stream, mx.data_ptr<float>(), my.data_ptr<float>(),
                     out.data_ptr<float>(), n);
  return out;
}

torch::Tensor stft_loss_bwd(torch::Tensor mx, torch::Tensor my,
                            torch::Tensor sums, torch::Tensor gouts) {
  auto gx = torch::empty_like(mx);
  const int64_t n = mx.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(stft_loss_bwd_kernel, dim3(gblocks(n)), dim3(256), 0,
                     stream, mx.data_ptr<float>(), my.data_ptr<float>(),
                     sums.data_ptr<float>(), gouts.data_ptr<float>(),
                     gx.data_ptr<float>(), n);
  return gx;
}

// table layout: [f*nt | r*nt | g*nt | numel*nt] int64 on device
torch::Tensor multi_l1_fwd(torch::Tensor tbl, torch::Tensor chunks,
                           int64_t nt, torch::ScalarType dt) {
  auto out = torch::zeros({1}, torch::TensorOptions()
                                   .dtype(torch::kFloat)
                                   .device(tbl.device()));
  int64_t* tp = tbl.data_ptr<int64_t>();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, dt, "multi_l1_fwd", [&] {
        hipLaunchKernelGGL((multi_l1_fwd_kernel<scalar_t>),
                           dim3((int)chunks.numel()), dim3(256), 0, stream,
                           (scalar_t**)tp, (scalar_t**)(tp + nt),
                           tp + 3 * nt, chunks.data_ptr<int64_t>(),
                           out.data_ptr<float>(), (int)nt);
      });
  return out;
}

void multi_l1_bwd(torch::Tensor tbl, torch::Tensor chunks, int64_t nt,
                  torch::Tensor gout, torch::ScalarType dt) {
  int64_t* tp = tbl.data_ptr<int64_t>();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, dt, "multi_l1_bwd", [&] {
        hipLaunchKernelGGL((multi_l1_bwd_kernel<scalar_t>),
                           dim3((int)chunks.numel()), dim3(256), 0, stream,
                           (scalar_t**)tp, (scalar_t**)(tp + nt),
                           (scalar_t**)(tp + 2 * nt), tp + 3 * nt,
                           chunks.data_ptr<int64_t>(),
                           gout.data_ptr<float>(), (int)nt);
      });
}
