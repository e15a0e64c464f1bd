// Shared device helpers for melgan_amd gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>

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

}  // namespace melgan
