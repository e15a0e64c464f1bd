// 256x256-tile bf16 GEMM tier (guide §5 "256² 8-phase" class structure,
// reconstructed):  C[M][N] = A[M][Kp] * Bt[N][Kp]^T
//
// vs the 128² glds engine (gemm.hip): one 512-thread block per 256²
// C-tile (8 waves = 2 per SIMD), BK=64 K-tiles double-buffered in
// 128 KiB LDS, RAW s_barrier + lgkmcnt(0) phase boundaries so the
// in-flight global_load_lds prefetch is NOT drained at every barrier
// (the 128² engine's __syncthreads() emits vmcnt(0) each K-step — the
// documented ~20% structural stall); the glds queue drains ONCE per
// K-tile at the buffer swap.  T2 bank swizzle (byte ^= (row&7)<<4,
// involution on the glds SOURCE address and the ds_read offset),
// T5 s_setprio around the MFMA clusters, T1 bijective XCD remap.
//
// Per wave: 128(M) x 64(N) output = 8x4 16² fragments; per K-tile
// 4 phases x 16 MFMA.  Each phase issues one half-tile (2 glds/thread)
// of the NEXT K-tile into the other buffer.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

namespace melgan {

typedef __bf16 bf16x8q __attribute__((ext_vector_type(8)));
typedef float f32x4q __attribute__((ext_vector_type(4)));

constexpr int QBM = 256, QBN = 256, QBK = 64;
constexpr int QROWB = QBK * 2;          // 128 B per image row
constexpr int QIMG = QBM * QBK;         // elements per operand image
                                        // (256 x 64 bf16 = 32 KiB)
// LDS: buf b at b*2*QIMG elements; A image first, B image second

DEV_INLINE int qswz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

// stage one 128-row half of an operand image: 2 glds of 16 B per thread
// (512 threads x 2 x 16 B = 16 KiB).  The glds LDS destination is the
// WAVE-UNIFORM chunk base (hardware writes lane l at base + l*16); the
// T2 swizzle is pre-applied to the per-lane SOURCE address (rule 21).
DEV_INLINE void qstage_half(const unsigned short* __restrict__ src,
                            int64_t row0, int64_t rows, int64_t ld,
                            int kcol0, unsigned short* lds_img,
                            int half, int wid, int lane) {
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int base16 = p * 512 + wid * 64;    // wave chunk (16-B units)
    const int off16 = base16 + lane;
    const int r = half * 128 + (off16 >> 3);  // image row
    const int s = off16 & 7;                  // 16-B slot in row
    int64_t srow = row0 + r;
    if (srow >= rows) srow = rows - 1;        // clamp (masked at C-write)
    const unsigned short* g =
        src + srow * ld + kcol0 + ((s ^ (r & 7)) << 3);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(lds_img +
                                                  (int64_t)base16 * 8 +
                                                  half * 128 * QBK),
        16, 0, 0);
  }
}

DEV_INLINE bf16x8q qread(const unsigned short* img, int r, int kelem) {
  // 8 bf16 at (row r, elements [kelem, kelem+8)); kelem mult of 8
  const int byte = r * QROWB + qswz(r, kelem * 2);
  return *(const bf16x8q*)((const char*)img + byte);
}

__global__ void __launch_bounds__(512, 2)
gemm_bt_256_kernel(const unsigned short* __restrict__ A,   // [M][Kp]
                   const unsigned short* __restrict__ Bt,  // [N][Kp]
                   const float* __restrict__ bias,
                   unsigned short* __restrict__ y,  // [Bb][M][To]
                   int M, int64_t N, int Kp, int To, int act, float slope,
                   int nwgx, int ilv) {
  // 128 KiB STATIC LDS (dynamic shared memory is capped at 64 KiB on
  // this runtime; static launches may use the full 160 KiB)
  __shared__ __attribute__((aligned(16))) unsigned short smem[4 * QIMG];

  // T1: bijective XCD-aware remap of the N-tile index
  int bx = blockIdx.x;
  {
    const int q = nwgx / 8, rr = nwgx % 8;
    if (q > 0) {
      const int xcd = bx % 8, idx = bx / 8;
      bx = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
      if (bx >= nwgx) bx = blockIdx.x;  // safety (non-multiple tail)
    }
  }
  const int m0 = blockIdx.y * QBM;
  const int64_t n0 = (int64_t)bx * QBN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;        // 0..1: M half
  const int wc = wid & 3;         // 0..3: N quarter
  const int row = lane & 15;
  const int kk0 = (lane >> 4) * 8;

  f32x4q acc[8][4] = {};

  const int ntiles = Kp / QBK;
  unsigned short* buf0 = smem;
  unsigned short* buf1 = smem + 2 * QIMG;

  // prologue: all 4 halves of K-tile 0 into buf0
  qstage_half(A, m0, M, Kp, 0, buf0, 0, wid, lane);
  qstage_half(A, m0, M, Kp, 0, buf0, 1, wid, lane);
  qstage_half(Bt, n0, N, Kp, 0, buf0 + QIMG, 0, wid, lane);
  qstage_half(Bt, n0, N, Kp, 0, buf0 + QIMG, 1, wid, lane);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < ntiles; ++t) {
    unsigned short* cur = (t & 1) ? buf1 : buf0;
    unsigned short* nxt = (t & 1) ? buf0 : buf1;
    const unsigned short* ia = cur;
    const unsigned short* ib = cur + QIMG;
    const bool pf = (t + 1) < ntiles;
    const int kc = (t + 1) * QBK;

    // ---- 4 phases: (ks, nhalf); 16 MFMA each ----
    bf16x8q af[8];
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int ks = ph >> 1;         // K sub-step (0: k 0..31, 1: 32..63)
      const int nh = ph & 1;          // which 2 of the 4 N fragments
      // prefetch one half of the next K-tile (A0,A1,B0,B1 over phases)
      if (pf) {
        if (ph == 0) qstage_half(A, m0, M, Kp, kc, nxt, 0, wid, lane);
        else if (ph == 1) qstage_half(A, m0, M, Kp, kc, nxt, 1, wid, lane);
        else if (ph == 2) qstage_half(Bt, n0, N, Kp, kc, nxt + QIMG, 0,
                                      wid, lane);
        else qstage_half(Bt, n0, N, Kp, kc, nxt + QIMG, 1, wid, lane);
      }
      // register fragments: A is shared by the two nh phases of a ks
      bf16x8q bf[2];
      if (nh == 0) {
#pragma unroll
        for (int mf = 0; mf < 8; ++mf)
          af[mf] = qread(ia, wr * 128 + mf * 16 + row, ks * 32 + kk0);
      }
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bf[nf] = qread(ib, wc * 64 + (nh * 2 + nf) * 16 + row,
                       ks * 32 + kk0);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[mf][nh * 2 + nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bf[nf], acc[mf][nh * 2 + nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    // buffer swap: this tile's prefetch (4 halves, 8 glds/thread) must
    // have LANDED in every wave before anyone reads `nxt`
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: C[m][n] -> y[b][gm][t], n = b*To + t
  const int col = lane & 15;
  const int row4 = (lane >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int64_t gn = n0 + wc * 64 + nf * 16 + col;
      if (gn >= N) continue;
      const int b = (int)(gn / To);
      const int t = (int)(gn - (int64_t)b * To);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gm = m0 + wr * 128 + mf * 16 + row4 + r;
        if (gm >= M) continue;
        float o = acc[mf][nf][r] + (bias ? bias[gm] : 0.f);
        o = apply_act(o, act, slope);
        union { __bf16 bb; unsigned short s; } cv{(__bf16)o};
        if (ilv <= 1) {
          y[((int64_t)b * M + gm) * To + t] = cv.s;
        } else {
          const int mo = M / ilv;
          const int phi = gm / mo;
          const int co = gm - phi * mo;
          y[((int64_t)b * mo + co) * To * ilv + t * ilv + phi] = cv.s;
        }
      }
    }
}

}  // namespace melgan

using namespace melgan;

static inline int q256cdiv(int64_t a, int64_t b) {
  return (int)((a + b - 1) / b);
}

bool gemm256_enabled() {
  static const bool disabled = []() {
    const char* e = getenv("MELGAN_AMD_NO_GEMM256");
    return e && e[0] == '1';
  }();
  return !disabled;
}

bool gemm256_eligible(int64_t M, int64_t N, int64_t Kp) {
  if (!(gemm256_enabled() && M >= 256 && N >= 256 && Kp % QBK == 0 &&
        Kp >= 128))
    return false;
  // one 512-thread block per CU: the chip needs ~256 tiles to fill
  // (under that the 128^2 engine's 4x block count wins)
  const int64_t tiles = ((M + 255) / 256) * ((N + 255) / 256);
  // at 1 block/CU (128 KiB LDS) the per-K-tile drain is exposed unless
  // multiple block-waves overlap; measured crossover ~2 waves of tiles,
  // M-strip shapes also lose (single-M-tile -> poor L2 A reuse)
  return tiles >= 448 && M >= 512;
}

// same contract as gemm.hip's engine call: y (Bb, M, To) pre-allocated
void gemm_bt_256(torch::Tensor A, torch::Tensor Bt, const float* bias_p,
                 torch::Tensor y, int64_t M, int64_t N, int64_t Kp,
                 int64_t To, int64_t act, double slope, int64_t ilv) {
  auto stream = at::hip::getCurrentHIPStream();
  const int nwgx = q256cdiv(N, QBN);
  dim3 grid(nwgx, q256cdiv(M, QBM), 1);
  hipLaunchKernelGGL(gemm_bt_256_kernel, grid, dim3(512), 0, stream,
                     (const unsigned short*)A.data_ptr(),
                     (const unsigned short*)Bt.data_ptr(), bias_p,
                     (unsigned short*)y.data_ptr(), (int)M, N, (int)Kp,
                     (int)To, (int)act, (float)slope, nwgx, (int)ilv);
}
