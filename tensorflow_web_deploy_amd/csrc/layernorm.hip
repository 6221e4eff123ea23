// Row-wise LayerNorm, bf16 in/out, f32 stats (gfx950).
// ViT-B/16 path: rows = N*tokens, C = 768.
//
// One wave per row, TWO passes over the row (second read is L1-hot:
// a row is ~1.5 KiB). Mean/variance accumulate in one read loop
// (var = E[x^2] - mean^2, f32). The earlier register-cached version
// spilled 76 B/lane to scratch (v[4] bf16x8 under lane guards) and ran
// 7x off the bandwidth roofline.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

typedef float f32x4_ln __attribute__((ext_vector_type(4)));

// Single-pass register variant for C <= NV*512 (ViT C=768 -> NV=2).
// Round-2 fix: the two-pass kernel below ran at 1.47 TB/s (18% of
// HBM) because gamma[c+e]/beta[c+e] were per-ELEMENT scalar global
// loads inside the row loop (1536 extra loads per row-pair) and X was
// read twice. Here the per-lane gamma/beta values (fixed indices) are
// hoisted into registers once per wave, and the row is cached in
// registers between the stats pass and the write (no lane-guarded
// array indexing -> no scratch spill; out-of-range lanes use clamped
// addresses and masked accumulation).
template <int NV>
__global__ void layernorm_reg_kernel(const __bf16* __restrict__ X,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     __bf16* __restrict__ Y, int64_t rows,
                                     int C, float eps) {
  const int lane = threadIdx.x & 63;
  const int wave_per_blk = blockDim.x >> 6;
  const int64_t wid = (int64_t)blockIdx.x * wave_per_blk +
                      (threadIdx.x >> 6);
  const float invC = 1.f / (float)C;

  float g[NV][8], bb[NV][8];
  bool valid[NV];
  int coff[NV];
#pragma unroll
  for (int j = 0; j < NV; ++j) {
    int c = j * 512 + lane * 8;
    valid[j] = c + 8 <= C;
    coff[j] = valid[j] ? c : 0;  // clamped: loads execute uniformly
    if (gamma) {
      f32x4_ln lo = *(const f32x4_ln*)(gamma + coff[j]);
      f32x4_ln hi = *(const f32x4_ln*)(gamma + coff[j] + 4);
#pragma unroll
      for (int e = 0; e < 4; ++e) { g[j][e] = lo[e]; g[j][4 + e] = hi[e]; }
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) g[j][e] = 1.f;
    }
    if (beta) {
      f32x4_ln lo = *(const f32x4_ln*)(beta + coff[j]);
      f32x4_ln hi = *(const f32x4_ln*)(beta + coff[j] + 4);
#pragma unroll
      for (int e = 0; e < 4; ++e) { bb[j][e] = lo[e]; bb[j][4 + e] = hi[e]; }
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) bb[j][e] = 0.f;
    }
  }

  const int64_t stride = (int64_t)gridDim.x * wave_per_blk * 2;
  for (int64_t row = wid * 2; row < rows; row += stride) {
    const bool two = row + 1 < rows;
    const __bf16* x0 = X + row * C;
    const __bf16* x1 = x0 + (two ? C : 0);
    bf16x8 v0[NV], v1[NV];
    float s0 = 0.f, q0 = 0.f, s1 = 0.f, q1 = 0.f;
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      v0[j] = *(const bf16x8*)(x0 + coff[j]);
      v1[j] = *(const bf16x8*)(x1 + coff[j]);
      const float m = valid[j] ? 1.f : 0.f;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float f0 = bf2f(v0[j][e]) * m;
        float f1 = bf2f(v1[j][e]) * m;
        s0 += f0;
        q0 += f0 * f0;
        s1 += f1;
        q1 += f1 * f1;
      }
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) {
      s0 += __shfl_xor(s0, off, 64);
      q0 += __shfl_xor(q0, off, 64);
      s1 += __shfl_xor(s1, off, 64);
      q1 += __shfl_xor(q1, off, 64);
    }
    const float m0 = s0 * invC;
    const float r0 = rsqrtf(fmaxf(q0 * invC - m0 * m0, 0.f) + eps);
    const float m1 = s1 * invC;
    const float r1 = rsqrtf(fmaxf(q1 * invC - m1 * m1, 0.f) + eps);
    __bf16* y0 = Y + row * C;
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      bf16x8 o0, o1;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        o0[e] = (__bf16)((bf2f(v0[j][e]) - m0) * r0 * g[j][e] + bb[j][e]);
        o1[e] = (__bf16)((bf2f(v1[j][e]) - m1) * r1 * g[j][e] + bb[j][e]);
      }
      if (valid[j]) {
        *(bf16x8*)(y0 + coff[j]) = o0;
        if (two) *(bf16x8*)(y0 + C + coff[j]) = o1;
      }
    }
  }
}

__global__ void layernorm_kernel(const __bf16* __restrict__ X,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 __bf16* __restrict__ Y, int64_t rows, int C,
                                 float eps) {
  // TWO rows per wave, interleaved: the sum/sumsq chains and the
  // shuffle-reduce of independent rows overlap (PMC: 61% issue-stall
  // on the one-row version)
  const int lane = threadIdx.x & 63;
  const int wave_per_blk = blockDim.x >> 6;
  const int64_t wid = (int64_t)blockIdx.x * wave_per_blk +
                      (threadIdx.x >> 6);
  const float invC = 1.f / (float)C;
  const int64_t stride = (int64_t)gridDim.x * wave_per_blk * 2;
  for (int64_t row = wid * 2; row < rows; row += stride) {
    const bool two = row + 1 < rows;
    const __bf16* x0 = X + row * C;
    const __bf16* x1 = x0 + (two ? C : 0);
    float s0 = 0.f, q0 = 0.f, s1 = 0.f, q1 = 0.f;
    for (int c = lane * 8; c < C; c += 512) {
      bf16x8 v0 = *(const bf16x8*)(x0 + c);
      bf16x8 v1 = *(const bf16x8*)(x1 + c);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float f0 = bf2f(v0[e]);
        float f1 = bf2f(v1[e]);
        s0 += f0;
        q0 += f0 * f0;
        s1 += f1;
        q1 += f1 * f1;
      }
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) {
      s0 += __shfl_xor(s0, off, 64);
      q0 += __shfl_xor(q0, off, 64);
      s1 += __shfl_xor(s1, off, 64);
      q1 += __shfl_xor(q1, off, 64);
    }
    const float m0 = s0 * invC;
    const float r0 = rsqrtf(fmaxf(q0 * invC - m0 * m0, 0.f) + eps);
    const float m1 = s1 * invC;
    const float r1 = rsqrtf(fmaxf(q1 * invC - m1 * m1, 0.f) + eps);
    __bf16* y0 = Y + row * C;
    for (int c = lane * 8; c < C; c += 512) {
      bf16x8 v0 = *(const bf16x8*)(x0 + c);
      bf16x8 v1 = *(const bf16x8*)(x1 + c);
      bf16x8 o0, o1;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float g = gamma ? gamma[c + e] : 1.f;
        float bb = beta ? beta[c + e] : 0.f;
        o0[e] = (__bf16)((bf2f(v0[e]) - m0) * r0 * g + bb);
        o1[e] = (__bf16)((bf2f(v1[e]) - m1) * r1 * g + bb);
      }
      *(bf16x8*)(y0 + c) = o0;
      if (two) *(bf16x8*)(y0 + C + c) = o1;
    }
  }
}

void launch_layernorm_bf16(const void* X, const float* gamma,
                           const float* beta, void* Y, int64_t rows, int C,
                           float eps, hipStream_t stream) {
  // two rows per wave, 4 waves per block; cover all rows in one sweep
  int64_t blocks = std::min<int64_t>((rows + 7) / 8, 8192);
  if (C % 8 == 0 && C <= 1024) {
    // register variant (see comment above): C <= NV*512
    if (C <= 512)
      hipLaunchKernelGGL(layernorm_reg_kernel<1>, dim3((int)blocks),
                         dim3(256), 0, stream, (const __bf16*)X, gamma,
                         beta, (__bf16*)Y, rows, C, eps);
    else
      hipLaunchKernelGGL(layernorm_reg_kernel<2>, dim3((int)blocks),
                         dim3(256), 0, stream, (const __bf16*)X, gamma,
                         beta, (__bf16*)Y, rows, C, eps);
    return;
  }
  hipLaunchKernelGGL(layernorm_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, (const __bf16*)X, gamma, beta, (__bf16*)Y, rows,
                     C, eps);
}
