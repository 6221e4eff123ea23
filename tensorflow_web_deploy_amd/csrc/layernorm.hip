// Row-wise LayerNorm, bf16 in/out, f32 stats (gfx950).
// ViT-B/16 path: rows = N*tokens, C = 768 (12 bf16x8 groups per lane-64 row).
// One wave per row; Welford-free two-pass-in-registers (single read).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

__global__ void layernorm_kernel(const __bf16* __restrict__ X,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 __bf16* __restrict__ Y, int64_t rows, int C,
                                 float eps) {
  const int lane = threadIdx.x & 63;
  const int wave_per_blk = blockDim.x >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * wave_per_blk +
                       (threadIdx.x >> 6);
  for (int64_t row = row0; row < rows;
       row += (int64_t)gridDim.x * wave_per_blk) {
    const __bf16* x = X + row * C;
    // C <= 64*8*MAXV elements held in registers across the two passes
    constexpr int MAXV = 4;  // up to C = 2048
    bf16x8 v[MAXV];
    float sum = 0.f;
    const int nv = C >> 9;  // full 512-element sweeps
    int c = lane * 8;
#pragma unroll
    for (int i = 0; i < MAXV; ++i) {
      if (i < nv || c < C) {
        v[i] = *(const bf16x8*)(x + c);
#pragma unroll
        for (int e = 0; e < 8; ++e) sum += bf2f(v[i][e]);
      }
      c += 512;
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) sum += __shfl_xor(sum, off, 64);
    const float mean = sum / (float)C;
    float var = 0.f;
    c = lane * 8;
#pragma unroll
    for (int i = 0; i < MAXV; ++i) {
      if (c < C) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float d = bf2f(v[i][e]) - mean;
          var += d * d;
        }
      }
      c += 512;
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) var += __shfl_xor(var, off, 64);
    const float rstd = rsqrtf(var / (float)C + eps);
    c = lane * 8;
#pragma unroll
    for (int i = 0; i < MAXV; ++i) {
      if (c < C) {
        bf16x8 o;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float g = gamma ? gamma[c + e] : 1.f;
          float bb = beta ? beta[c + e] : 0.f;
          o[e] = (__bf16)((bf2f(v[i][e]) - mean) * rstd * g + bb);
        }
        *(bf16x8*)(Y + row * C + c) = o;
      }
      c += 512;
    }
  }
}

void launch_layernorm_bf16(const void* X, const float* gamma,
                           const float* beta, void* Y, int64_t rows, int C,
                           float eps, hipStream_t stream) {
  // one wave per row, 4 waves per block
  int64_t blocks = std::min<int64_t>((rows + 3) / 4, 2048);
  hipLaunchKernelGGL(layernorm_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, (const __bf16*)X, gamma, beta, (__bf16*)Y, rows,
                     C, eps);
}
