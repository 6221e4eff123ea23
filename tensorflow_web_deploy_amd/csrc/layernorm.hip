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

__global__ void layernorm_kernel(const __bf16* __restrict__ X,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 __bf16* __restrict__ Y, int64_t rows, int C,
                                 float eps) {
  const int lane = threadIdx.x & 63;
  const int wave_per_blk = blockDim.x >> 6;
  const int64_t row0 = (int64_t)blockIdx.x * wave_per_blk +
                       (threadIdx.x >> 6);
  const float invC = 1.f / (float)C;
  for (int64_t row = row0; row < rows;
       row += (int64_t)gridDim.x * wave_per_blk) {
    const __bf16* x = X + row * C;
    float sum = 0.f, sq = 0.f;
    for (int c = lane * 8; c < C; c += 512) {
      bf16x8 v = *(const bf16x8*)(x + c);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float f = bf2f(v[e]);
        sum += f;
        sq += f * f;
      }
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) {
      sum += __shfl_xor(sum, off, 64);
      sq += __shfl_xor(sq, off, 64);
    }
    const float mean = sum * invC;
    const float var = fmaxf(sq * invC - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    __bf16* y = Y + row * C;
    for (int c = lane * 8; c < C; c += 512) {
      bf16x8 v = *(const bf16x8*)(x + c);
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float g = gamma ? gamma[c + e] : 1.f;
        float bb = beta ? beta[c + e] : 0.f;
        o[e] = (__bf16)((bf2f(v[e]) - mean) * rstd * g + bb);
      }
      *(bf16x8*)(y + c) = o;
    }
  }
}

void launch_layernorm_bf16(const void* X, const float* gamma,
                           const float* beta, void* Y, int64_t rows, int C,
                           float eps, hipStream_t stream) {
  // one wave per row, 4 waves per block; cover all rows in one sweep
  int64_t blocks = std::min<int64_t>((rows + 3) / 4, 8192);
  hipLaunchKernelGGL(layernorm_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, (const __bf16*)X, gamma, beta, (__bf16*)Y, rows,
                     C, eps);
}
