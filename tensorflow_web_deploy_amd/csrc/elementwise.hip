// Elementwise kernels, bf16 NHWC (gfx950). All memory-bound; vectorized
// 16 B/lane (guide G13). Grid-stride, capped grids (guide G11).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

static inline int grid_for(int64_t work) {
  return (int)std::max<int64_t>(
      1, std::min<int64_t>((work + 255) / 256, 2048));
}

// -------------------------------------------------------------- casts

__global__ void cast_f32_bf16_kernel(const float* __restrict__ X,
                                     __bf16* __restrict__ Y, int64_t n8,
                                     int64_t n) {
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n8 * 8; i < n; ++i) Y[i] = (__bf16)X[i];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    f32x4 a = *(const f32x4*)(X + i * 8);
    f32x4 b = *(const f32x4*)(X + i * 8 + 4);
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 4; ++e) o[e] = (__bf16)a[e];
#pragma unroll
    for (int e = 0; e < 4; ++e) o[4 + e] = (__bf16)b[e];
    *(bf16x8*)(Y + i * 8) = o;
  }
}

__global__ void cast_bf16_f32_kernel(const __bf16* __restrict__ X,
                                     float* __restrict__ Y, int64_t n8,
                                     int64_t n) {
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n8 * 8; i < n; ++i) Y[i] = bf2f(X[i]);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 v = *(const bf16x8*)(X + i * 8);
    f32x4 a, b;
#pragma unroll
    for (int e = 0; e < 4; ++e) a[e] = bf2f(v[e]);
#pragma unroll
    for (int e = 0; e < 4; ++e) b[e] = bf2f(v[4 + e]);
    *(f32x4*)(Y + i * 8) = a;
    *(f32x4*)(Y + i * 8 + 4) = b;
  }
}

void launch_cast_f32_bf16(const float* X, void* Y, int64_t n,
                          hipStream_t stream) {
  int64_t n8 = n / 8;
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3(grid_for(n8 + 1)), dim3(256),
                     0, stream, X, (__bf16*)Y, n8, n);
}

void launch_cast_bf16_f32(const void* X, float* Y, int64_t n,
                          hipStream_t stream) {
  int64_t n8 = n / 8;
  hipLaunchKernelGGL(cast_bf16_f32_kernel, dim3(grid_for(n8 + 1)), dim3(256),
                     0, stream, (const __bf16*)X, Y, n8, n);
}

// ------------------------------------------------- channel pad (stem)

// [rows][Cin] -> [rows][Cpad] with zero fill; lets every conv (C=1/3 stems
// included) run the implicit-GEMM kernel, which needs C % 8 == 0.
__global__ void channel_pad_kernel(const __bf16* __restrict__ X,
                                   __bf16* __restrict__ Y, int64_t rows,
                                   int Cin, int Cpad) {
  int64_t total = rows * Cpad;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % Cpad);
    int64_t r = i / Cpad;
    Y[i] = (c < Cin) ? X[r * Cin + c] : (__bf16)0.f;
  }
}

void launch_channel_pad_bf16(const void* X, void* Y, int64_t rows, int Cin,
                             int Cpad, hipStream_t stream) {
  int64_t total = rows * Cpad;
  hipLaunchKernelGGL(channel_pad_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, (const __bf16*)X, (__bf16*)Y, rows, Cin, Cpad);
}

// f32 [rows][Cin] -> bf16 [rows][Cpad] zero-padded (fused input cast +
// stem channel pad: saves one full activation pass)
__global__ void cast_pad_kernel(const float* __restrict__ X,
                                __bf16* __restrict__ Y, int64_t rows,
                                int Cin, int Cpad) {
  int64_t total = rows * Cpad;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % Cpad);
    int64_t r = i / Cpad;
    Y[i] = (c < Cin) ? (__bf16)X[r * Cin + c] : (__bf16)0.f;
  }
}

void launch_cast_f32_pad_bf16(const float* X, void* Y, int64_t rows,
                              int Cin, int Cpad, hipStream_t stream) {
  int64_t total = rows * Cpad;
  hipLaunchKernelGGL(cast_pad_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, X, (__bf16*)Y, rows, Cin, Cpad);
}

// --------------------------------------------------------- add (+relu)

__global__ void add_relu_kernel(const __bf16* __restrict__ A,
                                const __bf16* __restrict__ B,
                                __bf16* __restrict__ Y, int64_t n8,
                                int64_t n, int relu) {
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n8 * 8; i < n; ++i) {
      float v = bf2f(A[i]) + bf2f(B[i]);
      Y[i] = (__bf16)(relu && v < 0.f ? 0.f : v);
    }
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 a = *(const bf16x8*)(A + i * 8);
    bf16x8 b = *(const bf16x8*)(B + i * 8);
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = bf2f(a[e]) + bf2f(b[e]);
      o[e] = (__bf16)(relu && v < 0.f ? 0.f : v);
    }
    *(bf16x8*)(Y + i * 8) = o;
  }
}

void launch_add_relu_bf16(const void* A, const void* B, void* Y, int64_t n,
                          bool relu, hipStream_t stream) {
  int64_t n8 = n / 8;
  hipLaunchKernelGGL(add_relu_kernel, dim3(grid_for(n8 + 1)), dim3(256), 0,
                     stream, (const __bf16*)A, (const __bf16*)B, (__bf16*)Y,
                     n8, n, relu ? 1 : 0);
}

// ------------------------------------------------ bias + activation

// act: 0 = none, 1 = relu, 2 = relu6, 3 = tanh, 4 = sigmoid, 5 = gelu(erf)
__global__ void bias_act_kernel(const __bf16* __restrict__ X,
                                const float* __restrict__ bias,
                                __bf16* __restrict__ Y, int64_t rows, int C,
                                int act) {
  int c8 = C >> 3;
  int ct = C & 7;  // tail channels when C % 8 != 0
  int64_t total = rows * c8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cg = (int)(i % c8);
    int64_t off = (i / c8) * C + (int64_t)cg * 8;
    bf16x8 v = *(const bf16x8*)(X + off);
    f32x4 bl = bias ? *(const f32x4*)(bias + cg * 8) : f32x4{0, 0, 0, 0};
    f32x4 bh = bias ? *(const f32x4*)(bias + cg * 8 + 4) : f32x4{0, 0, 0, 0};
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float f = bf2f(v[e]) + (e < 4 ? bl[e] : bh[e - 4]);
      switch (act) {
        case 1: f = fmaxf(f, 0.f); break;
        case 2: f = fminf(fmaxf(f, 0.f), 6.f); break;
        case 3: f = tanhf(f); break;
        case 4: f = 1.f / (1.f + __expf(-f)); break;
        case 5: f = 0.5f * f * (1.f + erff(f * 0.70710678f)); break;
      }
      o[e] = (__bf16)f;
    }
    *(bf16x8*)(Y + off) = o;
  }
  if (ct) {
    int64_t ntail = rows * ct;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < ntail; i += (int64_t)gridDim.x * blockDim.x) {
      int c = c8 * 8 + (int)(i % ct);
      int64_t off = (i / ct) * C + c;
      float f = bf2f(X[off]) + (bias ? bias[c] : 0.f);
      switch (act) {
        case 1: f = fmaxf(f, 0.f); break;
        case 2: f = fminf(fmaxf(f, 0.f), 6.f); break;
        case 3: f = tanhf(f); break;
        case 4: f = 1.f / (1.f + __expf(-f)); break;
        case 5: f = 0.5f * f * (1.f + erff(f * 0.70710678f)); break;
      }
      Y[off] = (__bf16)f;
    }
  }
}

void launch_bias_act_bf16(const void* X, const float* bias, void* Y,
                          int64_t rows, int C, int act, hipStream_t stream) {
  int64_t total = rows * ((C >> 3) + (C & 7 ? 1 : 0));
  hipLaunchKernelGGL(bias_act_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, (const __bf16*)X, bias, (__bf16*)Y, rows, C,
                     act);
}

// --------------------------------------- scale*x + shift (unfused BN)

__global__ void scale_shift_kernel(const __bf16* __restrict__ X,
                                   const float* __restrict__ scale,
                                   const float* __restrict__ shift,
                                   __bf16* __restrict__ Y, int64_t rows,
                                   int C, int relu) {
  int c8 = C >> 3;
  int ct = C & 7;
  int64_t total = rows * c8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cg = (int)(i % c8);
    int64_t off = (i / c8) * C + (int64_t)cg * 8;
    bf16x8 v = *(const bf16x8*)(X + off);
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float f = bf2f(v[e]) * scale[cg * 8 + e] + shift[cg * 8 + e];
      o[e] = (__bf16)(relu && f < 0.f ? 0.f : f);
    }
    *(bf16x8*)(Y + off) = o;
  }
  if (ct) {
    int64_t ntail = rows * ct;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < ntail; i += (int64_t)gridDim.x * blockDim.x) {
      int c = c8 * 8 + (int)(i % ct);
      int64_t off = (i / ct) * C + c;
      float f = bf2f(X[off]) * scale[c] + shift[c];
      Y[off] = (__bf16)(relu && f < 0.f ? 0.f : f);
    }
  }
}

void launch_scale_shift_bf16(const void* X, const float* scale,
                             const float* shift, void* Y, int64_t rows, int C,
                             bool relu, hipStream_t stream) {
  int64_t total = rows * ((C >> 3) + (C & 7 ? 1 : 0));
  hipLaunchKernelGGL(scale_shift_kernel, dim3(grid_for(total)), dim3(256), 0,
                     stream, (const __bf16*)X, scale, shift, (__bf16*)Y, rows,
                     C, relu ? 1 : 0);
}
