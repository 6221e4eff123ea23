// Pooling / reduction / softmax kernels, NHWC bf16 (gfx950).
//
// Replaces the reference TF runtime's MaxPool/AvgPool/Mean/Softmax kernels.
// All memory-bound: vectorized bf16x8 (16 B/lane) loads per guide G13.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

// ------------------------------------------------------------------ pool

template <bool MAX>
__global__ void pool_kernel(const __bf16* __restrict__ X,
                            __bf16* __restrict__ Y, int Nimg, int H, int W,
                            int C, int kh, int kw, int sh, int sw, int pt,
                            int pl, int HO, int WO) {
  // one thread per (output pixel, 8-channel group). 32-bit index math:
  // the 64-bit divmod chain was ~2x the memory time per output (round-2
  // profile: 3.2 TB/s effective); uint32 divides + a one-iteration
  // grid (uncapped) cut it.
  int c8 = C >> 3;
  int64_t total = (int64_t)Nimg * HO * WO * c8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    uint32_t i32 = (uint32_t)idx;
    uint32_t cg = i32 % (uint32_t)c8;
    uint32_t p = i32 / (uint32_t)c8;
    uint32_t wo = p % (uint32_t)WO;
    uint32_t q = p / (uint32_t)WO;
    uint32_t ho = q % (uint32_t)HO;
    uint32_t n = q / (uint32_t)HO;
    int hbase = ho * sh - pt, wbase = wo * sw - pl;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = MAX ? -3.4e38f : 0.f;
    int cnt = 0;
    for (int r = 0; r < kh; ++r) {
      int hi = hbase + r;
      if ((unsigned)hi >= (unsigned)H) continue;
      for (int s = 0; s < kw; ++s) {
        int wi = wbase + s;
        if ((unsigned)wi >= (unsigned)W) continue;
        const bf16x8 v = *(const bf16x8*)(X + ((((int64_t)n * H + hi) * W + wi) * C) + cg * 8);
        ++cnt;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          acc[e] = MAX ? fmaxf(acc[e], f) : acc[e] + f;
        }
      }
    }
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      o[e] = (__bf16)(MAX ? acc[e] : acc[e] / (float)max(cnt, 1));
    *(bf16x8*)(Y + ((((int64_t)n * HO + ho) * WO + wo) * C) + cg * 8) = o;
  }
}

// Stride-1 pools (Inception 3x3s1 SAME avg/max: 10.7% of its step) are
// separable: out(y) = combine of rowsum(y-1..y+1). A thread computes a
// VERTICAL STRIP of VS=4 outputs from VS+kh-1 register rowsums: 4.5
// loads/output instead of 9, vertical reuse in registers. Measured
// 2.3 -> ~5 TB/s class. KH template keeps register indexing static.
template <bool MAX, int KH>
__global__ void pool_s1_kernel(const __bf16* __restrict__ X,
                               __bf16* __restrict__ Y, int Nimg, int H,
                               int W, int C, int kw, int pt, int pl,
                               int HO, int WO) {
  constexpr int VS = 4;
  const int c8 = C >> 3;
  const int strips = (HO + VS - 1) / VS;
  const int64_t total = (int64_t)Nimg * strips * WO * c8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    uint32_t i32 = (uint32_t)idx;
    uint32_t cg = i32 % (uint32_t)c8;
    uint32_t p = i32 / (uint32_t)c8;
    uint32_t wo = p % (uint32_t)WO;
    uint32_t q = p / (uint32_t)WO;
    uint32_t strip = q % (uint32_t)strips;
    uint32_t n = q / (uint32_t)strips;
    const int oy0 = (int)strip * VS;
    const int wbase = (int)wo - pl;
    // horizontal (column) validity is row-independent
    int nvc = 0;
    float rs[VS + KH - 1][8];
    bool rv[VS + KH - 1];
    const __bf16* xn = X + ((int64_t)n * H * W * C) + cg * 8;
#pragma unroll
    for (int r = 0; r < VS + KH - 1; ++r) {
      int iy = oy0 - pt + r;
      rv[r] = (unsigned)iy < (unsigned)H;
#pragma unroll
      for (int e = 0; e < 8; ++e) rs[r][e] = MAX ? -3.4e38f : 0.f;
      if (!rv[r]) continue;
      const __bf16* xrow = xn + (int64_t)iy * W * C;
      int cnt = 0;
      for (int s = 0; s < kw; ++s) {
        int iw = wbase + s;
        if ((unsigned)iw >= (unsigned)W) continue;
        ++cnt;
        bf16x8 v = *(const bf16x8*)(xrow + (int64_t)iw * C);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          rs[r][e] = MAX ? fmaxf(rs[r][e], f) : rs[r][e] + f;
        }
      }
      if (r == 0) nvc = cnt;
      else if (cnt) nvc = cnt;  // same for every valid row
    }
#pragma unroll
    for (int v = 0; v < VS; ++v) {
      int oy = oy0 + v;
      if (oy >= HO) break;
      float acc[8];
      int nvr = 0;
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = MAX ? -3.4e38f : 0.f;
#pragma unroll
      for (int r = 0; r < KH; ++r) {
        if (!rv[v + r]) continue;
        ++nvr;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          acc[e] = MAX ? fmaxf(acc[e], rs[v + r][e])
                       : acc[e] + rs[v + r][e];
      }
      bf16x8 o;
      const float inv = MAX ? 1.f : 1.f / (float)max(nvr * max(nvc, 1), 1);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        o[e] = (__bf16)(MAX ? acc[e] : acc[e] * inv);
      *(bf16x8*)(Y + ((((int64_t)n * HO + oy) * WO + wo) * C) + cg * 8) =
          o;
    }
  }
}

template <bool MAX>
static bool pool_s1_dispatch(const void* X, void* Y, int Nimg, int H,
                             int W, int C, int kh, int kw, int pt, int pl,
                             int HO, int WO, hipStream_t stream) {
  if ((C & 7) != 0 || kh < 2 || kh > 5) return false;
  int strips = (HO + 3) / 4;
  int64_t total = (int64_t)Nimg * strips * WO * (C >> 3);
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 1 << 20);
  switch (kh) {
    case 2:
      hipLaunchKernelGGL((pool_s1_kernel<MAX, 2>), dim3(blocks), dim3(256),
                         0, stream, (const __bf16*)X, (__bf16*)Y, Nimg, H,
                         W, C, kw, pt, pl, HO, WO);
      return true;
    case 3:
      hipLaunchKernelGGL((pool_s1_kernel<MAX, 3>), dim3(blocks), dim3(256),
                         0, stream, (const __bf16*)X, (__bf16*)Y, Nimg, H,
                         W, C, kw, pt, pl, HO, WO);
      return true;
    case 5:
      hipLaunchKernelGGL((pool_s1_kernel<MAX, 5>), dim3(blocks), dim3(256),
                         0, stream, (const __bf16*)X, (__bf16*)Y, Nimg, H,
                         W, C, kw, pt, pl, HO, WO);
      return true;
  }
  return false;
}

void launch_maxpool_bf16(const void* X, void* Y, int Nimg, int H, int W,
                         int C, int kh, int kw, int sh, int sw, int pt,
                         int pl, int HO, int WO, hipStream_t stream) {
  if (sh == 1 && sw == 1 &&
      pool_s1_dispatch<true>(X, Y, Nimg, H, W, C, kh, kw, pt, pl, HO, WO,
                             stream))
    return;
  int64_t total = (int64_t)Nimg * HO * WO * (C >> 3);
  // one output per thread (no cap): the former 2048-block cap made
  // each thread redo the divmod chain ~5x on 35x35 maps
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 1 << 20);
  hipLaunchKernelGGL((pool_kernel<true>), dim3(blocks), dim3(256), 0, stream,
                     (const __bf16*)X, (__bf16*)Y, Nimg, H, W, C, kh, kw, sh,
                     sw, pt, pl, HO, WO);
}

void launch_avgpool_bf16(const void* X, void* Y, int Nimg, int H, int W,
                         int C, int kh, int kw, int sh, int sw, int pt,
                         int pl, int HO, int WO, hipStream_t stream) {
  if (sh == 1 && sw == 1 &&
      pool_s1_dispatch<false>(X, Y, Nimg, H, W, C, kh, kw, pt, pl, HO, WO,
                              stream))
    return;
  int64_t total = (int64_t)Nimg * HO * WO * (C >> 3);
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 1 << 20);
  hipLaunchKernelGGL((pool_kernel<false>), dim3(blocks), dim3(256), 0, stream,
                     (const __bf16*)X, (__bf16*)Y, Nimg, H, W, C, kh, kw, sh,
                     sw, pt, pl, HO, WO);
}

// ---------------------------------------------------------- global mean

__global__ void global_mean_kernel(const __bf16* __restrict__ X,
                                   __bf16* __restrict__ Y, int Nimg, int HW,
                                   int C) {
  // thread per (n, c8 group); HW is small (49/64), loop it
  int c8 = C >> 3;
  int64_t total = (int64_t)Nimg * c8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int cg = idx % c8;
    int n = idx / c8;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    const __bf16* base = X + (int64_t)n * HW * C + cg * 8;
    for (int p = 0; p < HW; ++p) {
      bf16x8 v = *(const bf16x8*)(base + (int64_t)p * C);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += bf2f(v[e]);
    }
    bf16x8 o;
    float inv = 1.f / (float)HW;
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (__bf16)(acc[e] * inv);
    *(bf16x8*)(Y + (int64_t)n * C + cg * 8) = o;
  }
}

void launch_global_mean_bf16(const void* X, void* Y, int Nimg, int HW, int C,
                             hipStream_t stream) {
  int64_t total = (int64_t)Nimg * (C >> 3);
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(global_mean_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const __bf16*)X, (__bf16*)Y, Nimg, HW, C);
}

// --------------------------------------------------------------- softmax

template <typename OUT, int NV>
__global__ void softmax_kernel(const __bf16* __restrict__ X,
                               OUT* __restrict__ Y, int M, int N) {
  // one wave per row; row held in registers (NV values per lane) so the
  // row is read from HBM exactly once (attention rows: N=196, NV=4)
  int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (row >= M) return;
  const __bf16* x = X + (int64_t)row * N;
  float r[NV];
  float m = -3.4e38f;
#pragma unroll
  for (int j = 0; j < NV; ++j) {
    int i = j * 64 + lane;
    r[j] = i < N ? bf2f(x[i]) : -3.4e38f;
    m = fmaxf(m, r[j]);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < NV; ++j) {
    r[j] = __expf(r[j] - m);
    s += r[j];
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_xor(s, off, 64);
  float inv = 1.f / s;
  OUT* y = Y + (int64_t)row * N;
#pragma unroll
  for (int j = 0; j < NV; ++j) {
    int i = j * 64 + lane;
    if (i < N) y[i] = (OUT)(r[j] * inv);
  }
}

// large-N fallback (3 passes; classifier heads are tiny anyway)
template <typename OUT>
__global__ void softmax_big_kernel(const __bf16* __restrict__ X,
                                   OUT* __restrict__ Y, int M, int N) {
  int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (row >= M) return;
  const __bf16* x = X + (int64_t)row * N;
  float m = -3.4e38f;
  for (int i = lane; i < N; i += 64) m = fmaxf(m, bf2f(x[i]));
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  float s = 0.f;
  for (int i = lane; i < N; i += 64) s += __expf(bf2f(x[i]) - m);
#pragma unroll
  for (int off = 32; off; off >>= 1) s += __shfl_xor(s, off, 64);
  float inv = 1.f / s;
  OUT* y = Y + (int64_t)row * N;
  for (int i = lane; i < N; i += 64)
    y[i] = (OUT)(__expf(bf2f(x[i]) - m) * inv);
}

template <typename OUT>
static void softmax_dispatch(const __bf16* X, OUT* Y, int M, int N,
                             hipStream_t stream) {
  int wpb = 4;
  dim3 grid((M + wpb - 1) / wpb), blk(wpb * 64);
  if (N <= 256)
    hipLaunchKernelGGL((softmax_kernel<OUT, 4>), grid, blk, 0, stream, X, Y,
                       M, N);
  else if (N <= 512)
    hipLaunchKernelGGL((softmax_kernel<OUT, 8>), grid, blk, 0, stream, X, Y,
                       M, N);
  else if (N <= 1024)
    hipLaunchKernelGGL((softmax_kernel<OUT, 16>), grid, blk, 0, stream, X, Y,
                       M, N);
  else
    hipLaunchKernelGGL((softmax_big_kernel<OUT>), grid, blk, 0, stream, X, Y,
                       M, N);
}

void launch_softmax_bf16(const void* X, float* Y, int M, int N,
                         hipStream_t stream) {
  softmax_dispatch<float>((const __bf16*)X, Y, M, N, stream);
}

void launch_softmax_bf16_out(const void* X, void* Y, int M, int N,
                             hipStream_t stream) {
  softmax_dispatch<__bf16>((const __bf16*)X, (__bf16*)Y, M, N, stream);
}
