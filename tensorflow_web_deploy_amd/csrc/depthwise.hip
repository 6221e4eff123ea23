// depthwise.hip — NHWC depthwise conv for gfx950 (MobileNet family).
//
// Replaces the reference runtime's DepthwiseConv2dNative kernel. The op
// is HBM-bound (R*S reads + 1 write per output element, ~0.02 flop/B at
// 3x3), so the design goal is pure bandwidth: one thread per (pixel,
// 8-channel group) with the channel vector on the contiguous NHWC axis —
// every activation load and the store are full bf16x8 (dwordx4)
// accesses, and the tiny [R*S*C] filter stays resident in L2/L1.
// Epilogue fuses the folded-BN bias + relu/relu6 (+ optional residual)
// so MobileNet's depthwise stage is ONE kernel per block.
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

#include "kernels.h"

typedef __bf16 dwbf16x8 __attribute__((ext_vector_type(8)));

template <bool VEC>
__global__ __launch_bounds__(256) void dwconv_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ w,
    const float* __restrict__ bias, const __bf16* __restrict__ residual,
    __bf16* __restrict__ y, int H, int W, int C, int HO, int WO, int R,
    int S, int sh, int sw, int pt, int pl, int act, int c8, int total) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total) return;
  int c = (idx % c8) * 8;
  int p = idx / c8;  // flattened (n, ho, wo)
  int wo = p % WO;
  int t = p / WO;
  int ho = t % HO;
  int n = t / HO;
  const int nv = VEC ? 8 : min(8, C - c);

  float acc[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) acc[e] = 0.f;
  if (bias) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      if (VEC || e < nv) acc[e] = bias[c + e];
  }

  const int hb = ho * sh - pt;
  const int wb = wo * sw - pl;
  const __bf16* xn = x + ((int64_t)n * H * W * C) + c;
  for (int r = 0; r < R; ++r) {
    int ih = hb + r;
    if ((unsigned)ih >= (unsigned)H) continue;
    for (int s2 = 0; s2 < S; ++s2) {
      int iw = wb + s2;
      if ((unsigned)iw >= (unsigned)W) continue;
      const __bf16* xp = xn + ((int64_t)ih * W + iw) * C;
      const __bf16* wp = w + (r * S + s2) * C + c;
      if (VEC) {
        dwbf16x8 xv = *(const dwbf16x8*)xp;
        dwbf16x8 wv = *(const dwbf16x8*)wp;
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += (float)xv[e] * (float)wv[e];
      } else {
        for (int e = 0; e < nv; ++e)
          acc[e] += (float)xp[e] * (float)wp[e];
      }
    }
  }

  __bf16* yp = y + ((int64_t)p * C) + c;
  if (residual) {
    const __bf16* rp = residual + ((int64_t)p * C) + c;
    if (VEC) {
      dwbf16x8 rv = *(const dwbf16x8*)rp;
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += (float)rv[e];
    } else {
      for (int e = 0; e < nv; ++e) acc[e] += (float)rp[e];
    }
  }
  if (act == 1) {
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = acc[e] > 0.f ? acc[e] : 0.f;
  } else if (act == 2) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      acc[e] = acc[e] < 0.f ? 0.f : (acc[e] > 6.f ? 6.f : acc[e]);
  }
  if (VEC) {
    dwbf16x8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e) v[e] = (__bf16)acc[e];
    *(dwbf16x8*)yp = v;
  } else {
    for (int e = 0; e < nv; ++e) yp[e] = (__bf16)acc[e];
  }
}

// stride-1 two-pixel variant: a thread computes wo0 and wo0+1, so each
// input column loads ONCE for both (12 instead of 18 loads at 3x3 —
// NOTES_ROUND2 item 9 follow-up). VEC-only (C % 8 == 0).
__global__ __launch_bounds__(256) void dwconv2_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ w,
    const float* __restrict__ bias, const __bf16* __restrict__ residual,
    __bf16* __restrict__ y, int H, int W, int C, int HO, int WO, int R,
    int S, int pt, int pl, int act, int c8, int wo2, int total) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total) return;
  int c = (idx % c8) * 8;
  int p = idx / c8;  // flattened (n, ho, wpair)
  int wp = p % wo2;
  int t = p / wo2;
  int ho = t % HO;
  int n = t / HO;
  const int wo0 = wp * 2;
  const bool two = wo0 + 1 < WO;

  float a0[8], a1[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    float b = bias ? bias[c + e] : 0.f;
    a0[e] = b;
    a1[e] = b;
  }

  const int hb = ho - pt;
  const int wb = wo0 - pl;
  const __bf16* xn = x + ((int64_t)n * H * W * C) + c;
  for (int r = 0; r < R; ++r) {
    int ih = hb + r;
    if ((unsigned)ih >= (unsigned)H) continue;
    const __bf16* xrow = xn + (int64_t)ih * W * C;
    const __bf16* wrow = w + r * S * C + c;
    for (int j = 0; j < S + 1; ++j) {
      int iw = wb + j;
      if ((unsigned)iw >= (unsigned)W) continue;
      dwbf16x8 xv = *(const dwbf16x8*)(xrow + (int64_t)iw * C);
      if (j < S) {
        dwbf16x8 wv = *(const dwbf16x8*)(wrow + j * C);
#pragma unroll
        for (int e = 0; e < 8; ++e) a0[e] += (float)xv[e] * (float)wv[e];
      }
      if (j >= 1) {
        dwbf16x8 wv = *(const dwbf16x8*)(wrow + (j - 1) * C);
#pragma unroll
        for (int e = 0; e < 8; ++e) a1[e] += (float)xv[e] * (float)wv[e];
      }
    }
  }

  const int64_t p0 = (((int64_t)n * HO + ho) * WO + wo0);
  if (residual) {
    dwbf16x8 rv = *(const dwbf16x8*)(residual + p0 * C + c);
#pragma unroll
    for (int e = 0; e < 8; ++e) a0[e] += (float)rv[e];
    if (two) {
      dwbf16x8 r2 = *(const dwbf16x8*)(residual + (p0 + 1) * C + c);
#pragma unroll
      for (int e = 0; e < 8; ++e) a1[e] += (float)r2[e];
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    if (act == 1) {
      a0[e] = fmaxf(a0[e], 0.f);
      a1[e] = fmaxf(a1[e], 0.f);
    } else if (act == 2) {
      a0[e] = fminf(fmaxf(a0[e], 0.f), 6.f);
      a1[e] = fminf(fmaxf(a1[e], 0.f), 6.f);
    }
  }
  dwbf16x8 v0, v1;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    v0[e] = (__bf16)a0[e];
    v1[e] = (__bf16)a1[e];
  }
  *(dwbf16x8*)(y + p0 * C + c) = v0;
  if (two) *(dwbf16x8*)(y + (p0 + 1) * C + c) = v1;
}

void launch_dwconv_bf16(const void* x, const void* w, const float* bias,
                        const void* residual, void* y, int N, int H,
                        int W, int C, int HO, int WO, int R, int S,
                        int sh, int sw, int pt, int pl, int act,
                        hipStream_t stream) {
  int c8 = (C + 7) >> 3;
  // A/B'd NEGATIVE (MobileNetV2 51.1k -> 48.6k img/s): halving the
  // thread count costs more latency-hiding than the shared column
  // loads save (L2 was already absorbing the re-reads). Parked.
  static const bool dw2 = getenv("TWD_DW2") != nullptr;
  if (dw2 && (C & 7) == 0 && sh == 1 && sw == 1 && S <= 7) {
    int wo2 = (WO + 1) >> 1;
    int total = N * HO * wo2 * c8;
    hipLaunchKernelGGL(dwconv2_kernel, dim3((total + 255) / 256),
                       dim3(256), 0, stream, (const __bf16*)x,
                       (const __bf16*)w, bias, (const __bf16*)residual,
                       (__bf16*)y, H, W, C, HO, WO, R, S, pt, pl, act,
                       c8, wo2, total);
    return;
  }
  int total = N * HO * WO * c8;
  int blocks = (total + 255) / 256;
  if ((C & 7) == 0)
    hipLaunchKernelGGL(dwconv_kernel<true>, dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)x, (const __bf16*)w, bias,
                       (const __bf16*)residual, (__bf16*)y, H, W, C, HO,
                       WO, R, S, sh, sw, pt, pl, act, c8, total);
  else
    hipLaunchKernelGGL(dwconv_kernel<false>, dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)x, (const __bf16*)w, bias,
                       (const __bf16*)residual, (__bf16*)y, H, W, C, HO,
                       WO, R, S, sh, sw, pt, pl, act, c8, total);
}
