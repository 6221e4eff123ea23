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

void launch_dwconv_bf16(const void* x, const void* w, const float* bias,
                        const void* residual, void* y, int N, int H,
                        int W, int C, int HO, int WO, int R, int S,
                        int sh, int sw, int pt, int pl, int act,
                        hipStream_t stream) {
  int c8 = (C + 7) >> 3;
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
