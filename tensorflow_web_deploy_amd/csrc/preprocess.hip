// Image preprocess kernels (gfx950): bilinear resize + normalize + NHWC
// pack, uint8 RGB/gray -> f32 (or bf16) model input.
//
// Replaces the reference's PIL resize/np normalize: the CPU
// does only the inherently-serial JPEG entropy decode; pixels hit HBM
// once as uint8 and are resized/normalized on a side HIP stream.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>

#include "kernels.h"

// norm: 0 = raw [0,255], 1 = unit [0,1), 2 = inception [-1,1)
__global__ void resize_norm_kernel(const uint8_t* __restrict__ src, int sh,
                                   int sw, int sc, float* __restrict__ dst,
                                   int dh, int dw, int dc, int norm) {
  // PIL-style bilinear with half-pixel centers
  const float scale_y = (float)sh / dh;
  const float scale_x = (float)sw / dw;
  int64_t total = (int64_t)dh * dw * dc;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % dc;
    int x = (i / dc) % dw;
    int y = i / ((int64_t)dc * dw);
    float fy = (y + 0.5f) * scale_y - 0.5f;
    float fx = (x + 0.5f) * scale_x - 0.5f;
    int y0 = (int)floorf(fy), x0 = (int)floorf(fx);
    float wy = fy - y0, wx = fx - x0;
    int y1 = min(y0 + 1, sh - 1), x1 = min(x0 + 1, sw - 1);
    y0 = max(y0, 0); x0 = max(x0, 0);
    int cs = min(c, sc - 1);  // gray -> replicate
    float v00 = src[((int64_t)y0 * sw + x0) * sc + cs];
    float v01 = src[((int64_t)y0 * sw + x1) * sc + cs];
    float v10 = src[((int64_t)y1 * sw + x0) * sc + cs];
    float v11 = src[((int64_t)y1 * sw + x1) * sc + cs];
    float v = v00 * (1 - wy) * (1 - wx) + v01 * (1 - wy) * wx +
              v10 * wy * (1 - wx) + v11 * wy * wx;
    if (norm == 1) v *= (1.f / 255.f);
    else if (norm == 2) v = v * (1.f / 127.5f) - 1.f;
    dst[i] = v;
  }
}

void launch_resize_norm(const void* src, int sh, int sw, int sc, float* dst,
                        int dh, int dw, int dc, int norm,
                        hipStream_t stream) {
  int64_t total = (int64_t)dh * dw * dc;
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 1024);
  hipLaunchKernelGGL(resize_norm_kernel, dim3(blocks), dim3(256), 0, stream,
                     (const uint8_t*)src, sh, sw, sc, dst, dh, dw, dc, norm);
}

// Batched variant: all images packed in one arena; meta[i] = {byte_off,
// sh, sw, sc}. One launch per super-batch (the per-image Python loop
// was GIL-starved under load).
__global__ void resize_norm_batch_kernel(const uint8_t* __restrict__ arena,
                                         const int* __restrict__ meta,
                                         float* __restrict__ dst, int dh,
                                         int dw, int dc, int norm) {
  const int img = blockIdx.y;
  const uint8_t* src = arena + meta[img * 4 + 0];
  const int sh = meta[img * 4 + 1];
  const int sw = meta[img * 4 + 2];
  const int sc = meta[img * 4 + 3];
  const float scale_y = (float)sh / dh;
  const float scale_x = (float)sw / dw;
  float* out = dst + (int64_t)img * dh * dw * dc;
  int64_t total = (int64_t)dh * dw * dc;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % dc;
    int x = (i / dc) % dw;
    int y = i / ((int64_t)dc * dw);
    float fy = (y + 0.5f) * scale_y - 0.5f;
    float fx = (x + 0.5f) * scale_x - 0.5f;
    int y0 = (int)floorf(fy), x0 = (int)floorf(fx);
    float wy = fy - y0, wx = fx - x0;
    int y1 = min(y0 + 1, sh - 1), x1 = min(x0 + 1, sw - 1);
    y0 = max(y0, 0); x0 = max(x0, 0);
    int cs = min(c, sc - 1);
    float v00 = src[((int64_t)y0 * sw + x0) * sc + cs];
    float v01 = src[((int64_t)y0 * sw + x1) * sc + cs];
    float v10 = src[((int64_t)y1 * sw + x0) * sc + cs];
    float v11 = src[((int64_t)y1 * sw + x1) * sc + cs];
    float v = v00 * (1 - wy) * (1 - wx) + v01 * (1 - wy) * wx +
              v10 * wy * (1 - wx) + v11 * wy * wx;
    if (norm == 1) v *= (1.f / 255.f);
    else if (norm == 2) v = v * (1.f / 127.5f) - 1.f;
    out[i] = v;
  }
}

void launch_resize_norm_batch(const void* arena, const int* meta, float* dst,
                              int n, int dh, int dw, int dc, int norm,
                              hipStream_t stream) {
  int64_t per = (int64_t)dh * dw * dc;
  int blocks = (int)std::min<int64_t>((per + 255) / 256, 64);
  hipLaunchKernelGGL(resize_norm_batch_kernel, dim3(blocks, n), dim3(256), 0,
                     stream, (const uint8_t*)arena, meta, dst, dh, dw, dc,
                     norm);
}
