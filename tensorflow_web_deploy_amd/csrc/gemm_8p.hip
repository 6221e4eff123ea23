// gemm_8p.hip — 256x256 GEMM, guide §5 "8-phase template" geometry, v6.
//
// v6 insight (PMC round 2, profiles/round2_gemm_pmc.md): the v5
// reconstruction parked 42.5% of wave cycles at 4096^3 because its
// waves computed a 64x32 piece of EVERY C-quadrant and therefore read
// ALL FOUR LDS half-tiles each K-tile — every wave's schedule coupled
// to every staged half (3 vmcnt waits/tile). The template's geometry
// (guide: "per-wave output 128x64 = BM/WARPS_M x BN/WARPS_N", waves
// 2Mx4N) gives each wave a CONTIGUOUS 128x64 block, so a wave touches
// exactly ONE A-half (its wr) and ONE B-half (wc>>1):
//
//   * 512 thr / 8 waves; LDS 128 KiB = 2 dbuf x 4 half-tiles
//     (A rows 0-127/128-255, B cols 0-127/128-255), each [128][64] bf16.
//   * Wave (wr=w>>2, wc=w&3) owns C rows [wr*128,+128) cols [wc*64,+64):
//     acc[8][4] f32x4 (128 VGPR).
//   * K-tile = 4 phases, quadrant order (m,n) = (0,0),(1,0),(1,1),(0,1)
//     over the wave's OWN block: 16 MFMA each; ds_reads 12/8/4/0 with
//     A-set + B-pair register reuse (24 ds_read_b128 per tile).
//   * ONE stage per phase (2 glds: halves of tile t+1 in order
//     A0,B0,A1,B1) and ONE counted wait per K-tile: after staging
//     A0(t+1) at phase 0, vmcnt(2) proves all four halves of tile t
//     and leaves the new stage in flight. Never vmcnt(0) in the loop.
//   * Phase shape per guide: [reads] stage [wait P0] barrier
//     lgkmcnt(0) setprio(1) 16xMFMA setprio(0) barrier. P0's reads sit
//     AFTER its barrier (they are only proven by P0's own vmcnt).
//   * st_16x32 LDS swizzle on the glds SOURCE + ds_read address.
//
// Scope: linear A (dense/1x1 shapes), bias + act + optional residual,
// LDS-restage epilogue. Routed for large square-ish shapes when it
// beats gemm_conv_kernel_256 (see dispatch in gemm_conv.hip / bench
// history in NOTES).
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

#include "kernels.h"

typedef __bf16 bf16x8_8p __attribute__((ext_vector_type(8)));
typedef float f32x4_8p __attribute__((ext_vector_type(4)));

#define BM8 256
#define BK8 64

// LDS half-tile byte offsets: buf b at b*64KiB; A0,A1,B0,B1 each 16KiB
#define HALF_OFF(b, h) ((b) * 65536 + (h) * 16384)
// h index: 0=A0 1=A1 2=B0 3=B1

extern __shared__ __attribute__((aligned(16))) char smem8p[];

__global__ __launch_bounds__(512) void gemm_8p_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    __bf16* __restrict__ Cout, const float* __restrict__ bias,
    const __bf16* __restrict__ residual, int M, int N, int K, int Kp,
    int a_bytes, int tiles_m, int tiles_n, int act, int ldc) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  // XCD-aware bijective remap
  int nwg = tiles_m * tiles_n;
  int bid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int tn = bid / tiles_m;
  const int tm = bid - tn * tiles_m;
  const int row0 = tm * BM8;
  const int col0 = tn * BM8;

  const auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, 0, a_bytes, 0x00020000);
  const auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, 0, (int)(((int64_t)(N - 1) * Kp + Kp) * 2), 0x00020000);

  // staging geometry: half = [128 rows][64 elems]; glds j covers rows
  // j*64 + T/8, col-group T%8; st_16x32 swizzle applied on the SOURCE
  // so the LDS image is "swizzled-linear" and ds_reads un-swizzle.
  const int T = tid;
  const int s_colgrp = T & 7;
  auto stage_half = [&](int h, int t, int nt_) {
    const int kb = (t < nt_ ? t : nt_ - 1) * BK8;
    const int buf = t & 1;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int row = j * 64 + (T >> 3);  // 0..127 within the half
      int cg = s_colgrp ^ (((row >> 2) & 1) << 1);
      int32_t voff;
      if (h < 2) {
        int grow = row0 + h * 128 + row;
        voff = grow < M ? (grow * Kp + kb + cg * 8) * 2 : 0x7ffffff0;
      } else {
        int gcol = col0 + (h - 2) * 128 + row;
        voff = gcol < N ? (gcol * Kp + kb + cg * 8) * 2 : 0x7ffffff0;
      }
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          h < 2 ? rsrcA : rsrcB,
          (__attribute__((address_space(3))) void*)(
              smem8p + HALF_OFF(buf, h) + j * 8192 + wave * 1024),
          16, voff, 0, 0, 0);
    }
  };

  // swizzled ds_read byte address within a half: (row, elem-group cg)
  auto lds_b = [](int row, int cg) {
    return row * 128 + ((cg ^ (((row >> 2) & 1) << 1)) << 4);
  };

  const int wr = wave >> 2;        // A-half this wave reads
  const int wc = wave & 3;
  const int bh = wc >> 1;          // B-half this wave reads
  const int bcol0 = (wc & 1) * 64; // col strip within that half

  f32x4_8p acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      float b0 = 0.f;
      if (bias) {
        int col = col0 + wc * 64 + ni * 16 + (lane & 15);
        b0 = col < N ? bias[col] : 0.f;
      }
      acc[mi][ni] = f32x4_8p{b0, b0, b0, b0};
    }

  const int nt = Kp / BK8;

  // register frags: two A m-sets (rows mi*16, m-set s covers mi s*4..+4)
  // and one live B n-pair (cols ni*16, pair p covers ni p*2..+2)
  bf16x8_8p af[2][4][2];  // [m-set][mi'][kk]
  bf16x8_8p bf[2][2];     // [ni'][kk] — current pair only

  auto read_a = [&](int set, int buf) {
    const char* Ah = smem8p + HALF_OFF(buf, wr);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = (set * 4 + mi) * 16 + (lane & 15);
        af[set][mi][kk] = *(const bf16x8_8p*)(
            Ah + lds_b(row, kk * 4 + (lane >> 4)));
      }
  };
  auto read_b = [&](int pair, int buf) {
    const char* Bh = smem8p + HALF_OFF(buf, 2 + bh);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int row = bcol0 + (pair * 2 + ni) * 16 + (lane & 15);
        bf[ni][kk] = *(const bf16x8_8p*)(
            Bh + lds_b(row, kk * 4 + (lane >> 4)));
      }
  };
  auto mfma_q = [&](int mset, int npair) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mset * 4 + mi][npair * 2 + ni] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mset][mi][kk], bf[ni][kk],
                  acc[mset * 4 + mi][npair * 2 + ni], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // prologue: all four halves of tile 0; drain fully once
  stage_half(0, 0, nt);
  stage_half(2, 0, nt);
  stage_half(1, 0, nt);
  stage_half(3, 0, nt);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    // ---- phase 0: quadrant (m0, n0) ----
    stage_half(0, t + 1, nt);
    // ONE counted wait per K-tile: proves all four halves of tile t,
    // leaves the stage just issued (2 glds) in flight
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    read_a(0, buf);
    read_b(0, buf);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_q(0, 0);
    __builtin_amdgcn_s_barrier();
    // ---- phase 1: (m1, n0) ----
    read_a(1, buf);
    stage_half(2, t + 1, nt);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_q(1, 0);
    __builtin_amdgcn_s_barrier();
    // ---- phase 2: (m1, n1) ----
    read_b(1, buf);
    stage_half(1, t + 1, nt);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_q(1, 1);
    __builtin_amdgcn_s_barrier();
    // ---- phase 3: (m0, n1) ---- (no reads: af[0] + bf[1] both live)
    stage_half(3, t + 1, nt);
    __builtin_amdgcn_s_barrier();
    mfma_q(0, 1);
    __builtin_amdgcn_s_barrier();
  }
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // ---- epilogue: restage C through LDS for wide dwordx4 stores ----
  {
    __bf16* ct = (__bf16*)smem8p;  // [256][256] bf16 = 128 KiB
    const int rb = wr * 128;
    const int cb = wc * 64;
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int col = cb + ni * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = rb + mi * 16 + (lane >> 4) * 4 + r;
          float v = acc[mi][ni][r];
          if (residual) {
            int gr = row0 + row, gc = col0 + col;
            if (gr < M && gc < N)
              v += (float)residual[(int64_t)gr * ldc + gc];
          }
          if (act == 1)
            v = v > 0.f ? v : 0.f;
          else if (act == 2)
            v = v < 0.f ? 0.f : (v > 6.f ? 6.f : v);
          else if (act == 5)
            v = 0.5f * v * (1.f + erff(v * 0.70710678f));
          ct[row * 256 + col] = (__bf16)v;
        }
      }
    __builtin_amdgcn_s_barrier();
    const bool full = (row0 + 256 <= M) && (col0 + 256 <= N);
#pragma unroll 1
    for (int i = tid; i < 256 * 32; i += 512) {
      int row = i >> 5, ch = i & 31;
      int gr = row0 + row, gc = col0 + ch * 8;
      bf16x8_8p v = *(bf16x8_8p*)(ct + row * 256 + ch * 8);
      if (full) {
        *(bf16x8_8p*)(Cout + (int64_t)gr * ldc + gc) = v;
      } else if (gr < M) {
        int nv = N - gc;
        if (nv >= 8) {
          *(bf16x8_8p*)(Cout + (int64_t)gr * ldc + gc) = v;
        } else {
          for (int e = 0; e < 8; ++e)
            if (e < nv) Cout[(int64_t)gr * ldc + gc + e] = v[e];
        }
      }
    }
  }
}

void launch_gemm_8p(const void* A, const void* B, void* C,
                    const float* bias, const void* residual, int M,
                    int N, int K, int Kp, int act, int ldc,
                    hipStream_t stream) {
  if (ldc == 0) ldc = N;
  int tiles_m = (M + BM8 - 1) / BM8;
  int tiles_n = (N + BM8 - 1) / BM8;
  int64_t a_bytes64 = (int64_t)M * Kp * 2;
  int a_bytes = a_bytes64 > 0x7fffffff ? 0x7fffffff : (int)a_bytes64;
  hipLaunchKernelGGL(gemm_8p_kernel, dim3(tiles_m * tiles_n), dim3(512),
                     131072, stream, (const __bf16*)A, (const __bf16*)B,
                     (__bf16*)C, bias, (const __bf16*)residual, M, N, K,
                     Kp, a_bytes, tiles_m, tiles_n, act, ldc);
}
