// gemm_8p.hip — 256x256 GEMM on the guide's 8-phase quadrant schedule
// (cdna_hip_programming.md §5 "256² 8-phase template"), derived from
// the spec (no example source available):
//
//   * 512 threads / 8 waves; LDS 128 KiB = 2 dbuf x 4 HALF-tiles
//     (A rows 0-127 / 128-255, B cols 0-127 / 128-255), each
//     [128][64] bf16.
//   * Each K-tile runs as 4 phases; phase q computes C-QUADRANT
//     Q[q] (order (0,0),(1,0),(1,1),(0,1)) for the full K=64: the
//     phase touches ONLY A-half qr and B-half qc, so halves become
//     consumable progressively.
//   * Each phase stages ONE half-tile (2 glds) of tile t+1 in
//     first-need order A0,B0,A1,B1 and gates with a counted
//     s_waitcnt vmcnt(6) (phases 0-2; none at 3): the wait leaves
//     exactly the 3 newest half-tiles in flight — the property the
//     guide states as "3 half-tiles prefetched ahead".
//   * Per phase: stage -> vmcnt -> barrier -> 12 ds_read_b128 ->
//     lgkmcnt(0) -> setprio(1) 16 MFMA setprio(0) -> barrier.
//   * st_16x32 LDS swizzle (byte ^= ((byte>>9)&1)<<5, i.e. col-group
//     bit 1 flips with row bit 2), applied on the glds SOURCE address
//     and the ds_read address — the LDS image itself stays linear per
//     instruction (rule 21 generalized).
//
// Scope: MODE=linear A (dense/1x1), bias + act + optional residual,
// LDS-restage epilogue. UNROUTED: measured faster than
// gemm_conv_kernel_256 only at 8192^3 (1006 vs 954 TF) and slower at
// 4096^3 (875 vs 1130) and on model shapes — NOTES_ROUND2.md item 2
// records the five-variant exploration. bench/gemm8p_bench.py
// reproduces numerics + perf via the gemm8p_bf16 binding.
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

  // staging geometry: half = [128 rows][64 elems]; instruction j covers
  // rows j*64 + T/8 (T = global thread id), col-group T%8 (8 elems).
  // st_16x32 swizzle on the SOURCE: colgrp_src = colgrp ^ ((row>>2)&1)<<1
  const int T = tid;
  const int s_colgrp = T & 7;
  // stage half h of K-tile t into buffer t&1 (t may be nt: parity
  // still alternates, k reads clamp to the last tile — harmless
  // redundant loads on the final iteration). h: 0=A0 1=A1 2=B0 3=B1.
  //
  // glds writes lane l's 16 B at LDS base + l*16 (wave-uniform base);
  // for a [row][64-elem] half, (j*64 + w*8 + l/8)*128 + (l%8)*16 ==
  // j*8192 + w*1024 + l*16, so base = half + j*8192 + wave*1024 gives
  // the LINEAR image and the st_16x32 swizzle lives purely in the
  // per-lane SOURCE address.
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

  // quadrant order and wave sub-tile: within quadrant (128x128), wave
  // covers rows (w>>2)*64, cols (w&3)*32 -> 4 m-frags x 2 n-frags
  const int wr2 = wave >> 2;   // 0..1
  const int wc2 = wave & 3;    // 0..3
  const int QR[4] = {0, 1, 1, 0};
  const int QC[4] = {0, 0, 1, 1};
  const int SO[4] = {0, 2, 1, 3};  // stage order A0,B0,A1,B1

  f32x4_8p acc[4][4][2];  // [quadrant][mi][ni]
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        float b0 = 0.f;
        if (bias) {
          int col = col0 + QC[q] * 128 + wc2 * 32 + ni * 16 + (lane & 15);
          b0 = col < N ? bias[col] : 0.f;
        }
        acc[q][mi][ni] = f32x4_8p{b0, b0, b0, b0};
      }

  const int nt = Kp / BK8;
  // prologue: all four halves of tile 0
  stage_half(0, 0, nt);
  stage_half(2, 0, nt);
  stage_half(1, 0, nt);
  stage_half(3, 0, nt);

  // Fragment reuse across the quadrant order (0,0),(1,0),(1,1),(0,1):
  // q0 and q3 use the SAME A-frags (half 0, same wave rows), q1/q2
  // share half 1; q0/q1 share B-pair 0, q2/q3 pair 1. So each A set
  // and B pair is read ONCE per K-tile (24 ds_reads, not 40) and held
  // in registers. A sets are issued where the previous phase's MFMA
  // burst hides them; B pairs are the only exposed reads (4 each).
  bf16x8_8p af[2][4][2], bf[2][2];  // af[set][mi][kk], bf[ni][kk]

  auto issue_a = [&](int set, int buf) {
    const char* Ah = smem8p + HALF_OFF(buf, set);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wr2 * 64 + mi * 16 + (lane & 15);
        af[set][mi][kk] = *(const bf16x8_8p*)(
            Ah + lds_b(row, kk * 4 + (lane >> 4)));
      }
  };
  auto issue_b = [&](int pair, int buf) {
    const char* Bh = smem8p + HALF_OFF(buf, 2 + pair);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int row = wc2 * 32 + ni * 16 + (lane & 15);
        bf[ni][kk] = *(const bf16x8_8p*)(
            Bh + lds_b(row, kk * 4 + (lane >> 4)));
      }
  };

  auto mfma_q = [&](int q, int set) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[q][mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[set][mi][kk], bf[ni][kk], acc[q][mi][ni], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // prologue: verify tile0 A0/B0, sync, pre-read A set 0
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  issue_a(0, 0);

  // Waits as derived for the v3 schedule (sim in git history):
  //   q0: vmcnt(2) pre-barrier (t:A1 proven for q1); stage t+1:A0
  //       AFTER the barrier (its next-tile counterpart would overwrite
  //       the half the slowest wave still reads).
  //   q1: stage t+1:B0; vmcnt(4) (t:B1 proven for q2).
  //   q2: stage t+1:A1; no wait.
  //   q3: stage t+1:B1; vmcnt(4) (t+1:A0 proven -> pre-read next A0
  //       from the other buffer under q3's MFMAs).
  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    // ---- q0 ----
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    stage_half(0, t + 1, nt);
    issue_b(0, buf);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    issue_a(1, buf);  // half 1 for q1/q2, hidden under q0's MFMAs
    mfma_q(0, 0);
    __builtin_amdgcn_s_barrier();
    // ---- q1 ----
    stage_half(2, t + 1, nt);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_q(1, 1);
    __builtin_amdgcn_s_barrier();
    // ---- q2 ----
    stage_half(1, t + 1, nt);
    __builtin_amdgcn_s_barrier();
    issue_b(1, buf);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_q(2, 1);
    __builtin_amdgcn_s_barrier();
    // ---- q3 ----
    stage_half(3, t + 1, nt);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    mfma_q(3, 0);
    // next tile's A0 AFTER q3's MFMAs consume set 0 (prefetching it
    // above would create a register WAR the scheduler resolves by
    // serializing — wrong data cannot occur, but the MFMAs would wait
    // on next-tile loads; a 3rd A set doesn't fit in 256 VGPRs)
    issue_a(0, buf ^ 1);
    __builtin_amdgcn_s_barrier();
  }
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  // ---- epilogue: restage C through LDS for wide dwordx4 stores ----
  // (the scattered per-element form measured 888 vs 1010 TF at 4k/8k:
  // the 2-byte stores cost ~2x the loop's own margin at nt=64)
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  {
    __bf16* ct = (__bf16*)smem8p;  // [256][256] bf16 = 128 KiB
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int rb = QR[q] * 128 + wr2 * 64;
      const int cb = QC[q] * 128 + wc2 * 32;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          int col = cb + ni * 16 + (lane & 15);
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int row = rb + mi * 16 + (lane >> 4) * 4 + r;
            float v = acc[q][mi][ni][r];
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
