// MFMA implicit-GEMM kernel for gfx950 (CDNA4) — the conv2d/dense hot path.
//
// Replaces the reference's TF-runtime Conv2D/MatMul kernels (cuDNN/cuBLAS
// behind Session.run) with a hand-written CDNA4 kernel per the north star:
// "MFMA im2col-GEMM with LDS-staged input tiles".
//
// Computes  C[M][N] = act( A[M][K] @ B[N][K]^T + bias [+ residual] )
//   A: activations, NHWC-flattened rows (bf16). For conv, rows are gathered
//      im2col-style on the fly while staging (per-lane source addresses into
//      a buffer descriptor whose hardware OOB check supplies the zero
//      padding — no zero-page buffers, no branches in the load path).
//   B: weights pre-transposed to [N][K] (HWIO -> [O][R*S*C]), K padded to a
//      multiple of BK with zeros at load time (Python side).
//   C: bf16 [M][N] row-major = the next layer's NHWC activation.
//
// Structure (cdna_hip_programming.md §5 "step-3" verified shape):
//   128x128 tile, BK=64, 256 threads = 4 waves (each computing 64x64 via
//   4x4 fragments of v_mfma_f32_16x16x32_bf16), double-buffered LDS filled
//   by global_load_lds (buffer variant), XOR-swizzled LDS reads
//   (slot ^= row&7) with the inverse swizzle applied to the *source*
//   address (guide §5.4 rule 21), XCD-aware bijective block remap (T1),
//   epilogue staged through LDS for wide dwordx4 stores.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define BM 128
#define BN 128
#define BK 64
#define NTHREADS 256

// LDS: A/B tiles double buffered: [2][128][64] bf16 each = 32 KiB per op.
// The epilogue reuses the first 32 KiB as a [128][128] bf16 C tile.
#define LDS_BYTES (4 * BM * BK * 2)
#define A_OFF(buf) ((buf) * (BM * BK * 2))
#define B_OFF(buf) ((2 + (buf)) * (BM * BK * 2))

extern __shared__ __attribute__((aligned(16))) char smem[];

// n / d for n < 2^24, d < 2^16 with m = ceil(2^40/d)
__device__ __forceinline__ uint32_t mdiv(uint32_t n, uint64_t m) {
  return (uint32_t)(((uint64_t)n * m) >> 40);
}

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

// ---------------------------------------------------------------- kernel

struct ConvParams {
  // geometry (conv mode); unused in linear mode
  int C;        // input channels
  int H, W;     // input spatial
  int R, S;     // filter spatial
  int sh, sw;   // strides
  int pt, pl;   // top/left pad
  int HO, WO;   // output spatial
  uint64_t magic_C, magic_S, magic_HW, magic_WO;
};

// Templated tile kernel. TBM x TBN output tile, BK=64 K-steps, wave tile
// 64x64 => NWAVES = (TBM/64)*(TBN/64). Instantiated as (128,128) for
// general shapes and (256,64) for narrow-N layers (ResNet stage-1 1x1/3x3
// with Cout=64) where a 128-wide tile would waste half the MFMA work.
// MODE: 0 = linear A rows (dense / 1x1 stride-1 conv); 1 = im2col gather
template <int TBM, int TBN, int MODE, bool HAS_RES, int DEPTH = 2>
__global__ __launch_bounds__((TBM / 64) * (TBN / 64) * 64)
void gemm_conv_kernel(const __bf16* __restrict__ A,
                      const __bf16* __restrict__ B,
                      __bf16* __restrict__ Cout,
                      const float* __restrict__ bias,
                      const __bf16* __restrict__ residual,
                      int M, int N, int K, int Kp,
                      int a_bytes, int tiles_m, int tiles_n,
                      int relu, ConvParams cp,
                      int64_t sA = 0, int64_t sB = 0, int64_t sC = 0,
                      float* __restrict__ Cpart = nullptr,
                      int nt_slice = 1 << 28, int ldc = 0,
                      int lda = 0, int ldb = 0, int batch2 = 1,
                      int64_t sA2 = 0, int64_t sB2 = 0, int64_t sC2 = 0) {
  if (ldc == 0) ldc = N;  // C row stride (concat-slice / fused-qkv writes)
  if (lda == 0) lda = K;  // A row stride (strided-batch attention views)
  if (ldb == 0) ldb = Kp;
  constexpr int NWAVES = (TBM / 64) * (TBN / 64);
  constexpr int WC_N = TBN / 64;            // wave columns
  constexpr int CA = TBM / 8 / NWAVES;      // A-staging glds per wave
  constexpr int CB = TBN / 8 / NWAVES;      // B-staging glds per wave
  constexpr int ATILE = TBM * BK * 2;       // bytes
  constexpr int BTILE = TBN * BK * 2;
  // strided-batch support (attention: blockIdx.y = flattened (b1,b2))
  {
    const int bz = blockIdx.y;
    const int b1 = bz / batch2;
    const int b2 = bz - b1 * batch2;
    const int64_t offA = b1 * sA + b2 * sA2;
    const int64_t offB = b1 * sB + b2 * sB2;
    const int64_t offC = b1 * sC + b2 * sC2;
    A += offA;
    B += offB;
    Cout += offC;
    if (HAS_RES) residual += offC;
    a_bytes -= (int)(offA * 2);
  }
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  // ---- XCD-aware bijective remap (guide §5: T1, bijective variant) ----
  // tiles_n < 0 flags 4x4 tile clustering (L2 panel sharing in the
  // overflow regime — see gemm_conv_kernel_256's comment)
  const bool t2d = tiles_n < 0;
  if (t2d) tiles_n = -tiles_n;
  int nwg = tiles_m * tiles_n;
  int bid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  // tm fastest => consecutive blocks share the B panel (L2 reuse)
  int tn, tm;
  if (t2d) {
    const int bpr = tiles_m >> 2;
    const int blk = bid >> 4, wi = bid & 15;
    tm = (blk % bpr) * 4 + (wi & 3);
    tn = (blk / bpr) * 4 + (wi >> 2);
  } else {
    tn = bid / tiles_m;
    tm = bid - tn * tiles_m;
  }
  const int row0 = tm * TBM;
  const int col0 = tn * TBN;

  const auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, /*stride*/ 0, a_bytes, 0x00020000);
  const auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, 0, (int)(((int64_t)(N - 1) * ldb + Kp) * 2), 0x00020000);

  // ---- per-lane staging state ----
  // Staging instruction j covers tile rows j*8 + lane/8, lane slot =
  // lane&7; source element k = kb + (slot ^ (row&7))*8 (rule 21: the
  // XOR swizzle moves to the source address; LDS image stays linear).
  const int srow_base = (lane >> 3);        // 0..7
  const int sslot = lane & 7;
  int32_t a_base[CA];   // element offset of (row, k=0) or im2col base
  int32_t a_hb[CA], a_wb[CA];
  bool a_valid[CA];
#pragma unroll
  for (int i = 0; i < CA; ++i) {
    int j = wave * CA + i;
    int row_l = j * 8 + srow_base;
    int row_g = row0 + row_l;
    if (MODE == 0) {
      a_valid[i] = row_g < M;
      a_base[i] = row_g * lda;
    } else {
      a_valid[i] = row_g < M;
      uint32_t rg = (uint32_t)min(row_g, max(M - 1, 0));
      uint32_t n_img = mdiv(rg, cp.magic_HW);
      uint32_t rem = rg - n_img * (uint32_t)(cp.HO * cp.WO);
      uint32_t ho = mdiv(rem, cp.magic_WO);
      uint32_t wo = rem - ho * (uint32_t)cp.WO;
      int hb = (int)ho * cp.sh - cp.pt;
      int wb = (int)wo * cp.sw - cp.pl;
      a_hb[i] = hb;
      a_wb[i] = wb;
      a_base[i] = (((int)n_img * cp.H + hb) * cp.W + wb) * cp.C;
    }
  }
  // B rows (cols of C): same staging pattern
  int32_t b_voff[CB];
  bool b_valid[CB];
#pragma unroll
  for (int i = 0; i < CB; ++i) {
    int j = wave * CB + i;
    int row_l = j * 8 + srow_base;
    int col_g = col0 + row_l;
    b_valid[i] = col_g < N;
    b_voff[i] = col_g * ldb;
  }

  const int swz = srow_base & 7;  // row&7 of this lane's staged row

  // ---- accumulators (+ free bias init) ----
  const int wr = wave / WC_N, wc = wave % WC_N;  // wave tile (64x64)
  f32x4 acc[4][4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      float b0 = 0.f;
      if (bias) {
        int col = col0 + wc * 64 + ni * 16 + (lane & 15);
        if (col < N) b0 = bias[col];
      }
      acc[mi][ni] = {b0, b0, b0, b0};
    }

  // split-K: blockIdx.z selects a contiguous K-step slice
  const int nt_all = (K + BK - 1) / BK;
  const int kt0 = blockIdx.z * nt_slice;
  const int nt = min(nt_all - kt0, nt_slice);
  auto a_off = [&](int buf) { return buf * ATILE; };
  auto b_off = [&](int buf) { return DEPTH * ATILE + buf * BTILE; };

  // ---- staging helper (as lambdas so the main loop stays readable) ----
  auto stage = [&](int buf, int t) {
    const int kb = t * BK;
    const int k = kb + ((sslot ^ swz) << 3);
#pragma unroll
    for (int i = 0; i < CA; ++i) {
      int32_t voff;
      if (MODE == 0) {
        voff = (a_valid[i] && k < K) ? (a_base[i] + k) * 2 : 0x7ffffff0;
      } else {
        uint32_t rs = mdiv((uint32_t)k, cp.magic_C);
        uint32_t c = (uint32_t)k - rs * (uint32_t)cp.C;
        uint32_t r = mdiv(rs, cp.magic_S);
        uint32_t sx = rs - r * (uint32_t)cp.S;
        int hi = a_hb[i] + (int)r;
        int wi = a_wb[i] + (int)sx;
        bool ok = a_valid[i] && k < K && (uint32_t)hi < (uint32_t)cp.H &&
                  (uint32_t)wi < (uint32_t)cp.W;
        voff = ok ? (a_base[i] + ((int)r * cp.W + (int)sx) * cp.C + (int)c) * 2
                  : 0x7ffffff0;
      }
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          rsrcA,
          (__attribute__((address_space(3))) void*)(smem + a_off(buf) +
                                                    (wave * CA + i) * 1024),
          16, voff, 0, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < CB; ++i) {
      // B tile ([N][Kp] row-major; zero rows beyond N via OOB)
      int32_t voffB = b_valid[i] ? (b_voff[i] + k) * 2 : 0x7ffffff0;
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          rsrcB,
          (__attribute__((address_space(3))) void*)(smem + b_off(buf) +
                                                    (wave * CB + i) * 1024),
          16, voffB, 0, 0, 0);
    }
  };

  // swizzled LDS byte offset of (row, col8) in a [rows][64]-bf16 tile
  auto lds_byte = [](int row, int col8) {
    return row * (BK * 2) + ((col8 ^ (row & 7)) << 4);
  };

  auto compute = [&](int buf) {
    // register-pipelined: kk=1 fragments load before kk=0's MFMAs issue
    bf16x8 af[2][4], bfr[2][4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      int row = wr * 64 + mi * 16 + (lane & 15);
      af[0][mi] = *(bf16x8*)(smem + a_off(buf) +
                             lds_byte(row, (lane >> 4)));
    }
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int row = wc * 64 + ni * 16 + (lane & 15);
      bfr[0][ni] = *(bf16x8*)(smem + b_off(buf) +
                              lds_byte(row, (lane >> 4)));
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk == 0) {
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          int row = wr * 64 + mi * 16 + (lane & 15);
          af[1][mi] = *(bf16x8*)(smem + a_off(buf) +
                                 lds_byte(row, 4 + (lane >> 4)));
        }
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          int row = wc * 64 + ni * 16 + (lane & 15);
          bfr[1][ni] = *(bf16x8*)(smem + b_off(buf) +
                                  lds_byte(row, 4 + (lane >> 4)));
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk][mi], bfr[kk][ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  // ---- main loop: DEPTH-deep pipeline ----
  // Prefetch distance PF = DEPTH-1 K-tiles; per-iteration:
  //   counted wait (tile t landed, newer stages stay in flight)
  //   -> barrier -> stage tile t+PF -> compute tile t.
  // DEPTH=3 gives the narrow (256,64) tile (544 MFMA cycles/K-step)
  // two full iterations of latency cover (PMC: 44% barrier-parked at
  // DEPTH=2 on s1 3x3 — see profiles/).
  constexpr int PF = DEPTH - 1;
  constexpr int GPS = CA + CB;  // glds per stage per wave
#pragma unroll
  for (int i = 0; i < PF; ++i)
    if (i < nt) stage(i % DEPTH, kt0 + i);
  for (int t = 0; t < nt; ++t) {
    // stages newer than tile t still in flight at this point: the
    // prologue/earlier iterations staged through tile t+PF-1
    const int pending = min(nt - 1 - t, PF - 1);
    if (pending >= 2)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * GPS) : "memory");
    else if (pending == 1)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(GPS) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + PF < nt) stage((t + PF) % DEPTH, kt0 + t + PF);
    compute(t % DEPTH);
    // reads of buf t%DEPTH complete before the next barrier (compiler
    // lgkm waits precede the MFMAs); ordering for overwrite is the
    // head barrier of iteration t+1
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  if (Cpart != nullptr) {
    // split-K partial: scattered f32 stores to the slice slab (combined
    // with bias/residual/act by splitk_combine_kernel)
    float* slab = Cpart + (int64_t)blockIdx.z * M * N;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row_g = row0 + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
          int col_g = col0 + wc * 64 + ni * 16 + (lane & 15);
          if (row_g < M && col_g < N)
            slab[(int64_t)row_g * N + col_g] = acc[mi][ni][r];
        }
    return;
  }

  // ---- epilogue: stage C tile (bf16) through LDS for wide stores ----
  // D fragment layout (16x16x32): col = lane&15, row = (lane>>4)*4 + reg.
  __bf16* cl = (__bf16*)smem;  // [TBM][TBN]
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
        int col = wc * 64 + ni * 16 + (lane & 15);
        cl[row * TBN + col] = (__bf16)acc[mi][ni][r];
      }
    }
  __syncthreads();

  // write back: TBN/8 lanes per row, 8 bf16 (16 B) per lane
  constexpr int LPR = TBN / 8;                  // lanes per row
  constexpr int RPP = NWAVES * 64 / LPR;        // rows per pass
  const int wrow = tid / LPR;
  const int wcol8 = (tid % LPR) * 8;
#pragma unroll
  for (int p = 0; p < TBM / RPP; ++p) {
    int row_l = p * RPP + wrow;
    int row_g = row0 + row_l;
    int col_g = col0 + wcol8;
    if (row_g < M && col_g < N) {
      bf16x8 v = *(bf16x8*)(cl + row_l * TBN + wcol8);
      const int nvalid = min(8, N - col_g);
      if (HAS_RES) {
        if (nvalid == 8) {
          bf16x8 rv = *(const bf16x8*)(residual + (int64_t)row_g * ldc +
                                       col_g);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v[e] = (__bf16)(bf2f(v[e]) + bf2f(rv[e]));
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (e < nvalid)
              v[e] = (__bf16)(bf2f(v[e]) +
                              bf2f(residual[(int64_t)row_g * ldc + col_g +
                                            e]));
        }
      }
      if (relu == 1) {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = bf2f(v[e]) > 0.f ? v[e] : (__bf16)0.f;
      } else if (relu == 2) {  // relu6 (MobileNet blocks)
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(f < 0.f ? 0.f : (f > 6.f ? 6.f : f));
        }
      } else if (relu == 5) {  // erf-GELU fused epilogue (ViT MLP)
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(0.5f * f * (1.f + erff(f * 0.70710678f)));
        }
      }
      if (nvalid == 8) {
        *(bf16x8*)(Cout + (int64_t)row_g * ldc + col_g) = v;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (e < nvalid) Cout[(int64_t)row_g * ldc + col_g + e] = v[e];
      }
    }
  }
}

// ================================================================
// 256x256 pipelined kernel (8 waves, BK=64, counted vmcnt, raw barriers)
// for large layers — cdna_hip_programming.md §5 8-phase-template class.
// LDS: [2 buf][A0,A1,B0,B1 halves][128][64] bf16 = 128 KiB; 1 block/CU.
// Loads for K-tile t+1 stay in flight across the K-loop barriers
// (vmcnt(2) at each iteration head, never 0 in steady state).
// ================================================================

#define BM2 256
#define BN2 256
#define NTHREADS2 512
#define LDS2_BYTES (8 * 128 * BK * 2)  // 128 KiB
// half-tile h (0=A0,1=A1,2=B0,3=B1) of buffer buf
#define HALF_OFF(buf, h) (((buf) * 4 + (h)) * (128 * BK * 2))

template <int MODE, bool HAS_RES>
__global__ __launch_bounds__(NTHREADS2, 2)
void gemm_conv_kernel_256(const __bf16* __restrict__ A,
                          const __bf16* __restrict__ B,
                          __bf16* __restrict__ Cout,
                          const float* __restrict__ bias,
                          const __bf16* __restrict__ residual,
                          int M, int N, int K, int Kp,
                          int a_bytes, int tiles_m, int tiles_n,
                          int relu, ConvParams cp, int ldc = 0) {
  if (ldc == 0) ldc = N;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  // tiles_n < 0 flags 4x4 tile clustering (host A/B, TWD_TILE2D): an
  // XCD's contiguous logical range then walks 4x4 tile blocks, so its
  // co-resident workgroups share 4 A-panels AND 4 B-panels in L2
  // instead of 2 B + 16 A (column-major) — ~2.25x less HBM per K-tile
  // when the kernel is L2-miss-bound.
  const bool t2d = tiles_n < 0;
  if (t2d) tiles_n = -tiles_n;
  int nwg = tiles_m * tiles_n;
  int bid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  int tn, tm;
  if (t2d) {
    const int bpr = tiles_m >> 2;  // 4x4 blocks per block-row
    const int blk = bid >> 4, wi = bid & 15;
    const int bm = blk % bpr, bn = blk / bpr;
    tm = bm * 4 + (wi & 3);
    tn = bn * 4 + (wi >> 2);
  } else {
    tn = bid / tiles_m;
    tm = bid - tn * tiles_m;
  }
  const int row0 = tm * BM2;
  const int col0 = tn * BN2;

  const auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, 0, a_bytes, 0x00020000);
  const auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, 0, (int)((int64_t)N * Kp * 2), 0x00020000);

  // staging: half-tile = 128 rows x 64 cols = 16 KiB = 2 glds per wave.
  // glds i of half h covers rows (wave*2+i)*8 + lane/8 of that half.
  const int srow_base = lane >> 3;
  const int sslot = lane & 7;
  const int swz = srow_base & 7;
  // A halves: combo (h*2+i), rows h*128 + (wave*2+i)*8 + srow_base
  int32_t a_base[4], a_hb[4], a_wb[4];
  bool a_valid[4];
  int32_t b_voff[4];
  bool b_valid[4];
#pragma unroll
  for (int c2 = 0; c2 < 4; ++c2) {
    int h = c2 >> 1, i = c2 & 1;
    int row_l = h * 128 + (wave * 2 + i) * 8 + srow_base;
    int row_g = row0 + row_l;
    if (MODE == 0) {
      a_valid[c2] = row_g < M;
      a_base[c2] = row_g * K;
    } else {
      a_valid[c2] = row_g < M;
      uint32_t rg = (uint32_t)min(row_g, max(M - 1, 0));
      uint32_t n_img = mdiv(rg, cp.magic_HW);
      uint32_t rem = rg - n_img * (uint32_t)(cp.HO * cp.WO);
      uint32_t ho = mdiv(rem, cp.magic_WO);
      uint32_t wo = rem - ho * (uint32_t)cp.WO;
      int hb = (int)ho * cp.sh - cp.pt;
      int wb = (int)wo * cp.sw - cp.pl;
      a_hb[c2] = hb;
      a_wb[c2] = wb;
      a_base[c2] = (((int)n_img * cp.H + hb) * cp.W + wb) * cp.C;
    }
    int col_g = col0 + row_l;
    b_valid[c2] = col_g < N;
    b_voff[c2] = col_g * Kp;
  }

  const int wr = wave >> 2;  // 0..1: 128-row strip
  const int wc = wave & 3;   // 0..3: 64-col strip
  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      float b0 = 0.f;
      if (bias) {
        int col = col0 + wc * 64 + ni * 16 + (lane & 15);
        if (col < N) b0 = bias[col];
      }
      acc[mi][ni] = {b0, b0, b0, b0};
    }

  const int nt = (K + BK - 1) / BK;

  auto a_voffset = [&](int c2, int k) -> int32_t {
    if (MODE == 0) {
      return (a_valid[c2] && k < K) ? (a_base[c2] + k) * 2 : 0x7ffffff0;
    }
    uint32_t rs = mdiv((uint32_t)k, cp.magic_C);
    uint32_t c = (uint32_t)k - rs * (uint32_t)cp.C;
    uint32_t r = mdiv(rs, cp.magic_S);
    uint32_t s = rs - r * (uint32_t)cp.S;
    int hi = a_hb[c2] + (int)r;
    int wi = a_wb[c2] + (int)s;
    bool ok = a_valid[c2] && k < K && (uint32_t)hi < (uint32_t)cp.H &&
              (uint32_t)wi < (uint32_t)cp.W;
    return ok ? (a_base[c2] + ((int)r * cp.W + (int)s) * cp.C + (int)c) * 2
              : 0x7ffffff0;
  };

  // stage one half-tile (2 glds per wave) of K-tile t into buffer t&1
  auto stage_half = [&](int h, int t) {
    const int kb = t * BK;
    const int buf = t & 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int32_t voff;
      int k = kb + ((sslot ^ swz) << 3);
      if (h < 2) {
        voff = a_voffset(h * 2 + i, k);
      } else {
        int c2 = (h - 2) * 2 + i;
        voff = b_valid[c2] ? (b_voff[c2] + k) * 2 : 0x7ffffff0;
      }
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          h < 2 ? rsrcA : rsrcB,
          (__attribute__((address_space(3))) void*)(
              smem + HALF_OFF(buf, h) + (wave * 2 + i) * 1024),
          16, voff, 0, 0, 0);
    }
  };

  auto lds_byte2 = [](int row, int col8) {
    return row * (BK * 2) + ((col8 ^ (row & 7)) << 4);
  };

  // prologue: stage all of K-tile 0 (8 glds per wave)
  stage_half(0, 0); stage_half(1, 0); stage_half(2, 0); stage_half(3, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // Schedule: stage the WHOLE next tile at the top of each iteration so
  // its 8 glds have a full K-tile of MFMA work (~4k SIMD cycles) as
  // latency cover; one barrier + one counted drain per K-tile (the
  // previous schedule staged the last halves only ~1 phase ahead:
  // MfmaUtil 25%, see profiles/).
  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    const bool pf = (t + 1) < nt;
    if (pf) {
      // overwrite-safe: buf^1 was consumed in iteration t-1 and every
      // wave passed the barrier above
      stage_half(0, t + 1); stage_half(1, t + 1);
      stage_half(2, t + 1); stage_half(3, t + 1);
    }
    const char* Ahalf = smem + HALF_OFF(buf, wr);
    const char* Bhalf = smem + HALF_OFF(buf, 2 + (wc >> 1));
    const int brow0 = (wc & 1) * 64;  // 64-col strip within the B half

    // register-pipelined fragment reads: group g+1's LDS reads issue
    // before group g's MFMAs so the first MFMA of each group does not
    // stall on lgkm (PMC: 37% of wave cycles parked)
    {
      bf16x8 bfr[2][4], afr[2][4];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = brow0 + ni * 16 + (lane & 15);
        bfr[0][ni] = *(bf16x8*)(Bhalf + lds_byte2(row, (lane >> 4)));
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        int row = q * 16 + (lane & 15);
        afr[0][q] = *(bf16x8*)(Ahalf + lds_byte2(row, (lane >> 4)));
      }
#pragma unroll
      for (int g = 0; g < 4; ++g) {         // (kk,mh) = g>>1, g&1
        const int cur = g & 1;
        if (g < 3) {
          const int kk1 = (g + 1) >> 1;
          const int mh1 = (g + 1) & 1;
#pragma unroll
          for (int q = 0; q < 4; ++q) {
            int row = (mh1 * 4 + q) * 16 + (lane & 15);
            afr[cur ^ 1][q] = *(bf16x8*)(Ahalf +
                lds_byte2(row, kk1 * 4 + (lane >> 4)));
          }
          if (mh1 == 0) {
#pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
              int row = brow0 + ni * 16 + (lane & 15);
              bfr[1][ni] = *(bf16x8*)(Bhalf +
                  lds_byte2(row, kk1 * 4 + (lane >> 4)));
            }
          }
        }
        const int kk = g >> 1;
        const int mh = g & 1;
        (void)kk;
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int q = 0; q < 4; ++q)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[mh * 4 + q][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[cur][q], bfr[g >> 1][ni], acc[mh * 4 + q][ni], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
    // drain tile t+1's glds (covered by the MFMAs above) + LDS reads,
    // then sync so the next iteration may read/overwrite
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue through LDS (reuse all 128 KiB as a 256x256 bf16 tile)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  __bf16* cl = (__bf16*)smem;
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wr * 128 + mi * 16 + (lane >> 4) * 4 + r;
        int col = wc * 64 + ni * 16 + (lane & 15);
        cl[row * BN2 + col] = (__bf16)acc[mi][ni][r];
      }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // 512 threads: 32 lanes per row (16 B each), 16 rows/pass, 16 passes
  const int wrow = tid >> 5;
  const int wcol8 = (tid & 31) * 8;
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int row_l = p * 16 + wrow;
    int row_g = row0 + row_l;
    int col_g = col0 + wcol8;
    if (row_g < M && col_g < N) {
      bf16x8 v = *(bf16x8*)(cl + row_l * BN2 + wcol8);
      const int nvalid = min(8, N - col_g);
      if (HAS_RES) {
        if (nvalid == 8) {
          bf16x8 rv = *(const bf16x8*)(residual + (int64_t)row_g * ldc +
                                       col_g);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v[e] = (__bf16)(bf2f(v[e]) + bf2f(rv[e]));
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (e < nvalid)
              v[e] = (__bf16)(bf2f(v[e]) +
                              bf2f(residual[(int64_t)row_g * ldc + col_g +
                                            e]));
        }
      }
      if (relu == 1) {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = bf2f(v[e]) > 0.f ? v[e] : (__bf16)0.f;
      } else if (relu == 2) {  // relu6 (MobileNet blocks)
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(f < 0.f ? 0.f : (f > 6.f ? 6.f : f));
        }
      } else if (relu == 5) {  // erf-GELU fused epilogue (ViT MLP)
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(0.5f * f * (1.f + erff(f * 0.70710678f)));
        }
      }
      if (nvalid == 8) {
        *(bf16x8*)(Cout + (int64_t)row_g * ldc + col_g) = v;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (e < nvalid) Cout[(int64_t)row_g * ldc + col_g + e] = v[e];
      }
    }
  }
}

// ================================================================
// 256x256 PHASED variant (env TWD_256P=1): per K-tile, 4 phases of
// {quadrant ds_reads + one half-tile stage + barrier + lgkm + 16 MFMA +
// barrier} — the cdna_hip_programming.md §5 8-phase structure's
// role-split (T5's prerequisite), with B fragments held in registers
// across the tile and vmcnt counted once per K-tile.
// ================================================================
template <int MODE, bool HAS_RES>
__global__ __launch_bounds__(NTHREADS2, 2)
void gemm_conv_kernel_256p(const __bf16* __restrict__ A,
                           const __bf16* __restrict__ B,
                           __bf16* __restrict__ Cout,
                           const float* __restrict__ bias,
                           const __bf16* __restrict__ residual,
                           int M, int N, int K, int Kp,
                           int a_bytes, int tiles_m, int tiles_n,
                           int relu, ConvParams cp, int ldc = 0) {
  if (ldc == 0) ldc = N;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  int nwg = tiles_m * tiles_n;
  int bid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int tn = bid / tiles_m;
  const int tm = bid - tn * tiles_m;
  const int row0 = tm * BM2;
  const int col0 = tn * BN2;

  const auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, 0, a_bytes, 0x00020000);
  const auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, 0, (int)((int64_t)N * Kp * 2), 0x00020000);

  const int srow_base = lane >> 3;
  const int sslot = lane & 7;
  const int swz = srow_base & 7;
  int32_t a_base[4], a_hb[4], a_wb[4];
  bool a_valid[4];
  int32_t b_voff[4];
  bool b_valid[4];
#pragma unroll
  for (int c2 = 0; c2 < 4; ++c2) {
    int h = c2 >> 1, i = c2 & 1;
    int row_l = h * 128 + (wave * 2 + i) * 8 + srow_base;
    int row_g = row0 + row_l;
    if (MODE == 0) {
      a_valid[c2] = row_g < M;
      a_base[c2] = row_g * K;
    } else {
      a_valid[c2] = row_g < M;
      uint32_t rg = (uint32_t)min(row_g, max(M - 1, 0));
      uint32_t n_img = mdiv(rg, cp.magic_HW);
      uint32_t rem = rg - n_img * (uint32_t)(cp.HO * cp.WO);
      uint32_t ho = mdiv(rem, cp.magic_WO);
      uint32_t wo = rem - ho * (uint32_t)cp.WO;
      a_hb[c2] = (int)ho * cp.sh - cp.pt;
      a_wb[c2] = (int)wo * cp.sw - cp.pl;
      a_base[c2] = (((int)n_img * cp.H + a_hb[c2]) * cp.W + a_wb[c2]) * cp.C;
    }
    int col_g = col0 + row_l;
    b_valid[c2] = col_g < N;
    b_voff[c2] = col_g * Kp;
  }

  const int wr = wave >> 2, wc = wave & 3;
  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      float b0 = 0.f;
      if (bias) {
        int col = col0 + wc * 64 + ni * 16 + (lane & 15);
        if (col < N) b0 = bias[col];
      }
      acc[mi][ni] = {b0, b0, b0, b0};
    }

  const int nt = (K + BK - 1) / BK;

  auto a_voffset = [&](int c2, int k) -> int32_t {
    if (MODE == 0)
      return (a_valid[c2] && k < K) ? (a_base[c2] + k) * 2 : 0x7ffffff0;
    uint32_t rs = mdiv((uint32_t)k, cp.magic_C);
    uint32_t c = (uint32_t)k - rs * (uint32_t)cp.C;
    uint32_t r = mdiv(rs, cp.magic_S);
    uint32_t sx = rs - r * (uint32_t)cp.S;
    int hi = a_hb[c2] + (int)r;
    int wi = a_wb[c2] + (int)sx;
    bool ok = a_valid[c2] && k < K && (uint32_t)hi < (uint32_t)cp.H &&
              (uint32_t)wi < (uint32_t)cp.W;
    return ok ? (a_base[c2] + ((int)r * cp.W + (int)sx) * cp.C + (int)c) * 2
              : 0x7ffffff0;
  };

  auto stage_half = [&](int h, int t) {
    const int kb = t * BK;
    const int k = kb + ((sslot ^ swz) << 3);
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int32_t voff;
      if (h < 2) {
        voff = a_voffset(h * 2 + i, k);
      } else {
        int c2 = (h - 2) * 2 + i;
        voff = b_valid[c2] ? (b_voff[c2] + k) * 2 : 0x7ffffff0;
      }
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          h < 2 ? rsrcA : rsrcB,
          (__attribute__((address_space(3))) void*)(
              smem + HALF_OFF(t & 1, h) + (wave * 2 + i) * 1024),
          16, voff, 0, 0, 0);
    }
  };

  auto lds_byte2 = [](int row, int col8) {
    return row * (BK * 2) + ((col8 ^ (row & 7)) << 4);
  };

  stage_half(0, 0); stage_half(1, 0); stage_half(2, 0); stage_half(3, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < nt; ++t) {
    const int buf = t & 1;
    const bool pf = (t + 1) < nt;
    const char* Ahalf = smem + HALF_OFF(buf, wr);
    const char* Bhalf = smem + HALF_OFF(buf, 2 + (wc >> 1));
    const int brow0 = (wc & 1) * 64;

    bf16x8 bfr[2][4];  // [kk][ni], held across all 4 phases
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // quadrant p: m-frags {2p, 2p+1}, both kk
      bf16x8 afr[2][2];  // [mi-sub][kk]
      if (p == 0) {
        // verify tile t resident BEFORE any LDS read of this buffer
        if (pf) {
          stage_half(0, t + 1);
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni) {
            int row = brow0 + ni * 16 + (lane & 15);
            bfr[kk][ni] = *(bf16x8*)(Bhalf +
                lds_byte2(row, kk * 4 + (lane >> 4)));
          }
      }
#pragma unroll
      for (int q = 0; q < 2; ++q)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          int row = (p * 2 + q) * 16 + (lane & 15);
          afr[q][kk] = *(bf16x8*)(Ahalf +
              lds_byte2(row, kk * 4 + (lane >> 4)));
        }
      if (pf && p > 0) stage_half(p, t + 1);
      if (p > 0) __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int q = 0; q < 2; ++q)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[p * 2 + q][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[q][kk], bfr[kk][ni], acc[p * 2 + q][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  __bf16* cl = (__bf16*)smem;
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wr * 128 + mi * 16 + (lane >> 4) * 4 + r;
        int col = wc * 64 + ni * 16 + (lane & 15);
        cl[row * BN2 + col] = (__bf16)acc[mi][ni][r];
      }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  const int wrow = tid >> 5;
  const int wcol8 = (tid & 31) * 8;
#pragma unroll
  for (int pp = 0; pp < 16; ++pp) {
    int row_l = pp * 16 + wrow;
    int row_g = row0 + row_l;
    int col_g = col0 + wcol8;
    if (row_g < M && col_g < N) {
      bf16x8 v = *(bf16x8*)(cl + row_l * BN2 + wcol8);
      const int nvalid = min(8, N - col_g);
      if (HAS_RES) {
        if (nvalid == 8) {
          bf16x8 rv = *(const bf16x8*)(residual + (int64_t)row_g * ldc +
                                       col_g);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v[e] = (__bf16)(bf2f(v[e]) + bf2f(rv[e]));
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (e < nvalid)
              v[e] = (__bf16)(bf2f(v[e]) +
                              bf2f(residual[(int64_t)row_g * ldc + col_g +
                                            e]));
        }
      }
      if (relu == 1) {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = bf2f(v[e]) > 0.f ? v[e] : (__bf16)0.f;
      } else if (relu == 2) {  // relu6 (MobileNet blocks)
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(f < 0.f ? 0.f : (f > 6.f ? 6.f : f));
        }
      } else if (relu == 5) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = bf2f(v[e]);
          v[e] = (__bf16)(0.5f * f * (1.f + erff(f * 0.70710678f)));
        }
      }
      if (nvalid == 8) {
        *(bf16x8*)(Cout + (int64_t)row_g * ldc + col_g) = v;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (e < nvalid) Cout[(int64_t)row_g * ldc + col_g + e] = v[e];
      }
    }
  }
}

static bool use_256p() {
  static bool v = getenv("TWD_256P") != nullptr;
  return v;
}

// ---------------------------------------------------------------- host

static uint64_t magic40(uint32_t d) {
  if (d == 0) return 0;
  return ((((__uint128_t)1) << 40) + d - 1) / d;
}

// ------------------------------- split-K combine (elementwise) ---------
__global__ void splitk_combine_kernel(const float* __restrict__ Cpart,
                                      const float* __restrict__ bias,
                                      const __bf16* __restrict__ residual,
                                      __bf16* __restrict__ Cout, int64_t MN,
                                      int N, int splitk, int relu) {
  for (int64_t i8 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i8 * 8 < MN; i8 += (int64_t)gridDim.x * blockDim.x) {
    int64_t i = i8 * 8;
    int rem = (int)min((int64_t)8, MN - i);
    float v[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) v[e] = 0.f;
    for (int sidx = 0; sidx < splitk; ++sidx) {
      const float* slab = Cpart + (int64_t)sidx * MN + i;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (e < rem) v[e] += slab[e];
    }
    if (bias) {
      int col = (int)(i % N);  // N % 8 == 0 when this path is used
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (e < rem) v[e] += bias[col + e];
    }
    if (residual) {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        if (e < rem) v[e] += bf2f(residual[i + e]);
    }
#pragma unroll
    for (int e = 0; e < 8; ++e)
      if (e < rem) {
        float f = v[e];
        if (relu == 1 && f < 0.f) f = 0.f;
        else if (relu == 2) f = f < 0.f ? 0.f : (f > 6.f ? 6.f : f);
        else if (relu == 5) f = 0.5f * f * (1.f + erff(f * 0.70710678f));
        Cout[i + e] = (__bf16)f;
      }
  }
}

// Tile dispatch: 256^2 pipelined kernel when the grid still fills the
// 256-CU chip and the tile isn't mostly padding; a narrow (256,64) tile
// for N <= 96 layers; else the general (128,128) tile.
static int b256_min() {
  static int v = [] {
    const char* e = getenv("TWD_B256_MIN");
    return e ? atoi(e) : 128;  // measured: ViT proj/fc +11-16%
  }();
  return v;
}

static bool use_256(int M, int N, int K) {
  long b256 = ((M + BM2 - 1) / BM2) * (long)((N + BN2 - 1) / BN2);
  // At K = 64 (one K-step) the 256^2 tile is epilogue-dominated
  // (128 KiB LDS restage per block); the 128^2 tile wins there
  // (measured: 1x1 C64->256 189 -> 209 TF; K=128 already favors 256^2).
  return M >= 192 && N >= 192 && K >= 128 && b256 >= b256_min();
}

template <int TBM, int TBN, int MODE, int DEPTH = 2>
static void launch_tile(const void* A, const void* B, void* C,
                        const float* bias, const void* residual, int M,
                        int N, int K, int Kp, int64_t a_bytes, int act,
                        const ConvParams& cp, hipStream_t stream,
                        float* ws = nullptr, int splitk = 1, int ldc = 0) {
  constexpr int THREADS = (TBM / 64) * (TBN / 64) * 64;
  constexpr int LDS = DEPTH * (TBM + TBN) * BK * 2;
  constexpr int LDS_TOTAL = LDS > TBM * TBN * 2 ? LDS : TBM * TBN * 2;
  int tiles_m = (M + TBM - 1) / TBM;
  int tiles_n = (N + TBN - 1) / TBN;
  if (splitk > 1 && ws) {
    int nt = (K + BK - 1) / BK;
    int nt_slice = (nt + splitk - 1) / splitk;
    dim3 grid(tiles_m * tiles_n, 1, splitk);
    hipLaunchKernelGGL((gemm_conv_kernel<TBM, TBN, MODE, false, DEPTH>), grid,
                       dim3(THREADS), LDS_TOTAL, stream, (const __bf16*)A,
                       (const __bf16*)B, (__bf16*)C, nullptr,
                       (const __bf16*)nullptr, M, N, K, Kp, (int)a_bytes,
                       tiles_m, tiles_n, 0, cp, (int64_t)0, (int64_t)0,
                       (int64_t)0, ws, nt_slice, 0);
    int64_t MN = (int64_t)M * N;
    int blocks = (int)std::min<int64_t>((MN / 8 + 255) / 256, 2048);
    hipLaunchKernelGGL(splitk_combine_kernel, dim3(blocks), dim3(256), 0,
                       stream, ws, bias, (const __bf16*)residual,
                       (__bf16*)C, MN, N, splitk, act);
    return;
  }
  dim3 grid(tiles_m * tiles_n);
  static const char* t2dv = getenv("TWD_TILE2D");
  int tiles_n_arg = tiles_n;
  {
    const bool div4 = (tiles_m & 3) == 0 && (tiles_n & 3) == 0;
    const bool want = t2dv ? (t2dv[0] == '1')
                           : ((long)tiles_m * tiles_n >= 512);
    if (div4 && want) tiles_n_arg = -tiles_n;
  }
  if (residual)
    hipLaunchKernelGGL((gemm_conv_kernel<TBM, TBN, MODE, true, DEPTH>), grid,
                       dim3(THREADS), LDS_TOTAL, stream, (const __bf16*)A,
                       (const __bf16*)B, (__bf16*)C, bias,
                       (const __bf16*)residual, M, N, K, Kp, (int)a_bytes,
                       tiles_m, tiles_n_arg, act, cp, (int64_t)0,
                       (int64_t)0, (int64_t)0, nullptr, 1 << 28, ldc);
  else
    hipLaunchKernelGGL((gemm_conv_kernel<TBM, TBN, MODE, false, DEPTH>), grid,
                       dim3(THREADS), LDS_TOTAL, stream, (const __bf16*)A,
                       (const __bf16*)B, (__bf16*)C, bias,
                       (const __bf16*)nullptr, M, N, K, Kp, (int)a_bytes,
                       tiles_m, tiles_n_arg, act, cp, (int64_t)0,
                       (int64_t)0, (int64_t)0, nullptr, 1 << 28, ldc);
}

// How many K-slices should this shape use? (1 = no split)
int splitk_plan(int M, int N, int K) {
  if (use_256(M, N, K)) return 1;
  int TBN = N <= 96 ? 64 : 128;
  int TBM = N <= 96 ? 256 : 128;
  long blocks = ((M + TBM - 1) / TBM) * (long)((N + TBN - 1) / TBN);
  int nt = (K + BK - 1) / BK;
  // A/B override: TWD_SPLITK=n forces the split for underfilled shapes
  const char* fs = getenv("TWD_SPLITK");
  if (fs && blocks < 256 && nt >= 8) {
    int v = atoi(fs);
    if (v > 1) return v;
  }
  // Split only when the chip is badly underfilled: the f32 slab
  // round-trip costs ~bytes/6TBps, so moderate underfill (>=128 blocks,
  // or >=64 blocks at small K) is cheaper unsplit (measured: s3 3x3
  // regressed 30->45us at 196 blocks split 2).
  if (blocks >= 128 || nt < 4) return 1;
  if (blocks >= 64 && K < 3072) return 1;
  int sk = (int)((224 + blocks - 1) / blocks);
  if (sk > 8) sk = 8;
  while (sk > 1 && nt / sk < 2) --sk;
  return sk;
}

template <int MODE>
static void dispatch(const void* A, const void* B, void* C, const float* bias,
                     const void* residual, int M, int N, int K, int Kp,
                     int64_t a_bytes, int act, const ConvParams& cp,
                     hipStream_t stream, float* ws = nullptr,
                     int splitk = 1, int ldc = 0) {
  // 256^2-class shapes that underfill at 256^2 tiles (e.g. ViT
  // projections: M=12544, N=768 -> 147 blocks on 256 CUs) get a
  // (128,256) tile instead: same 8-wave pipeline, twice the blocks
  {
    long b256 = ((M + BM2 - 1) / BM2) * (long)((N + BN2 - 1) / BN2);
    if (M >= 192 && N >= 192 && K >= 128 && b256 < b256_min()
        && 2 * b256 >= 256 && getenv("TWD_128256") != nullptr) {
      // measured neutral-to-slightly-worse on ViT proj: opt-in only
      launch_tile<128, 256, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                     a_bytes, act, cp, stream, ws, 1, ldc);
      return;
    }
  }
  if (use_256(M, N, K)) {
    int tiles_m = (M + BM2 - 1) / BM2;
    int tiles_n = (N + BN2 - 1) / BN2;
    dim3 grid(tiles_m * tiles_n);
    // 4x4 tile clustering (see kernel comment): needs both tile dims
    // divisible by 4; negated tiles_n is the wire flag. Default ON in
    // the L2-overflow regime (nwg >= 512: 8192^3 measured 1030 -> 1293
    // TF, +26%); smaller grids already L2-fit and measured flat-to-
    // slightly-negative at 4096^3. TWD_TILE2D=0/1 forces.
    static const char* t2d = getenv("TWD_TILE2D");
    int tiles_n_arg = tiles_n;
    const bool div4 = (tiles_m & 3) == 0 && (tiles_n & 3) == 0;
    const bool want = t2d ? (t2d[0] == '1')
                          : ((long)tiles_m * tiles_n >= 512);
    if (div4 && want) tiles_n_arg = -tiles_n;
    if (use_256p()) {
      if (residual)
        hipLaunchKernelGGL((gemm_conv_kernel_256p<MODE, true>), grid,
                           dim3(NTHREADS2), LDS2_BYTES, stream,
                           (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                           bias, (const __bf16*)residual, M, N, K, Kp,
                           (int)a_bytes, tiles_m, tiles_n, act, cp, ldc);
      else
        hipLaunchKernelGGL((gemm_conv_kernel_256p<MODE, false>), grid,
                           dim3(NTHREADS2), LDS2_BYTES, stream,
                           (const __bf16*)A, (const __bf16*)B, (__bf16*)C,
                           bias, (const __bf16*)nullptr, M, N, K, Kp,
                           (int)a_bytes, tiles_m, tiles_n, act, cp, ldc);
      return;
    }
    if (residual)
      hipLaunchKernelGGL((gemm_conv_kernel_256<MODE, true>), grid,
                         dim3(NTHREADS2), LDS2_BYTES, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C, bias,
                         (const __bf16*)residual, M, N, K, Kp, (int)a_bytes,
                         tiles_m, tiles_n_arg, act, cp, ldc);
    else
      hipLaunchKernelGGL((gemm_conv_kernel_256<MODE, false>), grid,
                         dim3(NTHREADS2), LDS2_BYTES, stream,
                         (const __bf16*)A, (const __bf16*)B, (__bf16*)C, bias,
                         (const __bf16*)nullptr, M, N, K, Kp, (int)a_bytes,
                         tiles_m, tiles_n_arg, act, cp, ldc);
    return;
  }
  // DEPTH=3 measured WORSE here (120 KiB LDS -> 1 block/CU, losing the
  // cross-block overlap that was hiding half the staging latency):
  // stem 537->351 TF, s1 3x3 500->325. Keep DEPTH=2 + 2 blocks/CU.
  if (N <= 96) {
    // A/B harness for the wait-bound s1 class: (128,64) halves the
    // tile so 2-3 blocks/CU overlap more staging (TWD_S1=1: DEPTH 3 at
    // 2 blocks/CU; TWD_S1=2: DEPTH 2 at 3 blocks/CU)
    const char* s1 = getenv("TWD_S1");
    if (s1 && s1[0] == '1') {
      launch_tile<128, 64, MODE, 3>(A, B, C, bias, residual, M, N, K, Kp,
                                    a_bytes, act, cp, stream, ws, splitk,
                                    ldc);
      return;
    }
    if (s1 && s1[0] == '2') {
      launch_tile<128, 64, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                    a_bytes, act, cp, stream, ws, splitk,
                                    ldc);
      return;
    }
    launch_tile<256, 64, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                  a_bytes, act, cp, stream, ws, splitk, ldc);
    return;
  }
  // Mid-N (Inception 17x17, N=160/192) pad-waste routing to (128,64)
  // was A/B-tested and REGRESSED (inc1x7 308->298 TF, Inception e2e
  // 19.6k->19.3k img/s): the class is latency/wait-bound, not
  // MFMA-bound, so cutting padded-N waste buys nothing while the
  // extra A-panel re-read costs. Kept opt-in for future A/Bs.
  if (N > 96 && N <= 224 && getenv("TWD_N64") != nullptr) {
    launch_tile<128, 64, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                  a_bytes, act, cp, stream, ws, splitk,
                                  ldc);
    return;
  }
  // mid-size M underfills the chip at 128^2 (e.g. Inception 17x17
  // factorized convs: 290 blocks = ~1 block/CU); halve the M-tile to
  // double the grid when it buys >= ~1.5 blocks/CU
  long b128 = ((M + BM - 1) / BM) * (long)((N + BN - 1) / BN);
  // [48,128)-block shapes: (64,128) pays on gather convs (Inception
  // E/D blocks: +1.3% e2e) but loses on plain 1x1 GEMMs (ResNet
  // K=2048 2048->512: -3% e2e — the linear A re-reads cheaply, so the
  // bigger tile's fewer B re-reads win). Gate on the gather MODE.
  static const char* b64min_s = getenv("TWD_6412_MIN");
  const long b64min =
      b64min_s ? atol(b64min_s) : ((MODE == 1 || K >= 3072) ? 48 : 128);
  if (splitk == 1 && b128 >= b64min && b128 < 320 && getenv("TWD_NO_6412") == nullptr)
    launch_tile<64, 128, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                  a_bytes, act, cp, stream, ws, splitk, ldc);
  else
    launch_tile<128, 128, MODE, 2>(A, B, C, bias, residual, M, N, K, Kp,
                                   a_bytes, act, cp, stream, ws, splitk,
                                   ldc);
}

void launch_gemm_strided_bf16(const void* A, const void* B, void* C,
                              int batch1, int batch2, int M, int N, int K,
                              int lda, int ldb, int ldc, int64_t sA1,
                              int64_t sA2, int64_t sB1, int64_t sB2,
                              int64_t sC1, int64_t sC2, int64_t a_total,
                              hipStream_t stream) {
  ConvParams cp{};
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n, batch1 * batch2);
  hipLaunchKernelGGL((gemm_conv_kernel<128, 128, 0, false, 2>), grid,
                     dim3(NTHREADS), LDS_BYTES, stream, (const __bf16*)A,
                     (const __bf16*)B, (__bf16*)C, nullptr,
                     (const __bf16*)nullptr, M, N, K, K,
                     (int)(a_total * 2), tiles_m, tiles_n, 0, cp,
                     sA1, sB1, sC1, nullptr, 1 << 28, ldc, lda, ldb,
                     batch2, sA2, sB2, sC2);
}

void launch_gemm_batched_bf16(const void* A, const void* B, void* C,
                              int batch, int M, int N, int K, int Kp,
                              int64_t sA, int64_t sB, int64_t sC,
                              hipStream_t stream) {
  ConvParams cp{};
  int tiles_m = (M + BM - 1) / BM;
  int tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n, batch);
  hipLaunchKernelGGL((gemm_conv_kernel<128, 128, 0, false, 2>), grid,
                     dim3(NTHREADS), LDS_BYTES, stream, (const __bf16*)A,
                     (const __bf16*)B, (__bf16*)C, nullptr,
                     (const __bf16*)nullptr, M, N, K, Kp,
                     (int)(batch * M * (int64_t)K * 2), tiles_m, tiles_n, 0,
                     cp, sA, sB, sC);
}

void launch_gemm_bf16(const void* A, const void* B, void* C,
                      const float* bias, const void* residual, int M, int N,
                      int K, int Kp, int64_t a_bytes, int act,
                      hipStream_t stream, float* ws, int splitk, int ldc) {
  ConvParams cp{};
  dispatch<0>(A, B, C, bias, residual, M, N, K, Kp, a_bytes, act, cp,
              stream, ws, splitk, ldc);
}

void launch_conv2d_bf16(const void* X, const void* B, void* C,
                        const float* bias, const void* residual, int Nimg,
                        int H, int W, int Cin, int R, int S, int Cout, int sh,
                        int sw, int pt, int pl, int HO, int WO, int Kp,
                        int act, hipStream_t stream, float* ws,
                        int splitk, int ldc) {
  int M = Nimg * HO * WO;
  int K = R * S * Cin;
  // direct-conv path (conv_direct.hip): im2col-free for C%32==0
  // small/mid-channel R*S>=3 shapes (Inception stems / 35x35 / 17x17
  // classes — VALU-issue-bound on the gather path, see profiles/).
  // TWD_DIRECT=0 disables for A/B.
  {
    static const char* dis = getenv("TWD_DIRECT");
    if (!(dis && dis[0] == '0') && !residual && splitk <= 1 &&
        launch_conv_direct(X, B, C, bias, Nimg, H, W, Cin, R, S, Cout,
                           sh, sw, pt, pl, HO, WO, Kp, act, stream,
                           ldc ? ldc : Cout))
      return;
  }
  ConvParams cp;
  cp.C = Cin; cp.H = H; cp.W = W; cp.R = R; cp.S = S;
  cp.sh = sh; cp.sw = sw; cp.pt = pt; cp.pl = pl; cp.HO = HO; cp.WO = WO;
  cp.magic_C = magic40(Cin);
  cp.magic_S = magic40(S);
  cp.magic_HW = magic40(HO * WO);
  cp.magic_WO = magic40(WO);
  int64_t a_bytes = (int64_t)Nimg * H * W * Cin * 2;
  dispatch<1>(X, B, C, bias, residual, M, Cout, K, Kp, a_bytes, act, cp,
              stream, ws, splitk, ldc);
}
