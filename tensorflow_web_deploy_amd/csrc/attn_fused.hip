// Fused attention forward (flash-style) for ViT-class shapes (gfx950).
//
//   O[b,h] = softmax(Q[b,h] @ K[b,h]^T) @ V[b,h]
//   S (tokens) <= 1024, D = 64, bf16 in/out, f32 accumulators/stats.
//
// Replaces the unfused chain (QK^T GEMM -> softmax kernel -> V-transpose
// copy -> PV GEMM): the S x S score matrix never touches HBM.
//
// Structure (cdna_hip_programming.md §B attention, simplified for short
// sequences): one workgroup per (b,h); 4 waves, each owning 64 query
// rows; K and V^T staged in LDS once (no K/V tiling loop re-reads);
// per 32-key tile: QK^T MFMA (A@B^T fragments) -> online softmax on the
// D-fragment layout (16-lane-group shuffle row-reduce) -> P transposed
// through a wave-private LDS bounce -> PV MFMA against V^T (again plain
// A@B^T fragments). All MFMAs are v_mfma_f32_16x16x32_bf16 with the
// same fragment maps the GEMM kernel uses (validated there).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

#define AT_D 64          // head dim (fixed)
#define AT_QW 32         // query rows per wave (8 thin waves)
#define AT_QM (AT_QW / 16)  // m-fragments per wave
#define AT_KT 32         // keys per inner tile
#define AT_PLD 40        // P-bounce row stride (odd 16B slots)

extern __shared__ __attribute__((aligned(16))) char smem[];

// LDS layout (dynamic): K [Sp][64] bf16 (row-swizzled like the GEMM
// tiles), then Vt [64][Sp] bf16 (row padded +8 to break conflicts),
// then P-bounce [4 waves][64][AT_KT] bf16.
__global__ __launch_bounds__(512)
void attn_fused_kernel(const __bf16* __restrict__ Q,
                       const __bf16* __restrict__ K,
                       const __bf16* __restrict__ V,
                       __bf16* __restrict__ O,
                       int batch2, int S, int Sp,
                       int64_t sQ1, int64_t sQ2, int lqd,
                       int64_t sK1, int64_t sK2, int lkd,
                       int64_t sV1, int64_t sV2, int lvd,
                       int64_t sO1, int64_t sO2, int lod,
                       float scale) {
  const int b1 = blockIdx.x / batch2;
  const int b2 = blockIdx.x - b1 * batch2;
  Q += b1 * sQ1 + b2 * sQ2;
  K += b1 * sK1 + b2 * sK2;
  V += b1 * sV1 + b2 * sV2;
  O += b1 * sO1 + b2 * sO2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  const int vt_ld = Sp + 8;                  // Vt row stride (elements)
  __bf16* k_lds = (__bf16*)smem;             // [Sp][64], swizzled
  __bf16* vt_lds = k_lds + Sp * AT_D;        // [64][vt_ld]
  __bf16* p_lds = vt_lds + AT_D * vt_ld;     // [8][AT_QW][AT_PLD]

  // ---- stage K (swizzled rows, like the GEMM B tile) and Vt ----
  // Each thread loads 16B chunks of K/V rows; K goes in with the
  // slot^=(row&7) involution so ds_read_b128 fragments are spread.
  for (int idx = tid; idx < Sp * 8; idx += 512) {
    int row = idx >> 3;          // key index
    int slot = idx & 7;          // 16B slot within the 128B row
    bf16x8 kv;
    if (row < S) {
      kv = *(const bf16x8*)(K + (int64_t)row * lkd + slot * 8);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) kv[e] = (__bf16)0.f;
    }
    *(bf16x8*)(k_lds + row * AT_D + ((slot ^ (row & 7)) << 3)) = kv;
    // V transposed: v[row][d] -> vt[d][row]
    bf16x8 vv;
    if (row < S) {
      vv = *(const bf16x8*)(V + (int64_t)row * lvd + slot * 8);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) vv[e] = (__bf16)0.f;
    }
    // lane-rotated write order: without it the 8 lanes of a group all
    // write d-rows 8 apart (stride = a multiple of 32 banks) at the
    // same key column -> 8-way conflict on every ds_write_b16
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int dd = (e + lane) & 7;
      vt_lds[(slot * 8 + dd) * vt_ld + row] = vv[dd];
    }
  }

  // ---- load this wave's Q fragments (A-operand layout) ----
  // A frag (16x16x32): lane holds A[row = l&15][k = (l>>4)*8 .. +7].
  const int q0 = wave * AT_QW;
  bf16x8 qf[AT_QM][2];  // [m-frag][kk]
#pragma unroll
  for (int m = 0; m < AT_QM; ++m)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int row = q0 + m * 16 + (lane & 15);
      int d = kk * 32 + (lane >> 4) * 8;
      if (row < S) {
        qf[m][kk] = *(const bf16x8*)(Q + (int64_t)row * lqd + d);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          qf[m][kk][e] = (__bf16)(bf2f(qf[m][kk][e]) * scale);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) qf[m][kk][e] = (__bf16)0.f;
      }
    }

  __syncthreads();  // K/Vt resident

  // ---- online-softmax state ----
  // D-frag layout: for (m, reg r): q-row = m*16 + (lane>>4)*4 + r.
  float mrow[AT_QM][4], lrow[AT_QM][4];
#pragma unroll
  for (int m = 0; m < AT_QM; ++m)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      mrow[m][r] = -3.4e38f;
      lrow[m][r] = 0.f;
    }
  f32x4 oacc[AT_QM][4];  // [m][d-frag]: col d = n*16 + (l&15)
#pragma unroll
  for (int m = 0; m < AT_QM; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) oacc[m][n] = {0.f, 0.f, 0.f, 0.f};

  __bf16* pb = p_lds + wave * AT_QW * AT_PLD;

  const int ntile = Sp / AT_KT;
  for (int t = 0; t < ntile; ++t) {
    const int kbase = t * AT_KT;
    // ---- S-block = Q @ K_t^T : acc [4m][2n] ----
    f32x4 sacc[AT_QM][2];
#pragma unroll
    for (int m = 0; m < AT_QM; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n) sacc[m][n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      // B frag: lane holds K[key = kbase + n*16 + (l&15)]
      //                     [d = (l>>4)*8 .. +7] (swizzled read)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int krow = kbase + n * 16 + (lane & 15);
        int slot = kk * 4 + (lane >> 4);
        bf16x8 kf = *(bf16x8*)(k_lds + krow * AT_D +
                               ((slot ^ (krow & 7)) << 3));
#pragma unroll
        for (int m = 0; m < AT_QM; ++m)
          sacc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[m][kk], kf, sacc[m][n], 0, 0, 0);
      }
    }
    // ---- mask padded keys, row stats across the 16-lane group ----
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      int key = kbase + n * 16 + (lane & 15);
      if (key >= S)
#pragma unroll
        for (int m = 0; m < AT_QM; ++m)
          sacc[m][n] = {-3.4e38f, -3.4e38f, -3.4e38f, -3.4e38f};
    }
#pragma unroll
    for (int m = 0; m < AT_QM; ++m)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float mx = fmaxf(sacc[m][0][r], sacc[m][1][r]);
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off, 64));
        float mold = mrow[m][r];
        float mnew = fmaxf(mold, mx);
        float alpha = __expf(mold - mnew);  // exp(-inf)=0 first tile
        // rescale O and l
#pragma unroll
        for (int n = 0; n < 4; ++n) oacc[m][n][r] *= alpha;
        float p0 = __expf(sacc[m][0][r] - mnew);
        float p1 = __expf(sacc[m][1][r] - mnew);
        sacc[m][0][r] = p0;
        sacc[m][1][r] = p1;
        float rs = p0 + p1;
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          rs += __shfl_xor(rs, off, 64);
        lrow[m][r] = lrow[m][r] * alpha + rs;
        mrow[m][r] = mnew;
      }
    // ---- P -> wave-private LDS bounce (transpose to A layout) ----
    // write: q-row = m*16+(l>>4)*4+r, key-col = n*16+(l&15)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int m = 0; m < AT_QM; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qr = m * 16 + (lane >> 4) * 4 + r;
          int kc = n * 16 + (lane & 15);
          pb[qr * AT_PLD + kc] = (__bf16)sacc[m][n][r];
        }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // ---- PV: O += P @ Vt^T  (A = P [64q][32k], B = Vt [64d][32k]) ----
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      // B frag: lane holds Vt[d = n*16 + (l&15)][k = kbase+(l>>4)*8..+7]
      int drow = n * 16 + (lane & 15);
      const __bf16* vp = vt_lds + drow * vt_ld + kbase + (lane >> 4) * 8;
      bf16x8 vf = *(const bf16x8*)(vp);
#pragma unroll
      for (int m = 0; m < AT_QM; ++m) {
        // A frag: lane holds P[q = m*16 + (l&15)][k = (l>>4)*8 .. +7]
        bf16x8 pf = *(const bf16x8*)(pb + (m * 16 + (lane & 15)) * AT_PLD +
                                     (lane >> 4) * 8);
        oacc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, vf, oacc[m][n], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l, store (strided output view) ----
#pragma unroll
  for (int m = 0; m < AT_QM; ++m) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qr = q0 + m * 16 + (lane >> 4) * 4 + r;
      if (qr >= S) continue;
      float inv = 1.f / lrow[m][r];
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int d = n * 16 + (lane & 15);
        O[(int64_t)qr * lod + d] = (__bf16)(oacc[m][n][r] * inv);
      }
    }
  }
}

// ---------------------------------------------------------------------
// v2 (TWD_ATTN2=1): swapped-QK^T in-register softmax (guide T12 class).
// mfma(K, Q) puts the scores KEY-major / QUERY-per-lane, so the online
// softmax reduces 8 in-lane values + two stride-16/32 shuffles per
// query (vs 4 shuffle steps x 8 rows), and P reaches the PV A-fragment
// through cvt_pk + 4 ds_bpermute lane exchanges — the wave-private
// P-LDS bounce (32 addressed ds_write_b16 + 2 lgkm drains per tile)
// disappears. Fragment algebra in NOTES_ROUND3.md "T12 worked notes".
__global__ __launch_bounds__(512)
void attn_fused2_kernel(const __bf16* __restrict__ Q,
                        const __bf16* __restrict__ K,
                        const __bf16* __restrict__ V,
                        __bf16* __restrict__ O,
                        int batch2, int S, int Sp,
                        int64_t sQ1, int64_t sQ2, int lqd,
                        int64_t sK1, int64_t sK2, int lkd,
                        int64_t sV1, int64_t sV2, int lvd,
                        int64_t sO1, int64_t sO2, int lod,
                        float scale) {
  const int b1 = blockIdx.x / batch2;
  const int b2 = blockIdx.x - b1 * batch2;
  Q += b1 * sQ1 + b2 * sQ2;
  K += b1 * sK1 + b2 * sK2;
  V += b1 * sV1 + b2 * sV2;
  O += b1 * sO1 + b2 * sO2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);
  const int jq = lane >> 4;  // lane quarter

  const int vt_ld = Sp + 8;
  __bf16* k_lds = (__bf16*)smem;
  __bf16* vt_lds = k_lds + Sp * AT_D;

  // ---- stage K (swizzled) + Vt — identical to v1 ----
  for (int idx = tid; idx < Sp * 8; idx += 512) {
    int row = idx >> 3;
    int slot = idx & 7;
    bf16x8 kv;
    if (row < S) {
      kv = *(const bf16x8*)(K + (int64_t)row * lkd + slot * 8);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) kv[e] = (__bf16)0.f;
    }
    *(bf16x8*)(k_lds + row * AT_D + ((slot ^ (row & 7)) << 3)) = kv;
    bf16x8 vv;
    if (row < S) {
      vv = *(const bf16x8*)(V + (int64_t)row * lvd + slot * 8);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) vv[e] = (__bf16)0.f;
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int dd = (e + lane) & 7;
      vt_lds[(slot * 8 + dd) * vt_ld + row] = vv[dd];
    }
  }

  // ---- Q fragments (B-operand now; same per-lane map as v1's A) ----
  const int q0 = wave * AT_QW;
  bf16x8 qf[AT_QM][2];
#pragma unroll
  for (int m = 0; m < AT_QM; ++m)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int row = q0 + m * 16 + (lane & 15);
      int d = kk * 32 + (lane >> 4) * 8;
      if (row < S) {
        qf[m][kk] = *(const bf16x8*)(Q + (int64_t)row * lqd + d);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          qf[m][kk][e] = (__bf16)(bf2f(qf[m][kk][e]) * scale);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) qf[m][kk][e] = (__bf16)0.f;
      }
    }

  __syncthreads();

  // per-lane online-softmax state: ONE scalar per query group (the
  // lane's query is qn*16 + (lane&15))
  float m2[AT_QM], l2[AT_QM];
#pragma unroll
  for (int qn = 0; qn < AT_QM; ++qn) {
    m2[qn] = -3.4e38f;
    l2[qn] = 0.f;
  }
  f32x4 oacc[AT_QM][4];
#pragma unroll
  for (int m = 0; m < AT_QM; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) oacc[m][n] = {0.f, 0.f, 0.f, 0.f};

  const int ntile = Sp / AT_KT;
  for (int t = 0; t < ntile; ++t) {
    const int kbase = t * AT_KT;
    // ---- S^T block = K_t @ Q^T : sacc[km][qn], rows = keys ----
    f32x4 sacc[2][AT_QM];
#pragma unroll
    for (int km = 0; km < 2; ++km)
#pragma unroll
      for (int qn = 0; qn < AT_QM; ++qn) sacc[km][qn] = {0, 0, 0, 0};
#pragma unroll
    for (int km = 0; km < 2; ++km) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int krow = kbase + km * 16 + (lane & 15);
        int slot = kk * 4 + (lane >> 4);
        bf16x8 kf = *(bf16x8*)(k_lds + krow * AT_D +
                               ((slot ^ (krow & 7)) << 3));
#pragma unroll
        for (int qn = 0; qn < AT_QM; ++qn)
          sacc[km][qn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              kf, qf[qn][kk], sacc[km][qn], 0, 0, 0);
      }
    }
    // mask padded keys: key = kbase + km*16 + jq*4 + r
#pragma unroll
    for (int km = 0; km < 2; ++km)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int key = kbase + km * 16 + jq * 4 + r;
        if (key >= S)
#pragma unroll
          for (int qn = 0; qn < AT_QM; ++qn)
            sacc[km][qn][r] = -3.4e38f;
      }
    // ---- per-query stats + P (all in registers) ----
#pragma unroll
    for (int qn = 0; qn < AT_QM; ++qn) {
      float mx = sacc[0][qn][0];
#pragma unroll
      for (int km = 0; km < 2; ++km)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          mx = fmaxf(mx, sacc[km][qn][r]);
      mx = fmaxf(mx, __shfl_xor(mx, 16, 64));
      mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
      // defer-max (guide T13): keep the running max while the tile's
      // max stays within THR — skips the alpha exp, the broadcast and
      // the O rescale on most tiles. Exact modulo normalization scale
      // (P bounded by e^THR; f32 accumulators); the decision covers
      // THIS tile's P, taken wave-uniformly before exponentiation.
      const float mold = m2[qn];
      float mnew = mold, alpha = 1.f;
      if (__any(mx > mold + 8.f)) {
        mnew = fmaxf(mold, mx);
        alpha = __expf(mold - mnew);
        // rescale O rows: oacc rows are q' = (lane>>4)*4 + r; fetch
        // that query's alpha from the lane holding it (same quarter)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int src = (lane & 48) | ((jq * 4 + r) & 15);
          float a_r = __builtin_bit_cast(
              float, __builtin_amdgcn_ds_bpermute(
                         src << 2, __builtin_bit_cast(int, alpha)));
#pragma unroll
          for (int n = 0; n < 4; ++n) oacc[qn][n][r] *= a_r;
        }
      }
      float rs = 0.f;
#pragma unroll
      for (int km = 0; km < 2; ++km)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = __expf(sacc[km][qn][r] - mnew);
          sacc[km][qn][r] = p;
          rs += p;
        }
      rs += __shfl_xor(rs, 16, 64);
      rs += __shfl_xor(rs, 32, 64);
      l2[qn] = l2[qn] * alpha + rs;
      m2[qn] = mnew;
      // ---- pack P -> PV A-fragment via cvt_pk + bpermute ----
      // c[i]: bf16x2 of keys (jq*4 + 2i', +1) for i<2 (km=0) and
      // 16 + ... for i>=2 (km=1)
      uint32_t c[4];
#pragma unroll
      for (int km = 0; km < 2; ++km)
#pragma unroll
        for (int pr = 0; pr < 2; ++pr) {
          // plain RNE casts; the v_cvt_pk_bf16_f32 asm form corrupted
          // one packed half under register pressure (debug iteration 4)
          uint32_t lo16 = (uint32_t)__builtin_bit_cast(
              uint16_t, (__bf16)sacc[km][qn][pr * 2]);
          uint32_t hi16 = (uint32_t)__builtin_bit_cast(
              uint16_t, (__bf16)sacc[km][qn][pr * 2 + 1]);
          c[km * 2 + pr] = lo16 | (hi16 << 16);
        }
      // target lane jq' wants quarters {2(jq'&1), 2(jq'&1)+1} of
      // km = jq'>>1 at its own lane&15. bpermute contributions come
      // from the SOURCE lane, so permute every c-reg and select the
      // km half on the RECEIVER side (a source-side select picked the
      // wrong half for jq' in {1,2} — first-debug lesson).
      const int lo = ((lane & 15) | ((2 * (jq & 1)) << 4)) << 2;
      const int hi = ((lane & 15) | ((2 * (jq & 1) + 1) << 4)) << 2;
      uint32_t bl[4], bh[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        bl[i] = (uint32_t)__builtin_amdgcn_ds_bpermute(lo, (int)c[i]);
        bh[i] = (uint32_t)__builtin_amdgcn_ds_bpermute(hi, (int)c[i]);
      }
      const bool kmHi = jq >= 2;
      union {
        uint32_t u[4];
        bf16x8 v;
      } pf;
      pf.u[0] = kmHi ? bl[2] : bl[0];
      pf.u[1] = kmHi ? bl[3] : bl[1];
      pf.u[2] = kmHi ? bh[2] : bh[0];
      pf.u[3] = kmHi ? bh[3] : bh[1];
      // ---- PV: O[qn] += P @ Vt^T ----
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int drow = n * 16 + (lane & 15);
        bf16x8 vf = *(const bf16x8*)(vt_lds + drow * vt_ld + kbase +
                                     (lane >> 4) * 8);
        oacc[qn][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf.v, vf, oacc[qn][n], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l (broadcast inv like alpha), store ----
#pragma unroll
  for (int m = 0; m < AT_QM; ++m) {
    const float invl = 1.f / l2[m];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qr = q0 + m * 16 + jq * 4 + r;
      if (qr >= S) continue;
      int src = (lane & 48) | ((jq * 4 + r) & 15);
      float inv = __builtin_bit_cast(
          float, __builtin_amdgcn_ds_bpermute(
                     src << 2, __builtin_bit_cast(int, invl)));
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int d = n * 16 + (lane & 15);
        O[(int64_t)qr * lod + d] = (__bf16)(oacc[m][n][r] * inv);
      }
    }
  }
}

void launch_attn_fused_bf16(const void* Q, const void* K, const void* V,
                            void* O, int batch1, int batch2, int S,
                            int64_t sQ1, int64_t sQ2, int lqd, int64_t sK1,
                            int64_t sK2, int lkd, int64_t sV1, int64_t sV2,
                            int lvd, int64_t sO1, int64_t sO2, int lod,
                            float scale, hipStream_t stream) {
  int Sp = (S + AT_KT - 1) / AT_KT * AT_KT;
  int vt_ld = Sp + 8;
  size_t lds = (size_t)Sp * AT_D * 2 + (size_t)AT_D * vt_ld * 2 +
               8 * AT_QW * AT_PLD * 2;
  dim3 grid(batch1 * batch2);
  // v2 (swapped-QK^T in-register softmax) is the default: 39.5 vs
  // 62.1 us on the ViT shape, maxerr 6e-4 vs fp32. TWD_ATTN2=0 falls
  // back to the v1 P-LDS-bounce kernel.
  static const char* v2 = getenv("TWD_ATTN2");
  if (!(v2 && v2[0] == '0')) {
    hipLaunchKernelGGL(attn_fused2_kernel, grid, dim3(512), lds, stream,
                       (const __bf16*)Q, (const __bf16*)K,
                       (const __bf16*)V, (__bf16*)O, batch2, S, Sp, sQ1,
                       sQ2, lqd, sK1, sK2, lkd, sV1, sV2, lvd, sO1, sO2,
                       lod, scale);
    return;
  }
  hipLaunchKernelGGL(attn_fused_kernel, grid, dim3(512), lds, stream,
                     (const __bf16*)Q, (const __bf16*)K, (const __bf16*)V,
                     (__bf16*)O, batch2, S, Sp, sQ1, sQ2, lqd, sK1, sK2,
                     lkd, sV1, sV2, lvd, sO1, sO2, lod, scale);
}
