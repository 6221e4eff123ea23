// Direct (im2col-free) conv for small/mid channel counts, NHWC bf16.
//
// Why it exists (round-2 PMC, profiles/round2_inception.md): the
// implicit-GEMM kernel issues one buffer_load per 16 B of *im2col* data
// — every input byte is re-gathered R*S times with ~26 VALU of magic-
// division address math per load. On Inception's stems that makes the
// kernel VALU-ISSUE-bound: 7.6 VALU instructions per MFMA, 43% of wave
// cycles busy issuing, 213 TF. The 17x17 1x7/7x1 class shows the same
// signature (3.85 VALU/MFMA, 40% parked).
//
// This kernel loads each input byte ONCE: a workgroup owns (image n,
// band of TH output rows), stages the needed input rows into LDS with
// row-contiguous chunked buffer_load_lds (2 VALU per 1 KiB), then runs
// the conv as sum over (r,s) of shifted GEMMs: for each kernel tap the
// A-fragment is a *contiguous* 16-pixel LDS read, B-fragments stream
// from global (weights are L2-resident), and accumulators live in
// registers across all taps. MFMA v_mfma_f32_16x16x32_bf16, fragment
// conventions identical to gemm_conv.hip.
//
// Routing constraints (dispatch in launch_conv_direct): C % 32 == 0,
// 16 <= Cout (= N) and N % 16 == 0, no residual input. Padding is
// handled by staging the halo columns and zeroing OOB rows/cols in
// LDS (SAME) — frag reads then never mask.
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

#include "kernels.h"

typedef __bf16 bf16x8_cd __attribute__((ext_vector_type(8)));
typedef float f32x4_cd __attribute__((ext_vector_type(4)));

extern __shared__ __attribute__((aligned(16))) char smem_cd[];


// 40-bit magic division helpers (same scheme as gemm_conv.hip)
__device__ __forceinline__ uint32_t cd_mdiv(uint32_t x, uint64_t magic) {
  return (uint32_t)((x * magic) >> 40);
}

// MFW = m-fragments per wave, NF = n-fragments (Cout/16)
template <int MFW, int NF, int ACT>
__global__ __launch_bounds__(256) void conv_direct_kernel(
    const __bf16* __restrict__ X, const __bf16* __restrict__ W2,
    __bf16* __restrict__ Y, const float* __restrict__ bias,
    int Nimg, int H, int W, int C, int R, int S, int sh, int sw,
    int pt, int pl, int HO, int WO, int Cout, int Kp, int TH,
    int bands, int rowstride, int a_bytes, uint64_t magic_WO,
    int ldc) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  int nwg = Nimg * bands;
  int bid = blockIdx.x;
  {  // XCD-aware bijective remap
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int band = bid % bands;
  const int n = bid / bands;
  const int oy0 = band * TH;
  const int THb = min(TH, HO - oy0);
  const int MF = (THb * WO + 15) >> 4;  // m-frags in this band

  const auto rsrcX = __builtin_amdgcn_make_buffer_rsrc(
      (void*)X, 0, a_bytes, 0x00020000);

  // ---- stage the input band: TI rows x WI cols x C, row-contiguous.
  // LDS row ti holds input row iy0+ti, cols [-pl, -pl+WI). Padding:
  // C % 32 == 0 makes the pad/image byte boundaries (pl*C*2 and
  // (pl+W)*C*2) multiples of 64, so every 16 B chunk is either fully
  // pad or fully image — pad chunks use the OOB-voffset sentinel and
  // arrive as zeros. OOB rows (top/bottom SAME padding) zero the same
  // way.
  const int TI = (THb - 1) * sh + R;
  const int WI = (WO - 1) * sw + S;
  const int iy0 = oy0 * sh - pt;
  const int rowbytes = WI * C * 2;
  const int lb = pl * C * 2;                        // first image byte
  const int rbv = min(rowbytes, (pl + W) * C * 2);  // first right-pad byte
  const int64_t img_base = (int64_t)n * H * W * C * 2;
  for (int ti = 0; ti < TI; ++ti) {
    const int iy = iy0 + ti;
    const bool rowok = (unsigned)iy < (unsigned)H;
    // global byte offset of LDS col 0 (= image col -pl)
    const int rowbase =
        (int)(img_base + ((int64_t)iy * W - pl) * C * 2);
    for (int ch = tid * 16; ch < rowbytes; ch += 256 * 16) {
      const bool ok = rowok && ch >= lb && ch < rbv;
      int32_t voff = ok ? rowbase + ch : 0x7ffffff0;
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          rsrcX,
          (__attribute__((address_space(3))) void*)(
              smem_cd + ti * rowstride + ch),
          16, voff, 0, 0, 0);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // ---- per-wave m-fragments: wave handles mf = wave, wave+4, ...
  // px (flat within band) = mf*16 + lane%16 -> (oy_l, ox) via magic
  int px_oy[MFW], px_ox[MFW];
#pragma unroll
  for (int i = 0; i < MFW; ++i) {
    int mf = wave + i * 4;
    int px = min(mf * 16 + (lane & 15), THb * WO - 1);
    uint32_t oy = cd_mdiv((uint32_t)px, magic_WO);
    px_oy[i] = (int)oy;
    px_ox[i] = px - (int)oy * WO;
  }

  f32x4_cd acc[MFW][NF];
#pragma unroll
  for (int i = 0; i < MFW; ++i)
#pragma unroll
    for (int nf = 0; nf < NF; ++nf)
      acc[i][nf] = f32x4_cd{0.f, 0.f, 0.f, 0.f};

  const int kq = lane >> 4;       // k-quarter 0..3 (8 elems)
  const int KC = C >> 5;          // 32-wide k-steps per tap
  const int T = R * S * KC;       // fused (tap, k-step) loop count
  const int wrow = (lane & 15);   // weight row within an n-frag
  // double-buffered B prefetch with NAMED even/odd register sets (a
  // runtime-indexed frag array would spill to scratch — rule 20): the
  // loads for step t+1 issue before step t's MFMA burst so the
  // global(L2) latency hides under it — without this the kernel
  // read-use serializes at 1 wave/SIMD on small-map shapes (measured
  // 2x loss vs the GEMM path)
  bf16x8_cd bfA[NF], bfB[NF];
  // incremental (rs, kc) cursors: n = prefetch target, c = compute
  int rs_n = 0, kc_n = 0, rs_c = 0, kc_c = 0;
  // macro-stamped bodies so the frag arrays are always referenced by
  // NAME (a pointer/reference parameter would make them addressable
  // and spill to scratch)
#define CD_LOAD_B(bf)                                                   \
  do {                                                                  \
    const int kbase = ((rs_n * C) + kc_n * 32 + kq * 8) * 2;            \
    _Pragma("unroll") for (int nf = 0; nf < NF; ++nf) bf[nf] =          \
        *(const bf16x8_cd*)((const char*)W2 +                           \
                            (nf * 16 + wrow) * Kp * 2 + kbase);         \
    if (++kc_n == KC) {                                                 \
      kc_n = 0;                                                         \
      ++rs_n;                                                           \
    }                                                                   \
  } while (0)

#define CD_COMPUTE(bf)                                                  \
  do {                                                                  \
    const int r = rs_c / S, s = rs_c - (rs_c / S) * S;                  \
    const int abase = (kc_c * 32 + kq * 8) * 2;                         \
    bf16x8_cd af0, af1;                                                 \
    {                                                                   \
      const int row = px_oy[0] * sh + r;                                \
      const int col = px_ox[0] * sw + s;                                \
      af0 = *(const bf16x8_cd*)(smem_cd + row * rowstride +             \
                                col * C * 2 + abase);                   \
    }                                                                   \
    _Pragma("unroll") for (int i = 0; i < MFW; ++i) {                   \
      if (i + 1 < MFW) {                                                \
        const int row = px_oy[i + 1] * sh + r;                          \
        const int col = px_ox[i + 1] * sw + s;                          \
        if ((i & 1) == 0)                                               \
          af1 = *(const bf16x8_cd*)(smem_cd + row * rowstride +         \
                                    col * C * 2 + abase);               \
        else                                                            \
          af0 = *(const bf16x8_cd*)(smem_cd + row * rowstride +         \
                                    col * C * 2 + abase);               \
      }                                                                 \
      _Pragma("unroll") for (int nf = 0; nf < NF; ++nf) acc[i][nf] =    \
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(                      \
              (i & 1) ? af1 : af0, bf[nf], acc[i][nf], 0, 0, 0);        \
    }                                                                   \
    if (++kc_c == KC) {                                                 \
      kc_c = 0;                                                         \
      ++rs_c;                                                           \
    }                                                                   \
  } while (0)

  CD_LOAD_B(bfA);
  int t = 0;
  for (; t + 2 <= T; t += 2) {
    if (t + 1 < T) CD_LOAD_B(bfB);
    CD_COMPUTE(bfA);
    if (t + 2 < T) CD_LOAD_B(bfA);
    CD_COMPUTE(bfB);
  }
  if (t < T) CD_COMPUTE(bfA);  // odd tail
#undef CD_LOAD_B
#undef CD_COMPUTE
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();  // input band no longer needed

  // ---- epilogue: acc -> LDS [band px][Cout] bf16, then chunked,
  // ldc-strided global stores
  __bf16* ct = (__bf16*)smem_cd;
#pragma unroll
  for (int i = 0; i < MFW; ++i) {
    int mf = wave + i * 4;
    if (mf >= MF) break;
#pragma unroll
    for (int nf = 0; nf < NF; ++nf) {
      int col = nf * 16 + (lane & 15);
      float b0 = bias ? bias[col] : 0.f;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int px = mf * 16 + kq * 4 + e;
        if (px < THb * WO) {
          float v = acc[i][nf][e] + b0;
          if (ACT == 1) v = fmaxf(v, 0.f);
          if (ACT == 2) v = fminf(fmaxf(v, 0.f), 6.f);
          ct[px * Cout + col] = (__bf16)v;
        }
      }
    }
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int npx = THb * WO;
  const int pxbytes = Cout * 2;        // multiple of 32 (Cout % 16 == 0)
  const int chunks_per_px = pxbytes >> 4;
  const int64_t out_base = (((int64_t)n * HO + oy0) * WO) * ldc * 2;
  for (int idx = tid; idx < npx * chunks_per_px; idx += 256) {
    int px = idx / chunks_per_px;
    int ch = idx - px * chunks_per_px;
    *(bf16x8_cd*)((char*)Y + out_base + (int64_t)px * ldc * 2 + ch * 16) =
        *(const bf16x8_cd*)((const char*)ct + px * pxbytes + ch * 16);
  }
}

// host-side dispatch ------------------------------------------------

template <int MFW, int NF>
static void cd_launch_act(const void* X, const void* W2, void* Y,
                          const float* bias, int Nimg, int H, int W, int C,
                          int R, int S, int sh, int sw, int pt, int pl,
                          int HO, int WO, int Cout, int Kp, int TH,
                          int bands, int rowstride, int a_bytes,
                          uint64_t magic_WO, int ldc, int act, int lds,
                          int grid, hipStream_t stream) {
  switch (act) {
    case 1:
      hipLaunchKernelGGL((conv_direct_kernel<MFW, NF, 1>), dim3(grid),
                         dim3(256), lds, stream, (const __bf16*)X,
                         (const __bf16*)W2, (__bf16*)Y, bias, Nimg, H, W, C,
                         R, S, sh, sw, pt, pl, HO, WO, Cout, Kp, TH, bands,
                         rowstride, a_bytes, magic_WO, ldc);
      break;
    case 2:
      hipLaunchKernelGGL((conv_direct_kernel<MFW, NF, 2>), dim3(grid),
                         dim3(256), lds, stream, (const __bf16*)X,
                         (const __bf16*)W2, (__bf16*)Y, bias, Nimg, H, W, C,
                         R, S, sh, sw, pt, pl, HO, WO, Cout, Kp, TH, bands,
                         rowstride, a_bytes, magic_WO, ldc);
      break;
    default:
      hipLaunchKernelGGL((conv_direct_kernel<MFW, NF, 0>), dim3(grid),
                         dim3(256), lds, stream, (const __bf16*)X,
                         (const __bf16*)W2, (__bf16*)Y, bias, Nimg, H, W, C,
                         R, S, sh, sw, pt, pl, HO, WO, Cout, Kp, TH, bands,
                         rowstride, a_bytes, magic_WO, ldc);
  }
}

static uint64_t cd_magic40(uint32_t d) {
  if (d == 0) return 0;
  return ((((__uint128_t)1) << 40) + d - 1) / d;
}

// Returns true if the shape was handled (else caller uses the GEMM path)
bool launch_conv_direct(const void* X, const void* W2, void* Y,
                        const float* bias, int Nimg, int H, int W, int C,
                        int R, int S, int Cout, int sh, int sw, int pt,
                        int pl, int HO, int WO, int Kp, int act,
                        hipStream_t stream, int ldc) {
  // Routing: measured A/B (profiles/round2_inception.md) — the direct
  // kernel wins on BIG-MAP small-channel shapes (Inception stems,
  // 147^2 C=32: 103 -> 77 us) where the im2col gather is VALU-bound
  // and the weight set is tiny; it LOSES on small maps (17x17 1x7:
  // 21 -> 47 us — B-frag streaming at 1 wave/SIMD) and mid maps
  // (35x35: 27 -> 34 us). Keep it to the measured-win region;
  // TWD_DIRECT=2 opens the full C%32 surface for experiments.
  static const char* wide = getenv("TWD_DIRECT");
  const bool open_all = wide && wide[0] == '2';
  if (C % 32 != 0 || Cout % 16 != 0 || Cout < 16 || Cout > 192 ||
      R * S < 3 || act > 2)
    return false;
  // Cout<=48: the Cout=64 stem (conv2) measured 153 us via direct vs
  // 89 us on the GEMM path in-model (NF=4 halves MFW, and the wider
  // epilogue tile halves the LDS bands) — only the narrow-N stems win
  if (!open_all && !(C <= 64 && Cout <= 48 && WO >= 96))
    return false;
  const int NF = Cout / 16;
  // MFW from the VGPR budget: MFW * NF <= 24 and MFW <= 10
  int MFW = 24 / NF;
  if (MFW > 10) MFW = 10;
  if (MFW < 1) return false;
  // TH: per-band output rows. Constraints: m-frags <= MFW*4,
  // LDS band + epilogue tile <= ~80 KiB, grid >= ~256 wgs.
  int TH = (MFW * 4 * 16) / WO;
  if (TH < 1) return false;
  for (;; --TH) {
    if (TH < 1) return false;
    int TI = (TH - 1) * sh + R;
    int WI = (WO - 1) * sw + S;
    int rowstride = ((WI * C * 2 + 1023) / 1024) * 1024;
    int lds_in = TI * rowstride;
    int lds_out = TH * WO * Cout * 2;
    if (lds_in < lds_out) lds_in = lds_out;
    if (lds_in > (100 << 10)) continue;
    int bands = (HO + TH - 1) / TH;
    // accept; underfill guard: need enough workgroups to fill the chip
    if ((int64_t)Nimg * bands < 224 && TH > 1) continue;
    int64_t ab64 = (int64_t)Nimg * H * W * C * 2;
    if (ab64 > 0x7ffffff0) return false;  // rowbase is 32-bit
    uint64_t mwo = cd_magic40((uint32_t)WO);
    int grid = Nimg * bands;
    switch (NF) {
      case 2:
        cd_launch_act<10, 2>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                             pt, pl, HO, WO, Cout, Kp, TH, bands,
                             rowstride, (int)ab64, mwo, ldc, act, lds_in,
                             grid, stream);
        return true;
      case 4:
        cd_launch_act<6, 4>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                            pt, pl, HO, WO, Cout, Kp, TH, bands, rowstride,
                            (int)ab64, mwo, ldc, act, lds_in, grid, stream);
        return true;
      case 6:
        cd_launch_act<4, 6>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                            pt, pl, HO, WO, Cout, Kp, TH, bands, rowstride,
                            (int)ab64, mwo, ldc, act, lds_in, grid, stream);
        return true;
      case 8:
        cd_launch_act<3, 8>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                            pt, pl, HO, WO, Cout, Kp, TH, bands, rowstride,
                            (int)ab64, mwo, ldc, act, lds_in, grid, stream);
        return true;
      case 10:
        cd_launch_act<2, 10>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                             pt, pl, HO, WO, Cout, Kp, TH, bands,
                             rowstride, (int)ab64, mwo, ldc, act, lds_in,
                             grid, stream);
        return true;
      case 12:
        cd_launch_act<2, 12>(X, W2, Y, bias, Nimg, H, W, C, R, S, sh, sw,
                             pt, pl, HO, WO, Cout, Kp, TH, bands,
                             rowstride, (int)ab64, mwo, ldc, act, lds_in,
                             grid, stream);
        return true;
      default:
        return false;
    }
  }
}
