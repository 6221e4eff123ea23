// pybind11/torch bindings for the CDNA4 HIP kernels (native path of the
// engine). Pure HIP — no CUDA compatibility layer: streams come from
// c10::hip directly.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "kernels.h"

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// 2D output that may be a column-slice view: [M][N] with row stride ldc
int check_out2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 2 && t.stride(1) == 1, name,
              " must be 2D with contiguous rows");
  return (int)t.stride(0);
}

// C[M][N] = act(A[M][K] @ B[N][K]^T + bias [+ residual])
void gemm_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor C,
               c10::optional<torch::Tensor> bias,
               c10::optional<torch::Tensor> residual, int64_t K,
               int64_t act) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  int ldc = check_out2d(C, "C");
  int64_t M = C.size(0), N = C.size(1);
  int64_t Kp = B.size(1);
  TORCH_CHECK(B.size(0) >= N, "B rows < N");
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
  const float* bptr = nullptr;
  if (bias) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32, "bias must be f32");
    bptr = bias->data_ptr<float>();
  }
  const void* rptr = nullptr;
  if (residual) {
    check_bf16(*residual, "residual");
    rptr = residual->data_ptr();
  }
  int sk = ldc == (int)N ? splitk_plan((int)M, (int)N, (int)K) : 1;
  torch::Tensor ws;
  float* wptr = nullptr;
  if (sk > 1) {
    ws = torch::empty({(int64_t)sk * M * N},
                      torch::dtype(torch::kFloat32).device(A.device()));
    wptr = ws.data_ptr<float>();
  }
  launch_gemm_bf16(A.data_ptr(), B.data_ptr(), C.data_ptr(), bptr, rptr,
                   (int)M, (int)N, (int)K, (int)Kp,
                   A.numel() * 2, (int)act, cur_stream(), wptr, sk, ldc);
}

// X [N,H,W,C] bf16, W2 [Cout][Kp] (pre-transposed HWIO, K-padded)
void conv2d_bf16(torch::Tensor X, torch::Tensor W2, torch::Tensor Y,
                 c10::optional<torch::Tensor> bias,
                 c10::optional<torch::Tensor> residual, int64_t R, int64_t S,
                 int64_t sh, int64_t sw, int64_t pt, int64_t pl,
                 int64_t act) {
  check_bf16(X, "X");
  check_bf16(W2, "W");
  TORCH_CHECK(X.dim() == 4 && Y.dim() == 4);
  // Y may be a channel-slice of a concat buffer: last dim contiguous,
  // pixel stride = total concat channels
  TORCH_CHECK(Y.is_cuda() && Y.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(Y.stride(3) == 1);
  int ldc = (int)Y.stride(2);
  int Nimg = X.size(0), H = X.size(1), W = X.size(2), Cin = X.size(3);
  int HO = Y.size(1), WO = Y.size(2), Cout = Y.size(3);
  TORCH_CHECK(Cin % 8 == 0, "Cin must be a multiple of 8 (pad first)");
  const float* bptr = nullptr;
  if (bias) bptr = bias->data_ptr<float>();
  const void* rptr = nullptr;
  if (residual) {
    check_bf16(*residual, "residual");
    rptr = residual->data_ptr();
  }
  int M = Nimg * HO * WO;
  int K = (int)(R * S) * Cin;
  int sk = ldc == Cout ? splitk_plan(M, Cout, K) : 1;
  torch::Tensor ws;
  float* wptr = nullptr;
  if (sk > 1) {
    ws = torch::empty({(int64_t)sk * M * Cout},
                      torch::dtype(torch::kFloat32).device(X.device()));
    wptr = ws.data_ptr<float>();
  }
  launch_conv2d_bf16(X.data_ptr(), W2.data_ptr(), Y.data_ptr(), bptr, rptr,
                     Nimg, H, W, Cin, (int)R, (int)S, Cout, (int)sh, (int)sw,
                     (int)pt, (int)pl, HO, WO, (int)W2.size(1), (int)act,
                     cur_stream(), wptr, sk, ldc);
}

void maxpool_bf16(torch::Tensor X, torch::Tensor Y, int64_t kh, int64_t kw,
                  int64_t sh, int64_t sw, int64_t pt, int64_t pl) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  launch_maxpool_bf16(X.data_ptr(), Y.data_ptr(), X.size(0), X.size(1),
                      X.size(2), X.size(3), kh, kw, sh, sw, pt, pl, Y.size(1),
                      Y.size(2), cur_stream());
}

void avgpool_bf16(torch::Tensor X, torch::Tensor Y, int64_t kh, int64_t kw,
                  int64_t sh, int64_t sw, int64_t pt, int64_t pl) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  launch_avgpool_bf16(X.data_ptr(), Y.data_ptr(), X.size(0), X.size(1),
                      X.size(2), X.size(3), kh, kw, sh, sw, pt, pl, Y.size(1),
                      Y.size(2), cur_stream());
}

void global_mean_bf16(torch::Tensor X, torch::Tensor Y) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  launch_global_mean_bf16(X.data_ptr(), Y.data_ptr(), X.size(0),
                          X.size(1) * X.size(2), X.size(3), cur_stream());
}

void softmax_bf16(torch::Tensor X, torch::Tensor Y) {
  check_bf16(X, "X");
  TORCH_CHECK(Y.scalar_type() == torch::kFloat32);
  launch_softmax_bf16(X.data_ptr(), Y.data_ptr<float>(), X.size(0), X.size(1),
                      cur_stream());
}

void softmax_bf16_out(torch::Tensor X, torch::Tensor Y) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  launch_softmax_bf16_out(X.data_ptr(), Y.data_ptr(), X.size(0), X.size(1),
                          cur_stream());
}

void cast_f32_bf16(torch::Tensor X, torch::Tensor Y) {
  TORCH_CHECK(X.scalar_type() == torch::kFloat32 && X.is_contiguous());
  check_bf16(Y, "Y");
  launch_cast_f32_bf16(X.data_ptr<float>(), Y.data_ptr(), X.numel(),
                       cur_stream());
}

void cast_bf16_f32(torch::Tensor X, torch::Tensor Y) {
  check_bf16(X, "X");
  TORCH_CHECK(Y.scalar_type() == torch::kFloat32);
  launch_cast_bf16_f32(X.data_ptr(), Y.data_ptr<float>(), X.numel(),
                       cur_stream());
}

void cast_f32_pad_bf16(torch::Tensor X, torch::Tensor Y) {
  TORCH_CHECK(X.scalar_type() == torch::kFloat32 && X.is_contiguous());
  check_bf16(Y, "Y");
  int64_t rows = X.numel() / X.size(-1);
  launch_cast_f32_pad_bf16(X.data_ptr<float>(), Y.data_ptr(), rows,
                           X.size(-1), Y.size(-1), cur_stream());
}

void channel_pad_bf16(torch::Tensor X, torch::Tensor Y) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  int64_t rows = X.numel() / X.size(-1);
  launch_channel_pad_bf16(X.data_ptr(), Y.data_ptr(), rows, X.size(-1),
                          Y.size(-1), cur_stream());
}

void add_relu_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor Y,
                   bool relu) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  check_bf16(Y, "Y");
  launch_add_relu_bf16(A.data_ptr(), B.data_ptr(), Y.data_ptr(), A.numel(),
                       relu, cur_stream());
}

void bias_act_bf16(torch::Tensor X, c10::optional<torch::Tensor> bias,
                   torch::Tensor Y, int64_t act) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  int C = X.size(-1);
  const float* bptr = nullptr;
  if (bias) {
    TORCH_CHECK(bias->numel() == C, "bias size ", bias->numel(),
                " != channels ", C);
    bptr = bias->data_ptr<float>();
  }
  launch_bias_act_bf16(X.data_ptr(), bptr, Y.data_ptr(), X.numel() / C, C,
                       (int)act, cur_stream());
}

void scale_shift_bf16(torch::Tensor X, torch::Tensor scale,
                      torch::Tensor shift, torch::Tensor Y, bool relu) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  int C = X.size(-1);
  TORCH_CHECK(scale.numel() == C && shift.numel() == C,
              "scale/shift size mismatch vs channels ", C);
  launch_scale_shift_bf16(X.data_ptr(), scale.data_ptr<float>(),
                          shift.data_ptr<float>(), Y.data_ptr(),
                          X.numel() / C, C, relu, cur_stream());
}

void layernorm_bf16(torch::Tensor X, c10::optional<torch::Tensor> gamma,
                    c10::optional<torch::Tensor> beta, torch::Tensor Y,
                    double eps) {
  check_bf16(X, "X");
  check_bf16(Y, "Y");
  int C = X.size(-1);
  TORCH_CHECK(C % 8 == 0 && C <= 2048, "layernorm C must be %8 and <=2048");
  const float* g = gamma ? gamma->data_ptr<float>() : nullptr;
  const float* b = beta ? beta->data_ptr<float>() : nullptr;
  launch_layernorm_bf16(X.data_ptr(), g, b, Y.data_ptr(), X.numel() / C, C,
                        (float)eps, cur_stream());
}

// batched C[b] = A[b] @ B[b]^T  (A [batch,M,K], B [batch,N,K], C [batch,M,N])
void gemm_batched_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor C) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  check_bf16(C, "C");
  TORCH_CHECK(A.dim() == 3 && B.dim() == 3 && C.dim() == 3);
  int batch = A.size(0), M = A.size(1), K = A.size(2);
  int N = B.size(1);
  TORCH_CHECK(B.size(0) == batch && C.size(0) == batch);
  TORCH_CHECK(B.size(2) == K && C.size(1) == M && C.size(2) == N);
  TORCH_CHECK(K % 8 == 0, "batched gemm K must be a multiple of 8");
  launch_gemm_batched_bf16(A.data_ptr(), B.data_ptr(), C.data_ptr(), batch,
                           M, N, K, K, (int64_t)M * K, (int64_t)N * K,
                           (int64_t)M * N, cur_stream());
}

// src: uint8 [sh, sw, sc]; dst: f32 [dh, dw, dc] on the given stream
void resize_norm(torch::Tensor src, torch::Tensor dst, int64_t norm) {
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == torch::kUInt8
              && src.is_contiguous());
  TORCH_CHECK(dst.is_cuda() && dst.scalar_type() == torch::kFloat32
              && dst.is_contiguous());
  launch_resize_norm(src.data_ptr(), src.size(0), src.size(1), src.size(2),
                     dst.data_ptr<float>(), dst.size(0), dst.size(1),
                     dst.size(2), (int)norm, cur_stream());
}

// arena u8 (packed images), meta i32 [n,4] = {off, sh, sw, sc} on GPU,
// dst f32 [n, H, W, C]
void resize_norm_batch(torch::Tensor arena, torch::Tensor meta,
                       torch::Tensor dst, int64_t norm) {
  TORCH_CHECK(arena.is_cuda() && arena.scalar_type() == torch::kUInt8);
  TORCH_CHECK(meta.is_cuda() && meta.scalar_type() == torch::kInt32);
  TORCH_CHECK(dst.is_cuda() && dst.scalar_type() == torch::kFloat32
              && dst.is_contiguous());
  launch_resize_norm_batch(arena.data_ptr(), meta.data_ptr<int>(),
                           dst.data_ptr<float>(), dst.size(0), dst.size(1),
                           dst.size(2), dst.size(3), (int)norm,
                           cur_stream());
}

// Attention GEMM on 4D views: every tensor supplies its own strides.
// A: [B1, M, ...] with k contiguous; B: [B1, N-rows ...]; C likewise.
// Python passes logical dims + strides explicitly (views, zero-copy).
void gemm_strided_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                       int64_t batch1, int64_t batch2, int64_t M, int64_t N,
                       int64_t K, int64_t lda, int64_t ldb, int64_t ldc,
                       int64_t sA1, int64_t sA2, int64_t sB1, int64_t sB2,
                       int64_t sC1, int64_t sC2) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(C.is_cuda() && C.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(K % 8 == 0, "strided gemm K must be %8");
  launch_gemm_strided_bf16(A.data_ptr(), B.data_ptr(), C.data_ptr(),
                           (int)batch1, (int)batch2, (int)M, (int)N, (int)K,
                           (int)lda, (int)ldb, (int)ldc, sA1, sA2, sB1, sB2,
                           sC1, sC2, A.numel(), cur_stream());
}

// Fused attention: Q/K/V/O are 4D views [B1, S, B2, D] or [B1, B2, S, D]
// etc — every tensor supplies explicit strides; D must be 64, S <= 256.
void attn_fused_bf16(torch::Tensor Q, torch::Tensor K, torch::Tensor V,
                     torch::Tensor O, int64_t batch1, int64_t batch2,
                     int64_t S, int64_t sQ1, int64_t sQ2, int64_t lqd,
                     int64_t sK1, int64_t sK2, int64_t lkd, int64_t sV1,
                     int64_t sV2, int64_t lvd, int64_t sO1, int64_t sO2,
                     int64_t lod, double scale) {
  TORCH_CHECK(Q.is_cuda() && Q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(S <= 256, "fused attention supports S <= 256");
  launch_attn_fused_bf16(Q.data_ptr(), K.data_ptr(), V.data_ptr(),
                         O.data_ptr(), (int)batch1, (int)batch2, (int)S,
                         sQ1, sQ2, (int)lqd, sK1, sK2, (int)lkd, sV1, sV2,
                         (int)lvd, sO1, sO2, (int)lod, (float)scale,
                         cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_bf16", &gemm_bf16, "bf16 MFMA GEMM (A @ B^T + bias, act)");
  m.def("conv2d_bf16", &conv2d_bf16, "bf16 implicit-GEMM conv2d NHWC");
  m.def("maxpool_bf16", &maxpool_bf16);
  m.def("avgpool_bf16", &avgpool_bf16);
  m.def("global_mean_bf16", &global_mean_bf16);
  m.def("softmax_bf16", &softmax_bf16);
  m.def("softmax_bf16_out", &softmax_bf16_out);
  m.def("cast_f32_bf16", &cast_f32_bf16);
  m.def("cast_bf16_f32", &cast_bf16_f32);
  m.def("channel_pad_bf16", &channel_pad_bf16);
  m.def("cast_f32_pad_bf16", &cast_f32_pad_bf16);
  m.def("add_relu_bf16", &add_relu_bf16);
  m.def("bias_act_bf16", &bias_act_bf16);
  m.def("scale_shift_bf16", &scale_shift_bf16);
  m.def("layernorm_bf16", &layernorm_bf16);
  m.def("gemm_batched_bf16", &gemm_batched_bf16);
  m.def("gemm_strided_bf16", &gemm_strided_bf16);
  m.def("attn_fused_bf16", &attn_fused_bf16);
  m.def("dwconv_bf16", [](torch::Tensor X, torch::Tensor W2,
                          c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> residual,
                          torch::Tensor Y, int64_t R, int64_t S,
                          int64_t sh, int64_t sw, int64_t pt, int64_t pl,
                          int64_t act) {
    TORCH_CHECK(X.scalar_type() == torch::kBFloat16 && X.is_contiguous());
    TORCH_CHECK(Y.scalar_type() == torch::kBFloat16 && Y.is_contiguous());
    TORCH_CHECK(W2.scalar_type() == torch::kBFloat16 && W2.is_contiguous());
    const float* bptr = nullptr;
    if (bias) {
      TORCH_CHECK(bias->scalar_type() == torch::kFloat32);
      bptr = bias->data_ptr<float>();
    }
    const void* rptr = nullptr;
    if (residual) {
      TORCH_CHECK(residual->scalar_type() == torch::kBFloat16 &&
                  residual->is_contiguous());
      rptr = residual->data_ptr();
    }
    int N = X.size(0), H = X.size(1), Wd = X.size(2), C = X.size(3);
    int HO = Y.size(1), WO = Y.size(2);
    TORCH_CHECK(Y.size(3) == C && Y.size(0) == N);
    launch_dwconv_bf16(X.data_ptr(), W2.data_ptr(), bptr, rptr,
                       Y.data_ptr(), N, H, Wd, C, HO, WO, (int)R, (int)S,
                       (int)sh, (int)sw, (int)pt, (int)pl, (int)act,
                       c10::hip::getCurrentHIPStream());
  });
  m.def("gemm8p_bf16", [](torch::Tensor A, torch::Tensor B,
                          torch::Tensor C,
                          c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> residual,
                          int64_t K, int64_t act) {
    check_bf16(A, "A");
    check_bf16(B, "B");
    int ldc = check_out2d(C, "C");
    int64_t M = C.size(0), N = C.size(1), Kp = B.size(1);
    const float* bptr = nullptr;
    if (bias) bptr = bias->data_ptr<float>();
    const void* rptr = nullptr;
    if (residual) rptr = residual->data_ptr();
    launch_gemm_8p(A.data_ptr(), B.data_ptr(), C.data_ptr(), bptr, rptr,
                   (int)M, (int)N, (int)K, (int)Kp, (int)act, ldc,
                   c10::hip::getCurrentHIPStream());
  });
  m.def("jpeg_decode_slots", [](torch::Tensor arena, torch::Tensor planes,
                                torch::Tensor flags, int64_t n,
                                int64_t slot_bytes) {
    TORCH_CHECK(arena.scalar_type() == torch::kUInt8 &&
                arena.is_contiguous() && arena.is_cuda());
    TORCH_CHECK(planes.scalar_type() == torch::kUInt8 &&
                planes.is_cuda() &&
                planes.numel() >= n * slot_bytes);
    TORCH_CHECK(flags.scalar_type() == torch::kUInt8 && flags.is_cuda() &&
                flags.numel() >= n);
    launch_jpeg_decode(arena.data_ptr(), planes.data_ptr(),
                       flags.data_ptr(), (int)n, slot_bytes,
                       c10::hip::getCurrentHIPStream());
  });
  m.def("resize_norm", &resize_norm);
  m.def("resize_norm_batch", &resize_norm_batch);
  // Pin an externally-allocated host buffer (e.g. a /dev/shm mapping the
  // decode workers write into) so H2D from it is a true async DMA.
  m.def("host_register", [](int64_t ptr, int64_t bytes) {
    hipError_t e = hipHostRegister(reinterpret_cast<void*>(ptr),
                                   (size_t)bytes, hipHostRegisterDefault);
    return e == hipSuccess;
  });
  m.def("host_unregister", [](int64_t ptr) {
    (void)hipHostUnregister(reinterpret_cast<void*>(ptr));
  });
}
