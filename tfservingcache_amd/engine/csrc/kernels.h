// Host-side kernel launcher declarations. All activation buffers are
// bf16 (ushort) unless noted; integer buffers are int32. Launches are
// asynchronous on the given stream.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

namespace tfsc {

// elementwise fn codes (planner 'eltwise' op)
enum EltFn : int {
  ELT_ADD = 0, ELT_SUB, ELT_MUL, ELT_DIV, ELT_MAX, ELT_MIN, ELT_SQDIFF,
  ELT_RELU, ELT_TANH, ELT_SIGMOID, ELT_ERF, ELT_SQRT, ELT_RSQRT, ELT_EXP,
  ELT_NEG, ELT_SQUARE, ELT_GELU, ELT_RELU6,
};

// activation codes for fused epilogues
enum Act : int { ACT_NONE = 0, ACT_RELU, ACT_TANH, ACT_SIGMOID, ACT_GELU,
                 ACT_RELU6 };

constexpr int MAX_DIMS = 6;

struct BcastArgs {          // broadcast strides (elements), 0 = broadcast
  int ndim;
  int64_t dims[MAX_DIMS];
  int64_t sa[MAX_DIMS];
  int64_t sb[MAX_DIMS];
};

void launch_eltwise_unary(hipStream_t s, const ushort* x, ushort* y,
                          int64_t n, int fn);
void launch_eltwise_binary(hipStream_t s, const ushort* a, const ushort* b,
                           ushort* y, int64_t n_out, const BcastArgs& bc,
                           int fn);
void launch_bn_act(hipStream_t s, const ushort* x, const ushort* scale,
                   const ushort* shift, ushort* y, int64_t rows, int64_t c,
                   int act);
void launch_softmax(hipStream_t s, const ushort* x, ushort* y,
                    int64_t rows, int64_t cols);
void launch_layernorm(hipStream_t s, const ushort* x, const ushort* gamma,
                      const ushort* beta, ushort* y, int64_t rows,
                      int64_t cols, float eps);
void launch_reduce_mean_last(hipStream_t s, const ushort* x, ushort* y,
                             int64_t rows, int64_t cols);
// mean over axis 1 of [d0, d1, d2]
void launch_reduce_mean_mid(hipStream_t s, const ushort* x, ushort* y,
                            int64_t d0, int64_t d1, int64_t d2);
// global mean over H,W of NHWC
void launch_global_mean(hipStream_t s, const ushort* x, ushort* y,
                        int64_t n, int64_t hw, int64_t c);
void launch_pool(hipStream_t s, const ushort* x, ushort* y, bool is_max,
                 int N, int H, int W, int C, int Ho, int Wo, int kh, int kw,
                 int sh, int sw, int pt, int pl);
void launch_transpose(hipStream_t s, const ushort* x, ushort* y, int ndim,
                      const int64_t* out_dims, const int64_t* in_strides,
                      int64_t n_out);
// inverse of launch_transpose: LINEAR source -> STRIDED destination
// (concat writes each input into an offset slice of the output)
void launch_scatter(hipStream_t s, const ushort* x, ushort* y, int ndim,
                    const int64_t* in_dims, const int64_t* out_strides,
                    int64_t n_in);
void launch_gather_rows(hipStream_t s, const ushort* table, const int* idx,
                        ushort* y, int64_t n_idx, int64_t row_elems);
void launch_pad_nhwc(hipStream_t s, const ushort* x, ushort* y,
                     int N, int H, int W, int C, int pt, int pb, int pl,
                     int pr);
void launch_pad_last(hipStream_t s, const ushort* x, ushort* y,
                     int64_t rows, int c_in, int c_out);
void launch_f32_to_bf16(hipStream_t s, const float* x, ushort* y,
                        int64_t n);
// dtype casts between workspace tensors: 0 = i32 -> bf16,
// 1 = bf16 -> i32 (truncate toward zero, TF Cast semantics)
void launch_cast(hipStream_t s, const void* x, void* y, int64_t n,
                 int mode);
// rowwise argmax over the last dim: [rows, cols] bf16 -> i32 (first
// max wins on ties, TF semantics)
void launch_argmax_last(hipStream_t s, const ushort* x, int* y,
                        int64_t rows, int64_t cols);
void launch_bf16_to_f32(hipStream_t s, const ushort* x, float* y,
                        int64_t n);

// GEMM: C[M,N] = act(A[M,K] @ B[K,N] + bias [+ residual]).
// A row-major bf16, B row-major bf16 (pre-transposed at load if the graph
// wanted B^T), C bf16. f32 accumulate via MFMA.
void launch_gemm(hipStream_t s, const ushort* A, const ushort* B,
                 const ushort* bias, const ushort* residual, ushort* C,
                 int64_t M, int64_t N, int64_t K, int act, float alpha);

// Batched GEMM over leading batch dim: [bat, M, K] @ [bat, K, N] (operands
// contiguous; strides in elements; stride 0 broadcasts an operand).
void launch_batched_gemm(hipStream_t s, const ushort* A, const ushort* B,
                         ushort* C, int64_t bat, int64_t M, int64_t N,
                         int64_t K, int64_t strideA, int64_t strideB,
                         int64_t strideC, bool trans_b, float alpha);

// Fused implicit-GEMM NHWC Conv2D (C % 8 == 0):
//   y[N*Ho*Wo, Kc] = act(x (*) w + bias [+ residual])
// with w pre-transposed [Kc][pad64(R*S*C)] like launch_gemm's B. The
// A-tile gather happens in the GEMM's LDS staging (per-lane source
// address from the conv geometry; out-of-image patches read from
// `zeros`, a >=16-byte zeroed device buffer).
void launch_conv_igemm(hipStream_t s, const ushort* x, const ushort* w,
                       const ushort* bias, const ushort* residual,
                       const ushort* zeros, ushort* y,
                       int N, int H, int W, int C, int Kc, int R, int S,
                       int sh, int sw, int pt, int pl, int Ho, int Wo,
                       int k_pad, int act);

// im2col for NHWC conv: out[N*Ho*Wo, K_pad] where the first R*S*C columns
// hold the patch elements (r,s,c order, c fastest) and columns >= R*S*C
// are zero (K_pad is the GEMM's 64-multiple). Out-of-image patch elements
// are zero. The conv then runs as launch_gemm with pre-transposed weights
// [Kc][K_pad].
void launch_im2col(hipStream_t s, const ushort* x, ushort* y,
                   int N, int H, int W, int C, int R, int S,
                   int sh, int sw, int pt, int pl, int Ho, int Wo,
                   int k_pad);

// Depthwise NHWC conv (depth_multiplier == 1, MobileNet-class):
//   y[n,ho,wo,c] = act(sum_{r,s} x[n, ho*sh-pt+r, wo*sw-pl+s, c] * w[r,s,c]
//                      + bias[c])
// Memory-bound (no MFMA shape): vectorized 4x bf16 over the contiguous
// channel dim, grid-stride over N*Ho*Wo*C/4. w is the [R,S,C,1] master
// read as flat [R*S*C].
void launch_depthwise_conv(hipStream_t s, const ushort* x, const ushort* w,
                           const ushort* bias, ushort* y,
                           int N, int H, int W, int C, int R, int S,
                           int sh, int sw, int pt, int pl, int Ho, int Wo,
                           int act);

// fp8 (OCP e4m3) GEMM with rowwise dequant in the f32 epilogue:
// C[M,N] = act(acc(Aq @ Bq^T) * sa[m] * sb[n] + bias [+ residual]).
// Aq/Bq are e4m3 bytes with K padded to a multiple of 128; B^T layout
// [N][Kp] like launch_gemm's pre-transposed weights. 2x the bf16 MFMA
// rate on gfx950 (profiles/fp8_groundwork.md).
void launch_gemm_fp8(hipStream_t s, const uint8_t* A, const float* sa,
                     const uint8_t* B, const float* sb,
                     const ushort* bias, const ushort* residual,
                     ushort* C, int64_t M, int64_t N, int64_t K,
                     int act);

// rowwise activation quantization for the fp8 GEMM: bf16 [M,K] ->
// e4m3 [M,Kp] (zero-padded) + f32 amax/448 scale per row
void launch_quant_rowwise(hipStream_t s, const ushort* x, uint8_t* q,
                          float* scales, int64_t M, int64_t K,
                          int64_t Kp);

// Fused multi-head attention over the natural [B*S, H*D] QKV layout
// (flash-style online softmax; D must be 64). See ops/attention.hip.
void launch_attention(hipStream_t s, const ushort* Q, const ushort* K,
                      const ushort* V, ushort* O, int B, int S, int H,
                      int D_, float scale);

}  // namespace tfsc
