// Conv2D support kernels (NHWC).
//
// Convolution executes as im2col -> bf16 MFMA GEMM (gemm.hip) with the
// bias/BN/ReLU/residual epilogue fused into the GEMM. 1x1 stride-1 convs
// skip im2col entirely (pure GEMM over [N*H*W, C]) — that covers ~2/3 of
// ResNet-50's convolutions; the fused single-kernel implicit-GEMM path is
// the planned next optimization of this file.
#include "../common.h"
#include "../kernels.h"

namespace tfsc {

constexpr int TPB = 256;
constexpr int MAX_BLOCKS = 2048;

// vectorized path (C % 8 == 0): one thread per (m, r, s, c8-group) moves
// 8 contiguous bf16 (one short4_t pair = 16B) — 1/8 the index math and
// 16B coalesced loads/stores vs the scalar element kernel (im2col was
// 43-50% of ResNet kernel time before this).
__global__ void k_im2col_v8(const ushort* __restrict__ x,
                            ushort* __restrict__ y, int N, int H, int W,
                            int C, int R, int S, int sh, int sw, int pt,
                            int pl, int Ho, int Wo, int k_pad) {
  int rsc = R * S * C;
  int c8 = C >> 3;
  int64_t groups_per_m = (int64_t)R * S * c8;
  int64_t n_groups = (int64_t)N * Ho * Wo * groups_per_m;
  int64_t i0 = (int64_t)blockIdx.x * TPB + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * TPB;
  for (int64_t i = i0; i < n_groups; i += stride) {
    int g = int(i % groups_per_m);
    int64_t mm = i / groups_per_m;
    int c0 = (g % c8) << 3;
    int t = g / c8;
    int ss = t % S, r = t / S;
    int wo = int(mm % Wo);
    int64_t t2 = mm / Wo;
    int ho = int(t2 % Ho);
    int n = int(t2 / Ho);
    int hi = ho * sh + r - pt;
    int wi = wo * sw + ss - pl;
    int k = (r * S + ss) * C + c0;
    ushort* dst = y + mm * k_pad + k;
    if (hi >= 0 && hi < H && wi >= 0 && wi < W) {
      const ushort* src = x + (((int64_t)n * H + hi) * W + wi) * C + c0;
      *reinterpret_cast<uint4*>(dst) =
          *reinterpret_cast<const uint4*>(src);
    } else {
      *reinterpret_cast<uint4*>(dst) = uint4{0, 0, 0, 0};
    }
  }
}

// zero-fill of the k >= rsc padding columns (done once by a cheap kernel
// so the main copy kernels skip the tail test)
__global__ void k_im2col_padzero(ushort* __restrict__ y, int64_t rows,
                                 int rsc, int k_pad) {
  int tail = k_pad - rsc;
  int64_t n = rows * tail;
  int64_t i0 = (int64_t)blockIdx.x * TPB + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * TPB;
  for (int64_t i = i0; i < n; i += stride)
    y[(i / tail) * k_pad + rsc + i % tail] = 0;
}

// small-C path (e.g. the ResNet stem: C=3): one thread per (m, r) moves
// the whole S*C row-run (contiguous in x when fully inside the image).
__global__ void k_im2col_row(const ushort* __restrict__ x,
                             ushort* __restrict__ y, int N, int H, int W,
                             int C, int R, int S, int sh, int sw, int pt,
                             int pl, int Ho, int Wo, int k_pad) {
  int sc = S * C;
  int64_t n_rows = (int64_t)N * Ho * Wo * R;
  int64_t i0 = (int64_t)blockIdx.x * TPB + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * TPB;
  for (int64_t i = i0; i < n_rows; i += stride) {
    int r = int(i % R);
    int64_t mm = i / R;
    int wo = int(mm % Wo);
    int64_t t2 = mm / Wo;
    int ho = int(t2 % Ho);
    int n = int(t2 / Ho);
    int hi = ho * sh + r - pt;
    int wi0 = wo * sw - pl;
    ushort* dst = y + mm * k_pad + r * sc;
    if (hi < 0 || hi >= H) {
      for (int j = 0; j < sc; ++j) dst[j] = 0;
      continue;
    }
    const ushort* src = x + (((int64_t)n * H + hi) * W + wi0) * C;
    for (int ss = 0; ss < S; ++ss) {
      int wi = wi0 + ss;
      bool in = (wi >= 0 && wi < W);
      for (int c = 0; c < C; ++c)
        dst[ss * C + c] = in ? src[ss * C + c] : ushort(0);
    }
  }
}

void launch_im2col(hipStream_t s, const ushort* x, ushort* y,
                   int N, int H, int W, int C, int R, int S,
                   int sh, int sw, int pt, int pl, int Ho, int Wo,
                   int k_pad) {
  int64_t rows = (int64_t)N * Ho * Wo;
  int rsc = R * S * C;
  if (k_pad > rsc) {
    int64_t n = rows * (k_pad - rsc);
    int grid = int(ceil_div(n, (int64_t)TPB));
    if (grid > MAX_BLOCKS) grid = MAX_BLOCKS;
    hipLaunchKernelGGL(k_im2col_padzero, dim3(grid), dim3(TPB), 0, s,
                       y, rows, rsc, k_pad);
  }
  if (C % 8 == 0) {
    int64_t n = rows * R * S * (C / 8);
    int grid = int(ceil_div(n, (int64_t)TPB));
    if (grid > MAX_BLOCKS) grid = MAX_BLOCKS;
    hipLaunchKernelGGL(k_im2col_v8, dim3(grid), dim3(TPB), 0, s, x, y,
                       N, H, W, C, R, S, sh, sw, pt, pl, Ho, Wo, k_pad);
  } else {
    int64_t n = rows * R;
    int grid = int(ceil_div(n, (int64_t)TPB));
    if (grid > MAX_BLOCKS) grid = MAX_BLOCKS;
    hipLaunchKernelGGL(k_im2col_row, dim3(grid), dim3(TPB), 0, s, x, y,
                       N, H, W, C, R, S, sh, sw, pt, pl, Ho, Wo, k_pad);
  }
}

}  // namespace tfsc
