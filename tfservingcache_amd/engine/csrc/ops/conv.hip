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

__global__ void k_im2col(const ushort* __restrict__ x,
                         ushort* __restrict__ y, int N, int H, int W, int C,
                         int R, int S, int sh, int sw, int pt, int pl,
                         int Ho, int Wo, int k_pad) {
  // one thread per output element [m, k]; k (=r*S*C + s*C + c) is the
  // fastest dim -> coalesced writes; reads coalesce over c runs.
  int64_t n_out = (int64_t)N * Ho * Wo * k_pad;
  int64_t i0 = (int64_t)blockIdx.x * TPB + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * TPB;
  int rsc = R * S * C;
  for (int64_t i = i0; i < n_out; i += stride) {
    int k = int(i % k_pad);
    int64_t mm = i / k_pad;
    ushort v = 0;
    if (k < rsc) {
      int c = k % C;
      int t = k / C;
      int ss = t % S, r = t / S;
      int wo = int(mm % Wo);
      int64_t t2 = mm / Wo;
      int ho = int(t2 % Ho);
      int n = int(t2 / Ho);
      int hi = ho * sh + r - pt;
      int wi = wo * sw + ss - pl;
      if (hi >= 0 && hi < H && wi >= 0 && wi < W)
        v = x[(((int64_t)n * H + hi) * W + wi) * C + c];
    }
    y[i] = v;
  }
}

void launch_im2col(hipStream_t s, const ushort* x, ushort* y,
                   int N, int H, int W, int C, int R, int S,
                   int sh, int sw, int pt, int pl, int Ho, int Wo,
                   int k_pad) {
  int64_t n_out = (int64_t)N * Ho * Wo * k_pad;
  int64_t blocks = ceil_div(n_out, (int64_t)TPB);
  int grid = int(blocks < MAX_BLOCKS ? (blocks > 0 ? blocks : 1)
                                     : MAX_BLOCKS);
  hipLaunchKernelGGL(k_im2col, dim3(grid), dim3(TPB), 0, s, x, y,
                     N, H, W, C, R, S, sh, sw, pt, pl, Ho, Wo, k_pad);
}

}  // namespace tfsc
