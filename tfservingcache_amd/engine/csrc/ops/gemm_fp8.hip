// fp8 (OCP e4m3) MFMA GEMM for gfx950 — the 2x-rate serving dtype.
//
// C[M,N] = act(dequant(Aq[M,K] @ Bq^T[N,K]) + bias [+ residual])
//   with rowwise dequant:  C_real = acc * sa[m] * sb[n]
//
// Same 128x128 block-tile anatomy as gemm.hip's bf16 kernel (identical
// LDS image byte layout: 128-byte rows, 8x16B XOR-swizzled chunks,
// double-buffered glds staging) — but a K-tile is 128 fp8 elements in
// the same 128 bytes a bf16 tile spent on 64, so HBM/LDS traffic per
// FLOP halves, and each tile step is ONE
// __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4 (measured 2.0x the
// bf16 MFMA rate, profiles/fp8_groundwork.md). Unit e8m0 scales are
// passed to the MFMA; the real (non-power-of-2) rowwise scales are
// applied exactly in the f32 epilogue.
//
// Quantization recipe (the standard fp8 serving scheme, verified on HW
// to ~3e-6 in the groundwork probe):
//   * weights: per-output-channel (per row of B^T[N][K]) amax/448,
//     quantized once at model load (python side, torch fp8 cast);
//   * activations: per-row amax/448 on the fly (launch_quant_rowwise,
//     fused amax+quantize, one workgroup per row).
#include "../common.h"
#include "../kernels.h"

#include <stdexcept>

namespace tfsc {

namespace fp8 {

using i32x4_t = __attribute__((ext_vector_type(4))) int;
using i32x8_t = __attribute__((ext_vector_type(8))) int;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

constexpr int BK = 128;                   // fp8 elements per K tile
constexpr int TILE_BYTES_ROW = 128;       // bytes per LDS tile row

TFSC_DEV int xcd_swizzle8(int bid, int nblocks) {
  constexpr int NXCD = 8;
  if (nblocks < 2 * NXCD) return bid;
  int q = nblocks / NXCD, r = nblocks % NXCD;
  int xcd = bid % NXCD, idx = bid / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

TFSC_DEV int lds_off8(int row, int chunk) {   // byte offset into a tile
  return row * TILE_BYTES_ROW + ((chunk ^ (row & 7)) << 4);
}

// stage a [ROWS][128B] fp8 tile with 16-byte glds, XOR-swizzled source
template <int ROWS, int NWAVES>
TFSC_DEV void stage_tile_glds_u8(const uint8_t* __restrict__ src,
                                 int64_t ld, int row_limit, int row0,
                                 int k0, char* lds_tile, int wave,
                                 int lane) {
  constexpr int ROWS_PER_WAVE = ROWS / NWAVES;
  constexpr int N_GLDS = ROWS_PER_WAVE / 8;
  int r_in = (lane >> 3);
  int chunk = lane & 7;
  #pragma unroll
  for (int i = 0; i < N_GLDS; ++i) {
    int row = wave * ROWS_PER_WAVE + i * 8 + r_in;
    int grow = row0 + row;
    grow = grow < row_limit ? grow : row_limit;
    int chunk_src = chunk ^ (row & 7);
    const uint8_t* gptr = src + (int64_t)grow * ld + k0 + chunk_src * 16;
    char* lds_base = lds_tile + (wave * ROWS_PER_WAVE + i * 8) *
                     TILE_BYTES_ROW;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(gptr),
        reinterpret_cast<uint32_t*>(lds_base), 16, 0, 0);
  }
}

TFSC_DEV float act8(float v, int act) {
  switch (act) {
    case ACT_RELU: return v > 0.f ? v : 0.f;
    case ACT_TANH: return tanhf(v);
    case ACT_SIGMOID: return 1.f / (1.f + __expf(-v));
    case ACT_GELU: return 0.5f * v * (1.f + erff(v * 0.70710678f));
    case ACT_RELU6: return v < 0.f ? 0.f : (v > 6.f ? 6.f : v);
    default: return v;
  }
}

// epilogue with rowwise dequant: same restage-through-LDS pattern as
// gemm.hip's epilogue_store, plus sa[m]*sb[n] applied before bias/act
template <bool HAS_BIAS, bool HAS_RES, int NW>
TFSC_DEV void epilogue_fp8(f32x4_t (&acc)[4][4], char* smem,
                           const float* __restrict__ sa,
                           const float* __restrict__ sb,
                           const ushort* __restrict__ bias,
                           const ushort* __restrict__ residual,
                           ushort* __restrict__ Cb,
                           int M, int N, int m0, int n0, int wave,
                           int lane, int wm, int wn, int act) {
  __syncthreads();
  ushort* stage = reinterpret_cast<ushort*>(smem + wave * 8192);
  const int col_in = lane & 15;
  const int row_base = (lane >> 4) * 4;
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = ni * 16 + col_in;
      int gcol = n0 + wn * 64 + col;
      float bv = HAS_BIAS && gcol < N ? bf2f(bias[gcol]) : 0.f;
      float sbv = gcol < N ? sb[gcol] : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = mi * 16 + row_base + r;
        int grow = m0 + wm * 64 + row;
        float sav = grow < M ? sa[grow] : 0.f;
        stage[row * 64 + col] = f2bf(acc[mi][ni][r] * sav * sbv + bv);
      }
    }
  }
  __syncthreads();
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    int chunk_id = i * 64 + lane;
    int row = chunk_id >> 3;
    int chunk = chunk_id & 7;
    int grow = m0 + wm * 64 + row;
    int gcol0 = n0 + wn * 64 + chunk * 8;
    if (grow >= M) continue;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(stage + row * 64 +
                                                chunk * 8);
    int64_t goff = (int64_t)grow * N + gcol0;
    if (HAS_RES || act != ACT_NONE) {
      bf16x8 rv = {};
      if (HAS_RES && gcol0 + 7 < N)
        rv = *reinterpret_cast<const bf16x8*>(residual + goff);
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = bf2f((ushort)v[j]);
        if (HAS_RES) {
          if (gcol0 + 7 < N) x += bf2f((ushort)rv[j]);
          else if (gcol0 + j < N) x += bf2f(residual[goff + j]);
        }
        v[j] = (short)f2bf(act8(x, act));
      }
    }
    if (gcol0 + 7 < N) {
      *reinterpret_cast<bf16x8*>(Cb + goff) = v;
    } else {
      for (int j = 0; j < 8 && gcol0 + j < N; ++j)
        Cb[goff + j] = (ushort)v[j];
    }
  }
}

template <int WMW, int WNW, bool HAS_BIAS, bool HAS_RES>
__global__ __launch_bounds__(WMW * WNW * WAVE)
void gemm_fp8_kernel(const uint8_t* __restrict__ A,
                     const float* __restrict__ sa,
                     const uint8_t* __restrict__ B,
                     const float* __restrict__ sb,
                     const ushort* __restrict__ bias,
                     const ushort* __restrict__ residual,
                     ushort* __restrict__ C,
                     int M, int N, int K, int act, int n_tiles_m) {
  constexpr int BM_ = WMW * 64, BN_ = WNW * 64;
  constexpr int NW = WMW * WNW;
  constexpr int A_BYTES = BM_ * TILE_BYTES_ROW;
  constexpr int B_BYTES = BN_ * TILE_BYTES_ROW;
  __shared__ __attribute__((aligned(16))) char smem[2 * (A_BYTES + B_BYTES)];
  auto lds_a = [&](int buf) -> char* {
    return smem + buf * (A_BYTES + B_BYTES); };
  auto lds_b = [&](int buf) -> char* {
    return smem + buf * (A_BYTES + B_BYTES) + A_BYTES; };

  const int bid = xcd_swizzle8(blockIdx.x, gridDim.x);
  const int tile_m = bid % n_tiles_m;
  const int tile_n = bid / n_tiles_m;
  const int m0 = tile_m * BM_;
  const int n0 = tile_n * BN_;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wm = wave / WNW, wn = wave % WNW;

  f32x4_t acc[4][4] = {};
  const int n_ktiles = K / BK;

  stage_tile_glds_u8<BM_, NW>(A, K, M - 1, m0, 0, lds_a(0), wave, lane);
  stage_tile_glds_u8<BN_, NW>(B, K, N - 1, n0, 0, lds_b(0), wave, lane);

  int cur = 0;
  for (int kt = 0; kt < n_ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < n_ktiles) {
      int k0 = (kt + 1) * BK;
      stage_tile_glds_u8<BM_, NW>(A, K, M - 1, m0, k0, lds_a(cur ^ 1),
                                  wave, lane);
      stage_tile_glds_u8<BN_, NW>(B, K, N - 1, n0, k0, lds_b(cur ^ 1),
                                  wave, lane);
    }
    const char* at = lds_a(cur);
    const char* bt = lds_b(cur);
    const int frow = lane & 15;
    const int kgrp = lane >> 4;          // k-block of 32 bytes

    // fragment: lane (kgrp,frow) holds bytes [kgrp*32, kgrp*32+32) of
    // its row = two XOR-swizzled 16B chunks (2*kgrp, 2*kgrp+1)
    i32x8_t a_frag[4], b_frag[4];
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      int row = wm * 64 + mi * 16 + frow;
      i32x4_t lo = *reinterpret_cast<const i32x4_t*>(
          at + lds_off8(row, 2 * kgrp));
      i32x4_t hi = *reinterpret_cast<const i32x4_t*>(
          at + lds_off8(row, 2 * kgrp + 1));
      a_frag[mi] = i32x8_t{lo[0], lo[1], lo[2], lo[3],
                           hi[0], hi[1], hi[2], hi[3]};
    }
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int row = wn * 64 + ni * 16 + frow;
      i32x4_t lo = *reinterpret_cast<const i32x4_t*>(
          bt + lds_off8(row, 2 * kgrp));
      i32x4_t hi = *reinterpret_cast<const i32x4_t*>(
          bt + lds_off8(row, 2 * kgrp + 1));
      b_frag[ni] = i32x8_t{lo[0], lo[1], lo[2], lo[3],
                           hi[0], hi[1], hi[2], hi[3]};
    }
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0,
            0, 0x7F7F7F7F, 0, 0x7F7F7F7F);      // unit e8m0 scales
    __builtin_amdgcn_s_setprio(0);
    cur ^= 1;
  }

  epilogue_fp8<HAS_BIAS, HAS_RES, NW>(acc, smem, sa, sb, bias, residual,
                                      C, M, N, m0, n0, wave, lane, wm,
                                      wn, act);
}

// ---------------------------------------------------------------------------
// rowwise activation quantization: bf16 [M,K] -> e4m3 [M,Kp] + f32
// scale per row (amax/448). One workgroup per row chunk, fused
// amax-reduce + quantize (the row is re-read from L2, not HBM).
// ---------------------------------------------------------------------------
TFSC_DEV uint8_t f32_to_e4m3(float f) {
  // OCP e4m3fn: bias 7, no inf, max 448. Pure bit manipulation (the
  // frexpf/roundf version cost 16% of BERT GPU time); round-half-up
  // on the dropped 20 mantissa bits.
  uint32_t bits = __float_as_uint(f);
  uint8_t sign = uint8_t((bits >> 24) & 0x80);
  float a = fabsf(f);
  if (f != f) return 0x7F;
  if (a >= 448.f) return sign | 0x7E;
  int e = int((bits >> 23) & 0xFF) - 127;
  if (e < -6) {
    // subnormal: steps of 2^-9; a < 2^-6 so the product is exact range
    int mant = int(a * 512.f + 0.5f);
    if (mant > 7) return sign | 0x08;
    return sign | uint8_t(mant);
  }
  uint32_t mant23 = bits & 0x7FFFFF;
  uint32_t mant3 = (mant23 + 0x80000u) >> 20;     // round half up
  if (mant3 == 8) {
    mant3 = 0;
    ++e;
    if (e > 8) return sign | 0x7E;
  }
  return sign | uint8_t((e + 7) << 3) | uint8_t(mant3);
}

// one WAVE per row (no block barrier): amax via in-wave shfl, then
// each lane quantizes its strided elements. Values are re-read from L2
// for the quantize pass (rows are KB-scale).
__global__ __launch_bounds__(256)
void quant_rowwise_kernel(const ushort* __restrict__ x,
                          uint8_t* __restrict__ q,
                          float* __restrict__ scales,
                          int M, int K, int Kp) {
  int wv = threadIdx.x / WAVE;
  int lane = threadIdx.x % WAVE;
  int waves_total = gridDim.x * (256 / WAVE);
  for (int row = blockIdx.x * (256 / WAVE) + wv; row < M;
       row += waves_total) {
    const ushort* xr = x + (int64_t)row * K;
    float amax = 0.f;
    int k4 = K / 4;
    const short4_t* xr4 = reinterpret_cast<const short4_t*>(xr);
    for (int i = lane; i < k4; i += WAVE) {
      short4_t v = xr4[i];
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        amax = fmaxf(amax, fabsf(bf2f((ushort)v[j])));
    }
    for (int i = k4 * 4 + lane; i < K; i += WAVE)
      amax = fmaxf(amax, fabsf(bf2f(xr[i])));
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));
    float scale = amax > 0.f ? amax / 448.f : 1.f;
    float inv = 1.f / scale;
    if (lane == 0) scales[row] = scale;
    uint8_t* qr = q + (int64_t)row * Kp;
    // vector path: 4 bf16 in -> 4 e4m3 bytes out as one u32 store
    for (int i = lane; i < k4; i += WAVE) {
      short4_t v = xr4[i];
      uint32_t outw = 0;
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        outw |= uint32_t(f32_to_e4m3(bf2f((ushort)v[j]) * inv)) << (8 * j);
      reinterpret_cast<uint32_t*>(qr)[i] = outw;
    }
    for (int i = k4 * 4 + lane; i < Kp; i += WAVE)
      qr[i] = i < K ? f32_to_e4m3(bf2f(xr[i]) * inv) : 0;
  }
}

}  // namespace fp8

void launch_quant_rowwise(hipStream_t s, const ushort* x, uint8_t* q,
                          float* scales, int64_t M, int64_t K,
                          int64_t Kp) {
  int64_t waves_needed = M;
  int blocks = int(ceil_div(waves_needed, (int64_t)(256 / WAVE)));
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(fp8::quant_rowwise_kernel, dim3(blocks), dim3(256),
                     0, s, x, q, scales, int(M), int(K), int(Kp));
}

void launch_gemm_fp8(hipStream_t s, const uint8_t* A, const float* sa,
                     const uint8_t* B, const float* sb,
                     const ushort* bias, const ushort* residual,
                     ushort* C, int64_t M, int64_t N, int64_t K,
                     int act) {
  using namespace fp8;
  if (K % BK != 0)
    throw std::runtime_error("gemm_fp8: K must be a multiple of 128");
  auto blocks = [&](int bm, int bn) {
    return ceil_div(M, (int64_t)bm) * ceil_div(N, (int64_t)bn);
  };
  bool hb = bias != nullptr, hr = residual != nullptr;
  auto go = [&](auto kern, int bm, int bn) {
    int ntm = int(ceil_div(M, (int64_t)bm));
    int ntn = int(ceil_div(N, (int64_t)bn));
    dim3 grid(ntm * ntn);
    dim3 block((bm / 64) * (bn / 64) * WAVE);
    hipLaunchKernelGGL(kern, grid, block, 0, s, A, sa, B, sb, bias,
                       residual, C, int(M), int(N), int(K), act, ntm);
  };
  if (blocks(128, 128) >= 232) {
    if (hb && hr)  go(gemm_fp8_kernel<2, 2, true, true>, 128, 128);
    else if (hb)   go(gemm_fp8_kernel<2, 2, true, false>, 128, 128);
    else if (hr)   go(gemm_fp8_kernel<2, 2, false, true>, 128, 128);
    else           go(gemm_fp8_kernel<2, 2, false, false>, 128, 128);
  } else if (blocks(64, 128) >= 232) {
    if (hb && hr)  go(gemm_fp8_kernel<1, 2, true, true>, 64, 128);
    else if (hb)   go(gemm_fp8_kernel<1, 2, true, false>, 64, 128);
    else if (hr)   go(gemm_fp8_kernel<1, 2, false, true>, 64, 128);
    else           go(gemm_fp8_kernel<1, 2, false, false>, 64, 128);
  } else {
    if (hb && hr)  go(gemm_fp8_kernel<1, 1, true, true>, 64, 64);
    else if (hb)   go(gemm_fp8_kernel<1, 1, true, false>, 64, 64);
    else if (hr)   go(gemm_fp8_kernel<1, 1, false, true>, 64, 64);
    else           go(gemm_fp8_kernel<1, 1, false, false>, 64, 64);
  }
}

}  // namespace tfsc
