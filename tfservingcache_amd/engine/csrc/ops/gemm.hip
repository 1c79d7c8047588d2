// bf16 MFMA GEMM for gfx950 — the serving hot path.
//
// C[M,N] = act(alpha * A[M,K] @ B^T[N,K] + bias [+ residual])
//
// Structure follows the CDNA4 canonical GEMM anatomy
// (/opt/skills/guides/cdna_hip_programming.md §5): 128x128 block tile,
// BK=64, 4 waves (one 64x64 wave tile each), double-buffered LDS filled
// with 16-byte `global_load_lds` (lane-linear dest, XOR-swizzled SOURCE
// address + matching XOR on the ds_read — §5.4 rule 21 / T2), MFMA
// v_mfma_f32_16x16x32_bf16 accumulating f32, fused
// bias/activation/residual epilogue.
//
// B is expected TRANSPOSED ([N][K] row-major): the planner pre-transposes
// weights at model-load time so both A and B fragments read contiguously
// along K. A variant with B in [K][N] (register-staged transpose on the
// fly) serves the batched attention P@V case.
//
// Row indices are clamped at the staging loads (no out-of-bounds reads on
// partial tiles); stores are masked by (row < M && col < N). K must be a
// multiple of 64 — the planner zero-pads weights and im2col buffers.
#include "../common.h"
#include "../kernels.h"

#include <stdexcept>

namespace tfsc {

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int N_WAVES = 4;                // 2x2 wave grid of 64x64 tiles
constexpr int THREADS = N_WAVES * WAVE;   // 256

// XCD-aware blockIdx remap (guide T1, bijective variant): the dispatcher
// places block b on XCD b % 8, so consecutive tile ids (which share A/B
// panels) land on different per-XCD L2s; remapping gives each XCD a
// contiguous chunk of the grid. Applies when the kernel is HBM-bound
// (+10% on large GEMMs); bijective for any grid size.
TFSC_DEV int xcd_swizzle(int bid, int nblocks) {
  constexpr int NXCD = 8;
  if (nblocks < 2 * NXCD) return bid;
  int q = nblocks / NXCD, r = nblocks % NXCD;
  int xcd = bid % NXCD, idx = bid / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// LDS tile: [128 rows][64 cols] bf16, 128-byte rows = 8 chunks of 16B.
// XOR swizzle: chunk' = chunk ^ (row & 7)  (T2: <=2-way conflicts).
TFSC_DEV int lds_off(int row, int chunk) {   // byte offset into a tile
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

template <int ROWS, int NWAVES>
TFSC_DEV void stage_tile_glds(const ushort* __restrict__ src, int64_t ld,
                              int row_limit, int row0, int k0,
                              char* lds_tile, int wave, int lane) {
  // the block's NWAVES waves stage ROWS rows together (each glds covers
  // 8 rows x 64 cols = 1 KiB); lane l covers (row = 8*i + l/8,
  // chunk = l%8) of its wave's slice. The glds LDS destination is
  // wave-uniform-base + lane*16 (lane-linear), which matches the
  // [row][chunk] image exactly.
  constexpr int ROWS_PER_WAVE = ROWS / NWAVES;
  constexpr int N_GLDS = ROWS_PER_WAVE / 8;
  int r_in = (lane >> 3);           // 0..7
  int chunk = lane & 7;             // 0..7
  #pragma unroll
  for (int i = 0; i < N_GLDS; ++i) {
    int row = wave * ROWS_PER_WAVE + i * 8 + r_in;
    int grow = row0 + row;
    grow = grow < row_limit ? grow : row_limit;
    // source chunk is pre-swizzled so the LINEAR lds image holds the
    // swizzled layout (rule 21: swizzle source + read, never the dest)
    int chunk_src = chunk ^ (row & 7);
    const ushort* gptr = src + (int64_t)grow * ld + k0 + chunk_src * 8;
    // wave-uniform base for this glds: start of the wave's 8-row slice
    char* lds_base = lds_tile + (wave * ROWS_PER_WAVE + i * 8) * 128;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(gptr),
        reinterpret_cast<uint32_t*>(lds_base), 16, 0, 0);
  }
}

// register-staged transpose path for B given as [K][N] (attention P@V):
// each lane loads 8 bf16 along N (coalesced) and scatters them into the
// [N][K]-image LDS tile with the same XOR swizzle.
template <int ROWS, int NWAVES>
TFSC_DEV void stage_tile_transposed(const ushort* __restrict__ src,
                                    int64_t ld, int k_limit, int n0, int k0,
                                    ushort* lds_tile, int wave, int lane,
                                    int n_limit) {
  // tile wanted: rows = n (ROWS), cols = k (64). Source element (k, n).
  constexpr int NG = ROWS / 8;            // n-groups of 8
  constexpr int THREADS_ = NWAVES * WAVE;
  constexpr int PASSES = (ROWS * 64) / (THREADS_ * 8);
  int tid = wave * WAVE + lane;
  #pragma unroll
  for (int pass = 0; pass < PASSES; ++pass) {
    int idx = pass * THREADS_ + tid;      // covers k-major: 64 k x NG groups
    int k = idx / NG;                     // 0..63
    int ng = idx % NG;                    // n-group of 8
    int gk = k0 + k;
    gk = gk < k_limit ? gk : k_limit;
    const ushort* gptr = src + (int64_t)gk * ld + n0 + ng * 8;
    ushort vals[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      int gn = n0 + ng * 8 + j;
      vals[j] = (gn < n_limit) ? gptr[j] : ushort(0);
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      int n = ng * 8 + j;
      int byte_off = lds_off(n, k >> 3) + (k & 7) * 2;
      *reinterpret_cast<ushort*>(
          reinterpret_cast<char*>(lds_tile) + byte_off) = vals[j];
    }
  }
}

TFSC_DEV float act_apply(float v, int act) {
  switch (act) {
    case ACT_RELU: return v > 0.f ? v : 0.f;
    case ACT_TANH: return tanhf(v);
    case ACT_SIGMOID: return 1.f / (1.f + __expf(-v));
    case ACT_GELU: return 0.5f * v * (1.f + erff(v * 0.70710678f));
    case ACT_RELU6: return v < 0.f ? 0.f : (v > 6.f ? 6.f : v);
    default: return v;
  }
}

// Coalesced epilogue: the MFMA C/D fragment layout scatters a lane's
// values over 2-byte strides, so direct stores are scalar ushort
// (32-byte wave segments — issue- and coalescing-poor, cf. guide T21).
// Instead each wave restages its 64x64 tile (acc*alpha + bias, bf16)
// through its own 8 KB LDS region, then streams it out row-major with
// 16-byte lanes — and reads the optional residual coalesced in the same
// pass. One __syncthreads() guards the LDS reuse of the staging tiles.
template <bool HAS_BIAS, bool HAS_RES>
TFSC_DEV void epilogue_store(f32x4_t (&acc)[4][4], char* smem,
                             const ushort* __restrict__ bias,
                             const ushort* __restrict__ residual,
                             ushort* __restrict__ Cb,
                             int M, int N, int m0, int n0, int wave,
                             int lane, int wm, int wn, int act,
                             float alpha) {
  __syncthreads();              // tiles are dead; reuse as staging
  ushort* stage = reinterpret_cast<ushort*>(smem + wave * 8192);
  const int col_in = lane & 15;
  const int row_base = (lane >> 4) * 4;
  #pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = ni * 16 + col_in;
      float bv = HAS_BIAS ? bf2f(bias[n0 + wn * 64 + col]) : 0.f;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = mi * 16 + row_base + r;
        // pad column stride by 0 (64 cols x 2B = 128 B rows; lanes of a
        // 16-lane group write 2B at 2B stride -> conflict-free halves)
        stage[row * 64 + col] = f2bf(acc[mi][ni][r] * alpha + bv);
      }
    }
  }
  __syncthreads();
  // stream out: iteration i, lane l -> linear 16B chunk i*64+l of the
  // wave's 64x64 tile (8 chunks per row) — consecutive lanes write
  // consecutive chunks (coalesced 1 KB per wave-instruction)
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
        v[j] = (short)f2bf(act_apply(x, act));
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

// Geometry: WMW x WNW waves, each owning a 64x64 output tile
// (BM = WMW*64, BN = WNW*64). (2,2) = the full 128x128 tile; (1,2) and
// (1,1) keep medium shapes (e.g. BERT dense layers, attention batched
// GEMMs) from under-filling the 256-CU chip.
template <int WMW, int WNW, bool TRANS_B, bool HAS_BIAS, bool HAS_RES>
__global__ __launch_bounds__(WMW * WNW * WAVE)
void gemm_bf16_kernel(const ushort* __restrict__ A,
                      const ushort* __restrict__ B,
                      const ushort* __restrict__ bias,
                      const ushort* __restrict__ residual,
                      ushort* __restrict__ C,
                      int M, int N, int K, int act, float alpha,
                      int64_t strideA, int64_t strideB, int64_t strideC,
                      int n_tiles_m) {
  constexpr int BM_ = WMW * 64, BN_ = WNW * 64;
  constexpr int NW = WMW * WNW;
  constexpr int A_BYTES = BM_ * BK * 2, B_BYTES = BN_ * BK * 2;
  __shared__ __attribute__((aligned(16))) char smem[2 * (A_BYTES + B_BYTES)];
  auto lds_a = [&](int buf) -> char* {
    return smem + buf * (A_BYTES + B_BYTES); };
  auto lds_b = [&](int buf) -> char* {
    return smem + buf * (A_BYTES + B_BYTES) + A_BYTES; };

  const int bat = blockIdx.y;
  const ushort* Ab = A + bat * strideA;
  const ushort* Bb = B + bat * strideB;
  ushort* Cb = C + bat * strideC;

  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tile_m = bid % n_tiles_m;
  const int tile_n = bid / n_tiles_m;
  const int m0 = tile_m * BM_;
  const int n0 = tile_n * BN_;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wm = wave / WNW, wn = wave % WNW;

  f32x4_t acc[4][4] = {};

  const int n_ktiles = K / BK;

  // prologue: stage tile 0 into buf 0
  {
    stage_tile_glds<BM_, NW>(Ab, K, M - 1, m0, 0, lds_a(0), wave, lane);
    if (TRANS_B) {
      stage_tile_glds<BN_, NW>(Bb, K, N - 1, n0, 0, lds_b(0), wave, lane);
    } else {
      stage_tile_transposed<BN_, NW>(
          Bb, N, K - 1, n0, 0, reinterpret_cast<ushort*>(lds_b(0)), wave,
          lane, N);
    }
  }

  // 2-phase schedule (T3 minimum form, cdna_hip_programming.md §5.5):
  // one barrier per K-tile; the next tile's staging loads are issued
  // BEFORE this tile's ds_read+MFMA so the glds overlaps the compute.
  // __syncthreads() emits the vmcnt(0) drain for the in-flight glds.
  int cur = 0;
  for (int kt = 0; kt < n_ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < n_ktiles) {
      int k0 = (kt + 1) * BK;
      stage_tile_glds<BM_, NW>(Ab, K, M - 1, m0, k0, lds_a(cur ^ 1),
                               wave, lane);
      if (TRANS_B) {
        stage_tile_glds<BN_, NW>(Bb, K, N - 1, n0, k0, lds_b(cur ^ 1),
                                 wave, lane);
      } else {
        stage_tile_transposed<BN_, NW>(
            Bb, N, K - 1, n0, k0,
            reinterpret_cast<ushort*>(lds_b(cur ^ 1)), wave, lane, N);
      }
    }

    const char* at = lds_a(cur);
    const char* bt = lds_b(cur);
    const int frow = lane & 15;           // fragment row/col within 16
    const int kgrp = lane >> 4;           // 0..3

    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {      // two K=32 steps per 64-tile
      bf16x8_t a_frag[4], b_frag[4];
      const int chunk = ks * 4 + kgrp;
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int row = wm * 64 + mi * 16 + frow;
        a_frag[mi] = *reinterpret_cast<const bf16x8_t*>(
            at + lds_off(row, chunk));
      }
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int row = wn * 64 + ni * 16 + frow;
        b_frag[ni] = *reinterpret_cast<const bf16x8_t*>(
            bt + lds_off(row, chunk));
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    cur ^= 1;
  }

  epilogue_store<HAS_BIAS, HAS_RES>(acc, smem, bias, residual, Cb, M, N,
                                    m0, n0, wave, lane, wm, wn, act,
                                    alpha);
}

template <int WMW, int WNW, bool TRANS_B>
static void gemm_launch_geom(hipStream_t s, const ushort* A,
                             const ushort* B, const ushort* bias,
                             const ushort* residual, ushort* C,
                             int64_t bat, int64_t M, int64_t N, int64_t K,
                             int act, float alpha, int64_t sA, int64_t sB,
                             int64_t sC) {
  int ntm = int(ceil_div(M, WMW * 64)), ntn = int(ceil_div(N, WNW * 64));
  dim3 grid(ntm * ntn, (unsigned)bat);
  dim3 block(WMW * WNW * WAVE);
  bool hb = bias != nullptr, hr = residual != nullptr;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, block, 0, s, A, B, bias, residual, C,
                       (int)M, (int)N, (int)K, act, alpha, sA, sB, sC, ntm);
  };
  if (hb && hr)  launch(gemm_bf16_kernel<WMW, WNW, TRANS_B, true, true>);
  else if (hb)   launch(gemm_bf16_kernel<WMW, WNW, TRANS_B, true, false>);
  else if (hr)   launch(gemm_bf16_kernel<WMW, WNW, TRANS_B, false, true>);
  else           launch(gemm_bf16_kernel<WMW, WNW, TRANS_B, false, false>);
}

template <bool TRANS_B>
static void gemm_dispatch(hipStream_t s, const ushort* A, const ushort* B,
                          const ushort* bias, const ushort* residual,
                          ushort* C, int64_t bat, int64_t M, int64_t N,
                          int64_t K, int act, float alpha, int64_t sA,
                          int64_t sB, int64_t sC) {
  if (K % BK != 0)
    throw std::runtime_error("gemm: K must be a multiple of 64 (got " +
                             std::to_string(K) + ")");
  // geometry by fill: prefer 128x128 tiles; shrink while the grid
  // under-fills the chip (2 blocks/CU at 64KB LDS -> target >=512,
  // accept >=232 = ~1 block/CU)
  auto blocks = [&](int bm, int bn) {
    return bat * ceil_div(M, bm) * ceil_div(N, bn);
  };
  if (blocks(128, 128) >= 232) {
    gemm_launch_geom<2, 2, TRANS_B>(s, A, B, bias, residual, C, bat, M, N,
                                    K, act, alpha, sA, sB, sC);
  } else if (blocks(64, 128) >= 232) {
    gemm_launch_geom<1, 2, TRANS_B>(s, A, B, bias, residual, C, bat, M, N,
                                    K, act, alpha, sA, sB, sC);
  } else {
    gemm_launch_geom<1, 1, TRANS_B>(s, A, B, bias, residual, C, bat, M, N,
                                    K, act, alpha, sA, sB, sC);
  }
}

void launch_gemm(hipStream_t s, const ushort* A, const ushort* B,
                 const ushort* bias, const ushort* residual, ushort* C,
                 int64_t M, int64_t N, int64_t K, int act, float alpha) {
  gemm_dispatch<true>(s, A, B, bias, residual, C, 1, M, N, K, act, alpha,
                      0, 0, 0);
}

// ---------------------------------------------------------------------------
// Fused implicit-GEMM conv: same 128x128xBK=64 MFMA structure, but the
// A tile is GATHERED from the NHWC input during glds staging — each
// lane's 16-byte chunk maps to 8 contiguous channels of one patch
// element (requires C % 8 == 0 so chunks never straddle an (r,s)
// boundary); out-of-image/-range chunks read from a zeroed buffer so
// the glds issues unconditionally. Removes the im2col memory pass that
// was 43-50% of ResNet-50 kernel time.
// ---------------------------------------------------------------------------
struct ConvGeom {
  int H, W, C, R, S, sh, sw, pt, pl, Ho, Wo, rsc;
};

TFSC_DEV void stage_tile_conv_a(const ushort* __restrict__ x,
                                const ushort* __restrict__ zeros,
                                const ConvGeom g, int M, int m0, int k0,
                                char* lds_tile, int wave, int lane) {
  int r_in = (lane >> 3);
  int chunk = lane & 7;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = wave * 32 + i * 8 + r_in;
    int m = m0 + row;
    m = m < M - 1 ? m : M - 1;
    int chunk_src = chunk ^ (row & 7);
    int k = k0 + chunk_src * 8;           // start of the 8-channel chunk
    // decompose m -> (n, ho, wo) and k -> (r, s, c)
    int wo = m % g.Wo;
    int t = m / g.Wo;
    int ho = t % g.Ho;
    int n = t / g.Ho;
    int c = k % g.C;
    int t2 = k / g.C;
    int s = t2 % g.S;
    int r = t2 / g.S;
    int hi = ho * g.sh + r - g.pt;
    int wi = wo * g.sw + s - g.pl;
    bool ok = (k < g.rsc) & (hi >= 0) & (hi < g.H) & (wi >= 0) & (wi < g.W);
    const ushort* gptr = ok
        ? x + (((int64_t)n * g.H + hi) * g.W + wi) * g.C + c
        : zeros;
    char* lds_base = lds_tile + (wave * 32 + i * 8) * 128;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(gptr),
        reinterpret_cast<uint32_t*>(lds_base), 16, 0, 0);
  }
}

template <bool HAS_RES>
__global__ __launch_bounds__(THREADS)
void conv_igemm_kernel(const ushort* __restrict__ x,
                       const ushort* __restrict__ B,
                       const ushort* __restrict__ bias,
                       const ushort* __restrict__ residual,
                       const ushort* __restrict__ zeros,
                       ushort* __restrict__ Cout,
                       int M, int N, int K, int act, ConvGeom g,
                       int n_tiles_m) {
  __shared__ __attribute__((aligned(16))) char smem[4 * BM * BK * 2];
  auto lds_a = [&](int buf) -> char* { return smem + buf * 32768; };
  auto lds_b = [&](int buf) -> char* { return smem + 16384 + buf * 32768; };

  const int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const int tile_m = bid % n_tiles_m;
  const int tile_n = bid / n_tiles_m;
  const int m0 = tile_m * BM;
  const int n0 = tile_n * BN;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wm = wave >> 1, wn = wave & 1;

  f32x4_t acc[4][4] = {};
  const int n_ktiles = K / BK;

  stage_tile_conv_a(x, zeros, g, M, m0, 0, lds_a(0), wave, lane);
  stage_tile_glds<128, 4>(B, K, N - 1, n0, 0, lds_b(0), wave, lane);

  int cur = 0;
  for (int kt = 0; kt < n_ktiles; ++kt) {
    __syncthreads();
    if (kt + 1 < n_ktiles) {
      int k0 = (kt + 1) * BK;
      stage_tile_conv_a(x, zeros, g, M, m0, k0, lds_a(cur ^ 1), wave, lane);
      stage_tile_glds<128, 4>(B, K, N - 1, n0, k0, lds_b(cur ^ 1), wave,
                              lane);
    }
    const char* at = lds_a(cur);
    const char* bt = lds_b(cur);
    const int frow = lane & 15;
    const int kgrp = lane >> 4;
    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t a_frag[4], b_frag[4];
      const int chunk = ks * 4 + kgrp;
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8_t*>(
            at + lds_off(wm * 64 + mi * 16 + frow, chunk));
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b_frag[ni] = *reinterpret_cast<const bf16x8_t*>(
            bt + lds_off(wn * 64 + ni * 16 + frow, chunk));
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    cur ^= 1;
  }

  epilogue_store<true, HAS_RES>(acc, smem, bias, residual, Cout, M, N,
                                m0, n0, wave, lane, wm, wn, act, 1.0f);
}

void launch_conv_igemm(hipStream_t s, const ushort* x, const ushort* w,
                       const ushort* bias, const ushort* residual,
                       const ushort* zeros, ushort* y,
                       int N_img, int H, int W, int C, int Kc, int R,
                       int S, int sh, int sw, int pt, int pl, int Ho,
                       int Wo, int k_pad, int act) {
  if (C % 8 != 0)
    throw std::runtime_error("conv_igemm requires C % 8 == 0");
  if (k_pad % BK != 0)
    throw std::runtime_error("conv_igemm k_pad must be 64-aligned");
  ConvGeom g{H, W, C, R, S, sh, sw, pt, pl, Ho, Wo, R * S * C};
  int M = N_img * Ho * Wo;
  int ntm = int(ceil_div(M, BM)), ntn = int(ceil_div(Kc, BN));
  dim3 grid(ntm * ntn);
  if (residual)
    hipLaunchKernelGGL((conv_igemm_kernel<true>), grid, dim3(THREADS), 0,
                       s, x, w, bias, residual, zeros, y, M, Kc, k_pad,
                       act, g, ntm);
  else
    hipLaunchKernelGGL((conv_igemm_kernel<false>), grid, dim3(THREADS), 0,
                       s, x, w, bias, nullptr, zeros, y, M, Kc, k_pad,
                       act, g, ntm);
}

void launch_batched_gemm(hipStream_t s, const ushort* A, const ushort* B,
                         ushort* C, int64_t bat, int64_t M, int64_t N,
                         int64_t K, int64_t strideA, int64_t strideB,
                         int64_t strideC, bool trans_b, float alpha) {
  if (trans_b)
    gemm_dispatch<true>(s, A, B, nullptr, nullptr, C, bat, M, N, K,
                        ACT_NONE, alpha, strideA, strideB, strideC);
  else
    gemm_dispatch<false>(s, A, B, nullptr, nullptr, C, bat, M, N, K,
                         ACT_NONE, alpha, strideA, strideB, strideC);
}

}  // namespace tfsc
