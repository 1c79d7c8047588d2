// Fused multi-head attention (flash-style, online softmax) for gfx950.
//
//   out[b,s,h,:] = softmax(scale * Q[b,s,h,:] . K[b,:,h,:]^T) @ V[b,:,h,:]
//
// Consumes Q/K/V in their NATURAL layout [B*S, H*D] (the output of the
// QKV projection GEMMs) and writes out in the same layout — replacing
// the unfused plan (4 transposes + 2 batched GEMMs + standalone softmax
// per layer, ~31% of BERT kernel time) with ONE kernel and no S x S
// score materialization.
//
// Geometry: one workgroup (4 waves) per (b, h, 64-row q-block). Q block
// kept in registers (2 A-fragments per wave). Round-2 revision (the
// round-1 kernel measured 11% MFMA-busy / 44% wave-parked — barrier-
// and staging-latency bound): kv tiles are processed TWO per barrier
// from a 4-deep LDS ring (K and V^T 4 x 8 KB each), so each
// __syncthreads covers 128 kv columns of MFMA work and the online
// softmax runs once per 128 columns instead of per 64; the P bounce
// region overlaps the (dead after the prologue) Q tile. 80 KB LDS ->
// 2 workgroups/CU. Requires head_dim D == 64 (BERT-base class); the
// planner keeps the unfused path otherwise.
#include "../common.h"
#include "../kernels.h"

#include <stdexcept>

namespace tfsc {

namespace attn {

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

constexpr int D = 64;          // head dim
constexpr int QBLK = 64;       // q rows per workgroup (16 per wave)
constexpr int KVBLK = 64;      // kv rows per staged tile
constexpr int NW = 4;          // waves
constexpr int THREADS = NW * WAVE;

// shared 64x64 bf16 tile image with the gemm.hip chunk-XOR swizzle
TFSC_DEV int t_off(int row, int chunk) {
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

// [16][128] bf16 image (P bounce): row stride 256 B, 16 chunks
TFSC_DEV int p_off(int row, int chunk) {
  return row * 256 + ((chunk ^ (row & 7)) << 4);
}

// stage a [64][64] tile from rows of a [rows_total, H*D] matrix
// (row = s index, stride ld elements, cols = one head's D slice)
TFSC_DEV void stage_rows_glds(const ushort* __restrict__ src, int64_t ld,
                              int row0, int row_limit, char* lds_tile,
                              int wave, int lane) {
  int r_in = lane >> 3, chunk = lane & 7;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    int row = wave * 16 + i * 8 + r_in;
    int grow = row0 + row;
    grow = grow < row_limit ? grow : row_limit;
    int chunk_src = chunk ^ (row & 7);
    const ushort* gptr = src + (int64_t)grow * ld + chunk_src * 8;
    char* lds_base = lds_tile + (wave * 16 + i * 8) * 128;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const uint32_t*>(gptr),
        reinterpret_cast<uint32_t*>(lds_base), 16, 0, 0);
  }
}

// stage V^T: [d][kv] image from V rows [kv, H*D] (one vectorized 16 B
// global read per group, 8 scattered 2 B LDS writes)
TFSC_DEV void stage_vt(const ushort* __restrict__ src, int64_t ld,
                       int row0, int row_limit, ushort* lds_tile,
                       int wave, int lane) {
  int tid = wave * WAVE + lane;
  #pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = pass * THREADS + tid;       // 512 groups: kv(64) x dgrp(8)
    int kv = idx >> 3;
    int dg = idx & 7;
    int gkv = row0 + kv;
    gkv = gkv < row_limit ? gkv : row_limit;
    const ushort* gptr = src + (int64_t)gkv * ld + dg * 8;
    short8_t v = *reinterpret_cast<const short8_t*>(gptr);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      int d = dg * 8 + j;
      int byte_off = t_off(d, kv >> 3) + (kv & 7) * 2;
      *reinterpret_cast<ushort*>(
          reinterpret_cast<char*>(lds_tile) + byte_off) = ushort(v[j]);
    }
  }
}

// RING = staged kv-tile slots: 4 (pair prefetch, 80 KB LDS, 2 WG/CU)
// for long sequences; 2 (no prefetch needed when n_kv <= 2, 48 KB LDS,
// 3 WG/CU) for the S <= 128 serving shapes
template <int RING>
__global__ __launch_bounds__(THREADS)
void attention_kernel(const ushort* __restrict__ Q,
                      const ushort* __restrict__ K,
                      const ushort* __restrict__ V,
                      ushort* __restrict__ O,
                      int B, int S, int H, float scale, int n_qblk) {
  // LDS: K ring RINGx8KB | V^T ring RINGx8KB | q/p overlay 16 KB
  // (q tile uses the first 8 KB until its registers are loaded; the
  // per-wave [16][128] P bounce then reuses the whole 16 KB)
  __shared__ __attribute__((aligned(16))) char smem[8192 * (2 * RING + 2)];
  auto k_lds = [&](int buf) -> char* { return smem + buf * 8192; };
  auto vt_lds = [&](int buf) -> char* {
    return smem + RING * 8192 + buf * 8192;
  };
  char* q_lds = smem + 2 * RING * 8192;
  char* p_lds = smem + 2 * RING * 8192;  // 4 waves x 4 KB (after Q)

  const int flat = blockIdx.x;
  const int qb = flat % n_qblk;
  const int h = (flat / n_qblk) % H;
  const int b = flat / (n_qblk * H);
  const int64_t ld = (int64_t)H * D;
  const ushort* Qb = Q + (int64_t)b * S * ld + h * D;
  const ushort* Kb = K + (int64_t)b * S * ld + h * D;
  const ushort* Vb = V + (int64_t)b * S * ld + h * D;
  ushort* Ob = O + (int64_t)b * S * ld + h * D;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int frow = lane & 15;
  const int kgrp = lane >> 4;
  const int q0 = qb * QBLK;

  // ---- stage Q AND the first kv tile pair concurrently (disjoint
  // LDS; one barrier covers all five tiles), then pull this wave's 16
  // q rows into registers
  stage_rows_glds(Qb, ld, q0, S - 1, q_lds, wave, lane);
  stage_rows_glds(Kb, ld, 0, S - 1, k_lds(0), wave, lane);
  stage_vt(Vb, ld, 0, S - 1, reinterpret_cast<ushort*>(vt_lds(0)), wave,
           lane);
  stage_rows_glds(Kb, ld, KVBLK, S - 1, k_lds(1), wave, lane);
  stage_vt(Vb, ld, KVBLK, S - 1, reinterpret_cast<ushort*>(vt_lds(1)),
           wave, lane);
  __syncthreads();
  bf16x8_t qf[2];      // A-fragments for the 2 K=32 steps over D
  #pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    int row = wave * 16 + frow;
    qf[ks] = *reinterpret_cast<const bf16x8_t*>(
        q_lds + t_off(row, ks * 4 + kgrp));
  }
  __syncthreads();   // q_lds consumed; the region becomes the P bounce

  // ---- online softmax state: this lane covers rows wave*16 + kgrp*4+r
  float m_run[4], l_run[4];
  f32x4_t o_acc[4] = {};      // O fragments over d (4 x 16 cols)
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -3.0e38f;
    l_run[r] = 0.f;
  }

  const int n_kv = (S + KVBLK - 1) / KVBLK;
  const int n_pair = (n_kv + 1) / 2;

  for (int t = 0; t < n_pair; ++t) {
    // pair 0 was staged with Q (covered by the prologue barriers); a
    // barrier here covers the pair t prefetch issued in iteration t-1
    if (t > 0) __syncthreads();
    const int base = RING >= 4 ? (t & 1) * 2 : 0;  // slots of this pair
    if (RING >= 4 && t + 1 < n_pair) {
      const int nb = ((t + 1) & 1) * 2;
      stage_rows_glds(Kb, ld, (2 * t + 2) * KVBLK, S - 1, k_lds(nb),
                      wave, lane);
      stage_vt(Vb, ld, (2 * t + 2) * KVBLK, S - 1,
               reinterpret_cast<ushort*>(vt_lds(nb)), wave, lane);
      stage_rows_glds(Kb, ld, (2 * t + 3) * KVBLK, S - 1, k_lds(nb + 1),
                      wave, lane);
      stage_vt(Vb, ld, (2 * t + 3) * KVBLK, S - 1,
               reinterpret_cast<ushort*>(vt_lds(nb + 1)), wave, lane);
    }

    // ---- QK^T: scores[16 q rows][128 kv] = 8 x 1x4 fragments
    f32x4_t sc[8] = {};
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const char* kt = k_lds(base + half);
      #pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
              kt + t_off(ni * 16 + frow, ks * 4 + kgrp));
          sc[half * 4 + ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[ks], kf, sc[half * 4 + ni], 0, 0, 0);
        }
      }
    }
    // scale + mask the kv tail
    const int kv_base = t * 2 * KVBLK;
    #pragma unroll
    for (int ni = 0; ni < 8; ++ni) {
      int kv_col = kv_base + ni * 16 + frow;   // this lane's kv column
      bool valid = kv_col < S;
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        sc[ni][r] = valid ? sc[ni][r] * scale : -3.0e38f;
    }

    // ---- per-row max over the 128 kv columns (one pass per pair)
    float pmax[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float v = sc[0][r];
      #pragma unroll
      for (int ni = 1; ni < 8; ++ni) v = fmaxf(v, sc[ni][r]);
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, 16));
      pmax[r] = v;
    }
    // online update
    float alpha[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_run[r], pmax[r]);
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
    }
    // exponentiate P and row-sum
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      float sum = 0.f;
      #pragma unroll
      for (int ni = 0; ni < 8; ++ni) {
        sc[ni][r] = __expf(sc[ni][r] - m_run[r]);
        sum += sc[ni][r];
      }
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        sum += __shfl_xor(sum, off, 16);
      l_run[r] = l_run[r] * alpha[r] + sum;
    }
    // rescale O by alpha (rows of O fragments match reg index)
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        o_acc[ni][r] *= alpha[r];

    // ---- P (C-layout) -> LDS [16][128] -> A-fragment layout
    {
      ushort* pw = reinterpret_cast<ushort*>(p_lds + wave * 4096);
      #pragma unroll
      for (int ni = 0; ni < 8; ++ni) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = kgrp * 4 + r;            // 0..15
          int col = ni * 16 + frow;          // 0..127
          *reinterpret_cast<ushort*>(
              reinterpret_cast<char*>(pw) + p_off(row, col >> 3) +
              (col & 7) * 2) = f2bf(sc[ni][r]);
        }
      }
      // wave-local write->read; compiler orders via lgkmcnt
      const char* pr = reinterpret_cast<const char*>(pw);
      #pragma unroll
      for (int ks = 0; ks < 4; ++ks) {       // contraction over 128 kv
        bf16x8_t pf = *reinterpret_cast<const bf16x8_t*>(
            pr + p_off(frow, ks * 4 + kgrp));
        const char* vt = vt_lds(base + (ks >> 1));
        int ksub = ks & 1;
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
              vt + t_off(ni * 16 + frow, ksub * 4 + kgrp));
          o_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pf, vf, o_acc[ni], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: O /= l, store rows q0 + wave*16 + kgrp*4 + r
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = q0 + wave * 16 + kgrp * 4 + r;
    if (row >= S) continue;
    float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    ushort* orow = Ob + (int64_t)row * ld;
    #pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      orow[ni * 16 + frow] = f2bf(o_acc[ni][r] * inv);
  }
}

}  // namespace attn

void launch_attention(hipStream_t s, const ushort* Q, const ushort* K,
                      const ushort* V, ushort* O, int B, int S, int H,
                      int D_, float scale) {
  if (D_ != attn::D)
    throw std::runtime_error("fused attention requires head_dim == 64");
  int n_qblk = (S + attn::QBLK - 1) / attn::QBLK;
  dim3 grid(B * H * n_qblk);
  if (S <= 2 * attn::KVBLK)
    hipLaunchKernelGGL(attn::attention_kernel<2>, grid,
                       dim3(attn::THREADS), 0, s, Q, K, V, O, B, S, H,
                       scale, n_qblk);
  else
    hipLaunchKernelGGL(attn::attention_kernel<4>, grid,
                       dim3(attn::THREADS), 0, s, Q, K, V, O, B, S, H,
                       scale, n_qblk);
}

}  // namespace tfsc
