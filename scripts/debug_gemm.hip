// Standalone triage for the GEMM kernel pieces. Builds with:
//   hipcc --offload-arch=gfx950 -O2 scripts/debug_gemm.hip -o /tmp/dbg
// Runs three stages so a fault isolates: (1) glds tile staging echo,
// (2) single MFMA fragment check, (3) the full 128x128x64 gemm tile.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;
typedef float f32x4_t __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf2f(ushort v) {
  union { unsigned u; float f; } c; c.u = unsigned(v) << 16; return c.f;
}
__device__ __forceinline__ ushort f2bf(float f) {
  union { float f; unsigned u; } c; c.f = f;
  unsigned u = c.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;
  u += 0x7fffu + ((u >> 16) & 1u);
  return ushort(u >> 16);
}

__device__ __forceinline__ int lds_off(int row, int chunk) {
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

// stage a [128][64] bf16 tile with glds and write it back deswizzled
__global__ void stage_echo(const ushort* __restrict__ src, ushort* dst,
                           int M, int K) {
  __shared__ __attribute__((aligned(16))) char smem[16384];
  int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  int r_in = lane >> 3, chunk = lane & 7;
  for (int i = 0; i < 4; ++i) {
    int row = wave * 32 + i * 8 + r_in;
    int grow = row < M - 1 ? row : M - 1;
    int chunk_src = chunk ^ (row & 7);
    const ushort* gptr = src + (long)grow * K + chunk_src * 8;
    char* lds_base = smem + (wave * 32 + i * 8) * 128;
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned*>(gptr),
        reinterpret_cast<unsigned*>(lds_base), 16, 0, 0);
  }
  __syncthreads();
  // read back deswizzled: each thread copies 32 elems
  for (int e = threadIdx.x; e < 128 * 64; e += blockDim.x) {
    int row = e / 64, col = e % 64;
    int chunk2 = col / 8, within = col % 8;
    const ushort* p = reinterpret_cast<const ushort*>(
        smem + lds_off(row, chunk2)) + within;
    dst[e] = *p;
  }
}

// one MFMA: D = A(16x32) @ B(32x16) with fragment layout check
__global__ void mfma_once(const ushort* A, const ushort* Bt, float* D) {
  int lane = threadIdx.x;
  int frow = lane & 15, kgrp = lane >> 4;
  bf16x8_t a, b;
  for (int j = 0; j < 8; ++j) {
    // A[i][k]: i=frow, k=kgrp*8+j ; row-major [16][32]
    reinterpret_cast<ushort*>(&a)[j] = A[frow * 32 + kgrp * 8 + j];
    // B^T[j][k]: col j=frow, k likewise ; Bt is [16][32]
    reinterpret_cast<ushort*>(&b)[j] = Bt[frow * 32 + kgrp * 8 + j];
  }
  f32x4_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  // C layout: col=lane&15, row=(lane>>4)*4+r
  for (int r = 0; r < 4; ++r) {
    int row = (lane >> 4) * 4 + r, col = lane & 15;
    D[row * 16 + col] = acc[r];
  }
}

int main() {
  int M = 128, N = 128, K = 64;
  std::vector<ushort> hA(M * K);
  for (int i = 0; i < M * K; ++i) {
    float v = float((i * 37 % 113) - 56) / 56.0f;
    unsigned u; float f = v; __builtin_memcpy(&u, &f, 4);
    u += 0x7fffu + ((u >> 16) & 1u);
    hA[i] = ushort(u >> 16);
  }
  ushort *dA, *dOut;
  hipMalloc(&dA, M * K * 2);
  hipMalloc(&dOut, M * K * 2);
  hipMemcpy(dA, hA.data(), M * K * 2, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(stage_echo, dim3(1), dim3(256), 0, 0, dA, dOut, M, K);
  if (hipDeviceSynchronize() != hipSuccess) {
    printf("STAGE_ECHO: sync FAILED: %s\n",
           hipGetErrorString(hipGetLastError()));
    return 1;
  }
  std::vector<ushort> hOut(M * K);
  hipMemcpy(hOut.data(), dOut, M * K * 2, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < M * K && bad < 5; ++i)
    if (hOut[i] != hA[i]) {
      printf("stage mismatch at %d (row %d col %d): got %04x want %04x\n",
             i, i / 64, i % 64, hOut[i], hA[i]);
      ++bad;
    }
  printf("STAGE_ECHO: %s\n", bad ? "FAIL" : "OK");

  // mfma test
  std::vector<float> fa(16 * 32), fb(16 * 32);
  std::vector<ushort> ha(16 * 32), hb(16 * 32);
  auto tobf = [](float f) { unsigned u; __builtin_memcpy(&u, &f, 4);
                            u += 0x7fffu + ((u >> 16) & 1u);
                            return ushort(u >> 16); };
  for (int i = 0; i < 16 * 32; ++i) {
    fa[i] = float((i % 7) - 3) * 0.25f;
    fb[i] = float(((i * 3) % 11) - 5) * 0.125f;   // asymmetric
    ha[i] = tobf(fa[i]);
    hb[i] = tobf(fb[i]);
  }
  ushort *da, *db; float* dd;
  hipMalloc(&da, 16 * 32 * 2); hipMalloc(&db, 16 * 32 * 2);
  hipMalloc(&dd, 16 * 16 * 4);
  hipMemcpy(da, ha.data(), 16 * 32 * 2, hipMemcpyHostToDevice);
  hipMemcpy(db, hb.data(), 16 * 32 * 2, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(mfma_once, dim3(1), dim3(64), 0, 0, da, db, dd);
  if (hipDeviceSynchronize() != hipSuccess) {
    printf("MFMA: sync FAILED: %s\n", hipGetErrorString(hipGetLastError()));
    return 1;
  }
  std::vector<float> hd(16 * 16);
  hipMemcpy(hd.data(), dd, 16 * 16 * 4, hipMemcpyDeviceToHost);
  double maxerr = 0;
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      double want = 0;
      for (int k = 0; k < 32; ++k) want += double(fa[i * 32 + k]) *
                                           double(fb[j * 32 + k]);
      maxerr = fmax(maxerr, fabs(want - hd[i * 16 + j]));
    }
  printf("MFMA: maxerr=%g %s\n", maxerr, maxerr < 0.05 ? "OK" : "FAIL");
  return 0;
}
