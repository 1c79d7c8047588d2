// MX-fp8 (OCP e4m3) groundwork probe for the round-2 fp8 serving path.
//
// gfx950's only large-K low-precision MFMA is the block-scaled
//   __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4
//     (A: i32x8 = 32 fp8/lane, B: i32x8, C: f32x4,
//      cbsz imm = A fmt, blgp imm = B fmt (0 = fp8 e4m3),
//      opsel_a imm, scale_a (packed e8m0), opsel_b imm, scale_b)
// This probe (a) verifies the A/B fragment layout + unit-scale numerics
// against a host fp32 reference on dequantized operands, and (b)
// measures the register-resident MFMA rate vs the bf16 16x16x32 MFMA —
// the "2x bf16" headroom that motivates an MX-fp8 GEMM.
//
// Build:  hipcc --offload-arch=gfx950 -O3 scripts/fp8_probe.hip -o /tmp/fp8_probe
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdint>
#include <cstdlib>
#include <vector>

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef int i32x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define CHK(x)                                                        \
  do {                                                                \
    hipError_t e_ = (x);                                              \
    if (e_ != hipSuccess) {                                           \
      fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e_),    \
              __LINE__);                                              \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

// ---- host e4m3 (OCP e4m3fn: bias 7, no inf, max 448) ---------------------
static uint8_t f32_to_e4m3(float f) {
  if (f != f) return 0x7F;             // nan
  uint8_t sign = f < 0 ? 0x80 : 0;
  f = fabsf(f);
  if (f > 448.f) f = 448.f;
  if (f < 0.001953125f / 8.f) return sign;   // < half min subnormal
  int e;
  float m = frexpf(f, &e);             // f = m * 2^e, m in [0.5, 1)
  e -= 1;
  m *= 2.f;                            // m in [1, 2)
  if (e < -6) {                        // subnormal: 2^-6 * (mant/8)
    int mant = int(roundf(f / 0.001953125f * 8.f));  // 2^-9 steps
    if (mant > 7) return sign | 0x08;  // rounds up to first normal
    return sign | uint8_t(mant);
  }
  int mant = int(roundf((m - 1.f) * 8.f));
  if (mant == 8) {
    mant = 0;
    e += 1;
  }
  if (e > 8) return sign | 0x7E;       // saturate to 448
  return sign | uint8_t((e + 7) << 3) | uint8_t(mant);
}

static float e4m3_to_f32(uint8_t v) {
  int sign = v & 0x80 ? -1 : 1;
  int exp = (v >> 3) & 0xF;
  int mant = v & 7;
  if (exp == 0) return sign * ldexpf(float(mant) / 8.f, -6);
  if (exp == 15 && mant == 7) return sign * nanf("");
  return sign * ldexpf(1.f + float(mant) / 8.f, exp - 7);
}

// ---- numerics kernel: one 16x16x128 tile, unit scales --------------------
// Layout hypothesis (the 16x16x32 bf16 pattern scaled to 32 B/lane):
//   A[row][k]: lane = (k/32)*16 + row, byte = k%32
__global__ void k_fp8_tile(const uint8_t* A, const uint8_t* B,
                           float* D) {
  int lane = threadIdx.x;
  int row = lane & 15;
  int kblk = lane >> 4;
  i32x8 a, b;
  const uint8_t* ap = A + row * 128 + kblk * 32;
  const uint8_t* bp = B + row * 128 + kblk * 32;
  a = *reinterpret_cast<const i32x8*>(ap);
  b = *reinterpret_cast<const i32x8*>(bp);
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a, b, c, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
  // D layout (shape-determined): col = lane&15, row = (lane>>4)*4 + reg
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    D[(kblk * 4 + r) * 16 + row] = c[r];
}

// scale-semantics mapper: VARIANT selects an sa hypothesis
template <int VARIANT>
__global__ void k_fp8_tile_scaled(const uint8_t* A, const uint8_t* B,
                                  float* D) {
  int lane = threadIdx.x;
  int row = lane & 15;
  int kblk = lane >> 4;
  i32x8 a = *reinterpret_cast<const i32x8*>(A + row * 128 + kblk * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + row * 128 + kblk * 32);
  f32x4 c = {};
  int sa = 0x7F7F7F7F;
  if (VARIANT == 0) sa = 128;                       // uniform 2.0, byte0
  if (VARIANT == 1) sa = lane == 0 ? 128 : 127;     // 2.0 on lane0 only
  if (VARIANT == 2) sa = 126 + kblk;                // per-lane own-block
  if (VARIANT == 3)                                 // 4 packed, opsel-able
    sa = 126 | (127 << 8) | (128 << 16) | (129 << 24);
  int sb = 0x7F7F7F7F;
  if (VARIANT == 4) {                               // scale B too
    sa = 126 + kblk;
    sb = 126 + kblk;                                // B col=lane&15, blk
  }
  if (VARIANT >= 16 && VARIANT < 20)                // lane 16*(V-16) only
    sa = lane == 16 * (VARIANT - 16) ? 128 : 127;
  if (VARIANT == 6) {                               // per-ROW scales:
    sa = 126 + (row & 3);                           // every k-lane of a
    sb = 126 + ((row + 1) & 3);                     // row shares it
  }
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a, b, c, 0, 0, 0, sa, 0, sb);
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    D[(kblk * 4 + r) * 16 + row] = c[r];
}

// ---- rate kernels: register-resident MFMA loops --------------------------
template <int ITERS>
__global__ void k_fp8_rate(const uint8_t* A, float* out) {
  i32x8 a = *reinterpret_cast<const i32x8*>(A + threadIdx.x * 32);
  i32x8 b = a;
  f32x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  for (int i = 0; i < ITERS; ++i) {
    acc0 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc0, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    acc1 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc1, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    acc2 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc2, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    acc3 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc3, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
  }
  f32x4 s = acc0 + acc1 + acc2 + acc3;
  if (threadIdx.x == 0) out[blockIdx.x] = s[0] + s[1] + s[2] + s[3];
}

template <int ITERS>
__global__ void k_bf16_rate(const uint8_t* A, float* out) {
  bf16x8 a = *reinterpret_cast<const bf16x8*>(A + threadIdx.x * 16);
  bf16x8 b = a;
  f32x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  for (int i = 0; i < ITERS; ++i) {
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0);
  }
  f32x4 s = acc0 + acc1 + acc2 + acc3;
  if (threadIdx.x == 0) out[blockIdx.x] = s[0] + s[1] + s[2] + s[3];
}

int main() {
  // ---- numerics: asymmetric operands (guide G9) -----------------------
  std::vector<uint8_t> A(16 * 128), B(16 * 128);
  std::vector<float> Af(16 * 128), Bf(16 * 128);
  srand(7);
  for (int i = 0; i < 16 * 128; ++i) {
    float av = (rand() % 2000 - 1000) / 500.f;
    float bv = (rand() % 2000 - 1000) / 700.f + 0.3f;
    // avoid e4m3 subnormals (probing HW denormal behavior separately)
    if (fabsf(av) < 0.017f) av = 0.f;
    if (fabsf(bv) < 0.017f) bv = 0.3f;
    A[i] = f32_to_e4m3(av);
    B[i] = f32_to_e4m3(bv);
    Af[i] = e4m3_to_f32(A[i]);          // reference on dequantized
    Bf[i] = e4m3_to_f32(B[i]);
  }
  uint8_t *dA, *dB;
  float* dD;
  CHK(hipMalloc(&dA, A.size()));
  CHK(hipMalloc(&dB, B.size()));
  CHK(hipMalloc(&dD, 16 * 16 * 4));
  CHK(hipMemcpy(dA, A.data(), A.size(), hipMemcpyHostToDevice));
  CHK(hipMemcpy(dB, B.data(), B.size(), hipMemcpyHostToDevice));

  auto check2 = [&](const char* name) {
    std::vector<float> D(16 * 16);
    CHK(hipMemcpy(D.data(), dD, D.size() * 4, hipMemcpyDeviceToHost));
    double max_err = 0;
    for (int r = 0; r < 16; ++r)
      for (int c = 0; c < 16; ++c) {
        double want = 0, mag = 0;
        for (int k = 0; k < 128; ++k) {
          double sc = ldexp(1.0, 2 * ((k / 32) - 1));   // A and B scaled
          double t = double(Af[r * 128 + k]) *
                     double(Bf[c * 128 + k]) * sc;
          want += t;
          mag += fabs(t);
        }
        max_err = fmax(max_err, fabs(want - D[r * 16 + c]) /
                                    (mag + 1e-3));
      }
    printf("%-22s max_rel_err=%.3g  %s\n", name, max_err,
           max_err < 1e-4 ? "OK" : "FAIL");
    return max_err < 1e-4;
  };

  auto check = [&](const char* name, bool scaled) {
    std::vector<float> D(16 * 16);
    CHK(hipMemcpy(D.data(), dD, D.size() * 4, hipMemcpyDeviceToHost));
    double max_err = 0;
    for (int r = 0; r < 16; ++r)
      for (int c = 0; c < 16; ++c) {
        double want = 0;
        for (int k = 0; k < 128; ++k) {
          double sa = scaled ? ldexp(1.0, (k / 32) - 1) : 1.0;
          want += double(Af[r * 128 + k]) * sa * double(Bf[c * 128 + k]);
        }
        double mag = 0;
        for (int k = 0; k < 128; ++k)
          mag += fabs(double(Af[r * 128 + k]) *
                      double(Bf[c * 128 + k])) *
                 (scaled ? ldexp(1.0, (k / 32) - 1) : 1.0);
        // normalize by the dot's magnitude (f32 accumulation in HW vs
        // double here; cancellation makes |want| the wrong yardstick)
        max_err = fmax(max_err, fabs(want - D[r * 16 + c]) /
                                    (mag + 1e-3));
      }
    printf("%-22s max_rel_err=%.3g  %s\n", name, max_err,
           max_err < 1e-4 ? "OK" : "FAIL");
    return max_err < 1e-4;
  };

  hipLaunchKernelGGL(k_fp8_tile, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  CHK(hipDeviceSynchronize());
  bool ok = check("fp8 e4m3 unit-scale", false);
  // scale-semantics mapping: report observed D[0][0], D[4][0], D[0][1]
  // ratios vs the unit-scale result for each hypothesis
  std::vector<float> Dunit(16 * 16);
  CHK(hipMemcpy(Dunit.data(), dD, 16 * 16 * 4, hipMemcpyDeviceToHost));
  auto run_variant = [&](auto kern, const char* name) {
    hipLaunchKernelGGL(kern, dim3(1), dim3(64), 0, 0, dA, dB, dD);
    CHK(hipDeviceSynchronize());
    std::vector<float> Ds(16 * 16);
    CHK(hipMemcpy(Ds.data(), dD, 16 * 16 * 4, hipMemcpyDeviceToHost));
    printf("scale %-18s ratios r0c0=%.3f r4c0=%.3f r0c1=%.3f "
           "r8c0=%.3f r12c0=%.3f\n", name,
           Ds[0] / Dunit[0], Ds[4 * 16] / Dunit[4 * 16],
           Ds[1] / Dunit[1], Ds[8 * 16] / Dunit[8 * 16],
           Ds[12 * 16] / Dunit[12 * 16]);
  };
  run_variant(k_fp8_tile_scaled<0>, "uniform-128-b0");
  run_variant(k_fp8_tile_scaled<1>, "lane0-only-128");
  run_variant(k_fp8_tile_scaled<3>, "packed4-b0");

  // decode the lane->(row, K-block) scale mapping: doubling only lane
  // 16*j's scale adds S_{p(j)} (that block's dot) to row-0 cells;
  // match the observed delta against each block's host-computed dot
  for (int j = 0; j < 4; ++j) {
    switch (j) {
      case 0: hipLaunchKernelGGL(k_fp8_tile_scaled<16>, dim3(1),
                                 dim3(64), 0, 0, dA, dB, dD); break;
      case 1: hipLaunchKernelGGL(k_fp8_tile_scaled<17>, dim3(1),
                                 dim3(64), 0, 0, dA, dB, dD); break;
      case 2: hipLaunchKernelGGL(k_fp8_tile_scaled<18>, dim3(1),
                                 dim3(64), 0, 0, dA, dB, dD); break;
      case 3: hipLaunchKernelGGL(k_fp8_tile_scaled<19>, dim3(1),
                                 dim3(64), 0, 0, dA, dB, dD); break;
    }
    CHK(hipDeviceSynchronize());
    std::vector<float> Ds(16 * 16);
    CHK(hipMemcpy(Ds.data(), dD, 16 * 16 * 4, hipMemcpyDeviceToHost));
    // match the delta against every contiguous k-range start*g..+g
    // for granularities g in {8, 16, 32} (across columns jointly)
    printf("scale lane %2d:", 16 * j);
    for (int g = 8; g <= 32; g *= 2) {
      int best = -1;
      double bestd = 1e30;
      for (int s0 = 0; s0 + g <= 128; s0 += g) {
        double err = 0;
        for (int c = 0; c < 8; ++c) {
          double S = 0;
          for (int k = s0; k < s0 + g; ++k)
            S += double(Af[k]) * double(Bf[c * 128 + k]);
          double delta = Ds[c] - Dunit[c];
          err += fabs(delta - S);
        }
        if (err < bestd) {
          bestd = err;
          best = s0;
        }
      }
      printf("  g%d: k=[%d,%d) err %.3g", g, best, best + g, bestd);
    }
    printf("\n");
  }

  // verified semantics: lane 16*blk+row supplies the e8m0 A-scale for
  // (row, K-block) at the opsel byte (symmetric for B with col) —
  // full numeric check of the A-scaled and A+B-scaled forms
  // OPEN ITEM: per-32-block MX scales need the ISA's fragment K-order
  // (a K-permutation inside the 32-byte lane fragment; invisible to
  // unit-scale tests since A and B share it). The contiguous-k
  // hypothesis below FAILS by design — kept as documentation. Rowwise
  // scaling (next check) is permutation-invariant and verified.
  hipLaunchKernelGGL(k_fp8_tile_scaled<2>, dim3(1), dim3(64), 0, 0, dA,
                     dB, dD);
  CHK(hipDeviceSynchronize());
  check("fp8 MX blk (open item)", true);     // informational only
  // per-ROW (A) + per-COL (B) scales — the rowwise fp8 quantization
  // recipe; K-permutation-invariant, so usable without the ISA's
  // fragment K-order
  hipLaunchKernelGGL(k_fp8_tile_scaled<6>, dim3(1), dim3(64), 0, 0, dA,
                     dB, dD);
  CHK(hipDeviceSynchronize());
  {
    std::vector<float> D(16 * 16);
    CHK(hipMemcpy(D.data(), dD, D.size() * 4, hipMemcpyDeviceToHost));
    double max_err = 0;
    for (int r = 0; r < 16; ++r)
      for (int c = 0; c < 16; ++c) {
        double sa = ldexp(1.0, (r & 3) - 1);
        double sb = ldexp(1.0, ((c + 1) & 3) - 1);
        double want = 0, mag = 0;
        for (int k = 0; k < 128; ++k) {
          double t = double(Af[r * 128 + k]) *
                     double(Bf[c * 128 + k]) * sa * sb;
          want += t;
          mag += fabs(t);
        }
        max_err = fmax(max_err, fabs(want - D[r * 16 + c]) /
                                    (mag + 1e-3));
      }
    printf("%-22s max_rel_err=%.3g  %s\n", "fp8 rowwise-scaled",
           max_err, max_err < 1e-4 ? "OK" : "FAIL");
    ok &= max_err < 1e-4;
  }

  // ---- HW subnormal behavior: A = one subnormal byte, B = ones ------
  std::vector<uint8_t> As(16 * 128, 0), Bs(16 * 128, 0x38);  // 1.0
  for (int i = 0; i < 16 * 128; ++i) As[i] = 0x04;  // 2^-7 subnormal
  CHK(hipMemcpy(dA, As.data(), As.size(), hipMemcpyHostToDevice));
  CHK(hipMemcpy(dB, Bs.data(), Bs.size(), hipMemcpyHostToDevice));
  hipLaunchKernelGGL(k_fp8_tile, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  CHK(hipDeviceSynchronize());
  float d00;
  CHK(hipMemcpy(&d00, dD, 4, hipMemcpyDeviceToHost));
  printf("subnormal probe: 128 x (0x04=2^-7) x 1.0 -> HW %.6f, "
         "IEEE-style expect %.6f (0 => HW flushes e4m3 denormals)\n",
         d00, 128.0 * ldexpf(1.f, -7));
  CHK(hipMemcpy(dA, A.data(), A.size(), hipMemcpyHostToDevice));
  CHK(hipMemcpy(dB, B.data(), B.size(), hipMemcpyHostToDevice));

  // ---- rate (register-resident; 2048 blocks x 256 thr fills 256 CUs) --
  constexpr int ITERS = 4096, BLOCKS = 2048, THREADS = 256;
  float* dOut;
  CHK(hipMalloc(&dOut, BLOCKS * 4));
  auto time_one = [&](auto kern, double flop_per_mfma) {
    hipLaunchKernelGGL(kern, dim3(BLOCKS), dim3(THREADS), 0, 0, dA,
                       dOut);                       // warm
    CHK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    hipEventCreate(&t0);
    hipEventCreate(&t1);
    hipEventRecord(t0);
    hipLaunchKernelGGL(kern, dim3(BLOCKS), dim3(THREADS), 0, 0, dA,
                       dOut);
    hipEventRecord(t1);
    CHK(hipEventSynchronize(t1));
    float ms;
    hipEventElapsedTime(&ms, t0, t1);
    double mfmas = double(BLOCKS) * (THREADS / 64) * 4 * ITERS;
    return mfmas * flop_per_mfma / (ms * 1e-3) / 1e12;
  };
  double tf_fp8 = time_one(k_fp8_rate<ITERS>, 2.0 * 16 * 16 * 128);
  double tf_bf16 = time_one(k_bf16_rate<ITERS>, 2.0 * 16 * 16 * 32);
  printf("MFMA rate: MX-fp8 16x16x128 = %.0f TF, bf16 16x16x32 = %.0f "
         "TF (ratio %.2fx)\n", tf_fp8, tf_bf16, tf_fp8 / tf_bf16);
  return ok ? 0 : 1;
}
