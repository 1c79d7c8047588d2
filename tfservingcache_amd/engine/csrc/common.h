// Common device helpers for the CDNA4 (gfx950) serving kernels.
//
// All activation tensors are bf16 (stored as ushort words); integer
// tensors are i32; accumulation is f32. Written for MI355X only: wave64,
// MFMA via __builtin_amdgcn_mfma_*, LDS-tiled staging
// (see /opt/skills/guides/cdna_hip_programming.md).
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define TFSC_DEV __device__ __forceinline__

using bf16_t = ushort;

TFSC_DEV float bf2f(bf16_t v) {
  union { uint32_t u; float f; } c;
  c.u = uint32_t(v) << 16;
  return c.f;
}

// round-to-nearest-even f32 -> bf16
TFSC_DEV bf16_t f2bf(float f) {
  union { float f; uint32_t u; } c;
  c.f = f;
  uint32_t u = c.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return bf16_t(0x7fc0);  // NaN
  uint32_t lsb = (u >> 16) & 1u;
  u += 0x7fffu + lsb;
  return bf16_t(u >> 16);
}

TFSC_DEV uint32_t f2bf2(float lo, float hi) {
  return uint32_t(f2bf(lo)) | (uint32_t(f2bf(hi)) << 16);
}

// short4/short8 vector types for coalesced bf16 loads (G13: always
// vectorize bf16 — scalar bf16 loads are ~2-2.5x slower)
typedef short  short4_t  __attribute__((ext_vector_type(4)));
typedef short  short8_t  __attribute__((ext_vector_type(8)));
typedef float  f32x4     __attribute__((ext_vector_type(4)));
typedef float  f32x16    __attribute__((ext_vector_type(16)));
typedef short  bf16x4    __attribute__((ext_vector_type(4)));
typedef short  bf16x8    __attribute__((ext_vector_type(8)));

constexpr int WAVE = 64;

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(_e) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                   \
    }                                                                       \
  } while (0)

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }
