// Shared device-side helpers for the sparktorch_amd CDNA4 (gfx950) kernels.
// Pure HIP: no torch headers here — launchers take raw pointers + hipStream_t.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE 64

// bf16 carried as raw uint16 bits; conversions are explicit bit ops so we do
// not depend on hip_bf16 operator overloads inside hot loops.
typedef uint16_t bf16raw;

__device__ __forceinline__ float bf16_to_f32(bf16raw h) {
  union {
    uint32_t u;
    float f;
  } c;
  c.u = ((uint32_t)h) << 16;
  return c.f;
}

// round-to-nearest-even f32 -> bf16
__device__ __forceinline__ bf16raw f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } c;
  c.f = f;
  uint32_t u = c.u;
  uint32_t rounding = 0x7fff + ((u >> 16) & 1);
  u += rounding;
  return (bf16raw)(u >> 16);
}

typedef float floatx4 __attribute__((ext_vector_type(4)));
typedef float floatx16 __attribute__((ext_vector_type(16)));
typedef short shortx8 __attribute__((ext_vector_type(8)));
typedef short shortx4 __attribute__((ext_vector_type(4)));

__host__ __device__ __forceinline__ int64_t ceil_div_i64(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

#define HIP_CHECK_LAUNCH()                                         \
  do {                                                             \
    hipError_t err_ = hipGetLastError();                           \
    if (err_ != hipSuccess) return err_;                           \
  } while (0)
