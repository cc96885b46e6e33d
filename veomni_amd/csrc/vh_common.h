// Common device/host helpers for the gfx950 hot-path kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <cstdio>
#include <cstring>

// ---------------------------------------------------------------- error path
void vh_set_error(const char* fmt, ...);

#define VH_CHECK(cond, ...)                                                    \
  do {                                                                         \
    if (!(cond)) {                                                             \
      vh_set_error(__VA_ARGS__);                                               \
      return 1;                                                                \
    }                                                                          \
  } while (0)

#define VH_HIP(call)                                                           \
  do {                                                                         \
    hipError_t _e = (call);                                                    \
    if (_e != hipSuccess) {                                                    \
      vh_set_error("%s failed: %s", #call, hipGetErrorString(_e));             \
      return 2;                                                                \
    }                                                                          \
  } while (0)

// ---------------------------------------------------------------- bf16 utils
using bf16_t = uint16_t;

__device__ __forceinline__ float bf2f(bf16_t v) {
  union { uint32_t u; float f; } c;
  c.u = static_cast<uint32_t>(v) << 16;
  return c.f;
}

__device__ __forceinline__ bf16_t f2bf(float f) {
  // round-to-nearest-even, matching torch's fp32->bf16 conversion
  union { float f; uint32_t u; } c;
  c.f = f;
  uint32_t u = c.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return static_cast<bf16_t>((u >> 16) | 0x40);  // NaN
  uint32_t lsb = (u >> 16) & 1u;
  u += 0x7fffu + lsb;
  return static_cast<bf16_t>(u >> 16);
}

// vectorized 8 x bf16 (16 B)
struct bf16x8 { bf16_t v[8]; };
static_assert(sizeof(bf16x8) == 16, "");

typedef __attribute__((ext_vector_type(8))) __bf16 bf16v8;   // MFMA operand
typedef __attribute__((ext_vector_type(4))) float f32x4;      // MFMA acc (16x16)
typedef __attribute__((ext_vector_type(4))) int int32x4;

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// silu and its derivative in fp32 (matches torch aten::silu numerics class)
__device__ __forceinline__ float siluf(float x) { return x * sigmoidf_(x); }
__device__ __forceinline__ float dsiluf(float x) {
  float s = sigmoidf_(x);
  return s * (1.0f + x * (1.0f - s));
}

constexpr int kWave = 64;
