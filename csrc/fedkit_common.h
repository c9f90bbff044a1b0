// Common helpers for the fedkit CDNA4 (gfx950) kernels.
// All kernels in csrc/ are written for MI355X: wave64, 256 CUs / 8 XCDs,
// HBM3E-bound elementwise paths vectorized to 16 B/lane (guide G13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

static inline hipStream_t fedkit_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define FEDKIT_CHECK(x, msg) TORCH_CHECK((x), msg)

constexpr int kWave = 64;

static inline int grid_1d(long long total, int block, int cap = 2048) {
  long long g = (total + block - 1) / block;
  return (int)std::min<long long>(g, cap);
}

// dtype tags
template <typename T> struct is_bf16 { static constexpr bool value = false; };
template <> struct is_bf16<__hip_bfloat16> { static constexpr bool value = true; };

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ void from_f32(float v, float& out) { out = v; }
__device__ __forceinline__ void from_f32(float v, __hip_bfloat16& out) {
  out = __float2bfloat16(v);
}

__device__ __forceinline__ float fedkit_elu_f(float x) {
  return x > 0.f ? x : __expf(x) - 1.f;
}

#define DISPATCH_F32_BF16(TENSOR, NAME, ...)                                 \
  do {                                                                         \
    if ((TENSOR).scalar_type() == at::kFloat) {                                \
      using scalar_t = float;                                                  \
      __VA_ARGS__;                                                             \
    } else if ((TENSOR).scalar_type() == at::kBFloat16) {                      \
      using scalar_t = __hip_bfloat16;                                         \
      __VA_ARGS__;                                                             \
    } else {                                                                   \
      TORCH_CHECK(false, NAME ": unsupported dtype ", (TENSOR).scalar_type()); \
    }                                                                          \
  } while (0)
