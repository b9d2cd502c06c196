// dtype dispatch + VecIO shared by the host launchers / kernels.
#pragma once

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

// Map torch scalar types onto the kernel types (float / bf16 / f16).
#define DDP_DISPATCH_FLOAT(TYPE, NAME, ...)                         \
  [&] {                                                             \
    switch (TYPE) {                                                 \
      case torch::kFloat32: {                                       \
        using scalar_t = float;                                     \
        return __VA_ARGS__();                                       \
      }                                                             \
      case torch::kBFloat16: {                                      \
        using scalar_t = bf16;                                      \
        return __VA_ARGS__();                                       \
      }                                                             \
      case torch::kHalf: {                                          \
        using scalar_t = _Float16;                                  \
        return __VA_ARGS__();                                       \
      }                                                             \
      default:                                                      \
        TORCH_CHECK(false, NAME, ": unsupported dtype ", TYPE);     \
    }                                                               \
  }()

// float <-> T conversions used by generic kernels
template <typename T>
DEV_INLINE float to_f(T v);
template <>
DEV_INLINE float to_f<float>(float v) {
  return v;
}
template <>
DEV_INLINE float to_f<bf16>(bf16 v) {
  return __bfloat162float(v);
}
template <>
DEV_INLINE float to_f<_Float16>(_Float16 v) {
  return (float)v;
}

template <typename T>
DEV_INLINE T to_t(float v);
template <>
DEV_INLINE float to_t<float>(float v) {
  return v;
}
template <>
DEV_INLINE bf16 to_t<bf16>(float v) {
  return __float2bfloat16(v);
}
template <>
DEV_INLINE _Float16 to_t<_Float16>(float v) {
  return (_Float16)v;
}

// 16-byte vectorized access helpers
template <typename T>
struct VecIO;

template <>
struct VecIO<float> {
  static constexpr int kPerLane = 4;  // 16 B
  using Vec = float4v;
  DEV_INLINE static float get(const Vec& v, int i) { return v[i]; }
  DEV_INLINE static void set(Vec& v, int i, float x) { v[i] = x; }
};

template <>
struct VecIO<bf16> {
  static constexpr int kPerLane = 8;  // 16 B
  using Vec = short8;
  DEV_INLINE static float get(const Vec& v, int i) { return bfbits2f(v[i]); }
  DEV_INLINE static void set(Vec& v, int i, float x) { v[i] = f2bfbits(x); }
};

template <>
struct VecIO<_Float16> {
  static constexpr int kPerLane = 8;  // 16 B
  using Vec = half8;
  DEV_INLINE static float get(const Vec& v, int i) { return (float)v[i]; }
  DEV_INLINE static void set(Vec& v, int i, float x) { v[i] = (_Float16)x; }
};
