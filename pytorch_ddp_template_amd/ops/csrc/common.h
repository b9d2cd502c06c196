// Common device-side helpers for the CDNA4 (gfx950) kernels.
//
// Conventions (see /opt/skills guides; SURVEY.md §2b for the op inventory):
//  * wave = 64 lanes; blocks are multiples of 64 (256 default).
//  * bf16 memory I/O is always vectorized (>= 8 bf16 / 16 B per lane where
//    the layout permits); fp32 accumulate everywhere.
//  * memory-bound kernels use grid-stride loops capped at ~2048 blocks
//    (256 CUs x 8 blocks) so the scheduler keeps every XCD busy.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV_INLINE __device__ __forceinline__

constexpr int kWave = 64;

using bf16 = __hip_bfloat16;

// 16-byte vector types for wide loads/stores
typedef float float4v __attribute__((ext_vector_type(4)));
typedef short short8 __attribute__((ext_vector_type(8)));
typedef _Float16 half8 __attribute__((ext_vector_type(8)));

DEV_INLINE float bf2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 f2bf(float v) { return __float2bfloat16(v); }

// bf16 <-> float via raw bits (for packed short8 processing)
DEV_INLINE float bfbits2f(short u) {
  union {
    unsigned int i;
    float f;
  } x;
  x.i = ((unsigned int)(unsigned short)u) << 16;
  return x.f;
}
DEV_INLINE short f2bfbits(float f) {
  union {
    float f;
    unsigned int i;
  } x;
  x.f = f;
  // round-to-nearest-even
  unsigned int lsb = (x.i >> 16) & 1u;
  unsigned int rounded = x.i + 0x7fffu + lsb;
  return (short)(rounded >> 16);
}

// grid-stride loop helper
#define GRID_STRIDE(i, n)                                              \
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; \
       i < (n); i += (long long)gridDim.x * blockDim.x)

DEV_INLINE int lane_id() { return threadIdx.x & (kWave - 1); }
DEV_INLINE int wave_id() { return threadIdx.x >> 6; }

// wave-wide f32 sum (64 lanes) — butterfly: the result is valid in EVERY
// lane (shfl_down cascades are lane-0-only and silently wrong for callers
// that broadcast the reduction, e.g. row softmax / LayerNorm)
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block reduce via LDS (blockDim.x <= 1024); every thread returns the sum
template <int BLOCK>
DEV_INLINE float block_reduce_sum(float v, float* lds_scratch /*BLOCK/64*/) {
  float w = wave_reduce_sum(v);
  if (lane_id() == 0) lds_scratch[wave_id()] = w;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < BLOCK / kWave; ++i) total += lds_scratch[i];
  return total;
}

static inline int grid_1d(long long n, int block, int cap = 2048) {
  long long g = (n + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return (int)g;
}
