// Normalization + softmax kernels (CDNA4, gfx950).
//
// BatchNorm NHWC ([M=N*H*W, C] channel-last: per-channel stats are coalesced
// 64-lane stripe reductions), LayerNorm (row-wise, ViT), row softmax (+pre-
// scale, attention).  All reductions accumulate fp32; BN fwd optionally
// fuses the ReLU epilogue (conv->bn->relu chains write memory once).
//
// batchnorm/layernorm "CDNA4 kernels, LDS reductions".

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace nrm {

// ---- BN pass 1: per-channel sum / sumsq ----
// Thread t owns channel slot c0 = (t % (C/8))*8 and strides rows; every
// load is a 16-B vector and consecutive threads cover consecutive channel
// slots, so the sweep is fully coalesced and HBM-bound.  Per-block LDS
// accumulation, then one global atomic per channel per block.
constexpr int kBnMaxC = 2048;  // largest channel count (ResNet-50 layer4)

template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ ws,
                                long long M, int C) {
  // ws layout: [2 * gridDim.x][C] — per-block plain-store partials (row b =
  // block b's sums, row gridDim.x + b its sumsqs).  No global atomics: with
  // 2048 blocks the old per-block atomic tail hammered a C-sized array from
  // every block and dominated small layers; the finalize pass reduces the
  // workspace instead.
  __shared__ float ssum[kBnMaxC];
  __shared__ float ssq[kBnMaxC];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    ssum[c] = 0.f;
    ssq[c] = 0.f;
  }
  __syncthreads();
  using IO = VecIO<T>;
  const int slots = C / 8;  // C % 8 == 0 (host checks)
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const int c0 = (int)(tid % slots) * 8;
  const long long row0 = tid / slots;
  const long long rstride = nthreads / slots;
  float a0 = 0, a1 = 0, a2 = 0, a3 = 0, a4 = 0, a5 = 0, a6 = 0, a7 = 0;
  float q0 = 0, q1 = 0, q2 = 0, q3 = 0, q4 = 0, q5 = 0, q6 = 0, q7 = 0;
  // When nthreads % slots != 0 the trailing threads have row0 == rstride and
  // would re-accumulate rows already owned by the row0 == 0 threads (double
  // counting).  They skip the sweep but still reach the barrier + epilogue.
  if (row0 < rstride)
  for (long long m = row0; m < M; m += rstride) {
    const T* p = x + m * C + c0;
    float v[8];
    if (IO::kPerLane == 8) {
      auto vec = *reinterpret_cast<const typename VecIO<T>::Vec*>(p);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = IO::get(vec, j);
    } else {  // f32: two 16-B loads
      auto va = *reinterpret_cast<const float4v*>(p);
      auto vb = *reinterpret_cast<const float4v*>(p + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        v[j] = va[j];
        v[4 + j] = vb[j];
      }
    }
    a0 += v[0]; a1 += v[1]; a2 += v[2]; a3 += v[3];
    a4 += v[4]; a5 += v[5]; a6 += v[6]; a7 += v[7];
    q0 += v[0]*v[0]; q1 += v[1]*v[1]; q2 += v[2]*v[2]; q3 += v[3]*v[3];
    q4 += v[4]*v[4]; q5 += v[5]*v[5]; q6 += v[6]*v[6]; q7 += v[7]*v[7];
  }
  float a[8] = {a0,a1,a2,a3,a4,a5,a6,a7};
  float q[8] = {q0,q1,q2,q3,q4,q5,q6,q7};
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&ssum[c0 + j], a[j]);
    atomicAdd(&ssq[c0 + j], q[j]);
  }
  __syncthreads();
  float* wsum = ws + (long long)blockIdx.x * C;
  float* wsq = ws + ((long long)gridDim.x + blockIdx.x) * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    wsum[c] = ssum[c];
    wsq[c] = ssq[c];
  }
}

// ---- BN pass 2: finalize mean/rstd + update running stats ----
// one block per channel; 256 threads stripe the workspace rows then tree-
// reduce in LDS (a serial per-thread loop over ~1024 partials sat on the
// critical path between the stats and norm kernels).
__global__ void bn_finalize_kernel(const float* __restrict__ ws, int nblocks,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float momentum, float eps, long long M,
                                   int C) {
  __shared__ float r1[256], r2[256];
  const int c = blockIdx.x;
  float s = 0.f, sq = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
    s += ws[(long long)b * C + c];
    sq += ws[((long long)nblocks + b) * C + c];
  }
  r1[threadIdx.x] = s;
  r2[threadIdx.x] = sq;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) {
      r1[threadIdx.x] += r1[threadIdx.x + w];
      r2[threadIdx.x] += r2[threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x != 0) return;
  s = r1[0];
  sq = r2[0];
  const float mu = s / (float)M;
  const float var = fmaxf(sq / (float)M - mu * mu, 0.f);
  mean[c] = mu;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    const long long denom = M > 1 ? M - 1 : 1;
    const float unbiased = var * ((float)M / (float)denom);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---- BN pass 3: normalize (+optional fused relu, +optional residual) ----
// Vectorized: each thread owns one 8-channel slot and strides rows, so the
// per-channel scale/shift loads hoist out of the row loop.  ADD fuses the
// residual skip into this pass (y = relu(bn(x) + addend)): the separate
// add_relu pass (read bn-out + skip, write out) disappears — one fewer
// full-tensor read+write per residual block.
template <typename T, bool RELU, bool ADD = false>
__global__ void bn_norm_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const T* __restrict__ gamma,
                               const T* __restrict__ beta, long long M, int C,
                               const T* __restrict__ addend = nullptr) {
  using IO = VecIO<T>;
  const int slots = C / 8;
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const int c0 = (int)(tid % slots) * 8;
  const long long row0 = tid / slots;
  const long long rstride = nthreads / slots;
  if (row0 >= rstride) return;  // trailing threads would duplicate rows
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = rstd[c0 + j] * to_f(gamma[c0 + j]);
    sh[j] = to_f(beta[c0 + j]) - mean[c0 + j] * sc[j];
  }
  for (long long m = row0; m < M; m += rstride) {
    if (IO::kPerLane == 8) {
      auto v = *reinterpret_cast<const typename VecIO<T>::Vec*>(x + m * C + c0);
      typename VecIO<T>::Vec va{};
      if (ADD)
        va = *reinterpret_cast<const typename VecIO<T>::Vec*>(
            addend + m * C + c0);
      typename VecIO<T>::Vec o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float r = IO::get(v, j) * sc[j] + sh[j];
        if (ADD) r += IO::get(va, j);
        IO::set(o, j, RELU ? fmaxf(r, 0.f) : r);
      }
      *reinterpret_cast<typename VecIO<T>::Vec*>(y + m * C + c0) = o;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float r = to_f(x[m * C + c0 + j]) * sc[j] + sh[j];
        if (ADD) r += to_f(addend[m * C + c0 + j]);
        y[m * C + c0 + j] = to_t<T>(RELU ? fmaxf(r, 0.f) : r);
      }
    }
  }
}

// ---- BN inference (running stats) ----
template <typename T, bool RELU>
__global__ void bn_infer_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const float* __restrict__ rmean,
                                const float* __restrict__ rvar,
                                const T* __restrict__ gamma,
                                const T* __restrict__ beta, float eps,
                                long long M, int C) {
  const long long total = M * C;
  GRID_STRIDE(i, total) {
    const int c = (int)(i % C);
    float rs = rsqrtf(rvar[c] + eps);
    float v = (to_f(x[i]) - rmean[c]) * rs * to_f(gamma[c]) + to_f(beta[c]);
    if (RELU) v = fmaxf(v, 0.f);
    y[i] = to_t<T>(v);
  }
}

// ---- BN bwd pass 1: dgamma = sum dy*xhat, dbeta = sum dy ----
// Same coalesced 8-channel-per-thread layout as bn_stats_kernel.
// RELU: the forward fused relu(bn(x)); dy is masked by y > 0 inline (the
// separate relu_bwd pass and its extra tensor round-trip disappear).
// Partials go to ws ([2 * gridDim.x][C], like bn_stats) — no global atomics.
template <typename T, bool RELU>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ ws, long long M,
                                    int C) {
  __shared__ float sdg[kBnMaxC];
  __shared__ float sdb[kBnMaxC];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    sdg[c] = 0.f;
    sdb[c] = 0.f;
  }
  __syncthreads();
  using IO = VecIO<T>;
  const int slots = C / 8;
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const int c0 = (int)(tid % slots) * 8;
  const long long row0 = tid / slots;
  const long long rstride = nthreads / slots;
  float mu[8], rs[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[c0 + j];
    rs[j] = rstd[c0 + j];
  }
  float dg[8] = {}, db[8] = {};
  // row0 >= rstride threads would double-count rows (nthreads % slots != 0);
  // they skip the sweep but still reach the barrier + workspace store.
  if (row0 < rstride)
  for (long long m = row0; m < M; m += rstride) {
    const T* pg = dy + m * C + c0;
    const T* px = x + m * C + c0;
    float g[8], v[8];
    if (IO::kPerLane == 8) {
      auto vg = *reinterpret_cast<const typename VecIO<T>::Vec*>(pg);
      auto vx = *reinterpret_cast<const typename VecIO<T>::Vec*>(px);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        g[j] = IO::get(vg, j);
        v[j] = IO::get(vx, j);
      }
      if (RELU) {
        auto vy = *reinterpret_cast<const typename VecIO<T>::Vec*>(
            y + m * C + c0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (!(IO::get(vy, j) > 0.f)) g[j] = 0.f;
      }
    } else {
      auto ga = *reinterpret_cast<const float4v*>(pg);
      auto gb = *reinterpret_cast<const float4v*>(pg + 4);
      auto xa = *reinterpret_cast<const float4v*>(px);
      auto xb = *reinterpret_cast<const float4v*>(px + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        g[j] = ga[j]; g[4 + j] = gb[j];
        v[j] = xa[j]; v[4 + j] = xb[j];
      }
      if (RELU) {
        auto ya = *reinterpret_cast<const float4v*>(y + m * C + c0);
        auto yb = *reinterpret_cast<const float4v*>(y + m * C + c0 + 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (!(ya[j] > 0.f)) g[j] = 0.f;
          if (!(yb[j] > 0.f)) g[4 + j] = 0.f;
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      db[j] += g[j];
      dg[j] += g[j] * (v[j] - mu[j]) * rs[j];
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&sdg[c0 + j], dg[j]);
    atomicAdd(&sdb[c0 + j], db[j]);
  }
  __syncthreads();
  float* wdg = ws + (long long)blockIdx.x * C;
  float* wdb = ws + ((long long)gridDim.x + blockIdx.x) * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    wdg[c] = sdg[c];
    wdb[c] = sdb[c];
  }
}

// reduce the bwd-stats workspace into dgamma/dbeta (block-per-channel tree
// reduce, like bn_finalize_kernel); also emits the grads in the param dtype
// so no separate f32->bf16 cast kernel runs per BN layer.
template <typename T>
__global__ void bn_bwd_finalize_kernel(const float* __restrict__ ws,
                                       int nblocks,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       T* __restrict__ dgamma16,
                                       T* __restrict__ dbeta16, int C) {
  __shared__ float r1[256], r2[256];
  const int c = blockIdx.x;
  float dg = 0.f, db = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
    dg += ws[(long long)b * C + c];
    db += ws[((long long)nblocks + b) * C + c];
  }
  r1[threadIdx.x] = dg;
  r2[threadIdx.x] = db;
  __syncthreads();
  for (int w = 128; w > 0; w >>= 1) {
    if (threadIdx.x < w) {
      r1[threadIdx.x] += r1[threadIdx.x + w];
      r2[threadIdx.x] += r2[threadIdx.x + w];
    }
    __syncthreads();
  }
  if (threadIdx.x != 0) return;
  dgamma[c] = r1[0];
  dbeta[c] = r2[0];
  if (dgamma16) {
    dgamma16[c] = to_t<T>(r1[0]);
    dbeta16[c] = to_t<T>(r2[0]);
  }
}

// ---- BN bwd pass 2: dx (vectorized 8-channel slots) ----
template <typename T, bool RELU>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const T* __restrict__ y, T* __restrict__ dx,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const T* __restrict__ gamma,
                                 const float* __restrict__ dgamma,
                                 const float* __restrict__ dbeta, long long M,
                                 int C) {
  using IO = VecIO<T>;
  const float invM = 1.f / (float)M;
  const int slots = C / 8;
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const int c0 = (int)(tid % slots) * 8;
  const long long row0 = tid / slots;
  const long long rstride = nthreads / slots;
  float mu[8], rs[8], gm[8], dgv[8], dbv[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[c0 + j];
    rs[j] = rstd[c0 + j];
    gm[j] = to_f(gamma[c0 + j]);
    dgv[j] = dgamma[c0 + j];
    dbv[j] = dbeta[c0 + j];
  }
  for (long long m = row0; m < M; m += rstride) {
    if (IO::kPerLane == 8) {
      auto vg = *reinterpret_cast<const typename VecIO<T>::Vec*>(dy + m * C + c0);
      auto vx = *reinterpret_cast<const typename VecIO<T>::Vec*>(x + m * C + c0);
      typename VecIO<T>::Vec vy{};
      if (RELU)
        vy = *reinterpret_cast<const typename VecIO<T>::Vec*>(y + m * C + c0);
      typename VecIO<T>::Vec o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = IO::get(vg, j);
        if (RELU && !(IO::get(vy, j) > 0.f)) g = 0.f;
        float xh = (IO::get(vx, j) - mu[j]) * rs[j];
        IO::set(o, j, gm[j] * rs[j] * (g - invM * (dbv[j] + xh * dgv[j])));
      }
      *reinterpret_cast<typename VecIO<T>::Vec*>(dx + m * C + c0) = o;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = to_f(dy[m * C + c0 + j]);
        if (RELU && !(to_f(y[m * C + c0 + j]) > 0.f)) g = 0.f;
        float xh = (to_f(x[m * C + c0 + j]) - mu[j]) * rs[j];
        dx[m * C + c0 + j] =
            to_t<T>(gm[j] * rs[j] * (g - invM * (dbv[j] + xh * dgv[j])));
      }
    }
  }
}

// ---- LayerNorm fwd: one block per row ----
template <typename T>
__global__ void ln_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                              const T* __restrict__ gamma,
                              const T* __restrict__ beta,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, long long M,
                              int D, float eps) {
  // one WAVE per row, 4 rows per block: rows are short (ViT D=768) and a
  // 256-thread block per row serialized on block-wide reductions
  const long long m = (long long)blockIdx.x * 4 + wave_id();
  if (m >= M) return;
  const int lane = lane_id();
  const T* row = x + m * D;
  T* yrow = y + m * D;
  float s = 0.f, ss = 0.f;
  for (int d = lane; d < D; d += 64) {
    float v = to_f(row[d]);
    s += v;
    ss += v * v;
  }
  s = wave_reduce_sum(s);
  ss = wave_reduce_sum(ss);
  const float mu = s / D;
  const float var = fmaxf(ss / D - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[m] = mu;
    rstd_out[m] = rs;
  }
  for (int d = lane; d < D; d += 64) {
    float v = (to_f(row[d]) - mu) * rs * to_f(gamma[d]) + to_f(beta[d]);
    yrow[d] = to_t<T>(v);
  }
}

// ---- LayerNorm bwd: dx per row + atomically accumulated dgamma/dbeta ----
template <typename T>
__global__ void ln_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                              const T* __restrict__ gamma,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              T* __restrict__ dx, long long M, int D) {
  const long long m = (long long)blockIdx.x * 4 + wave_id();
  if (m >= M) return;
  const int lane = lane_id();
  const T* dyr = dy + m * D;
  const T* xr = x + m * D;
  T* dxr = dx + m * D;
  const float mu = mean[m], rs = rstd[m];
  float s1 = 0.f, s2 = 0.f;
  for (int d = lane; d < D; d += 64) {
    float g = to_f(dyr[d]) * to_f(gamma[d]);
    float xh = (to_f(xr[d]) - mu) * rs;
    s1 += g;
    s2 += g * xh;
  }
  s1 = wave_reduce_sum(s1);
  s2 = wave_reduce_sum(s2);
  const float invD = 1.f / D;
  for (int d = lane; d < D; d += 64) {
    float g = to_f(dyr[d]) * to_f(gamma[d]);
    float xh = (to_f(xr[d]) - mu) * rs;
    dxr[d] = to_t<T>(rs * (g - invD * (s1 + xh * s2)));
  }
}

// ---- LN bwd params: dgamma/dbeta column sums -> per-block ws partials ----
// Same coalesced 8-col-slot layout as bn_bwd_stats (the old per-row atomic
// accumulation was ~38M global atomics onto a D-sized array per ViT layer);
// per-row mean/rstd are scalar loads per row iteration.
template <typename T>
__global__ void ln_bwd_param_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ ws, long long M,
                                    int D) {
  __shared__ float sdg[kBnMaxC];
  __shared__ float sdb[kBnMaxC];
  for (int c = threadIdx.x; c < D; c += blockDim.x) {
    sdg[c] = 0.f;
    sdb[c] = 0.f;
  }
  __syncthreads();
  using IO = VecIO<T>;
  const int slots = D / 8;
  const long long tid = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long nthreads = (long long)gridDim.x * blockDim.x;
  const int c0 = (int)(tid % slots) * 8;
  const long long row0 = tid / slots;
  const long long rstride = nthreads / slots;
  float dg[8] = {}, db[8] = {};
  // For ViT-B D=768 (slots=96) a 1024-block launch leaves 262144 % 96 = 64
  // trailing threads with row0 == rstride that would re-accumulate rows
  // ≡ 0 (mod rstride) — deterministic dgamma/dbeta overcount at production
  // M.  Skip the sweep for them; they still join the barrier + store.
  if (row0 < rstride)
  for (long long m = row0; m < M; m += rstride) {
    const float mu = mean[m], rs = rstd[m];
    const T* pg = dy + m * D + c0;
    const T* px = x + m * D + c0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float g = to_f(pg[j]);
      const float xh = (to_f(px[j]) - mu) * rs;
      db[j] += g;
      dg[j] += g * xh;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&sdg[c0 + j], dg[j]);
    atomicAdd(&sdb[c0 + j], db[j]);
  }
  __syncthreads();
  float* wdg = ws + (long long)blockIdx.x * D;
  float* wdb = ws + ((long long)gridDim.x + blockIdx.x) * D;
  for (int c = threadIdx.x; c < D; c += blockDim.x) {
    wdg[c] = sdg[c];
    wdb[c] = sdb[c];
  }
}

// ---- row softmax fwd (pre-scale) ----
// One WAVE per row, 4 rows per block (attention rows are short — S=197 —
// so a 256-thread block per row left 3/4 of the block idle and serialized
// on block-wide reductions; wave shuffles reduce in-register).
template <typename T>
__global__ void softmax_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   long long M, int D, float scale) {
  const long long m = (long long)blockIdx.x * 4 + wave_id();
  if (m >= M) return;
  const int lane = lane_id();
  const T* row = x + m * D;
  T* yrow = y + m * D;
  float mx = -INFINITY;
  for (int d = lane; d < D; d += 64) mx = fmaxf(mx, to_f(row[d]) * scale);
  mx = wave_reduce_max(mx);
  float s = 0.f;
  for (int d = lane; d < D; d += 64) s += __expf(to_f(row[d]) * scale - mx);
  s = wave_reduce_sum(s);
  const float inv = 1.f / s;
  for (int d = lane; d < D; d += 64)
    yrow[d] = to_t<T>(__expf(to_f(row[d]) * scale - mx) * inv);
}

// ---- row softmax bwd: dx = scale * y * (dy - sum(dy*y)) ----
template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y,
                                   T* __restrict__ dx, long long M, int D,
                                   float scale) {
  const long long m = (long long)blockIdx.x * 4 + wave_id();
  if (m >= M) return;
  const int lane = lane_id();
  const T* dyr = dy + m * D;
  const T* yr = y + m * D;
  T* dxr = dx + m * D;
  float dot = 0.f;
  for (int d = lane; d < D; d += 64) dot += to_f(dyr[d]) * to_f(yr[d]);
  dot = wave_reduce_sum(dot);
  for (int d = lane; d < D; d += 64)
    dxr[d] = to_t<T>(scale * to_f(yr[d]) * (to_f(dyr[d]) - dot));
}

}  // namespace nrm

// ======================= host launchers ==================================

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  c10::optional<torch::Tensor> running_mean,
                                  c10::optional<torch::Tensor> running_var,
                                  double momentum, double eps, bool relu) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous());
  long long M = x.size(0);
  int C = (int)x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto rstd = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  TORCH_CHECK(C % 8 == 0 && C <= 2048,
              "bn kernels need C % 8 == 0 and C <= 2048 (got ", C, ")");
  dim3 sgrid(grid_1d(M * C / 8, 256, 1024));
  auto ws = torch::empty({2 * (long long)sgrid.x, C}, f32);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_fwd", [&] {
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    hipLaunchKernelGGL((nrm::bn_stats_kernel<scalar_t>), sgrid, dim3(256), 0,
                       stream, xp, ws.data_ptr<float>(), M, C);
    hipLaunchKernelGGL(
        nrm::bn_finalize_kernel, dim3(C), dim3(256), 0, stream,
        ws.data_ptr<float>(), (int)sgrid.x, mean.data_ptr<float>(),
        rstd.data_ptr<float>(),
        running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
        running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
        (float)momentum, (float)eps, M, C);
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const auto* bp = reinterpret_cast<const scalar_t*>(beta.data_ptr());
    auto* yp = reinterpret_cast<scalar_t*>(y.data_ptr());
    if (relu)
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, true>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         xp, yp, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         gp, bp, M, C);
    else
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, false>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         xp, yp, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         gp, bp, M, C);
  });
  return {y, mean, rstd};
}

// BN forward from a precomputed partial workspace (the producing conv's
// epilogue wrote per-block sums/sumsqs of its output into ws [2*nb][C]) —
// the stats pass over x disappears entirely.
std::vector<torch::Tensor> bn_fwd_ws(torch::Tensor x, torch::Tensor gamma,
                                     torch::Tensor beta, torch::Tensor ws,
                                     c10::optional<torch::Tensor> running_mean,
                                     c10::optional<torch::Tensor> running_var,
                                     double momentum, double eps, bool relu,
                                     c10::optional<torch::Tensor> addend) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous());
  long long M = x.size(0);
  int C = (int)x.size(1);
  TORCH_CHECK(ws.size(1) == C && ws.size(0) % 2 == 0);
  const int nb = (int)(ws.size(0) / 2);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto rstd = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  TORCH_CHECK(C % 8 == 0 && C <= 2048);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_fwd_ws", [&] {
    hipLaunchKernelGGL(
        nrm::bn_finalize_kernel, dim3(C), dim3(256), 0, stream,
        ws.data_ptr<float>(), nb, mean.data_ptr<float>(),
        rstd.data_ptr<float>(),
        running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
        running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
        (float)momentum, (float)eps, M, C);
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const auto* bp = reinterpret_cast<const scalar_t*>(beta.data_ptr());
    auto* yp = reinterpret_cast<scalar_t*>(y.data_ptr());
    if (addend.has_value()) {
      TORCH_CHECK(relu, "bn_fwd_ws addend is only fused with relu=true");
      TORCH_CHECK(addend->is_contiguous() && addend->numel() == x.numel() &&
                  addend->scalar_type() == x.scalar_type());
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, true, true>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         xp, yp, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         gp, bp, M, C,
                         reinterpret_cast<const scalar_t*>(addend->data_ptr()));
    } else if (relu)
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, true>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         xp, yp, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         gp, bp, M, C);
    else
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, false>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         xp, yp, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         gp, bp, M, C);
  });
  return {y, mean, rstd};
}

torch::Tensor bn_infer(torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
                       torch::Tensor rmean, torch::Tensor rvar, double eps,
                       bool relu) {
  long long M = x.size(0);
  int C = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_infer", [&] {
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const auto* bp = reinterpret_cast<const scalar_t*>(beta.data_ptr());
    auto* yp = reinterpret_cast<scalar_t*>(y.data_ptr());
    if (relu)
      hipLaunchKernelGGL((nrm::bn_infer_kernel<scalar_t, true>),
                         dim3(grid_1d(M * C, 256)), dim3(256), 0, stream, xp,
                         yp, rmean.data_ptr<float>(), rvar.data_ptr<float>(),
                         gp, bp, (float)eps, M, C);
    else
      hipLaunchKernelGGL((nrm::bn_infer_kernel<scalar_t, false>),
                         dim3(grid_1d(M * C, 256)), dim3(256), 0, stream, xp,
                         yp, rmean.data_ptr<float>(), rvar.data_ptr<float>(),
                         gp, bp, (float)eps, M, C);
  });
  return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd,
                                  c10::optional<torch::Tensor> y_relu) {
  long long M = x.size(0);
  int C = (int)x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dgamma = torch::empty({C}, f32);
  auto dbeta = torch::empty({C}, f32);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream();
  TORCH_CHECK(C % 8 == 0 && C <= 2048,
              "bn kernels need C % 8 == 0 and C <= 2048 (got ", C, ")");
  const bool relu = y_relu.has_value();
  dim3 sgrid(grid_1d(M * C / 8, 256, 1024));
  auto ws = torch::empty({2 * (long long)sgrid.x, C}, f32);
  // grads returned in the param dtype (cast fused into the finalize)
  const bool wide = gamma.scalar_type() == torch::kFloat32;
  auto dgamma_out =
      wide ? dgamma : torch::empty({C}, x.options().dtype(gamma.scalar_type()));
  auto dbeta_out =
      wide ? dbeta : torch::empty({C}, x.options().dtype(gamma.scalar_type()));
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_bwd", [&] {
    const auto* dyp = reinterpret_cast<const scalar_t*>(dy.data_ptr());
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const auto* yp =
        relu ? reinterpret_cast<const scalar_t*>(y_relu->data_ptr()) : nullptr;
    if (relu)
      hipLaunchKernelGGL((nrm::bn_bwd_stats_kernel<scalar_t, true>), sgrid,
                         dim3(256), 0, stream, dyp, xp, yp,
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         ws.data_ptr<float>(), M, C);
    else
      hipLaunchKernelGGL((nrm::bn_bwd_stats_kernel<scalar_t, false>), sgrid,
                         dim3(256), 0, stream, dyp, xp, yp,
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         ws.data_ptr<float>(), M, C);
    hipLaunchKernelGGL((nrm::bn_bwd_finalize_kernel<scalar_t>), dim3(C),
                       dim3(256), 0, stream, ws.data_ptr<float>(),
                       (int)sgrid.x, dgamma.data_ptr<float>(),
                       dbeta.data_ptr<float>(),
                       wide ? nullptr
                            : reinterpret_cast<scalar_t*>(dgamma_out.data_ptr()),
                       wide ? nullptr
                            : reinterpret_cast<scalar_t*>(dbeta_out.data_ptr()),
                       C);
    if (relu)
      hipLaunchKernelGGL((nrm::bn_bwd_dx_kernel<scalar_t, true>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         dyp, xp, yp, reinterpret_cast<scalar_t*>(dx.data_ptr()),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), gp,
                         dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), M,
                         C);
    else
      hipLaunchKernelGGL((nrm::bn_bwd_dx_kernel<scalar_t, false>),
                         dim3(grid_1d(M * C / 8, 256)), dim3(256), 0, stream,
                         dyp, xp, yp, reinterpret_cast<scalar_t*>(dx.data_ptr()),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), gp,
                         dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), M,
                         C);
  });
  return {dx, dgamma_out, dbeta_out};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor beta, double eps) {
  long long M = x.size(0);
  int D = (int)x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({M}, f32);
  auto rstd = torch::empty({M}, f32);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(x.scalar_type(), "ln_fwd", [&] {
    hipLaunchKernelGGL((nrm::ln_fwd_kernel<scalar_t>),
                       dim3((unsigned)((M + 3) / 4)), dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       reinterpret_cast<const scalar_t*>(gamma.data_ptr()),
                       reinterpret_cast<const scalar_t*>(beta.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), M, D,
                       (float)eps);
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor gamma,
                                         torch::Tensor mean,
                                         torch::Tensor rstd) {
  long long M = x.size(0);
  int D = (int)x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dgamma = torch::empty({D}, f32);
  auto dbeta = torch::empty({D}, f32);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream();
  TORCH_CHECK(D % 8 == 0 && D <= 2048,
              "ln bwd needs D % 8 == 0 and D <= 2048 (got ", D, ")");
  dim3 sgrid(grid_1d(M * D / 8, 256, 1024));
  auto ws = torch::empty({2 * (long long)sgrid.x, D}, f32);
  const bool wide = gamma.scalar_type() == torch::kFloat32;
  auto dgamma_out =
      wide ? dgamma : torch::empty({D}, x.options().dtype(gamma.scalar_type()));
  auto dbeta_out =
      wide ? dbeta : torch::empty({D}, x.options().dtype(gamma.scalar_type()));
  DDP_DISPATCH_FLOAT(x.scalar_type(), "ln_bwd", [&] {
    const auto* dyp = reinterpret_cast<const scalar_t*>(dy.data_ptr());
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    hipLaunchKernelGGL((nrm::ln_bwd_kernel<scalar_t>),
                       dim3((unsigned)((M + 3) / 4)), dim3(256), 0, stream,
                       dyp, xp,
                       reinterpret_cast<const scalar_t*>(gamma.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), M, D);
    hipLaunchKernelGGL((nrm::ln_bwd_param_kernel<scalar_t>), sgrid, dim3(256),
                       0, stream, dyp, xp, mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), ws.data_ptr<float>(), M, D);
    hipLaunchKernelGGL((nrm::bn_bwd_finalize_kernel<scalar_t>), dim3(D),
                       dim3(256), 0, stream, ws.data_ptr<float>(),
                       (int)sgrid.x, dgamma.data_ptr<float>(),
                       dbeta.data_ptr<float>(),
                       wide ? nullptr
                            : reinterpret_cast<scalar_t*>(dgamma_out.data_ptr()),
                       wide ? nullptr
                            : reinterpret_cast<scalar_t*>(dbeta_out.data_ptr()),
                       D);
  });
  return {dx, dgamma_out, dbeta_out};
}

torch::Tensor softmax_fwd(torch::Tensor x, double scale) {
  long long M = x.size(0);
  int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(x.scalar_type(), "softmax_fwd", [&] {
    hipLaunchKernelGGL((nrm::softmax_fwd_kernel<scalar_t>), dim3((unsigned)((M + 3) / 4)),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()), M, D,
                       (float)scale);
  });
  return y;
}

torch::Tensor softmax_bwd(torch::Tensor dy, torch::Tensor y, double scale) {
  long long M = y.size(0);
  int D = (int)y.size(1);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(y.scalar_type(), "softmax_bwd", [&] {
    hipLaunchKernelGGL((nrm::softmax_bwd_kernel<scalar_t>), dim3((unsigned)((M + 3) / 4)),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<const scalar_t*>(y.data_ptr()),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), M, D,
                       (float)scale);
  });
  return dx;
}
