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

// ---- BN pass 1: per-channel sum / sumsq (atomics over M-split grid) ----
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, float* __restrict__ sum,
                                float* __restrict__ sumsq, long long M, int C) {
  const int c = blockIdx.x * kWave + lane_id();
  if (c >= C) return;
  const int nw = blockDim.x / kWave;
  const long long rows_per = (M + gridDim.y - 1) / gridDim.y;
  const long long m0 = blockIdx.y * rows_per;
  const long long m1 = min(M, m0 + rows_per);
  float s = 0.f, ss = 0.f;
  for (long long m = m0 + wave_id(); m < m1; m += nw) {
    float v = to_f(x[m * C + c]);
    s += v;
    ss += v * v;
  }
  atomicAdd(&sum[c], s);
  atomicAdd(&sumsq[c], ss);
}

// ---- BN pass 2: finalize mean/rstd + update running stats ----
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float momentum, float eps, long long M,
                                   int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mu = sum[c] / (float)M;
  const float var = fmaxf(sumsq[c] / (float)M - mu * mu, 0.f);
  mean[c] = mu;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    const long long denom = M > 1 ? M - 1 : 1;
    const float unbiased = var * ((float)M / (float)denom);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---- BN pass 3: normalize (+optional fused relu) ----
template <typename T, bool RELU>
__global__ void bn_norm_kernel(const T* __restrict__ x, T* __restrict__ y,
                               const float* __restrict__ mean,
                               const float* __restrict__ rstd,
                               const T* __restrict__ gamma,
                               const T* __restrict__ beta, long long M, int C) {
  const long long total = M * C;
  GRID_STRIDE(i, total) {
    const int c = (int)(i % C);
    float v = (to_f(x[i]) - mean[c]) * rstd[c] * to_f(gamma[c]) + to_f(beta[c]);
    if (RELU) v = fmaxf(v, 0.f);
    y[i] = to_t<T>(v);
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
template <typename T>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ dgamma,
                                    float* __restrict__ dbeta, long long M,
                                    int C) {
  const int c = blockIdx.x * kWave + lane_id();
  if (c >= C) return;
  const int nw = blockDim.x / kWave;
  const long long rows_per = (M + gridDim.y - 1) / gridDim.y;
  const long long m0 = blockIdx.y * rows_per;
  const long long m1 = min(M, m0 + rows_per);
  const float mu = mean[c], rs = rstd[c];
  float dg = 0.f, db = 0.f;
  for (long long m = m0 + wave_id(); m < m1; m += nw) {
    float g = to_f(dy[m * C + c]);
    float xh = (to_f(x[m * C + c]) - mu) * rs;
    dg += g * xh;
    db += g;
  }
  atomicAdd(&dgamma[c], dg);
  atomicAdd(&dbeta[c], db);
}

// ---- BN bwd pass 2: dx ----
template <typename T>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x, T* __restrict__ dx,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const T* __restrict__ gamma,
                                 const float* __restrict__ dgamma,
                                 const float* __restrict__ dbeta, long long M,
                                 int C) {
  const float invM = 1.f / (float)M;
  const long long total = M * C;
  GRID_STRIDE(i, total) {
    const int c = (int)(i % C);
    const float mu = mean[c], rs = rstd[c];
    float g = to_f(dy[i]);
    float xh = (to_f(x[i]) - mu) * rs;
    float v = to_f(gamma[c]) * rs *
              (g - invM * (dbeta[c] + xh * dgamma[c]));
    dx[i] = to_t<T>(v);
  }
}

// ---- LayerNorm fwd: one block per row ----
template <typename T>
__global__ void ln_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                              const T* __restrict__ gamma,
                              const T* __restrict__ beta,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, int D, float eps) {
  __shared__ float lds[4];
  const long long m = blockIdx.x;
  const T* row = x + m * D;
  T* yrow = y + m * D;
  float s = 0.f, ss = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float v = to_f(row[d]);
    s += v;
    ss += v * v;
  }
  s = block_reduce_sum<256>(s, lds);
  __syncthreads();
  ss = block_reduce_sum<256>(ss, lds);
  const float mu = s / D;
  const float var = fmaxf(ss / D - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[m] = mu;
    rstd_out[m] = rs;
  }
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
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
                              T* __restrict__ dx, float* __restrict__ dgamma,
                              float* __restrict__ dbeta, int D) {
  __shared__ float lds[4];
  const long long m = blockIdx.x;
  const T* dyr = dy + m * D;
  const T* xr = x + m * D;
  T* dxr = dx + m * D;
  const float mu = mean[m], rs = rstd[m];
  float s1 = 0.f, s2 = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float g = to_f(dyr[d]) * to_f(gamma[d]);
    float xh = (to_f(xr[d]) - mu) * rs;
    s1 += g;
    s2 += g * xh;
  }
  s1 = block_reduce_sum<256>(s1, lds);
  __syncthreads();
  s2 = block_reduce_sum<256>(s2, lds);
  const float invD = 1.f / D;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float g = to_f(dyr[d]) * to_f(gamma[d]);
    float xh = (to_f(xr[d]) - mu) * rs;
    dxr[d] = to_t<T>(rs * (g - invD * (s1 + xh * s2)));
    // dgamma/dbeta accumulated per row into f32 buffers (atomics)
    atomicAdd(&dgamma[d], to_f(dyr[d]) * xh);
    atomicAdd(&dbeta[d], to_f(dyr[d]));
  }
}

// ---- row softmax fwd (pre-scale) ----
template <typename T>
__global__ void softmax_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int D, float scale) {
  __shared__ float lds[4];
  const long long m = blockIdx.x;
  const T* row = x + m * D;
  T* yrow = y + m * D;
  float mx = -INFINITY;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    mx = fmaxf(mx, to_f(row[d]) * scale);
  // block max via wave reduce + lds
  float wmax = wave_reduce_max(mx);
  if (lane_id() == 0) lds[wave_id()] = wmax;
  __syncthreads();
  float bmax = fmaxf(fmaxf(lds[0], lds[1]), fmaxf(lds[2], lds[3]));
  __syncthreads();
  float s = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    s += __expf(to_f(row[d]) * scale - bmax);
  s = block_reduce_sum<256>(s, lds);
  const float inv = 1.f / s;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    yrow[d] = to_t<T>(__expf(to_f(row[d]) * scale - bmax) * inv);
}

// ---- row softmax bwd: dx = scale * y * (dy - sum(y*dy)) ----
template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y, T* __restrict__ dx,
                                   int D, float scale) {
  __shared__ float lds[4];
  const long long m = blockIdx.x;
  const T* dyr = dy + m * D;
  const T* yr = y + m * D;
  T* dxr = dx + m * D;
  float dot = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    dot += to_f(dyr[d]) * to_f(yr[d]);
  dot = block_reduce_sum<256>(dot, lds);
  for (int d = threadIdx.x; d < D; d += blockDim.x)
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
  auto sum = torch::zeros({C}, f32);
  auto sumsq = torch::zeros({C}, f32);
  auto mean = torch::empty({C}, f32);
  auto rstd = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 sgrid((C + kWave - 1) / kWave,
             (unsigned)std::min<long long>(64, (M + 4095) / 4096) );
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_fwd", [&] {
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    hipLaunchKernelGGL((nrm::bn_stats_kernel<scalar_t>), sgrid, dim3(256), 0,
                       stream, xp, sum.data_ptr<float>(),
                       sumsq.data_ptr<float>(), M, C);
    hipLaunchKernelGGL(
        nrm::bn_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0, stream,
        sum.data_ptr<float>(), sumsq.data_ptr<float>(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(),
        running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
        running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
        (float)momentum, (float)eps, M, C);
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    const auto* bp = reinterpret_cast<const scalar_t*>(beta.data_ptr());
    auto* yp = reinterpret_cast<scalar_t*>(y.data_ptr());
    if (relu)
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, true>),
                         dim3(grid_1d(M * C, 256)), dim3(256), 0, stream, xp,
                         yp, mean.data_ptr<float>(), rstd.data_ptr<float>(), gp,
                         bp, M, C);
    else
      hipLaunchKernelGGL((nrm::bn_norm_kernel<scalar_t, false>),
                         dim3(grid_1d(M * C, 256)), dim3(256), 0, stream, xp,
                         yp, mean.data_ptr<float>(), rstd.data_ptr<float>(), gp,
                         bp, M, C);
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
                                  torch::Tensor rstd) {
  long long M = x.size(0);
  int C = (int)x.size(1);
  auto f32 = x.options().dtype(torch::kFloat32);
  auto dgamma = torch::zeros({C}, f32);
  auto dbeta = torch::zeros({C}, f32);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 sgrid((C + kWave - 1) / kWave,
             (unsigned)std::min<long long>(64, (M + 4095) / 4096));
  DDP_DISPATCH_FLOAT(x.scalar_type(), "bn_bwd", [&] {
    const auto* dyp = reinterpret_cast<const scalar_t*>(dy.data_ptr());
    const auto* xp = reinterpret_cast<const scalar_t*>(x.data_ptr());
    const auto* gp = reinterpret_cast<const scalar_t*>(gamma.data_ptr());
    hipLaunchKernelGGL((nrm::bn_bwd_stats_kernel<scalar_t>), sgrid, dim3(256),
                       0, stream, dyp, xp, mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dgamma.data_ptr<float>(),
                       dbeta.data_ptr<float>(), M, C);
    hipLaunchKernelGGL((nrm::bn_bwd_dx_kernel<scalar_t>),
                       dim3(grid_1d(M * C, 256)), dim3(256), 0, stream, dyp,
                       xp, reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), gp,
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), M, C);
  });
  return {dx, dgamma, dbeta};
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
    hipLaunchKernelGGL((nrm::ln_fwd_kernel<scalar_t>), dim3((unsigned)M),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       reinterpret_cast<const scalar_t*>(gamma.data_ptr()),
                       reinterpret_cast<const scalar_t*>(beta.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), D,
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
  auto dgamma = torch::zeros({D}, f32);
  auto dbeta = torch::zeros({D}, f32);
  auto dx = torch::empty_like(dy);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(x.scalar_type(), "ln_bwd", [&] {
    hipLaunchKernelGGL((nrm::ln_bwd_kernel<scalar_t>), dim3((unsigned)M),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<const scalar_t*>(gamma.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), D);
  });
  return {dx, dgamma, dbeta};
}

torch::Tensor softmax_fwd(torch::Tensor x, double scale) {
  long long M = x.size(0);
  int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(x.scalar_type(), "softmax_fwd", [&] {
    hipLaunchKernelGGL((nrm::softmax_fwd_kernel<scalar_t>), dim3((unsigned)M),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()), D,
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
    hipLaunchKernelGGL((nrm::softmax_bwd_kernel<scalar_t>), dim3((unsigned)M),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<const scalar_t*>(y.data_ptr()),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), D,
                       (float)scale);
  });
  return dx;
}
