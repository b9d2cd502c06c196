#include "hip/hip_runtime.h"
// Loss kernels (CDNA4, gfx950).
//
// * Fused cross-entropy: log-softmax + NLL in one pass (mean reduction),
//   saving the row logsumexp for a one-pass backward — the op BASELINE.json
//   names for the classification configs.
// * MSE — the reference template's criterion (reference ddp.py:164,222).
//
// Backward kernels read the incoming dLoss as a DEVICE scalar so the train
// loop never syncs the host on loss readback (SURVEY.md §3.2 calls out the
// reference's per-step loss.item() stall).

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace loss {

// one block per row: lse + per-row loss contribution (atomic mean)
template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              float* __restrict__ loss_out,
                              float* __restrict__ lse_out, int D) {
  __shared__ float lds[4];
  const long long m = blockIdx.x;
  const T* row = logits + m * D;
  float mx = -INFINITY;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    mx = fmaxf(mx, to_f(row[d]));
  float wmax = wave_reduce_max(mx);
  if (lane_id() == 0) lds[wave_id()] = wmax;
  __syncthreads();
  float bmax = fmaxf(fmaxf(lds[0], lds[1]), fmaxf(lds[2], lds[3]));
  __syncthreads();
  float s = 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x)
    s += __expf(to_f(row[d]) - bmax);
  s = block_reduce_sum<256>(s, lds);
  if (threadIdx.x == 0) {
    const float lse = bmax + __logf(s);
    lse_out[m] = lse;
    const float nll = lse - to_f(row[target[m]]);
    atomicAdd(loss_out, nll / (float)gridDim.x);  // mean over rows
  }
}

// dlogits = (softmax - onehot) * (dloss / M); dloss read on device
template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              T* __restrict__ dlogits, long long M, int D) {
  const float scale = dloss[0] / (float)M;
  const long long total = M * D;
  GRID_STRIDE(i, total) {
    const long long m = i / D;
    const int d = (int)(i % D);
    float sm = __expf(to_f(logits[i]) - lse[m]);
    if ((long long)d == target[m]) sm -= 1.f;
    dlogits[i] = to_t<T>(sm * scale);
  }
}

// mse: mean((p - t)^2); partial sums via block reduce + atomic
template <typename T>
__global__ void mse_fwd_kernel(const T* __restrict__ p, const T* __restrict__ t,
                               float* __restrict__ out, long long n) {
  __shared__ float lds[4];
  float acc = 0.f;
  GRID_STRIDE(i, n) {
    float d = to_f(p[i]) - to_f(t[i]);
    acc += d * d;
  }
  float total = block_reduce_sum<256>(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, total / (float)n);
}

template <typename T>
__global__ void mse_bwd_kernel(const T* __restrict__ p, const T* __restrict__ t,
                               const float* __restrict__ dloss,
                               T* __restrict__ dp, long long n) {
  const float scale = 2.f * dloss[0] / (float)n;
  GRID_STRIDE(i, n) {
    dp[i] = to_t<T>(scale * (to_f(p[i]) - to_f(t[i])));
  }
}

}  // namespace loss

// ======================= host launchers ==================================

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor target) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(target.scalar_type() == torch::kInt64);
  long long M = logits.size(0);
  int D = (int)logits.size(1);
  auto f32 = logits.options().dtype(torch::kFloat32);
  auto out = torch::zeros({}, f32);
  auto lse = torch::empty({M}, f32);
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(logits.scalar_type(), "ce_fwd", [&] {
    hipLaunchKernelGGL((loss::ce_fwd_kernel<scalar_t>), dim3((unsigned)M),
                       dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                       target.data_ptr<int64_t>(), out.data_ptr<float>(),
                       lse.data_ptr<float>(), D);
  });
  return {out, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor target,
                     torch::Tensor lse, torch::Tensor dloss) {
  long long M = logits.size(0);
  int D = (int)logits.size(1);
  auto dl = torch::empty_like(logits);
  auto stream = c10::hip::getCurrentHIPStream();
  auto dloss_f = dloss.to(torch::kFloat32);
  DDP_DISPATCH_FLOAT(logits.scalar_type(), "ce_bwd", [&] {
    hipLaunchKernelGGL((loss::ce_bwd_kernel<scalar_t>),
                       dim3(grid_1d(M * D, 256)), dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                       target.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       dloss_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dl.data_ptr()), M, D);
  });
  return dl;
}

torch::Tensor mse_fwd(torch::Tensor p, torch::Tensor t) {
  long long n = p.numel();
  auto out = torch::zeros({}, p.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  DDP_DISPATCH_FLOAT(p.scalar_type(), "mse_fwd", [&] {
    hipLaunchKernelGGL((loss::mse_fwd_kernel<scalar_t>),
                       dim3(grid_1d(n, 256)), dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(p.data_ptr()),
                       reinterpret_cast<const scalar_t*>(t.data_ptr()),
                       out.data_ptr<float>(), n);
  });
  return out;
}

torch::Tensor mse_bwd(torch::Tensor p, torch::Tensor t, torch::Tensor dloss) {
  long long n = p.numel();
  auto dp = torch::empty_like(p);
  auto stream = c10::hip::getCurrentHIPStream();
  auto dloss_f = dloss.to(torch::kFloat32);
  DDP_DISPATCH_FLOAT(p.scalar_type(), "mse_bwd", [&] {
    hipLaunchKernelGGL((loss::mse_bwd_kernel<scalar_t>),
                       dim3(grid_1d(n, 256)), dim3(256), 0, stream,
                       reinterpret_cast<const scalar_t*>(p.data_ptr()),
                       reinterpret_cast<const scalar_t*>(t.data_ptr()),
                       dloss_f.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dp.data_ptr()), n);
  });
  return dp;
}
