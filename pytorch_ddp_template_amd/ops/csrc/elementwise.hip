// Elementwise / data-movement kernels (CDNA4, gfx950).
//
// All memory-bound: the design target is the HBM3E ceiling (~6.3 TB/s
// measured), so every bf16 access is vectorized to 16 B/lane (guide G13 —
// hipcc does not auto-vectorize bf16) and grids are capped + grid-strided.
//
// Covers: ReLU fwd/bwd (reference model.py:12), fused
// residual add+relu, GELU (ViT), bias-grad column sum, 2-D transpose,
// global avg-pool + max-pool (ResNet), and the conv2d dgrad helpers
// (weight rotation, zero-stuffing for strided dgrad).

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace ew {

// ---------------- generic vectorized unary/binary over bf16/f32 ----------

struct ReluOp {
  DEV_INLINE float operator()(float a) const { return fmaxf(a, 0.f); }
};
struct GeluOp {  // exact erf form (matches torch F.gelu default)
  DEV_INLINE float operator()(float a) const {
    return 0.5f * a * (1.f + erff(a * 0.70710678118654752440f));
  }
};

template <typename T, typename Op>
__global__ void unary_kernel(const T* __restrict__ in, T* __restrict__ out,
                             long long n, Op op) {
  using IO = VecIO<T>;
  constexpr int P = IO::kPerLane;
  const long long nvec = n / P;
  const auto* vin = reinterpret_cast<const typename IO::Vec*>(in);
  auto* vout = reinterpret_cast<typename IO::Vec*>(out);
  GRID_STRIDE(i, nvec) {
    typename IO::Vec v = vin[i];
#pragma unroll
    for (int j = 0; j < P; ++j) IO::set(v, j, op(IO::get(v, j)));
    vout[i] = v;
  }
  const long long base = nvec * P;
  GRID_STRIDE(i, n - base) {
    out[base + i] = to_t<T>(op(to_f(in[base + i])));
  }
}

template <typename T, typename Op2>
__global__ void binary_kernel(const T* __restrict__ a, const T* __restrict__ b,
                              T* __restrict__ out, long long n, Op2 op) {
  using IO = VecIO<T>;
  constexpr int P = IO::kPerLane;
  const long long nvec = n / P;
  const auto* va = reinterpret_cast<const typename IO::Vec*>(a);
  const auto* vb = reinterpret_cast<const typename IO::Vec*>(b);
  auto* vo = reinterpret_cast<typename IO::Vec*>(out);
  GRID_STRIDE(i, nvec) {
    typename IO::Vec x = va[i], y = vb[i], o;
#pragma unroll
    for (int j = 0; j < P; ++j) IO::set(o, j, op(IO::get(x, j), IO::get(y, j)));
    vo[i] = o;
  }
  const long long base = nvec * P;
  GRID_STRIDE(i, n - base) {
    out[base + i] = to_t<T>(op(to_f(a[base + i]), to_f(b[base + i])));
  }
}

struct AddReluOp {
  DEV_INLINE float operator()(float a, float b) const {
    return fmaxf(a + b, 0.f);
  }
};
struct AddOp {
  DEV_INLINE float operator()(float a, float b) const { return a + b; }
};
struct ReluBwdOp {  // (dy, y) -> dy * (y > 0)
  DEV_INLINE float operator()(float dy, float y) const {
    return y > 0.f ? dy : 0.f;
  }
};
struct GeluBwdOp {  // (dy, x) -> dy * (Phi(x) + x phi(x))
  DEV_INLINE float operator()(float dy, float x) const {
    float cdf = 0.5f * (1.f + erff(x * 0.70710678118654752440f));
    float pdf = 0.39894228040143267794f * __expf(-0.5f * x * x);
    return dy * (cdf + x * pdf);
  }
};

// ---------------- column sum (bias gradient): dy[M,N] -> f32 [N] ---------
template <typename T>
__global__ void col_sum_kernel(const T* __restrict__ dy,
                               float* __restrict__ out, long long M,
                               long long N) {
  const long long col = (long long)blockIdx.x * kWave + lane_id();
  if (col >= N) return;
  const int nw = blockDim.x / kWave;
  const long long rows_per_blk = (M + gridDim.y - 1) / gridDim.y;
  const long long m0 = blockIdx.y * rows_per_blk;
  const long long m1 = min(M, m0 + rows_per_blk);
  float acc = 0.f;
  for (long long m = m0 + wave_id(); m < m1; m += nw)
    acc += to_f(dy[m * N + col]);
  atomicAdd(&out[col], acc);
}

// ---------------- batched 2-D transpose: [B, M, N] -> [B, N, M] ----------
template <typename T>
__global__ void transpose_kernel(const T* __restrict__ in, T* __restrict__ out,
                                 int M, int N) {
  constexpr int TILE = 64;
  __shared__ T tile[TILE][TILE + 2];
  const long long b = blockIdx.z;
  const T* src = in + b * (long long)M * N;
  T* dst = out + b * (long long)M * N;
  const int n0 = blockIdx.x * TILE;
  const int m0 = blockIdx.y * TILE;
  for (int idx = threadIdx.x; idx < TILE * TILE; idx += blockDim.x) {
    int r = idx / TILE, c = idx % TILE;
    int m = m0 + r, n = n0 + c;
    tile[r][c] = (m < M && n < N) ? src[(long long)m * N + n] : to_t<T>(0.f);
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < TILE * TILE; idx += blockDim.x) {
    int r = idx / TILE, c = idx % TILE;  // output-local (row=n, col=m)
    int n = n0 + r, m = m0 + c;
    if (n < N && m < M) dst[(long long)n * M + m] = tile[c][r];
  }
}

// ---------------- global average pool NHWC: [N,H,W,C] -> [N,C] -----------
template <typename T>
__global__ void avgpool_global_kernel(const T* __restrict__ x,
                                      T* __restrict__ y, long long HW, int C) {
  __shared__ float partial[4][64];
  const int c = blockIdx.x * kWave + lane_id();
  const int n = blockIdx.y;
  const int w = wave_id();
  const int nw = blockDim.x / kWave;
  const T* base = x + (long long)n * HW * C;
  float acc = 0.f;
  if (c < C)
    for (long long i = w; i < HW; i += nw) acc += to_f(base[i * C + c]);
  partial[w][lane_id()] = acc;
  __syncthreads();
  if (w == 0 && c < C) {
    float total = 0.f;
    for (int i = 0; i < nw; ++i) total += partial[i][lane_id()];
    y[(long long)n * C + c] = to_t<T>(total / (float)HW);
  }
}

template <typename T>
__global__ void avgpool_global_bwd_kernel(const T* __restrict__ dy,
                                          T* __restrict__ dx, long long HW,
                                          int C) {
  const int n = blockIdx.y;
  const float inv = 1.f / (float)HW;
  const T* g = dy + (long long)n * C;
  T* base = dx + (long long)n * HW * C;
  const long long total = HW * C;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    int c = (int)(i % C);
    base[i] = to_t<T>(to_f(g[c]) * inv);
  }
}

// ---------------- max pool NHWC (kernel k, stride s, pad p) --------------
template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int* __restrict__ idx, int N, int H, int W,
                                   int C, int HO, int WO, int K, int S, int P) {
  const long long total = (long long)N * HO * WO * C;
  GRID_STRIDE(i, total) {
    int c = (int)(i % C);
    long long t = i / C;
    int wo = (int)(t % WO);
    t /= WO;
    int ho = (int)(t % HO);
    int n = (int)(t / HO);
    float best = -INFINITY;
    int best_idx = 0;
    for (int r = 0; r < K; ++r) {
      int hi = ho * S - P + r;
      if (hi < 0 || hi >= H) continue;
      for (int s = 0; s < K; ++s) {
        int wi = wo * S - P + s;
        if (wi < 0 || wi >= W) continue;
        float v = to_f(x[(((long long)n * H + hi) * W + wi) * C + c]);
        if (v > best) {
          best = v;
          best_idx = hi * W + wi;
        }
      }
    }
    y[i] = to_t<T>(best);
    idx[i] = best_idx;
  }
}

template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ dy,
                                   const int* __restrict__ idx,
                                   float* __restrict__ dx_f32, int N, int H,
                                   int W, int C, int HO, int WO) {
  const long long total = (long long)N * HO * WO * C;
  GRID_STRIDE(i, total) {
    int c = (int)(i % C);
    long long t = i / C;
    t /= WO;
    t /= HO;  // n
    int n = (int)t;
    long long off = ((long long)n * H * W + idx[i]) * C + c;
    atomicAdd(&dx_f32[off], to_f(dy[i]));
  }
}

template <typename T>
__global__ void cast_from_f32_kernel(const float* __restrict__ in,
                                     T* __restrict__ out, long long n) {
  GRID_STRIDE(i, n) out[i] = to_t<T>(in[i]);
}

// -------- conv dgrad helpers --------
// weight rotation: w[K,R,S,C] -> wr[C,R,S,K], wr[c, r, s, k] = w[k, R-1-r, S-1-s, c]
template <typename T>
__global__ void weight_rot_kernel(const T* __restrict__ w, T* __restrict__ wr,
                                  int K, int R, int S, int C) {
  const long long total = (long long)K * R * S * C;
  GRID_STRIDE(i, total) {
    int k = (int)(i % K);
    long long t = i / K;
    int s = (int)(t % S);
    t /= S;
    int r = (int)(t % R);
    int c = (int)(t / R);
    wr[i] = w[((((long long)k * R + (R - 1 - r)) * S + (S - 1 - s)) * C + c)];
  }
}

// zero-stuff dy for strided dgrad:
// [N,HO,WO,K] -> [N,(HO-1)*s+1+opad_h,(WO-1)*s+1+opad_w,K]
// (opad covers inputs the strided forward never reached when
//  (H+2p-R) % stride != 0 — transposed-conv output_padding semantics)
template <typename T>
__global__ void zero_stuff_kernel(const T* __restrict__ dy, T* __restrict__ out,
                                  int N, int HO, int WO, int K, int S, int HS,
                                  int WS) {
  const long long total = (long long)N * HS * WS * K;
  GRID_STRIDE(i, total) {
    int k = (int)(i % K);
    long long t = i / K;
    int ws = (int)(t % WS);
    t /= WS;
    int hs = (int)(t % HS);
    int n = (int)(t / HS);
    T v = to_t<T>(0.f);
    if (hs % S == 0 && ws % S == 0 && hs / S < HO && ws / S < WO)
      v = dy[(((long long)n * HO + hs / S) * WO + ws / S) * K + k];
    out[i] = v;
  }
}

// -------- im2col (generic conv fallback: non-pow2 C, e.g. the 3-ch stem) --
// x[N,H,W,C] -> cols[M, Kpad] where M = N*HO*WO, K = R*S*C, Kpad >= K
// (zero tail so the GEMM kernels' K%8 requirement holds).
template <typename T>
__global__ void im2col_kernel(const T* __restrict__ x, T* __restrict__ cols,
                              int N, int H, int W, int C, int R, int S, int st,
                              int pad, int HO, int WO, int Kpad) {
  const int K = R * S * C;
  const long long M = (long long)N * HO * WO;
  const long long total = M * Kpad;
  GRID_STRIDE(i, total) {
    const long long m = i / Kpad;
    const int k = (int)(i % Kpad);
    T v = to_t<T>(0.f);
    if (k < K) {
      const int c = k % C;
      const int rs = k / C;
      const int r = rs / S, s = rs % S;
      long long t = m;
      const int wo = (int)(t % WO);
      t /= WO;
      const int ho = (int)(t % HO);
      const int n = (int)(t / HO);
      const int hi = ho * st - pad + r;
      const int wi = wo * st - pad + s;
      if (hi >= 0 && hi < H && wi >= 0 && wi < W)
        v = x[(((long long)n * H + hi) * W + wi) * C + c];
    }
    cols[i] = v;
  }
}

}  // namespace ew

// ======================= host launchers ==================================

#define CHECK_GPU(x) \
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous GPU")

namespace {
template <typename T, typename Op>
void launch_unary(const torch::Tensor& in, torch::Tensor& out, Op op) {
  long long n = in.numel();
  int blocks = grid_1d(n / VecIO<T>::kPerLane + 1, 256);
  hipLaunchKernelGGL((ew::unary_kernel<T, Op>), dim3(blocks), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(),
                     reinterpret_cast<const T*>(in.data_ptr()),
                     reinterpret_cast<T*>(out.data_ptr()), n, op);
}

template <typename T, typename Op>
void launch_binary(const torch::Tensor& a, const torch::Tensor& b,
                   torch::Tensor& out, Op op) {
  long long n = a.numel();
  int blocks = grid_1d(n / VecIO<T>::kPerLane + 1, 256);
  hipLaunchKernelGGL((ew::binary_kernel<T, Op>), dim3(blocks), dim3(256), 0,
                     c10::hip::getCurrentHIPStream(),
                     reinterpret_cast<const T*>(a.data_ptr()),
                     reinterpret_cast<const T*>(b.data_ptr()),
                     reinterpret_cast<T*>(out.data_ptr()), n, op);
}
}  // namespace

torch::Tensor relu_fwd(torch::Tensor x) {
  CHECK_GPU(x);
  auto y = torch::empty_like(x);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "relu_fwd", [&] {
    launch_unary<scalar_t>(x, y, ew::ReluOp{});
  });
  return y;
}

torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y) {
  CHECK_GPU(dy);
  CHECK_GPU(y);
  auto dx = torch::empty_like(dy);
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "relu_bwd", [&] {
    launch_binary<scalar_t>(dy, y, dx, ew::ReluBwdOp{});
  });
  return dx;
}

// a += b, in place (residual skip-grad accumulation in the fused blocks —
// one read fewer than the eager out-of-place a + b the round-1 path used,
// and the whole r18 backward's kernel time stays in-tree)
torch::Tensor add_(torch::Tensor a, torch::Tensor b) {
  CHECK_GPU(a);
  CHECK_GPU(b);
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && a.numel() == b.numel(),
              "add_: contiguous same-numel tensors only");
  TORCH_CHECK(a.scalar_type() == b.scalar_type(), "add_: dtype mismatch");
  DDP_DISPATCH_FLOAT(a.scalar_type(), "add_", [&] {
    launch_binary<scalar_t>(a, b, a, ew::AddOp{});
  });
  return a;
}

torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b) {
  CHECK_GPU(a);
  CHECK_GPU(b);
  auto y = torch::empty_like(a);
  DDP_DISPATCH_FLOAT(a.scalar_type(), "add_relu", [&] {
    launch_binary<scalar_t>(a, b, y, ew::AddReluOp{});
  });
  return y;
}

torch::Tensor gelu_fwd(torch::Tensor x) {
  CHECK_GPU(x);
  auto y = torch::empty_like(x);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "gelu_fwd", [&] {
    launch_unary<scalar_t>(x, y, ew::GeluOp{});
  });
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x) {
  CHECK_GPU(dy);
  CHECK_GPU(x);
  auto dx = torch::empty_like(dy);
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "gelu_bwd", [&] {
    launch_binary<scalar_t>(dy, x, dx, ew::GeluBwdOp{});
  });
  return dx;
}

torch::Tensor col_sum(torch::Tensor dy) {
  CHECK_GPU(dy);
  TORCH_CHECK(dy.dim() == 2, "col_sum expects 2-D");
  long long M = dy.size(0), N = dy.size(1);
  auto out = torch::zeros({N}, dy.options().dtype(torch::kFloat32));
  // measured: the 8-col vectorized variant LOST 2x (fewer blocks for a
  // latency-bound row loop + 8x the atomic tail) — scalar lanes win here
  dim3 grid((N + kWave - 1) / kWave,
            (unsigned)std::min<long long>(64, (M + 255) / 256));
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "col_sum", [&] {
    hipLaunchKernelGGL((ew::col_sum_kernel<scalar_t>), grid, dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       out.data_ptr<float>(), M, N);
  });
  return out;
}

torch::Tensor transpose2d(torch::Tensor x) {
  CHECK_GPU(x);
  TORCH_CHECK(x.dim() == 2 || x.dim() == 3, "transpose2d expects 2-D/3-D");
  bool batched = x.dim() == 3;
  long long B = batched ? x.size(0) : 1;
  int M = (int)x.size(batched ? 1 : 0), N = (int)x.size(batched ? 2 : 1);
  auto out = batched ? torch::empty({B, N, M}, x.options())
                     : torch::empty({N, M}, x.options());
  dim3 grid((N + 63) / 64, (M + 63) / 64, (unsigned)B);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "transpose2d", [&] {
    hipLaunchKernelGGL((ew::transpose_kernel<scalar_t>), grid, dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(out.data_ptr()), M, N);
  });
  return out;
}

torch::Tensor avgpool_global(torch::Tensor x) {
  CHECK_GPU(x);
  TORCH_CHECK(x.dim() == 4, "avgpool_global expects NHWC");
  int N = (int)x.size(0), C = (int)x.size(3);
  long long HW = (long long)x.size(1) * x.size(2);
  auto y = torch::empty({N, C}, x.options());
  dim3 grid((C + kWave - 1) / kWave, N);
  DDP_DISPATCH_FLOAT(x.scalar_type(), "avgpool_global", [&] {
    hipLaunchKernelGGL((ew::avgpool_global_kernel<scalar_t>), grid, dim3(256),
                       0, c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()), HW, C);
  });
  return y;
}

torch::Tensor avgpool_global_bwd(torch::Tensor dy, int64_t H, int64_t W) {
  CHECK_GPU(dy);
  int N = (int)dy.size(0), C = (int)dy.size(1);
  long long HW = H * W;
  auto dx = torch::empty({(long long)N, H, W, (long long)C}, dy.options());
  dim3 grid(grid_1d(HW * C, 256, 256), N);
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "avgpool_global_bwd", [&] {
    hipLaunchKernelGGL((ew::avgpool_global_bwd_kernel<scalar_t>), grid,
                       dim3(256), 0, c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), HW, C);
  });
  return dx;
}

std::vector<torch::Tensor> maxpool2d_fwd(torch::Tensor x, int64_t k, int64_t s,
                                         int64_t p) {
  CHECK_GPU(x);
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      C = (int)x.size(3);
  int HO = (H + 2 * (int)p - (int)k) / (int)s + 1;
  int WO = (W + 2 * (int)p - (int)k) / (int)s + 1;
  auto y = torch::empty({N, HO, WO, C}, x.options());
  auto idx = torch::empty({N, HO, WO, C}, x.options().dtype(torch::kInt32));
  long long total = (long long)N * HO * WO * C;
  DDP_DISPATCH_FLOAT(x.scalar_type(), "maxpool_fwd", [&] {
    hipLaunchKernelGGL((ew::maxpool_fwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(y.data_ptr()),
                       idx.data_ptr<int>(), N, H, W, C, HO, WO, (int)k, (int)s,
                       (int)p);
  });
  return {y, idx};
}

torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                            int64_t W) {
  CHECK_GPU(dy);
  CHECK_GPU(idx);
  int N = (int)dy.size(0), HO = (int)dy.size(1), WO = (int)dy.size(2),
      C = (int)dy.size(3);
  auto dx32 = torch::zeros({(long long)N, H, W, (long long)C},
                           dy.options().dtype(torch::kFloat32));
  long long total = (long long)N * HO * WO * C;
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "maxpool_bwd", [&] {
    hipLaunchKernelGGL((ew::maxpool_bwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       idx.data_ptr<int>(), dx32.data_ptr<float>(), N, (int)H,
                       (int)W, C, HO, WO);
  });
  if (dy.scalar_type() == torch::kFloat32) return dx32;
  auto dx = torch::empty_like(dx32, dx32.options().dtype(dy.scalar_type()));
  long long n = dx32.numel();
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "cast", [&] {
    hipLaunchKernelGGL((ew::cast_from_f32_kernel<scalar_t>),
                       dim3(grid_1d(n, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(), dx32.data_ptr<float>(),
                       reinterpret_cast<scalar_t*>(dx.data_ptr()), n);
  });
  return dx;
}

torch::Tensor weight_rot(torch::Tensor w) {
  CHECK_GPU(w);
  int K = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2),
      C = (int)w.size(3);
  auto wr = torch::empty({C, R, S, K}, w.options());
  long long total = (long long)K * R * S * C;
  DDP_DISPATCH_FLOAT(w.scalar_type(), "weight_rot", [&] {
    hipLaunchKernelGGL((ew::weight_rot_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(w.data_ptr()),
                       reinterpret_cast<scalar_t*>(wr.data_ptr()), K, R, S, C);
  });
  return wr;
}

torch::Tensor im2col(torch::Tensor x, int64_t R, int64_t S, int64_t stride,
                     int64_t pad, int64_t Kpad) {
  CHECK_GPU(x);
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      C = (int)x.size(3);
  int HO = (H + 2 * (int)pad - (int)R) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - (int)S) / (int)stride + 1;
  long long M = (long long)N * HO * WO;
  auto cols = torch::empty({M, Kpad}, x.options());
  long long total = M * Kpad;
  DDP_DISPATCH_FLOAT(x.scalar_type(), "im2col", [&] {
    hipLaunchKernelGGL((ew::im2col_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(x.data_ptr()),
                       reinterpret_cast<scalar_t*>(cols.data_ptr()), N, H, W,
                       C, (int)R, (int)S, (int)stride, (int)pad, HO, WO,
                       (int)Kpad);
  });
  return cols;
}

torch::Tensor zero_stuff(torch::Tensor dy, int64_t s, int64_t opad_h,
                         int64_t opad_w) {
  CHECK_GPU(dy);
  int N = (int)dy.size(0), HO = (int)dy.size(1), WO = (int)dy.size(2),
      K = (int)dy.size(3);
  int HS = (HO - 1) * (int)s + 1 + (int)opad_h;
  int WS = (WO - 1) * (int)s + 1 + (int)opad_w;
  auto out = torch::empty({N, HS, WS, K}, dy.options());
  long long total = (long long)N * HS * WS * K;
  DDP_DISPATCH_FLOAT(dy.scalar_type(), "zero_stuff", [&] {
    hipLaunchKernelGGL((ew::zero_stuff_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0,
                       c10::hip::getCurrentHIPStream(),
                       reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                       reinterpret_cast<scalar_t*>(out.data_ptr()), N, HO, WO,
                       K, (int)s, HS, WS);
  });
  return out;
}
