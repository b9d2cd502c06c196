#include "hip/hip_runtime.h"
// fp32 MFMA GEMM family for gfx950 — exact f32 at the 157 TF f32 vector rate
// via v_mfma_f32_16x16x4_f32 (no TF32/xf32 on CDNA4; this IS bitwise f32).
//
// Mirrors gemm_bf16.hip's NT/TN/conv entry points for fp32 tensors (the
// reference template trains fp32 by default — ddp.py has no autocast — and
// the HIP-vs-CPU numerics tests run fp32).  Structure: 128x128 tile, BK=32,
// register staging into +4-padded LDS (conflict-free b32 fragment reads),
// single buffer, one stage/compute barrier pair per K-step.  Correctness
// over peak: the bf16 path is the throughput path.

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace g32 {

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int ROW = BK + 4;  // padded LDS row (elements)
constexpr int THREADS = 256;

struct ConvMeta {
  int H, W, C_log2, S, R, stride, pad;
  int HO, WO;
};

enum { MODE_PLAIN = 0, MODE_CONV = 1 };

template <int MODE, bool RELU, bool HAS_BIAS>
__global__ __launch_bounds__(THREADS) void gemm_nt_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, const float* __restrict__ bias, int M, int N, int K,
    long long strideA, long long strideB, long long strideC, ConvMeta cm) {
  __shared__ __attribute__((aligned(16))) float lds[2 * BM * ROW];  // A image then B image
  float* ldsA = lds;
  float* ldsB = lds + BM * ROW;

  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;
  const long long batch = blockIdx.z;
  A += batch * strideA;
  B += batch * strideB;
  C += batch * strideC;

  // staging geometry: lane -> row = tid/8 (+32 per pass), c0 = (tid%8)*4
  const int srow = threadIdx.x >> 3;
  const int sc0 = (threadIdx.x & 7) * 4;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;
  const int fr = lane & 15;
  const int fs = lane >> 4;

  f32x4 acc[4][4] = {};
  const int KT = (K + BK - 1) / BK;

  for (int kt = 0; kt < KT; ++kt) {
    const int kb = kt * BK;
    // ---- stage A ----
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rl = srow + p * 32;
      const int gm = m0 + rl;
      const int gk = kb + sc0;
      f32x4 v = {};
      if (gk < K) {
        if (MODE == MODE_CONV) {
          if (gm < M) {
            int t = gm;
            const int wo = t % cm.WO;
            t /= cm.WO;
            const int ho = t % cm.HO;
            const int n = t / cm.HO;
            const int c0 = gk & ((1 << cm.C_log2) - 1);
            const int rs = gk >> cm.C_log2;
            const int rr = rs / cm.S, ss = rs % cm.S;
            const int hi = ho * cm.stride - cm.pad + rr;
            const int wi = wo * cm.stride - cm.pad + ss;
            if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W)
              v = *reinterpret_cast<const f32x4*>(
                  A + (((long long)n * cm.H + hi) * cm.W + wi) *
                          (1LL << cm.C_log2) +
                  c0);
          }
        } else {
          if (gm < M) v = *reinterpret_cast<const f32x4*>(A + (long long)gm * K + gk);
        }
      }
      *reinterpret_cast<f32x4*>(&ldsA[rl * ROW + sc0]) = v;
      // ---- stage B ----
      const int gn = n0 + rl;
      f32x4 vb = {};
      if (gn < N && gk < K)
        vb = *reinterpret_cast<const f32x4*>(B + (long long)gn * K + gk);
      *reinterpret_cast<f32x4*>(&ldsB[rl * ROW + sc0]) = vb;
    }
    __syncthreads();
    // ---- MFMA ----
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      float af[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        af[i] = ldsA[(wm + i * 16 + fr) * ROW + kk * 4 + fs];
        bfr[i] = ldsB[(wn + i * 16 + fr) * ROW + kk * 4 + fs];
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int col = n0 + wn + ni * 16 + fr;
    if (col >= N) continue;
    float bv = HAS_BIAS ? bias[col] : 0.f;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm + mi * 16 + fs * 4 + r;
        if (row >= M) continue;
        float v = acc[mi][ni][r] + bv;
        if (RELU) v = fmaxf(v, 0.f);
        C[(long long)row * N + col] = v;
      }
    }
  }
}

// TN: C[I,J] += sum_m A[m,I] B[m,J]  (f32, atomics over split-M)
template <int MODE>
__global__ __launch_bounds__(THREADS) void gemm_tn_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int Mtot, int I, int J, int r, int s, long long ldc,
    long long coff, ConvMeta cm) {
  constexpr int BI = 64, BJ = 64, BMC = 32;
  constexpr int TROW = BMC + 4;
  __shared__ __attribute__((aligned(16))) float lds[2 * BI * TROW];
  float* ldsA = lds;
  float* ldsB = lds + BI * TROW;

  const int i0 = blockIdx.y * BI;
  const int j0 = blockIdx.x * BJ;
  const int n_chunks = (Mtot + BMC - 1) / BMC;
  const int per_z = (n_chunks + gridDim.z - 1) / gridDim.z;
  const int ch0 = blockIdx.z * per_z;
  const int ch1 = min(n_chunks, ch0 + per_z);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;
  const int fr = lane & 15;
  const int fs = lane >> 4;

  // staging: chunk [32 m][64 cols] f32; lane -> m = tid>>4 (+16/pass), c0=(tid&15)*4
  const int sm_base = threadIdx.x >> 4;
  const int sc0 = (threadIdx.x & 15) * 4;

  f32x4 acc[2][2] = {};

  for (int ch = ch0; ch < ch1; ++ch) {
    const int mbase = ch * BMC;
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int sm = sm_base + p * 16;
      const int gm = mbase + sm;
      // A
      f32x4 va = {};
      if (gm < Mtot && i0 + sc0 < I) {
        const long long off = (long long)gm * I + i0 + sc0;
        if (off + 4 <= (long long)Mtot * I)
          va = *reinterpret_cast<const f32x4*>(A + off);
        else {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (off + j < (long long)Mtot * I) va[j] = A[off + j];
        }
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) ldsA[(sc0 + j) * TROW + sm] = va[j];
      // B
      f32x4 vb = {};
      if (MODE == MODE_CONV) {
        if (gm < Mtot) {
          int t = gm;
          const int wo = t % cm.WO;
          t /= cm.WO;
          const int ho = t % cm.HO;
          const int n = t / cm.HO;
          const int hi = ho * cm.stride - cm.pad + r;
          const int wi = wo * cm.stride - cm.pad + s;
          if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W)
            vb = *reinterpret_cast<const f32x4*>(
                B + (((long long)n * cm.H + hi) * cm.W + wi) *
                        (1LL << cm.C_log2) +
                j0 + sc0);
        }
      } else if (gm < Mtot && j0 + sc0 < J) {
        const long long off = (long long)gm * J + j0 + sc0;
        if (off + 4 <= (long long)Mtot * J)
          vb = *reinterpret_cast<const f32x4*>(B + off);
        else {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (off + j < (long long)Mtot * J) vb[j] = B[off + j];
        }
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) ldsB[(sc0 + j) * TROW + sm] = vb[j];
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BMC / 4; ++kk) {
      float af[2], bfr[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        af[i] = ldsA[(wm + i * 16 + fr) * TROW + kk * 4 + fs];
        bfr[i] = ldsB[(wn + i * 16 + fr) * TROW + kk * 4 + fs];
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int col = j0 + wn + ni * 16 + fr;
    if (col >= J) continue;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = i0 + wm + mi * 16 + fs * 4 + rr;
        if (row >= I) continue;
        atomicAdd(&C[coff + (long long)row * ldc + col], acc[mi][ni][rr]);
      }
  }
}

}  // namespace g32

// ======================= host-side ======================================

torch::Tensor bmm_nt_f32(torch::Tensor A, torch::Tensor B,
                         c10::optional<torch::Tensor> bias, bool relu) {
  const bool batched = A.dim() == 3;
  long long bsz = batched ? A.size(0) : 1;
  int M = (int)A.size(batched ? 1 : 0), K = (int)A.size(batched ? 2 : 1);
  int N = (int)B.size(batched ? 1 : 0);
  TORCH_CHECK(K % 4 == 0, "f32 GEMM needs K % 4 == 0 (host pads)");
  auto C = batched ? torch::empty({bsz, M, N}, A.options())
                   : torch::empty({M, N}, A.options());
  dim3 grid((N + g32::BN - 1) / g32::BN, (M + g32::BM - 1) / g32::BM,
            (unsigned)bsz);
  g32::ConvMeta cm{};
  auto stream = c10::hip::getCurrentHIPStream();
  const float* bias_p = bias.has_value() ? bias->data_ptr<float>() : nullptr;
  long long sA = batched ? (long long)M * K : 0;
  long long sB = batched ? (long long)N * K : 0;
  long long sC = batched ? (long long)M * N : 0;
#define LAUNCH_NT32(RELU, HB)                                          \
  hipLaunchKernelGGL((g32::gemm_nt_f32_kernel<g32::MODE_PLAIN, RELU, HB>), \
                     grid, dim3(g32::THREADS), 0, stream,              \
                     A.data_ptr<float>(), B.data_ptr<float>(),         \
                     C.data_ptr<float>(), bias_p, M, N, K, sA, sB, sC, cm)
  if (relu) {
    if (bias_p) LAUNCH_NT32(true, true);
    else LAUNCH_NT32(true, false);
  } else {
    if (bias_p) LAUNCH_NT32(false, true);
    else LAUNCH_NT32(false, false);
  }
#undef LAUNCH_NT32
  return C;
}

torch::Tensor conv2d_fwd_f32(torch::Tensor x, torch::Tensor w,
                             c10::optional<torch::Tensor> bias, int64_t stride,
                             int64_t pad, bool relu) {
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      Cin = (int)x.size(3);
  int Kout = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  int cl = 0;
  while ((1 << cl) < Cin) ++cl;
  TORCH_CHECK((1 << cl) == Cin && cl >= 2, "conv f32 needs pow2 C >= 4");
  int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - S) / (int)stride + 1;
  int M = N * HO * WO, K = R * S * Cin;
  auto y = torch::empty({N, HO, WO, Kout}, x.options());
  dim3 grid((Kout + g32::BN - 1) / g32::BN, (M + g32::BM - 1) / g32::BM, 1);
  g32::ConvMeta cm{H, W, cl, S, R, (int)stride, (int)pad, HO, WO};
  auto stream = c10::hip::getCurrentHIPStream();
  const float* bias_p = bias.has_value() ? bias->data_ptr<float>() : nullptr;
#define LAUNCH_CV32(RELU, HB)                                           \
  hipLaunchKernelGGL((g32::gemm_nt_f32_kernel<g32::MODE_CONV, RELU, HB>), \
                     grid, dim3(g32::THREADS), 0, stream,               \
                     x.data_ptr<float>(), w.data_ptr<float>(),          \
                     y.data_ptr<float>(), bias_p, M, Kout, K, 0, 0, 0, cm)
  if (relu) {
    if (bias_p) LAUNCH_CV32(true, true);
    else LAUNCH_CV32(true, false);
  } else {
    if (bias_p) LAUNCH_CV32(false, true);
    else LAUNCH_CV32(false, false);
  }
#undef LAUNCH_CV32
  return y;
}

torch::Tensor bmm_tn_f32(torch::Tensor A, torch::Tensor B) {
  const bool batched = A.dim() == 3;
  long long bsz = batched ? A.size(0) : 1;
  int M = (int)A.size(batched ? 1 : 0), I = (int)A.size(batched ? 2 : 1);
  int J = (int)B.size(batched ? 2 : 1);
  auto C = batched ? torch::zeros({bsz, I, J}, A.options())
                   : torch::zeros({I, J}, A.options());
  auto stream = c10::hip::getCurrentHIPStream();
  int n_chunks = (M + 31) / 32;
  int tiles = ((J + 63) / 64) * ((I + 63) / 64);
  int z = std::max(1, std::min(n_chunks, 2048 / std::max(1, tiles) + 1));
  g32::ConvMeta cm{};
  for (long long b = 0; b < bsz; ++b) {
    dim3 grid((J + 63) / 64, (I + 63) / 64, z);
    hipLaunchKernelGGL((g32::gemm_tn_f32_kernel<g32::MODE_PLAIN>), grid,
                       dim3(g32::THREADS), 0, stream,
                       A.data_ptr<float>() + b * (long long)M * I,
                       B.data_ptr<float>() + b * (long long)M * J,
                       C.data_ptr<float>() + b * (long long)I * J, M, I, J, 0,
                       0, (long long)J, 0, cm);
  }
  return C;
}

torch::Tensor conv2d_wgrad_f32(torch::Tensor dy, torch::Tensor x,
                               int64_t stride, int64_t pad, int64_t R,
                               int64_t S) {
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      Cin = (int)x.size(3);
  int HO = (int)dy.size(1), WO = (int)dy.size(2), Kout = (int)dy.size(3);
  int cl = 0;
  while ((1 << cl) < Cin) ++cl;
  TORCH_CHECK((1 << cl) == Cin && cl >= 2, "conv wgrad f32 needs pow2 C");
  int M = N * HO * WO;
  auto dw = torch::zeros({(long long)Kout, R, S, (long long)Cin}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  g32::ConvMeta cm{H, W, cl, (int)S, (int)R, (int)stride, (int)pad, HO, WO};
  int n_chunks = (M + 31) / 32;
  int tiles = ((Cin + 63) / 64) * ((Kout + 63) / 64);
  int z = std::max(1, std::min(n_chunks, 2048 / std::max(1, tiles) + 1));
  for (int r = 0; r < (int)R; ++r)
    for (int s = 0; s < (int)S; ++s) {
      dim3 grid((Cin + 63) / 64, (Kout + 63) / 64, z);
      hipLaunchKernelGGL((g32::gemm_tn_f32_kernel<g32::MODE_CONV>), grid,
                         dim3(g32::THREADS), 0, stream, dy.data_ptr<float>(),
                         x.data_ptr<float>(), dw.data_ptr<float>(), M, Kout,
                         Cin, r, s, (long long)R * S * Cin,
                         (long long)(r * (int)S + s) * Cin, cm);
    }
  return dw;
}
