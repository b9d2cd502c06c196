// Direct 3x3 stride-1 convolution from an LDS spatial halo (gfx950).
//
// The implicit-GEMM NT kernel re-gathers each x element up to 9 times
// through global_load_lds (one per (r,s) tap of the im2col K dimension) —
// measured ~45% of the ResNet-18 step.  This kernel stages the x halo for
// a 128-output-row block ONCE ((G+2) image rows x (W+2) columns x C, zero
// padded) and the MFMA A-fragments read the taps straight out of it with
// computed LDS addresses; only the weight tile streams per k-step.
//
// Gates (host): R=S=3, stride=1, pad=1, W==WO pow2, H==HO pow2, pow2 C,
// (HO*WO) % 128 == 0 (a block never spans two images), halo <= 48 KiB —
// i.e. the spatially-large early layers, which dominate the NT pool.
//
// Bank behavior: consecutive m rows advance the halo address by C*2 bytes
// (a multiple of 64 banks for C>=32), so the 8-halfword c-chunks are
// XOR-swizzled by the halo position (chunk' = chunk ^ (pos & (chunks-1)))
// in both the staging writes and the fragment reads.

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace ch {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

template <typename T16>
struct MM;
template <>
struct MM<bf16> {
  using vec = bf16x8;
  DEV_INLINE static f32x16 mma32(vec a, vec b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct MM<_Float16> {
  using vec = f16x8;
  DEV_INLINE static f32x16 mma32(vec a, vec b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
  }
};

constexpr int BM = 128;              // output rows per block
constexpr int THREADS = 256;
constexpr int HALO_MAX = 24576;      // halfwords (48 KiB)

// weight tile: [BNT rows][32 k] like the NT kernel's B tile, 3 buffers.
// BMT: output rows per block (256 for narrow-N layers: more MFMA work per
// addressed fragment in this issue-bound regime).
template <typename T16, int BNT, bool EXTRAS, int BMT = BM>
__global__ __launch_bounds__(THREADS) void conv_halo_kernel(
    const T16* __restrict__ x, const T16* __restrict__ w,
    T16* __restrict__ y, const T16* __restrict__ zpad, int M, int N, int K,
    int W, int C, int lgW, int lgC, int lgHO, int HOim,
    float* __restrict__ stats_ws, int ws_nblocks) {
  constexpr int BK = 32;
  constexpr int BTILE = BNT * BK;  // halfwords per weight buffer
  __shared__ __attribute__((aligned(16))) T16 lds[HALO_MAX + 3 * BTILE];
  T16* halo = lds;
  T16* bt = lds + HALO_MAX;

  const int flat_id = (int)(blockIdx.y * gridDim.x + blockIdx.x);
  const int m0 = (int)blockIdx.y * BMT;
  const int n0 = (int)blockIdx.x * BNT;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int G = BMT >> lgW;           // output rows in this block
  const int Wp2 = W + 2;
  const int HROWS = G + 2;
  const int halo_hw = HROWS * Wp2 * C;  // halfwords used
  const int cch = C >> 3;               // 8-halfword chunks per position
  // image-local coordinates of the block (never spans two images: gated)
  const int grow = m0 >> lgW;                    // global row index
  const int n_img = grow >> lgHO;
  const int ho0 = grow & ((1 << lgHO) - 1);

  // ---- stage the halo once: per-lane gather, zero-padded edges ----
  {
    const int units = (halo_hw + 511) / 512;  // 1 KiB (512 hw) per glds
    for (int u = wave; u < units; u += 4) {
      const int idx8 = u * 64 + lane;  // 8-halfword piece index
      const long long e0 = (long long)idx8 * 8;
      const T16* src = zpad;
      T16* dst = halo + e0;
      if (e0 < halo_hw) {
        const int pos = (int)(e0 >> lgC);        // (hrow, wcol) position
        const int cslot = (int)(e0 & (C - 1)) >> 3;
        const int hrow = pos / Wp2;              // small; compiler magic ok
        const int wcol = pos - hrow * Wp2;
        const int hi = ho0 + hrow - 1;
        const int wi = wcol - 1;
        // swizzled source chunk for this LDS slot
        const int csrc = (cslot ^ (pos & (cch - 1))) << 3;
        if (hi >= 0 && hi < HOim && wi >= 0 && wi < W)
          src = x + ((((long long)n_img * HOim + hi) * W + wi) << lgC) + csrc;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  }

  // ---- weight staging (NT-style rows of [N, K], kswz chunk swizzle) ----
  const int rl = (wave * 2) * 16 + (lane >> 2);  // two 16-row chunks / wave
  const int kp = lane & 3;
  auto stage_w = [&](int buf, int kt) {
    const int k_base = kt * BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = rl + i * 16;
      if (row >= BNT) break;
      const int swz = kp ^ (((row) >> 2) & 3);
      const int gk = k_base + swz * 8;
      const int gn = n0 + row;
      const T16* src = (gn < N && gk < K) ? w + (long long)gn * K + gk : zpad;
      T16* dst = bt + buf * BTILE + (wave * 2 + i) * 512;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  const int KT = (K + BK - 1) / BK;
  constexpr int MI32 = BMT / 128;  // 32-row tiles per wave (wave dim = 2)
  const int wm = (wave >> 1) * (BMT / 2), wn = (wave & 1) * (BNT / 2);
  const int r32 = lane & 31;
  const int ks = lane >> 5;
  constexpr int NI32 = BNT / 64;
  f32x16 acc[2 * MI32][NI32] = {};

  // per-lane A-row geometry (fixed over the k loop)
  int seg_[2 * MI32], wo_[2 * MI32];
#pragma unroll
  for (int mi = 0; mi < 2 * MI32; ++mi) {
    const int m = wm + mi * 32 + r32;
    seg_[mi] = m >> lgW;
    wo_[mi] = m & (W - 1);
  }

  stage_w(0, 0);
  if (KT > 1) stage_w(1, 1);
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  if (KT > 1)
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();  // halo + first weight tile ready

  for (int kt = 0; kt < KT; ++kt) {
    const int buf = kt % 3;
    if (kt + 2 < KT) stage_w((kt + 2) % 3, kt + 2);
    const T16* bbase = bt + buf * BTILE;
    using vec16 = typename MM<T16>::vec;
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      // k-window of this MFMA: 16 halfwords at kt*32 + kh*16; the lane
      // reads 8 at +ks*8
      const int kw = kt * BK + kh * 16 + ks * 8;
      const int rs = kw >> lgC;
      const int r = (rs * 21846) >> 16;  // /3 for rs < 9
      const int s = rs - r * 3;
      const int c0 = kw & (C - 1);
      vec16 af[2 * MI32];
#pragma unroll
      for (int mi = 0; mi < 2 * MI32; ++mi) {
        const int pos = (seg_[mi] + r) * Wp2 + wo_[mi] + s;
        const int cslot = (c0 >> 3) ^ (pos & (cch - 1));
        af[mi] = *reinterpret_cast<const vec16*>(
            halo + ((long long)pos << lgC) + (cslot << 3));
      }
#pragma unroll
      for (int ni = 0; ni < NI32; ++ni) {
        const int rb = wn + ni * 32 + r32;
        const int chunk = kh * 2 + ks;
        const int swz = chunk ^ ((rb >> 2) & 3);
        vec16 bf = *reinterpret_cast<const vec16*>(
            bbase + rb * 32 + swz * 8);
#pragma unroll
        for (int mi = 0; mi < 2 * MI32; ++mi)
          acc[mi][ni] = MM<T16>::mma32(af[mi], bf, acc[mi][ni]);
      }
    }
    if (kt + 1 < KT) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (kt + 2 < KT)
        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
  }

  // ---- epilogue: store y (+ optional BN stat partials, NT-compatible) ----
  float col_sum[NI32] = {}, col_sq[NI32] = {};
#pragma unroll
  for (int ni = 0; ni < NI32; ++ni) {
    const int col = n0 + wn + ni * 32 + r32;
    if (col >= N) continue;
#pragma unroll
    for (int mi = 0; mi < 2 * MI32; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row =
            m0 + wm + mi * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
        if (row >= M) continue;
        const float v = acc[mi][ni][reg];
        y[(long long)row * N + col] = to_t<T16>(v);
        if (EXTRAS && stats_ws) {
          col_sum[ni] += v;
          col_sq[ni] += v * v;
        }
      }
    }
  }
  if (EXTRAS && stats_ws) {
    const long long vrow = (long long)flat_id * 2 + (wave >> 1);
    const long long wsrow = vrow % ws_nblocks;
    // If ANY block of this launch wraps, all blocks accumulate atomically
    // into the pre-zeroed workspace (plain stores by row owners race with
    // wrapped blocks' atomics — block execution order is unordered).
    const bool wrap =
        2ll * gridDim.x * gridDim.y > (long long)ws_nblocks || gridDim.z > 1;
#pragma unroll
    for (int ni = 0; ni < NI32; ++ni) {
      float sv = col_sum[ni], qv = col_sq[ni];
      sv += __shfl_xor(sv, 32);
      qv += __shfl_xor(qv, 32);
      const int col = n0 + wn + ni * 32 + r32;
      if (ks == 0 && col < N) {
        if (wrap) {
          atomicAdd(&stats_ws[wsrow * N + col], sv);
          atomicAdd(&stats_ws[((long long)ws_nblocks + wsrow) * N + col], qv);
        } else {
          stats_ws[wsrow * N + col] = sv;
          stats_ws[((long long)ws_nblocks + wsrow) * N + col] = qv;
        }
      }
    }
  }
}

}  // namespace ch

// host entry: returns true (and writes y) when the halo fast path applies.
// Caller falls back to the generic NT kernel otherwise.
bool conv2d_fwd_halo(torch::Tensor x, torch::Tensor w, torch::Tensor y,
                     torch::Tensor zp, int64_t stride, int64_t pad,
                     float* stats_ws, int ws_nblocks) {
  const int N_ = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            C = (int)x.size(3);
  const int Kout = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  if (!(R == 3 && S == 3 && stride == 1 && pad == 1)) return false;
  auto lg = [](int v) {
    int l = 0;
    while ((1 << l) < v) ++l;
    return ((1 << l) == v) ? l : -1;
  };
  const int lgW = lg(W), lgH = lg(H), lgC = lg(C);
  if (lgW < 2 || lgH < 0 || lgC < 3) return false;
  if ((long long)H * W % ch::BM != 0) return false;  // block spans one image
  const int M = N_ * H * W, K = 9 * C;
  const int BNT = Kout <= 64 ? 64 : 128;
  // narrow-N layers take 256 output rows per block: 2x the MFMA work per
  // addressed fragment (this regime is instruction-issue bound)
  // (the template BMT must match: narrow-N is compiled at 256 only)
  const int BMT = BNT == 64 ? 256 : ch::BM;
  const long long halo_hw = (long long)((BMT >> lgW) + 2) * (W + 2) * C;
  // +512: the last glds unit's lane-linear writes may run past the
  // used region (their sources are zpad) — keep them inside the carve
  if (halo_hw + 512 > ch::HALO_MAX) return false;
  if ((long long)H * W % BMT != 0) return false;
  dim3 grid((Kout + BNT - 1) / BNT, (M + BMT - 1) / BMT, 1);
  auto stream = c10::hip::getCurrentHIPStream();
  const bool ex = stats_ws != nullptr;
  auto do_launch = [&](auto tag, auto bntc, auto exc) {
    using scalar_t = decltype(tag);
    constexpr int BMTv = decltype(bntc)::value == 64 ? 256 : ch::BM;
    hipLaunchKernelGGL(
        (ch::conv_halo_kernel<scalar_t, decltype(bntc)::value,
                              decltype(exc)::value, BMTv>),
        grid, dim3(ch::THREADS), 0, stream,
        reinterpret_cast<const scalar_t*>(x.data_ptr()),
        reinterpret_cast<const scalar_t*>(w.data_ptr()),
        reinterpret_cast<scalar_t*>(y.data_ptr()),
        reinterpret_cast<const scalar_t*>(zp.data_ptr()), M, Kout, K, W, C,
        lgW, lgC, lgH, H, stats_ws, ws_nblocks);
  };
  auto pick = [&](auto tag) {
    if (BNT == 64) {
      if (ex)
        do_launch(tag, std::integral_constant<int, 64>{}, std::true_type{});
      else
        do_launch(tag, std::integral_constant<int, 64>{}, std::false_type{});
    } else {
      if (ex)
        do_launch(tag, std::integral_constant<int, 128>{}, std::true_type{});
      else
        do_launch(tag, std::integral_constant<int, 128>{}, std::false_type{});
    }
  };
  if (x.scalar_type() == torch::kBFloat16)
    pick(bf16{});
  else if (x.scalar_type() == torch::kHalf)
    pick(_Float16{});
  else
    return false;
  return true;
}
