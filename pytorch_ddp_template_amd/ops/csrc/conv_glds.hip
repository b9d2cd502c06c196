// Implicit-GEMM conv forward on the glds 256-tile structure (gfx950).
//
// y[M][Kout] = im2col(x)[M][RSC] @ w[Kout][RSC]^T  (NHWC, weight (K,R,S,C))
// — the weight tensor IS a contiguous [Kout][R*S*C] NT operand, and with
// C % 64 == 0 every 64-deep K-tile lives inside ONE tap (r,s), so the
// A-side im2col gather is a single per-row base + 16B chunk: exactly the
// glds staging pattern of gemm_plain.hip, with the gather folded into the
// per-lane SOURCE address (glds destinations stay lane-linear).
//
// Same structure as gemm_nt_plain256 (measured ~2x the older NT conv
// kernels' TF on plain shapes): BK=64, two LDS buffers, one barrier per
// K-tile with the next tile's glds issued before the MFMAs, source-side
// XOR k-slot swizzle, mfma_f32_16x16x32_bf16, XCD-bijective remap,
// LDS-staged coalesced epilogue.  Optional BN-stats epilogue (partial
// col sums/sumsq) matching the NT/halo workspace convention so the fused
// conv+BN path (bn_fwd_ws) consumes it unchanged.
//
// Eligibility (host): bf16, pow2 HO/WO, C % 64 == 0, M % BM == 0,
// Kout % BN == 0.  Tiles: BN=256/128 at BM=256, BN=64 at BM=512 — always
// 8 waves, per-wave 128 M-rows (MI=8), NI = BN/64 capped at 4.
// Covers every non-stem ResNet-18/CIFAR conv (fwd and, via the rotated-
// weight identity, stride-1 dgrad).  Reference scope: SURVEY §2b conv row.

#include <torch/extension.h>

#include "common.h"
#include "dispatch.h"

namespace cg {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BK = 64;
constexpr int THREADS = 512;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ int kswz(int row, int kbyte) {
  return kbyte ^ (((row >> 2) & 1) << 5);
}

struct Geom {
  int H, W, C, lgC, HO, WO, stride, pad, S, RS;  // S = kernel width
  // transposed-conv (stride-s dgrad) gather: the logical image is the
  // zero-STUFFED dy — coordinates valid only on the stride grid, source
  // indexed by the un-stuffed position.  lgSt=0 -> plain conv (H,W = real
  // source dims).
  int lgSt = 0;
};

template <int BM, int BN, bool STATS, int MINB = 1>
__global__ __launch_bounds__(THREADS, MINB) void conv_fwd_glds_kernel(
    const __hip_bfloat16* __restrict__ x, const __hip_bfloat16* __restrict__ w,
    __hip_bfloat16* __restrict__ y, const __hip_bfloat16* __restrict__ zpad,
    int M, int N /*Kout*/, int K /*RSC*/, Geom g,
    float* __restrict__ stats_ws, int ws_nblocks) {
  __shared__ __hip_bfloat16 smem[2 * (BM + BN) * BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  long long nwg = (long long)gridDim.x * gridDim.y;
  long long bid = (long long)blockIdx.y * gridDim.x + blockIdx.x;
  {
    const long long q = nwg >> 3, r = nwg & 7;
    const int xcd = (int)(bid & 7);
    const long long o = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + o;
  }
  const int n0 = (int)(bid % gridDim.x) * BN;
  const long long m0 = (bid / gridDim.x) * BM;
  const long long flat_id = bid;

  // ---- staging geometry (v1 pattern): per round 64 rows; wave w rows
  // [w*8, +8), lane: row += lane>>3, chunk = lane&7.
  const int srow = (tid >> 3) & 7;
  const int schunk = lane & 7;
  constexpr int RA = BM / 64;  // A rounds per K-tile
  constexpr int RB = BN / 64;  // B rounds

  // Per-row gather bases hoisted out of the K-loop: this thread stages the
  // same M-rows every tile.  hi/wi validity is per-tap, so keep the
  // unshifted coordinates.
  long long nbase[RA];
  int hv[RA], wv[RA];
#pragma unroll
  for (int r = 0; r < RA; ++r) {
    // hoisted out of the K-loop, so plain div/mod (non-pow2 spatial: the
    // ResNet-50 224-input pyramid is 56/28/14/7) costs nothing hot
    const long long m = m0 + r * 64 + wave * 8 + srow;
    const int wo = (int)(m % g.WO);
    const long long t = m / g.WO;
    const int ho = (int)(t % g.HO);
    const long long n = t / g.HO;
    hv[r] = ho * g.stride - g.pad;
    wv[r] = wo * g.stride - g.pad;
    nbase[r] = n * g.H * (long long)g.W;
  }
  const int smask = (1 << g.lgSt) - 1;

  auto stage = [&](int buf, int kt) {
    __hip_bfloat16* sa = smem + buf * ((BM + BN) * BK);
    __hip_bfloat16* sb = sa + BM * BK;
    const int tap = (kt << 6) >> g.lgC;
    const int c0 = (kt << 6) & (g.C - 1);
    const int rt = tap / g.S, st = tap - rt * g.S;
#pragma unroll
    for (int r = 0; r < RA; ++r) {
      const int row = r * 64 + wave * 8 + srow;
      const int kk = c0 + kswz(row, schunk * 16) / 2;
      const __hip_bfloat16* src = zpad;
      const int hs = hv[r] + rt, ws = wv[r] + st;
      if ((unsigned)(hs >> g.lgSt) < (unsigned)g.H &&
          (unsigned)(ws >> g.lgSt) < (unsigned)g.W &&
          ((hs | ws) & smask) == 0)
        src = x + ((nbase[r] + (long long)(hs >> g.lgSt) * g.W +
                    (ws >> g.lgSt))
                   << g.lgC) +
              kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)(
              sa + (r * 64 + wave * 8) * BK),
          16, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < RB; ++r) {
      const int row = r * 64 + wave * 8 + srow;
      const long long kk = (long long)kt * BK + kswz(row, schunk * 16) / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(
              w + (long long)(n0 + row) * K + kk),
          (__attribute__((address_space(3))) unsigned int*)(
              sb + (r * 64 + wave * 8) * BK),
          16, 0, 0);
    }
  };

  // wave grid: always 128 M-rows per wave (MI=8)
  constexpr int NWM = BM / 128;          // waves along M (2 or 4)
  constexpr int NWN = 8 / NWM;           // waves along N
  constexpr int NI = BN / (16 * NWN);    // 16-col frags per wave (4 or 2)
  const int wm = (wave / NWN) * 128;
  const int wn = (wave % NWN) * (NI * 16);
  const int frow = lane & 15;
  const int fkb = (lane >> 4) * 16;

  f32x4 acc[8][NI];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NI; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int NT = K / BK;
  stage(0, 0);
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < NT; ++kt) {
    if (kt + 1 < NT) stage(cur ^ 1, kt + 1);
    const __hip_bfloat16* sa = smem + cur * ((BM + BN) * BK);
    const __hip_bfloat16* sb = sa + BM * BK;
    bf16x8 bf[NI][2];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wn + ni * 16 + frow;
        bf[ni][ks] = *(const bf16x8*)((const char*)(sb + row * BK) +
                                      kswz(row, ks * 64 + fkb));
      }
#pragma unroll
    for (int mi = 0; mi < 8; ++mi) {
      bf16x8 af[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm + mi * 16 + frow;
        af[ks] = *(const bf16x8*)((const char*)(sa + row * BK) +
                                  kswz(row, ks * 64 + fkb));
      }
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        acc[mi][ni] = mfma16(af[0], bf[ni][0], acc[mi][ni]);
        acc[mi][ni] = mfma16(af[1], bf[ni][1], acc[mi][ni]);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- BN-stats partials (before the bf16 staging): per column, sum and
  // sum-of-squares over the block's BM rows.  C/D map: col = lane&15,
  // row-group = lane>>4 — reduce over the 4 row-groups via shfl.
  if (STATS && stats_ws) {
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      float sv = 0.f, qv = 0.f;
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const float v = acc[mi][ni][reg];
          sv += v;
          qv += v * v;
        }
      sv += __shfl_xor(sv, 16);
      qv += __shfl_xor(qv, 16);
      sv += __shfl_xor(sv, 32);
      qv += __shfl_xor(qv, 32);
      if (lane < 16) {
        const int col = n0 + wn + ni * 16 + frow;
        // big grids always wrap the capped workspace -> atomics
        // (pre-zeroed by the host, same convention as the NT kernel)
        const long long wsrow =
            ((flat_id * NWM + wave / NWN)) % ws_nblocks;
        atomicAdd(&stats_ws[wsrow * N + col], sv);
        atomicAdd(&stats_ws[((long long)ws_nblocks + wsrow) * N + col], qv);
      }
    }
  }

  // ---- epilogue: stage the y tile through LDS for coalesced rows
  __hip_bfloat16* cs = smem;  // [BM][BN] bf16
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      const int col = wn + ni * 16 + frow;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = wm + mi * 16 + (lane >> 4) * 4 + reg;
        cs[row * BN + col] = f2bf(acc[mi][ni][reg]);
      }
    }
  __syncthreads();
  constexpr int CPR = BN / 8;        // 16B chunks per row
  constexpr int RPR = THREADS / CPR; // rows per round
  const int wrow = tid / CPR, wchunk = tid % CPR;
#pragma unroll
  for (int r = 0; r < BM / RPR; ++r) {
    const int row = r * RPR + wrow;
    *(short8*)((char*)(y + (m0 + row) * N + n0) + wchunk * 16) =
        *(const short8*)((const char*)(cs + row * BN) + wchunk * 16);
  }
}

}  // namespace cg

// Host entry: returns true when handled.  y must be pre-allocated
// [N,HO,WO,Kout]; stats_ws (optional) pre-zeroed [2*ws_nblocks, Kout].
bool conv2d_fwd_glds_ex(const torch::Tensor& x, const torch::Tensor& w,
                        torch::Tensor& y, const torch::Tensor& zp,
                        int64_t stride, int64_t pad, float* stats_ws,
                        int ws_nblocks, int lgSt) {
  static const char* e = getenv("PDT_CONV_GLDS");
  if (e && e[0] == '0') return false;
  if (x.scalar_type() != torch::kBFloat16) return false;
  const int N_ = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            C = (int)x.size(3);
  const int Kout = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  if (C < 64 || (C & (C - 1))) return false;
  const int HO = (int)y.size(1), WO = (int)y.size(2);
  auto pow2l = [](int v) {
    int l = 0;
    while ((1 << l) < v) ++l;
    return ((1 << l) == v) ? l : -1;
  };
  const int lgC = pow2l(C);
  const long long M = (long long)N_ * HO * WO;
  const int K = R * S * C;
  static const char* e_bn = getenv("PDT_CG_BN");
  int BN = Kout % 256 == 0 ? 256 : Kout % 128 == 0 ? 128 : 64;
  if (e_bn && e_bn[0] == '6') BN = 64;  // force the 2-block 64-wide route
  if (Kout % 64) return false;
  int BM = BN == 64 ? 512 : 256;
  if (M % BM) {
    if (BM == 512 && M % 256 == 0) { BM = 256; BN = 64; }
    else return false;
  }
  if (M / BM > 2147483647LL / 8) return false;
  cg::Geom g{H, W, C, lgC, HO, WO, (int)stride, (int)pad, S, R * S, lgSt};
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(Kout / BN, (unsigned)(M / BM));
  const auto* xp = reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  const auto* wp = reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  auto* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr());
  const auto* zpp = reinterpret_cast<const __hip_bfloat16*>(zp.data_ptr());
#define LAUNCH_CG(BMv, BNv, MB)                                              \
  do {                                                                       \
    if (stats_ws)                                                            \
      hipLaunchKernelGGL((cg::conv_fwd_glds_kernel<BMv, BNv, true, MB>),     \
                         grid, dim3(cg::THREADS), 0, stream, xp, wp, yp,     \
                         zpp, (int)M, Kout, K, g, stats_ws, ws_nblocks);     \
    else                                                                     \
      hipLaunchKernelGGL((cg::conv_fwd_glds_kernel<BMv, BNv, false, MB>),    \
                         grid, dim3(cg::THREADS), 0, stream, xp, wp, yp,     \
                         zpp, (int)M, Kout, K, g, nullptr, 0);               \
  } while (0)
  // Kout=64: A/B between one 512x64 block/CU and two co-resident 256x64
  // blocks (80 KB LDS each — exactly 2/CU; better glds latency hiding)
  static const char* e_l1 = getenv("PDT_CG_N64");
  const bool n64_2b = !(e_l1 && e_l1[0] == '5');  // "5" -> 512x64 route
  if (BN == 64 && n64_2b && M % 256 == 0) {
    dim3 grid64(Kout / 64, (unsigned)(M / 256));
    grid = grid64;
    BM = 256;
  }
  if (BN == 256) LAUNCH_CG(256, 256, 1);
  else if (BN == 128) LAUNCH_CG(256, 128, 1);
  else if (BM == 512) LAUNCH_CG(512, 64, 1);
  else LAUNCH_CG(256, 64, 2);
#undef LAUNCH_CG
  return true;
}

bool conv2d_fwd_glds(const torch::Tensor& x, const torch::Tensor& w,
                     torch::Tensor& y, const torch::Tensor& zp,
                     int64_t stride, int64_t pad, float* stats_ws,
                     int ws_nblocks) {
  return conv2d_fwd_glds_ex(x, w, y, zp, stride, pad, stats_ws, ws_nblocks,
                            0);
}
