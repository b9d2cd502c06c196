// bf16 MFMA GEMM family for gfx950 (CDNA4) — the compute core.
//
// One NT kernel (C[M,N] = A[M,K] @ B[N,K]^T, both operands K-contiguous, the
// natural MFMA layout) carries:
//   * linear forward  (Y = X W^T; W stored [N,K] — reference model.py:11,13),
//   * linear dgrad    (dX = dY W, via the small W transpose),
//   * conv2d fwd/dgrad as implicit GEMM (NHWC; the A operand is gathered
//     im2col-style at LDS-staging time — MODE_CONV loader below),
//   * batched bmm_nt for the ViT attention path.
// A TN kernel (C[I,J] = sum_m A[m,I] B[m,J], fp32 out, split-M atomics)
// carries linear/conv weight gradients.
//
// Structure (guide §5 "canonical CDNA GEMM", 2-phase schedule):
//   128x128 tile, BK=32, 4 waves (2x2), 64x64 per wave = 4x4 fragments of
//   v_mfma_f32_16x16x32_bf16; global->LDS staging via
//   __builtin_amdgcn_global_load_lds width 16 (the compiler never auto-emits
//   it); double-buffered LDS, one barrier per K-step; XOR swizzle on the
//   k-slot (applied to the per-lane SOURCE address + the LDS read — never
//   the LDS destination, which glds writes lane-linearly); bijective
//   XCD-aware block swizzle for L2 locality.
//
// Out-of-range rows/chunks (M/N/K tails, conv padding) are redirected to a
// 64-byte zero page so the glds path needs no branches: zeros flow through
// the MFMA harmlessly.  K must be a multiple of 8 (host pads when not).

#include <torch/extension.h>

#include <unordered_map>

#include "common.h"
#include "dispatch.h"

namespace g16 {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// 16-bit dtype traits: fragment vector type + the matching MFMA intrinsic.
template <typename T16>
struct M16;
template <>
struct M16<bf16> {
  using vec = bf16x8;
  DEV_INLINE static f32x4 mma(vec a, vec b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
  // 32x32x16 is gfx950's full-rate bf16 shape (~2.5 PF vs ~2.0 for
  // 16x16x32) and does 2x the FLOPs per instruction — issue-bound kernels
  // want it.  A/B: lane holds 8 k at k = (lane>>5)*8, row/col = lane&31.
  // C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
  DEV_INLINE static f32x16 mma32(vec a, vec b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct M16<_Float16> {
  using vec = f16x8;
  DEV_INLINE static f32x4 mma(vec a, vec b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
  DEV_INLINE static f32x16 mma32(vec a, vec b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
  }
};

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int THREADS = 256;
constexpr int TILE_BYTES = BM * BK * 2;  // 8 KiB per operand tile

// swizzle: image[row][kp] holds global chunk kp ^ swz(row); involution.
DEV_INLINE int kswz(int row, int kp) { return kp ^ ((row >> 2) & 3); }

struct ConvMeta {
  int H, W, C_log2, S, R, stride, pad;
  int HO, WO;
  // dgrad zero-stuffing, folded into the im2col gather: the logical [H,W]
  // image is the stride-`ss` zero-stuffing of a physical [SH,SW] source
  // (dy).  ss <= 1 means dense; brace-inits that omit these get 0 = dense,
  // so no materialized zero-stuffed tensor (or its 2x HBM round trip) exists.
  int ss, SH, SW;
};

enum { MODE_PLAIN = 0, MODE_CONV = 1, MODE_CONVJ = 2 };

// ---------------------------------------------------------------- NT -----
// BNT: the N tile (128, or 64 when N <= 64 — ResNet layer1 Kout, C=64
// dgrads — so half the MFMAs aren't wasted on zero-page columns).
// Optional epilogue extras (runtime pointers, null = off):
//  * stats_ws [2*nblocks][N]: per-block BN partial sums/sumsqs of the
//    written outputs (the BN stats pass then never re-reads the tensor);
//  * addend (+addend_mask): C = acc + (mask>0 ? addend : 0) — fuses the
//    residual skip-gradient add (and its ReLU mask) into conv dgrad.
// EXTRAS=false compiles the epilogue extras out entirely (they cost
// registers/codegen even when the pointers are null).
template <typename T16, int MODE, bool RELU, bool HAS_BIAS, int BNT = BN,
          bool EXTRAS = false, int NBUF = 3>
__global__ __launch_bounds__(THREADS) void gemm_nt_bf16_kernel(
    const T16* __restrict__ A, const T16* __restrict__ B,
    T16* __restrict__ C, const T16* __restrict__ bias,
    const T16* __restrict__ zpad, int M, int N, int K, long long strideA,
    long long strideB, long long strideC, ConvMeta cm,
    float* __restrict__ stats_ws, int ws_nblocks,
    const T16* __restrict__ addend, const T16* __restrict__ addend_mask) {
  constexpr int NI = BNT / 32;              // B fragments per wave
  constexpr int BUF_BYTES = TILE_BYTES + BNT * 64;  // A tile + B tile
  // 3 LDS buffers + raw s_barrier with counted vmcnt waits: __syncthreads()
  // with glds in flight emits vmcnt(0) and drains the NEXT k-step's loads,
  // collapsing the pipeline (guide §6: -16..20%% at GEMM scale).  With 3
  // buffers each barrier only waits for loads issued TWO steps back.
  __shared__ __attribute__((aligned(16))) char smem[NBUF * BUF_BYTES];

  // ----- block swizzle (bijective XCD remap over the x*y grid) -----
  int nwg = gridDim.x * gridDim.y;
  int id = blockIdx.y * gridDim.x + blockIdx.x;
  const int flat_id = id;  // stats_ws row (pre-swizzle: any bijection works)
  if (nwg >= 16) {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = id & 7, idx = id >> 3;
    id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int tx = id % gridDim.x;  // N tile
  const int ty = id / gridDim.x;  // M tile
  const int m0 = ty * BM, n0 = tx * BNT;

  const long long batch = blockIdx.z;
  A += batch * strideA;
  B += batch * strideB;
  C += batch * strideC;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // ----- per-instr staging geometry (fixed across the K loop) -----
  // Each wave issues 2 glds per operand tile; instr chunk = 16 rows x 64 B.
  // lane -> (row_in_chunk = lane>>2, kp = lane&3)
  const int rl_a[2] = {(wave * 2 + 0) * 16 + (lane >> 2),
                       (wave * 2 + 1) * 16 + (lane >> 2)};
  const int kp = lane & 3;

  // A-side per-row precompute (conv: NHWC source decomposition)
  int a_n[2], a_hb[2], a_wb[2];
  bool a_ok[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int gm = m0 + rl_a[i];
    if (MODE == MODE_CONV) {
      a_ok[i] = gm < M;
      if (a_ok[i]) {
        int t = gm;
        const int wo = t % cm.WO;
        t /= cm.WO;
        const int ho = t % cm.HO;
        a_n[i] = t / cm.HO;
        a_hb[i] = ho * cm.stride - cm.pad;
        a_wb[i] = wo * cm.stride - cm.pad;
      }
    } else {
      a_ok[i] = gm < M;
    }
  }

  const int KT = (K + BK - 1) / BK;
  // magic reciprocal for the per-k-step rs/S decode: a runtime integer
  // divide is a ~40-cycle VALU sequence and sat inside the staging loop
  const int rcpS = MODE == MODE_CONV ? 65536 / cm.S + 1 : 0;

  // ----- staging -----
  auto stage = [&](int buf, int kt) {
    const int k_base = kt * BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      // ---- A tile ----
      const int swz = kswz(rl_a[i], kp);
      const int gk = k_base + swz * 8;  // first element of this 16B chunk
      const T16* src = zpad;
      if (MODE == MODE_CONV) {
        if (a_ok[i] && gk < K) {
          const int c0 = gk & ((1 << cm.C_log2) - 1);
          const int rs = gk >> cm.C_log2;
          const int r = (rs * rcpS) >> 16, s = rs - r * cm.S;
          const int hi = a_hb[i] + r, wi = a_wb[i] + s;
          if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W) {
            if (cm.ss <= 1) {
              src = A + (((long long)a_n[i] * cm.H + hi) * cm.W + wi) *
                            (1LL << cm.C_log2) +
                    c0;
            } else {  // stuffed dgrad source: only every ss-th point is real
              const int hs = hi / cm.ss, ws = wi / cm.ss;
              if (hi == hs * cm.ss && wi == ws * cm.ss && hs < cm.SH &&
                  ws < cm.SW)
                src = A + (((long long)a_n[i] * cm.SH + hs) * cm.SW + ws) *
                              (1LL << cm.C_log2) +
                      c0;
            }
          }
        }
      } else {
        if (a_ok[i] && gk < K) src = A + (long long)(m0 + rl_a[i]) * K + gk;
      }
      char* ldsA = &smem[buf * BUF_BYTES + (wave * 2 + i) * 1024];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)ldsA, 16, 0, 0);
    }
    // ---- B tile (always plain rows of [N,K]; BNT rows, BNT/64 units per
    // wave so BNT 64/128/256 all cover exactly — 256 is the wide-N route
    // that halves the A-operand re-reads on big plain GEMMs) ----
    constexpr int NBU = BNT >= 64 ? BNT / 64 : 1;
#pragma unroll
    for (int i = 0; i < NBU; ++i) {
      const int rlb = (wave * NBU + i) * 16 + (lane >> 2);
      const int gn = n0 + rlb;
      const int swzb = kswz(rlb, kp);
      const int gkb = k_base + swzb * 8;
      const T16* srcb =
          (gn < N && gkb < K) ? B + (long long)gn * K + gkb : zpad;
      char* ldsB =
          &smem[buf * BUF_BYTES + TILE_BYTES + (wave * NBU + i) * 1024];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)srcb,
          (__attribute__((address_space(3))) unsigned int*)ldsB, 16, 0, 0);
    }
  };

  // ----- main loop (32x32x16 MFMA: full-rate shape, half the instrs) -----
  constexpr int NI32 = BNT / 64;  // 32-wide B tiles per wave
  f32x16 acc[2][NI32] = {};
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (BNT / 2);
  const int r32 = lane & 31;  // row (A) / col (B, D) within a 32-tile
  const int ks = lane >> 5;   // k half-slot (8 halfwords)

  // per-wave glds count of one stage() call (the vmcnt budget at barriers):
  // 2 A units + BNT/64 B units, uniform across waves
  constexpr int STAGE_GLDS = 2 + (BNT >= 64 ? BNT / 64 : 1);
#define NT_WAIT_STAGE()                                        \
  do {                                                         \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");         \
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(STAGE_GLDS)       \
                 : "memory");                                  \
  } while (0)
#define NT_WAIT_ALL()                                          \
  do {                                                         \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");         \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");           \
  } while (0)

  stage(0, 0);
  if (NBUF >= 3) {
    if (KT > 1) stage(1, 1);
    if (KT > 1) NT_WAIT_STAGE(); else NT_WAIT_ALL();
  } else {
    NT_WAIT_ALL();
  }
  asm volatile("s_barrier" ::: "memory");
  for (int kt = 0; kt < KT; ++kt) {
    const int buf = kt % NBUF;
    if (NBUF >= 3) {
      if (kt + 2 < KT) stage((kt + 2) % NBUF, kt + 2);
    } else {
      if (kt + 1 < KT) stage((kt + 1) & 1, kt + 1);
    }
    const char* baseA = &smem[buf * BUF_BYTES];
    const char* baseB = baseA + TILE_BYTES;
    using vec16 = typename M16<T16>::vec;
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {  // two k-16 halves of the BK=32 step
      const int chunk = kh * 2 + ks;
      vec16 af[2], bf[NI32];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int ra = wm + i * 32 + r32;
        af[i] = *reinterpret_cast<const vec16*>(
            baseA + ra * 64 + kswz(ra, chunk) * 16);
      }
#pragma unroll
      for (int i = 0; i < NI32; ++i) {
        const int rb = wn + i * 32 + r32;
        bf[i] = *reinterpret_cast<const vec16*>(
            baseB + rb * 64 + kswz(rb, chunk) * 16);
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI32; ++ni)
          acc[mi][ni] = M16<T16>::mma32(af[mi], bf[ni], acc[mi][ni]);
    }
    if (kt + 1 < KT) {
      if (NBUF >= 3 && kt + 2 < KT) NT_WAIT_STAGE(); else NT_WAIT_ALL();
      asm volatile("s_barrier" ::: "memory");
    }
  }
#undef NT_WAIT_STAGE
#undef NT_WAIT_ALL

  // ----- epilogue: bias + relu (+addend) + bf16 store (+BN stat partials) --
  // C/D layout of 32x32x16: col = lane&31, row = (reg&3)+8*(reg>>2)+4*ks
  float col_sum[NI32] = {}, col_sq[NI32] = {};
#pragma unroll
  for (int ni = 0; ni < NI32; ++ni) {
    const int col = n0 + wn + ni * 32 + r32;
    if (col >= N) continue;
    float bv = HAS_BIAS ? to_f(bias[col]) : 0.f;
    // gather the residual addend (+mask) up front: independent loads issue
    // back-to-back and overlap one memory latency
    float av[2][16];
    if (EXTRAS && addend) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          const int row =
              m0 + wm + mi * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
          const long long idx = (long long)row * N + col;
          av[mi][reg] = row < M ? to_f(addend[idx]) : 0.f;
          if (addend_mask && row < M && !(to_f(addend_mask[idx]) > 0.f))
            av[mi][reg] = 0.f;
        }
    }
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row =
            m0 + wm + mi * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
        if (row >= M) continue;
        float v = acc[mi][ni][reg] + bv;
        if (RELU) v = fmaxf(v, 0.f);
        if (EXTRAS && addend) v += av[mi][reg];
        C[(long long)row * N + col] = to_t<T16>(v);
        if (EXTRAS && stats_ws) {
          col_sum[ni] += v;
          col_sq[ni] += v * v;
        }
      }
    }
  }
  if (EXTRAS && stats_ws) {
    // barrier-free: lanes l and l^32 share a column (ks differs) — one
    // shuffle reduces them; each wave PAIR (wave>>1 disambiguates waves
    // sharing a column range at different row halves) stores its slice into
    // its own workspace row (ws_nblocks = 2 * grid blocks).
    // ws rows are capped (host: ws_nblocks <= 8192); blocks beyond the cap
    // wrap around and accumulate atomically (big-M launches would otherwise
    // scale the workspace and its zero-fill/finalize with M)
    const long long vrow = (long long)flat_id * 2 + (wave >> 1);
    const long long wsrow = vrow % ws_nblocks;
    // If ANY vrow in this launch wraps (2*nwg > ws_nblocks), every block
    // accumulates atomically into the pre-zeroed workspace: block execution
    // order is not guaranteed, so a wrapped block's atomicAdd landing before
    // the row owner's plain store would be silently overwritten.  (z-batched
    // blocks share flat ids, so gridDim.z > 1 forces atomics too.)
    const bool wrap = 2ll * nwg > (long long)ws_nblocks || gridDim.z > 1;
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

// ---------------------------------------------------------------- TN -----
// C[I,J] (+)= sum_m A[m,I] * B[m,J], fp32 accumulation into global memory
// via atomics (grid.z splits the M range).  Used for dW = dY^T X.
// Staging transposes [m][i] chunks into padded LDS images [i][m] via
// register staging + lane-paired b32 writes (glds cannot transpose).
//
// MODE_CONV gathers the B operand (im2col of x) on the fly for conv wgrad;
// the j-tile then lives inside one (r,s) slice: j = c0_tile + c.
template <typename T16, int MODE>
__global__ __launch_bounds__(THREADS) void gemm_tn_bf16_kernel(
    const T16* __restrict__ A, const T16* __restrict__ B,
    float* __restrict__ C, int Mtot, int I, int J, int r, int s,
    long long ldc, long long coff, ConvMeta cm) {
  // C element (i, j) lives at C[coff + i*ldc + j] — conv wgrad writes a
  // (r,s) slice of dw[Kout][R*S*C], so ldc != J there.
  constexpr int BI = 64, BJ = 64, BMC = 32;  // m-chunk
  // padded LDS row: +8 keeps 16-B alignment for the b128 fragment reads
  // (row stride 80 B) while staying bank-conflict-free (banks r*20 mod 64
  // are distinct over any 16 consecutive rows).
  constexpr int ROW = BMC + 8;
  using vec16 = typename M16<T16>::vec;
  __shared__ __attribute__((aligned(16))) T16 lds[2 * BI * ROW];
  T16* ldsA = lds;
  T16* ldsB = lds + BI * ROW;

  const int i0 = blockIdx.y * BI;
  const int j0 = blockIdx.x * BJ;

  // split-M: this block handles chunk range [c0, c1)
  const int n_chunks = (Mtot + BMC - 1) / BMC;
  const int per_z = (n_chunks + gridDim.z - 1) / gridDim.z;
  const int ch0 = blockIdx.z * per_z;
  const int ch1 = min(n_chunks, ch0 + per_z);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  // staging geometry: [32 m][64 cols] bf16 = 32 rows x 128 B; 256 lanes x
  // 16 B covers it in one pass: lane -> m = tid>>3, col0 = (tid&7)*8.
  const int sm = threadIdx.x >> 3;        // m row in chunk (0..31)
  const int sc0 = (threadIdx.x & 7) * 8;  // first col of this 16B piece

  f32x4 acc[2][2] = {};
  const int wm = (wave >> 1) * 32, wn = (wave & 1) * 32;
  const int fr = lane & 15;
  const int fs = lane >> 4;

  for (int ch = ch0; ch < ch1; ++ch) {
    const int mbase = ch * BMC;
    const int gm = mbase + sm;
    // ---- load + transpose-stage A chunk ([m][I] -> image [i][m]) ----
    {
      vec16 v = {};
      if (gm < Mtot && i0 + sc0 < I) {
        const long long off = (long long)gm * I + i0 + sc0;
        if (off + 8 <= (long long)Mtot * I) {
          v = *reinterpret_cast<const vec16*>(A + off);
        } else {  // last-row partial chunk: element-wise guarded load
#pragma unroll
          for (int j = 0; j < 8; ++j)
            if (off + j < (long long)Mtot * I) v[j] = A[off + j];
        }
      }
      // pair lanes (sm even/odd) to write b32 [i][m..m+1]
      short8 mine = *reinterpret_cast<short8*>(&v);
      short8 other;
#pragma unroll
      for (int j = 0; j < 8; ++j) other[j] = __shfl_xor((int)mine[j], 8);
      if (((threadIdx.x >> 3) & 1) == 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned int pack = ((unsigned short)mine[j]) |
                              (((unsigned int)(unsigned short)other[j]) << 16);
          *reinterpret_cast<unsigned int*>(&ldsA[(sc0 + j) * ROW + sm]) = pack;
        }
      }
    }
    // ---- load + transpose-stage B chunk ----
    {
      vec16 v = {};
      if (MODE == MODE_CONV) {
        if (gm < Mtot) {
          int t = gm;
          const int wo = t % cm.WO;
          t /= cm.WO;
          const int ho = t % cm.HO;
          const int n = t / cm.HO;
          const int hi = ho * cm.stride - cm.pad + r;
          const int wi = wo * cm.stride - cm.pad + s;
          const int c = j0 + sc0;
          if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W &&
              c < (1 << cm.C_log2))
            v = *reinterpret_cast<const vec16*>(
                B + (((long long)n * cm.H + hi) * cm.W + wi) *
                        (1LL << cm.C_log2) +
                c);
        }
      } else {
        if (gm < Mtot && j0 + sc0 < J) {
          const long long off = (long long)gm * J + j0 + sc0;
          if (off + 8 <= (long long)Mtot * J) {
            v = *reinterpret_cast<const vec16*>(B + off);
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j)
              if (off + j < (long long)Mtot * J) v[j] = B[off + j];
          }
        }
      }
      short8 mine = *reinterpret_cast<short8*>(&v);
      short8 other;
#pragma unroll
      for (int j = 0; j < 8; ++j) other[j] = __shfl_xor((int)mine[j], 8);
      if (((threadIdx.x >> 3) & 1) == 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned int pack = ((unsigned short)mine[j]) |
                              (((unsigned int)(unsigned short)other[j]) << 16);
          *reinterpret_cast<unsigned int*>(&ldsB[(sc0 + j) * ROW + sm]) = pack;
        }
      }
    }
    __syncthreads();
    // ---- MFMA: k dimension = m (BMC=32 -> one 16x16x32 per frag) ----
    vec16 af[2], bfr[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      af[i] = *reinterpret_cast<const vec16*>(
          &ldsA[(wm + i * 16 + fr) * ROW + fs * 8]);
      bfr[i] = *reinterpret_cast<const vec16*>(
          &ldsB[(wn + i * 16 + fr) * ROW + fs * 8]);
    }
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = M16<T16>::mma(af[mi], bfr[ni], acc[mi][ni]);
    __syncthreads();
  }

  // ---- accumulate into C (f32) ----
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int col = j0 + wn + ni * 16 + fr;
    if (col >= J) continue;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = i0 + wm + mi * 16 + fs * 4 + rr;
        if (row >= I) continue;
        atomicAdd(&C[coff + (long long)row * ldc + col], acc[mi][ni][rr]);
      }
    }
  }
}

// ------------------------------------------------- TN conv wgrad (tr) ----
// dw accumulation, all R*S taps fused in ONE launch, built on two gfx950
// hardware paths instead of VALU shuffle-transposes:
//   * global_load_lds stages each operand tile m-major: one [32 m][16 ch]
//     image (1 KiB) per 16-channel group, written lane-linearly (lane l
//     covers m = l>>1, 8-ch half l&1) — zero VALU, zero register traffic;
//   * ds_read_b64_tr_b16 reads MFMA fragments from those images with a free
//     hardware transpose.  Per-lane address
//       img + (fs*8 + ((lane&15)>>2))*32B + (lane&3)*8B     (fs = lane>>4)
//     and a second read at +128B deliver img[fs*8 .. fs*8+7][lane&15] in
//     natural k order (semantics established empirically by
//     tools/tr_probe.hip: out[4a+d][j] = elem d at the 8B-aligned address
//     of lane 4j+a within each 16-lane group).
// Per 32-m chunk a workgroup stages 1+TAPS tiles (4*(1+TAPS) glds) and runs
// 4*TAPS MFMAs per thread between barriers; the old shuffle kernel ran 12
// MFMAs per ~100 staging instructions and was issue-bound (SQ_WAIT_INST_ANY
// 49%).  Output: f32 atomicAdd, host keeps total blocks small.
// MODE_PLAIN reuses the same machinery for the plain TN GEMM
// C[I,J] += sum_m A[m,I] B[m,J] (linear / ViT weight grads): the B gather is
// then just rows of [M,J] and TAPS must be 1.  J == Cin for conv mode.
// MODE_CONVJ (TAPS == 1) puts ALL taps of a generic RxS conv wgrad on the
// J axis instead: dw[Kout][R*S*Cin] is ONE TN GEMM where logical column
// j = tap*Cin + c gathers x at that tap's spatial shift (tap = j >> C_log2,
// r = tap/S, s = tap%S).  One launch stages each dy chunk grid.x times
// instead of R*S times — the ResNet-50 stem (7x7 s2, Cin-padded 8) was 49
// per-tap TN launches re-reading all of dy, 28% of its whole train step.
// Batched via grid.z = nbatch * zsplit (blockIdx.z / zsplit selects the
// batch, % zsplit the split-M slice) — ViT attention backward runs ~1.5K
// small TN GEMMs per call and a host-side per-batch launch loop was 70%%
// of its step time.
// TI/TJ: MFMA fragments along I/J for the whole BLOCK tile — 2 gives 64x64
// (the conv wgrad default; TAPS=9 already fills the accumulators), 4 gives
// 128x128 (TAPS=1 large plain GEMMs, e.g. ViT linear weight grads, where
// 64x64 tiles were instruction-bound at 4 MFMAs per staged chunk), 8 with
// WI=2/WJ=4 gives 256x256 on 8 waves (512 threads) — the r2 route for the
// big ViT shapes: at 128x128 each operand is re-read ~6x from HBM and the
// kernel sits on that traffic roofline (~450-510 TF measured, and rocBLAS
// lands in the same band); 256x256 halves the re-reads.
// WI/WJ: the wave grid (WI*WJ waves per block) splitting the block tile.
// NBUF: LDS staging depth.  3 enables counted vmcnt waits (stage chunk n+2
// while chunk n computes; the barrier only waits for loads issued one chunk
// back) — essential at 1 workgroup/CU where __syncthreads()'s vmcnt(0)
// would drain the just-prefetched chunk and expose full HBM latency every
// 32 m-rows.  2 keeps the original drain-at-barrier behavior (fine for the
// conv variants where 2 co-resident WGs hide each other's stalls).
template <typename T16, int TAPS, int MODE = MODE_CONV, int TI = 2,
          int TJ = 2, int WI = 2, int WJ = 2, int NBUF = 2, int FUSE = 0>
__global__ __launch_bounds__(WI * WJ * 64, WI * WJ > 4 ? 1 : 2)
void gemm_wgrad_tr_kernel(
    const T16* __restrict__ dy, const T16* __restrict__ x,
    float* __restrict__ dw, const T16* __restrict__ zpad, int Mtot,
    int I /*Kout*/, int J, long long ldc, ConvMeta cm, long long sA,
    long long sB, long long sC, int zsplit,
    float* __restrict__ colsum = nullptr /* f32 [I]: bias grad of dy */) {
  constexpr int BI = 32 * TI, BJ = 32 * TJ, BMC = 32;
  constexpr int IMG = 32 * 16;   // elements per [32 m][16 ch] image
  constexpr int AIMGS = BI / 16, BIMGS = BJ / 16;
  constexpr int TILE_A = AIMGS * IMG, TILE_B = BIMGS * IMG;
  constexpr int NUNITS = AIMGS + TAPS * BIMGS;  // glds image stages / chunk
  static_assert(NBUF == 2 || NUNITS % (WI * WJ) == 0,
                "counted vmcnt waits need a uniform per-wave glds count");
  typedef short v4s __attribute__((ext_vector_type(4)));
  using vec16 = typename M16<T16>::vec;
  __shared__ __attribute__((aligned(16)))
      T16 lds[NBUF * (TILE_A + TAPS * TILE_B)];

  // XCD-contiguous remap: the 8 XCDs have PRIVATE 4 MB L2s, and the default
  // round-robin dispatch places the `tiles` blocks that share a zidx (same
  // m-range → same A/B chunk reads) across all 8 — every shared chunk line
  // is fetched from HBM once PER XCD.  Remapping logical tile ids so each
  // XCD owns a contiguous x-fastest/z-slowest run keeps same-chunk readers
  // on one L2.
  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  {
    const int gx = gridDim.x, gy = gridDim.y;
    const long long nwg = (long long)gx * gy * gridDim.z;
    long long id = ((long long)bz * gy + by) * gx + bx;
    if (nwg >= 16) {
      const long long q = nwg >> 3, r = nwg & 7;
      const int xcd = (int)(id & 7);
      const long long idx = id >> 3;
      id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    bx = (int)(id % gx);
    const long long t = id / gx;
    by = (int)(t % gy);
    bz = (int)(t / gy);
  }
  const int Cin = J;
  const int i0 = by * BI;
  const int j0 = bx * BJ;
  const int batch = bz / zsplit;
  const int zidx = bz % zsplit;
  dy += (long long)batch * sA;
  x += (long long)batch * sB;
  dw += (long long)batch * sC;
  const int n_chunks = (Mtot + BMC - 1) / BMC;
  const int per_z = (n_chunks + zsplit - 1) / zsplit;
  const int ch0 = zidx * per_z;
  const int ch1 = min(n_chunks, ch0 + per_z);
  if (ch0 >= ch1) return;  // uniform per block

  constexpr int NW = WI * WJ;  // waves per block
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave / WJ) * (BI / WI), wn = (wave % WJ) * (BJ / WJ);
  const int fr = lane & 15;
  const int fs = lane >> 4;
  const int sm = lane >> 1;        // staged m row this lane covers
  const int sh8 = (lane & 1) * 8;  // 8-channel half within the 32-B row

  f32x16 acc[TI / WI][TJ / WJ][TAPS] = {};  // 32x32 tiles

  // magic reciprocal for the CONVJ tap -> (r,s) decode (S <= 256)
  const int rcpS = MODE == MODE_PLAIN ? 0 : 65536 / cm.S + 1;

  // ---- stage chunk ch into LDS buffer buf: 4*NOPS glds, no VALU pack ----
  auto stage = [&](int buf, int ch) {
    const int gm = ch * BMC + sm;
    const bool mok = gm < Mtot;
    int n = 0, hb = 0, wb = 0;
    if (MODE != MODE_PLAIN && mok) {
      int t = gm;
      const int wo = t % cm.WO;
      t /= cm.WO;
      const int ho = t % cm.HO;
      n = t / cm.HO;
      hb = ho * cm.stride - cm.pad;
      wb = wo * cm.stride - cm.pad;
    }
    T16* base = lds + buf * (TILE_A + TAPS * TILE_B);
    // wave w issues units u = w, w+NW, ... of the NUNITS image stages
    for (int u = wave; u < NUNITS; u += NW) {
      const T16* src = zpad;
      T16* dst;
      if (u < AIMGS) {
        const int ig = u;
        const int ii = i0 + ig * 16 + sh8;
        if (mok && ii < I) src = dy + (long long)gm * I + ii;
        dst = base + ig * IMG;
      } else if (MODE == MODE_PLAIN) {
        const int ig = u - AIMGS;  // TAPS == 1
        const int jj = j0 + ig * 16 + sh8;
        if (mok && jj < J) src = x + (long long)gm * J + jj;
        dst = base + TILE_A + ig * IMG;
      } else if (MODE == MODE_CONVJ) {
        const int ig = u - AIMGS;  // TAPS == 1
        // logical col jc = tap*Cin_phys + c; each lane's 8-elem group stays
        // inside one tap because Cin_phys is pow2 >= 8 and jc % 8 == 0
        const int jc = j0 + ig * 16 + sh8;
        const int tap = jc >> cm.C_log2;
        const int r = (tap * rcpS) >> 16;
        const int s = tap - r * cm.S;
        const int hi = hb + r, wi = wb + s;
        if (mok && jc < J && hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W)
          src = x +
                (((long long)n * cm.H + hi) * cm.W + wi) * (1LL << cm.C_log2) +
                (jc & ((1 << cm.C_log2) - 1));
        dst = base + TILE_A + ig * IMG;
      } else {
        const int tap = (u - AIMGS) / BIMGS;
        const int ig = (u - AIMGS) % BIMGS;
        // TAPS==9 is only launched for 3x3 kernels, so the decode is
        // compile-time when the loop unrolls
        const int r = TAPS == 1 ? 0 : tap / 3;
        const int s = TAPS == 1 ? 0 : tap % 3;
        const int hi = hb + r, wi = wb + s;
        const int jj = j0 + ig * 16 + sh8;
        if (mok && hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W && jj < Cin)
          src = x + (((long long)n * cm.H + hi) * cm.W + wi) * Cin + jj;
        dst = base + TILE_A + (tap * BIMGS + ig) * IMG;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  // tr-read addressing for 32x32x16 fragments: lane L covers column
  // (L&31) -> image (L>>4)&1 at in-image col L&15; k half-slot ks = L>>5
  // selects the 8-m window.  Two reads (offset 0/+128B) give m +0..7; the
  // second MFMA of a 32-m chunk reads +512B (16 rows further).
  const int ks = lane >> 5;
  const unsigned tr_lane_off =
      (unsigned)((ks * 8 + ((lane & 15) >> 2)) * 32 + (lane & 3) * 8);
  const int img_sel = (lane >> 4) & 1;  // which 16-ch image of a 32-col pair
#define LDS_BYTE(p)                                           \
  ((unsigned)(unsigned long long)(__attribute__((            \
      address_space(3))) const T16*)(p))

  constexpr int PER_WAVE_GLDS = (NUNITS + WI * WJ - 1) / (WI * WJ);
  // wait until at most `n` STAGES are still in flight (exact count: waiting
  // for more than are actually outstanding would let an older, unlanded
  // stage through)
  auto wg_wait_stages = [&](int n) {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    switch (n) {
      case 0:
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        break;
      case 1:
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(PER_WAVE_GLDS) : "memory");
        break;
      case 2:
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * PER_WAVE_GLDS)
                     : "memory");
        break;
      default:
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(3 * PER_WAVE_GLDS)
                     : "memory");
        break;
    }
  };

  stage(0, ch0);
  if (NBUF >= 3) {
    // fill the prefetch pipeline NBUF-1 deep, then wait for stage 0 only
    int issued = 1;
    for (int p = 1; p < NBUF - 1 && ch0 + p < ch1; ++p) {
      stage(p, ch0 + p);
      ++issued;
    }
    wg_wait_stages(issued - 1);
    asm volatile("s_barrier" ::: "memory");
  } else {
    __syncthreads();
  }

  constexpr int TI32 = TI / WI, TJ32 = TJ / WJ;
  float bsum[TI32] = {};
  for (int ch = ch0; ch < ch1; ++ch) {
    const int buf = (ch - ch0) % NBUF;
    const bool more = ch + 1 < ch1;
    if (NBUF >= 3) {
      if (ch + NBUF - 1 < ch1)
        stage((ch - ch0 + NBUF - 1) % NBUF, ch + NBUF - 1);
    } else if (more) {
      stage(buf ^ 1, ch + 1);  // glds latency hides under the MFMAs
    }
    const T16* base = lds + buf * (TILE_A + TAPS * TILE_B);

    if constexpr (FUSE && TAPS == 1 && TI / WI == 2 && TJ / WJ == 2) {
      // Fused-issue path: all 16 tr reads of the chunk go out in ONE asm
      // block with a single lgkmcnt wait.  The generic path below pays a
      // full lgkmcnt(0) drain per 4-read fragment block (4 serial LDS
      // round-trips per chunk against only 8 MFMAs) — PMC showed the plain
      // TN kernel wait-bound at ~6x SQ_BUSY with the MFMA pipe 16% busy.
      const T16* tb = base + TILE_A;
      const unsigned a0 =
          LDS_BYTE(base + ((wm >> 4) + img_sel) * IMG) + tr_lane_off;
      const unsigned a1 =
          LDS_BYTE(base + ((wm >> 4) + 2 + img_sel) * IMG) + tr_lane_off;
      const unsigned b0 =
          LDS_BYTE(tb + ((wn >> 4) + img_sel) * IMG) + tr_lane_off;
      const unsigned b1 =
          LDS_BYTE(tb + ((wn >> 4) + 2 + img_sel) * IMG) + tr_lane_off;
      v4s r[16];
      asm volatile(
          "ds_read_b64_tr_b16 %0, %16 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %16 offset:128\n\t"
          "ds_read_b64_tr_b16 %2, %16 offset:512\n\t"
          "ds_read_b64_tr_b16 %3, %16 offset:640\n\t"
          "ds_read_b64_tr_b16 %4, %17 offset:0\n\t"
          "ds_read_b64_tr_b16 %5, %17 offset:128\n\t"
          "ds_read_b64_tr_b16 %6, %17 offset:512\n\t"
          "ds_read_b64_tr_b16 %7, %17 offset:640\n\t"
          "ds_read_b64_tr_b16 %8, %18 offset:0\n\t"
          "ds_read_b64_tr_b16 %9, %18 offset:128\n\t"
          "ds_read_b64_tr_b16 %10, %18 offset:512\n\t"
          "ds_read_b64_tr_b16 %11, %18 offset:640\n\t"
          "ds_read_b64_tr_b16 %12, %19 offset:0\n\t"
          "ds_read_b64_tr_b16 %13, %19 offset:128\n\t"
          "ds_read_b64_tr_b16 %14, %19 offset:512\n\t"
          "ds_read_b64_tr_b16 %15, %19 offset:640\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]), "=&v"(r[4]),
            "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7]), "=&v"(r[8]), "=&v"(r[9]),
            "=&v"(r[10]), "=&v"(r[11]), "=&v"(r[12]), "=&v"(r[13]),
            "=&v"(r[14]), "=&v"(r[15])
          : "v"(a0), "v"(a1), "v"(b0), "v"(b1));
      vec16 af2[2][2], bf2[2][2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          reinterpret_cast<v4s*>(&af2[mi][kh])[0] = r[mi * 4 + kh * 2];
          reinterpret_cast<v4s*>(&af2[mi][kh])[1] = r[mi * 4 + kh * 2 + 1];
          reinterpret_cast<v4s*>(&bf2[mi][kh])[0] = r[8 + mi * 4 + kh * 2];
          reinterpret_cast<v4s*>(&bf2[mi][kh])[1] = r[8 + mi * 4 + kh * 2 + 1];
        }
      if (colsum && bx == 0 && wn == 0) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          float a = 0.f;
#pragma unroll
          for (int kh = 0; kh < 2; ++kh) {
            const vec16 v = af2[mi][kh];
#pragma unroll
            for (int j = 0; j < 8; ++j) a += (float)v[j];
          }
          bsum[mi] += a;
        }
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          acc[mi][ni][0] =
              M16<T16>::mma32(af2[mi][0], bf2[ni][0], acc[mi][ni][0]);
          acc[mi][ni][0] =
              M16<T16>::mma32(af2[mi][1], bf2[ni][1], acc[mi][ni][0]);
        }
      if (more) {
        if (NBUF >= 3) {
          wg_wait_stages(min(ch + NBUF - 1, ch1 - 1) - (ch + 1));
          asm volatile("s_barrier" ::: "memory");
        } else {
          __syncthreads();
        }
      }
      continue;
    }

    // A fragments (dy): per 32-col tile mi, per 16-m half kh
    vec16 af[TI32][2];
#pragma unroll
    for (int mi = 0; mi < TI32; ++mi) {
      const unsigned a0 = LDS_BYTE(base + ((wm >> 4) + mi * 2 + img_sel) * IMG) +
                          tr_lane_off;
      v4s l0, h0, l1, h1;
      // "=&v" (early-clobber) is load-bearing: without it LLVM may alias an
      // output pair with an address input, and since ds_read results land
      // asynchronously (lgkmcnt), a later read in the block can consume a
      // clobbered address — a cold-launch-timing-dependent corruption.
      asm volatile(
          "ds_read_b64_tr_b16 %0, %4 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %4 offset:128\n\t"
          "ds_read_b64_tr_b16 %2, %4 offset:512\n\t"
          "ds_read_b64_tr_b16 %3, %4 offset:640\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(l0), "=&v"(h0), "=&v"(l1), "=&v"(h1)
          : "v"(a0));
      reinterpret_cast<v4s*>(&af[mi][0])[0] = l0;
      reinterpret_cast<v4s*>(&af[mi][0])[1] = h0;
      reinterpret_cast<v4s*>(&af[mi][1])[0] = l1;
      reinterpret_cast<v4s*>(&af[mi][1])[1] = h1;
      if (colsum && bx == 0 && wn == 0) {
        // bias grad rides along: the staged dy tile is already in
        // registers; out-of-range m/i lanes were zero-page staged.
        // Only one J-tile column of blocks and only wn==0 waves contribute
        // (everyone else re-reads the same A tile).
        float a = 0.f;
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          const vec16 v = af[mi][kh];
#pragma unroll
          for (int j = 0; j < 8; ++j) a += (float)v[j];
        }
        bsum[mi] += a;
      }
    }
#pragma unroll
    for (int tap = 0; tap < TAPS; ++tap) {
      const T16* tb = base + TILE_A + tap * TILE_B;
#pragma unroll
      for (int ni = 0; ni < TJ32; ++ni) {
        const unsigned b0 =
            LDS_BYTE(tb + ((wn >> 4) + ni * 2 + img_sel) * IMG) + tr_lane_off;
        v4s l0, h0, l1, h1;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %4 offset:0\n\t"
            "ds_read_b64_tr_b16 %1, %4 offset:128\n\t"
            "ds_read_b64_tr_b16 %2, %4 offset:512\n\t"
            "ds_read_b64_tr_b16 %3, %4 offset:640\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(l0), "=&v"(h0), "=&v"(l1), "=&v"(h1)
            : "v"(b0));
        vec16 bf0, bf1;
        reinterpret_cast<v4s*>(&bf0)[0] = l0;
        reinterpret_cast<v4s*>(&bf0)[1] = h0;
        reinterpret_cast<v4s*>(&bf1)[0] = l1;
        reinterpret_cast<v4s*>(&bf1)[1] = h1;
#pragma unroll
        for (int mi = 0; mi < TI32; ++mi) {
          acc[mi][ni][tap] = M16<T16>::mma32(af[mi][0], bf0, acc[mi][ni][tap]);
          acc[mi][ni][tap] = M16<T16>::mma32(af[mi][1], bf1, acc[mi][ni][tap]);
        }
      }
    }
    if (more) {
      if (NBUF >= 3) {
        // only wait for the stage one chunk back; newer prefetches keep
        // flying through the barrier
        wg_wait_stages(min(ch + NBUF - 1, ch1 - 1) - (ch + 1));
        asm volatile("s_barrier" ::: "memory");
      } else {
        __syncthreads();  // drains the glds (vmcnt) + publishes buf^1
      }
    }
  }

#undef LDS_BYTE

  // ---- writeback: f32 atomics into dw[I][taps][Cin] slices ----
  // (static tap order: a runtime-rotated index into acc would demote the
  // accumulators to scratch)
  if (colsum && bx == 0 && wn == 0) {
#pragma unroll
    for (int mi = 0; mi < TI32; ++mi) {
      float v = bsum[mi] + __shfl_xor(bsum[mi], 32);  // join the ks pair
      const int col = i0 + wm + mi * 32 + (lane & 31);
      if (ks == 0 && col < I) atomicAdd(&colsum[col], v);
    }
  }

  // C/D layout of 32x32x16: col = lane&31, row = (reg&3)+8*(reg>>2)+4*ks
#pragma unroll
  for (int tap = 0; tap < TAPS; ++tap) {
    const long long coff = (long long)tap * Cin;
#pragma unroll
    for (int ni = 0; ni < TJ32; ++ni) {
      const int col = j0 + wn + ni * 32 + (lane & 31);
      if (col >= Cin) continue;
#pragma unroll
      for (int mi = 0; mi < TI32; ++mi)
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          const int row =
              i0 + wm + mi * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
          if (row >= I) continue;
          atomicAdd(&dw[(long long)row * ldc + coff + col],
                    acc[mi][ni][tap][reg]);
        }
    }
  }
}


// ------------------------------------------ TN plain, 128-B-row images ----
// Plain TN GEMM (C[I,J] += sum_m A[m,I] B[m,J]) with [32 m][64 ch] LDS
// images (128-B rows).  The gemm_wgrad_tr_kernel above stages [32 m][16 ch]
// images — 32 B per m-row per glds — so every staged line is a 32-B sliver
// of a 64-B(+) memory line: TCC counters on the ViT wgrad shapes showed
// 19M 64-B read requests per dispatch (4x the useful bytes) and the kernel
// issue/latency-bound at ~425 TF.  128-B rows cut the request count 4x.
// Everything else (tr-read fragments, f32 atomic writeback, zsplit, XCD
// remap, optional fused bias grad) matches the kernel above.
template <typename T16, int TI = 4, int TJ = 4, int WI = 2, int WJ = 2,
          int MINB = 3>
__global__ __launch_bounds__(WI * WJ * 64, MINB) void gemm_tn_plain_kernel(
    const T16* __restrict__ dy, const T16* __restrict__ x,
    float* __restrict__ dw, const T16* __restrict__ zpad, int Mtot, int I,
    int J, long long ldc, long long sA, long long sB, long long sC,
    int zsplit, float* __restrict__ colsum = nullptr) {
  constexpr int BI = 32 * TI, BJ = 32 * TJ, BMC = 32;
  constexpr int IMG = 32 * 64;  // elements per [32 m][64 ch] image
  constexpr int AIMGS = BI / 64, BIMGS = BJ / 64;
  static_assert(BI % 64 == 0 && BJ % 64 == 0, "128-B-row images");
  constexpr int TILE_A = AIMGS * IMG, TILE_B = BIMGS * IMG;
  constexpr int NUNITS = (AIMGS + BIMGS) * 4;  // 8-row glds units per chunk
  constexpr int NW = WI * WJ;
  typedef short v4s __attribute__((ext_vector_type(4)));
  using vec16 = typename M16<T16>::vec;
  __shared__ __attribute__((aligned(16))) T16 lds[2 * (TILE_A + TILE_B)];

  // XCD-contiguous remap (private per-XCD L2s; see gemm_wgrad_tr_kernel)
  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  {
    const int gx = gridDim.x, gy = gridDim.y;
    const long long nwg = (long long)gx * gy * gridDim.z;
    long long id = ((long long)bz * gy + by) * gx + bx;
    if (nwg >= 16) {
      const long long q = nwg >> 3, r = nwg & 7;
      const int xcd = (int)(id & 7);
      const long long idx = id >> 3;
      id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    bx = (int)(id % gx);
    const long long t = id / gx;
    by = (int)(t % gy);
    bz = (int)(t / gy);
  }
  const int i0 = by * BI;
  const int j0 = bx * BJ;
  const int batch = bz / zsplit;
  const int zidx = bz % zsplit;
  dy += (long long)batch * sA;
  x += (long long)batch * sB;
  dw += (long long)batch * sC;
  const int n_chunks = (Mtot + BMC - 1) / BMC;
  const int per_z = (n_chunks + zsplit - 1) / zsplit;
  const int ch0 = zidx * per_z;
  const int ch1 = min(n_chunks, ch0 + per_z);
  if (ch0 >= ch1) return;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave / WJ) * (BI / WI), wn = (wave % WJ) * (BJ / WJ);
  const int srow = lane >> 3;        // staged m row within the 8-row unit
  const int sch8 = (lane & 7) * 8;   // 8-ch group within the 128-B row

  f32x16 acc[TI / WI][TJ / WJ] = {};

  // stage chunk ch into buffer buf: every unit reads 8 rows x 128 B
  // CONTIGUOUS from the operand (full memory lines)
  auto stage = [&](int buf, int ch) {
    const int gm0 = ch * BMC;
    T16* base = lds + buf * (TILE_A + TILE_B);
    for (int u = wave; u < NUNITS; u += NW) {
      const int img = u >> 2;           // image index (A then B)
      const int r0 = (u & 3) * 8;       // 8-row block
      const int gm = gm0 + r0 + srow;
      const T16* src = zpad;
      T16* dst;
      if (img < AIMGS) {
        const int ii = i0 + img * 64 + sch8;
        if (gm < Mtot && ii < I) src = dy + (long long)gm * I + ii;
        dst = base + img * IMG + r0 * 64;
      } else {
        const int jj = j0 + (img - AIMGS) * 64 + sch8;
        if (gm < Mtot && jj < J) src = x + (long long)gm * J + jj;
        dst = base + TILE_A + (img - AIMGS) * IMG + r0 * 64;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  // tr-read fragment addressing for 128-B rows: per-lane row term * 128 B;
  // reads at +4 rows (512 B) and +16 rows (2048 B) walk the 32-m chunk
  const int ks = lane >> 5;
  const unsigned tr_lane_off =
      (unsigned)((ks * 8 + ((lane & 15) >> 2)) * 128 + (lane & 3) * 8);
  const int img_sel = (lane >> 4) & 1;
#define LDS_BYTE(p)                                           \
  ((unsigned)(unsigned long long)(__attribute__((            \
      address_space(3))) const T16*)(p))

  stage(0, ch0);
  __syncthreads();

  constexpr int TI32 = TI / WI, TJ32 = TJ / WJ;
  float bsum[TI32] = {};
  for (int ch = ch0; ch < ch1; ++ch) {
    const int buf = (ch - ch0) & 1;
    const bool more = ch + 1 < ch1;
    if (more) stage(buf ^ 1, ch + 1);
    const T16* base = lds + buf * (TILE_A + TILE_B);

    vec16 af[TI32][2];
#pragma unroll
    for (int mi = 0; mi < TI32; ++mi) {
      const int c = wm + mi * 32;
      const unsigned a0 =
          LDS_BYTE(base + (c >> 6) * IMG + (c & 63) + img_sel * 16) +
          tr_lane_off;
      v4s l0, h0, l1, h1;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %4 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %4 offset:512\n\t"
          "ds_read_b64_tr_b16 %2, %4 offset:2048\n\t"
          "ds_read_b64_tr_b16 %3, %4 offset:2560\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(l0), "=&v"(h0), "=&v"(l1), "=&v"(h1)
          : "v"(a0));
      reinterpret_cast<v4s*>(&af[mi][0])[0] = l0;
      reinterpret_cast<v4s*>(&af[mi][0])[1] = h0;
      reinterpret_cast<v4s*>(&af[mi][1])[0] = l1;
      reinterpret_cast<v4s*>(&af[mi][1])[1] = h1;
      if (colsum && bx == 0 && wn == 0) {
        float a = 0.f;
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          const vec16 v = af[mi][kh];
#pragma unroll
          for (int j = 0; j < 8; ++j) a += (float)v[j];
        }
        bsum[mi] += a;
      }
    }
#pragma unroll
    for (int ni = 0; ni < TJ32; ++ni) {
      const int c = wn + ni * 32;
      const unsigned b0 =
          LDS_BYTE(base + TILE_A + (c >> 6) * IMG + (c & 63) + img_sel * 16) +
          tr_lane_off;
      v4s l0, h0, l1, h1;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %4 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %4 offset:512\n\t"
          "ds_read_b64_tr_b16 %2, %4 offset:2048\n\t"
          "ds_read_b64_tr_b16 %3, %4 offset:2560\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(l0), "=&v"(h0), "=&v"(l1), "=&v"(h1)
          : "v"(b0));
      vec16 bf0, bf1;
      reinterpret_cast<v4s*>(&bf0)[0] = l0;
      reinterpret_cast<v4s*>(&bf0)[1] = h0;
      reinterpret_cast<v4s*>(&bf1)[0] = l1;
      reinterpret_cast<v4s*>(&bf1)[1] = h1;
#pragma unroll
      for (int mi = 0; mi < TI32; ++mi) {
        acc[mi][ni] = M16<T16>::mma32(af[mi][0], bf0, acc[mi][ni]);
        acc[mi][ni] = M16<T16>::mma32(af[mi][1], bf1, acc[mi][ni]);
      }
    }
    if (more) __syncthreads();
  }
#undef LDS_BYTE

  if (colsum && bx == 0 && wn == 0) {
#pragma unroll
    for (int mi = 0; mi < TI32; ++mi) {
      float v = bsum[mi] + __shfl_xor(bsum[mi], 32);
      const int col = i0 + wm + mi * 32 + (lane & 31);
      if (ks == 0 && col < I) atomicAdd(&colsum[col], v);
    }
  }
#pragma unroll
  for (int ni = 0; ni < TJ32; ++ni) {
    const int col = j0 + wn + ni * 32 + (lane & 31);
    if (col >= J) continue;
#pragma unroll
    for (int mi = 0; mi < TI32; ++mi)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row = i0 + wm + mi * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
        if (row >= I) continue;
        atomicAdd(&dw[(long long)row * ldc + col], acc[mi][ni][reg]);
      }
  }
}

}  // namespace g16

// ======================= host-side helpers ===============================

namespace {

torch::Tensor& zero_page(const torch::Device& dev, torch::ScalarType st) {
  static std::unordered_map<int, torch::Tensor> cache;
  int idx = dev.index() * 4 + (st == torch::kBFloat16 ? 0 : 1);
  auto it = cache.find(idx);
  if (it == cache.end()) {
    it = cache.emplace(idx, torch::zeros({64}, torch::dtype(st).device(dev)))
             .first;
  }
  return it->second;
}

struct NtExtras {
  float* stats_ws = nullptr;  // [2*nblocks][N], pre-zeroed
  int ws_nblocks = 0;
  const void* addend = nullptr;
  const void* addend_mask = nullptr;
};

template <typename t16, int MODE>
void launch_nt16(const torch::Tensor& A, const torch::Tensor& B,
                 torch::Tensor& C, const c10::optional<torch::Tensor>& bias,
                 bool relu, const torch::Tensor& zp, dim3 grid, int M, int N,
                 int K, long long sA, long long sB, long long sC,
                 g16::ConvMeta cm, NtExtras ex = {},
                 bool wide_default = false) {
  auto stream = c10::hip::getCurrentHIPStream();
  const t16* bias_p =
      bias.has_value() ? reinterpret_cast<const t16*>(bias->data_ptr())
                       : nullptr;
  // narrow-N variant: BNT=64 halves the B tile so no MFMA computes
  // zero-page columns (ResNet layer1 Kout=64 fwd, C=64 dgrads).
  const bool narrow = N <= 64;
  if (narrow) grid.x = (N + 63) / 64;
  // wide-N variant: BNT=256 halves the A-operand re-reads and doubles the
  // MFMAs per barrier window.  Measured: NEUTRAL on plain linear shapes
  // (510 vs 525 TF on ViT mlp1 — the 73.7 KB LDS drops co-residency and
  // cancels the traffic win) but a clear WIN for the conv modes where the
  // A side is an im2col gather that is expensive to re-read: ResNet-18
  // bench 106.2k -> 108.4k, ResNet-50 7621 -> 8006 samples/s.  Default:
  // on for conv modes with Kout >= 256, off for MODE_PLAIN.
  // PDT_NT_BN=256 / 128 forces on / off for A/B runs.
  static const char* e_ntbn = getenv("PDT_NT_BN");
  // NOTE: the epilogue-extras instantiations (stats workspace / residual
  // addend — the fused conv+BN path) are built at BNT=128 only, so wide-N
  // must not halve their grid: doing so left half of C unwritten
  // (uninitialized-memory NaNs from the second step on — found by the
  // round-2 step-1 NaN bisect, tools/graph_dbg2.py).
  const bool extras_early = ex.stats_ws != nullptr || ex.addend != nullptr;
  // Post-fix defaults: ON only where the call site asks (conv FORWARD
  // without epilogue extras — the im2col A-side re-read argument and the
  // ViT patchify win are real there).  The round-2 blanket conv-mode
  // "win" was an artifact of the unwritten-half-of-C bug above; with the
  // grid corrected, wide-N on DGRADs loses e2e (r18 103.0k vs 105.8k).
  const bool wide_n = (e_ntbn ? e_ntbn[0] == '2' : wide_default) &&
                      !narrow && !extras_early && N >= 256 && M >= 4096;
  if (wide_n) grid.x = (N + 255) / 256;
  // 2-buffer wide-N: same 49 KB footprint as BNT=128x3buf (unchanged
  // co-residency) with 2x the MFMAs per barrier window.  PDT_NT_NBUF2=1.
  static const char* e_nb2 = getenv("PDT_NT_NBUF2");
  const bool ntbuf2 = e_nb2 && e_nb2[0] == '1';
  const bool extras = ex.stats_ws != nullptr || ex.addend != nullptr;
  if (extras) {
    TORCH_CHECK(!relu && !bias_p,
                "epilogue extras only instantiated for relu=false, bias=none");
    if (narrow)
      hipLaunchKernelGGL(
          (g16::gemm_nt_bf16_kernel<t16, MODE, false, false, 64, true>), grid,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()),
          reinterpret_cast<t16*>(C.data_ptr()), nullptr,
          reinterpret_cast<const t16*>(zp.data_ptr()), M, N, K, sA, sB, sC,
          cm, ex.stats_ws, ex.ws_nblocks,
          reinterpret_cast<const t16*>(ex.addend),
          reinterpret_cast<const t16*>(ex.addend_mask));
    else
      hipLaunchKernelGGL(
          (g16::gemm_nt_bf16_kernel<t16, MODE, false, false, g16::BN, true>),
          grid, dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()),
          reinterpret_cast<t16*>(C.data_ptr()), nullptr,
          reinterpret_cast<const t16*>(zp.data_ptr()), M, N, K, sA, sB, sC,
          cm, ex.stats_ws, ex.ws_nblocks,
          reinterpret_cast<const t16*>(ex.addend),
          reinterpret_cast<const t16*>(ex.addend_mask));
    return;
  }
#define LAUNCH_NT16(RELU, HB)                                                 \
  do {                                                                        \
    if (narrow)                                                               \
      hipLaunchKernelGGL((g16::gemm_nt_bf16_kernel<t16, MODE, RELU, HB, 64>), \
                         grid, dim3(g16::THREADS), 0, stream,                 \
                         reinterpret_cast<const t16*>(A.data_ptr()),          \
                         reinterpret_cast<const t16*>(B.data_ptr()),          \
                         reinterpret_cast<t16*>(C.data_ptr()), bias_p,        \
                         reinterpret_cast<const t16*>(zp.data_ptr()), M, N,   \
                         K, sA, sB, sC, cm, ex.stats_ws, ex.ws_nblocks,       \
                         reinterpret_cast<const t16*>(ex.addend),             \
                         reinterpret_cast<const t16*>(ex.addend_mask));       \
    else if (wide_n && ntbuf2)                                                \
      hipLaunchKernelGGL(                                                     \
          (g16::gemm_nt_bf16_kernel<t16, MODE, RELU, HB, 256, false, 2>),     \
          grid, dim3(g16::THREADS), 0, stream,                                \
          reinterpret_cast<const t16*>(A.data_ptr()),                         \
          reinterpret_cast<const t16*>(B.data_ptr()),                         \
          reinterpret_cast<t16*>(C.data_ptr()), bias_p,                       \
          reinterpret_cast<const t16*>(zp.data_ptr()), M, N, K, sA, sB, sC,   \
          cm, ex.stats_ws, ex.ws_nblocks,                                     \
          reinterpret_cast<const t16*>(ex.addend),                            \
          reinterpret_cast<const t16*>(ex.addend_mask));                      \
    else if (wide_n)                                                          \
      hipLaunchKernelGGL(                                                     \
          (g16::gemm_nt_bf16_kernel<t16, MODE, RELU, HB, 256>), grid,         \
          dim3(g16::THREADS), 0, stream,                                      \
          reinterpret_cast<const t16*>(A.data_ptr()),                         \
          reinterpret_cast<const t16*>(B.data_ptr()),                         \
          reinterpret_cast<t16*>(C.data_ptr()), bias_p,                       \
          reinterpret_cast<const t16*>(zp.data_ptr()), M, N, K, sA, sB, sC,   \
          cm, ex.stats_ws, ex.ws_nblocks,                                     \
          reinterpret_cast<const t16*>(ex.addend),                            \
          reinterpret_cast<const t16*>(ex.addend_mask));                      \
    else                                                                      \
      hipLaunchKernelGGL((g16::gemm_nt_bf16_kernel<t16, MODE, RELU, HB>),     \
                         grid, dim3(g16::THREADS), 0, stream,                 \
                         reinterpret_cast<const t16*>(A.data_ptr()),          \
                         reinterpret_cast<const t16*>(B.data_ptr()),          \
                         reinterpret_cast<t16*>(C.data_ptr()), bias_p,        \
                         reinterpret_cast<const t16*>(zp.data_ptr()), M, N,   \
                         K, sA, sB, sC, cm, ex.stats_ws, ex.ws_nblocks,       \
                         reinterpret_cast<const t16*>(ex.addend),             \
                         reinterpret_cast<const t16*>(ex.addend_mask));       \
  } while (0)
  if (relu) {
    if (bias_p) LAUNCH_NT16(true, true);
    else LAUNCH_NT16(true, false);
  } else {
    if (bias_p) LAUNCH_NT16(false, true);
    else LAUNCH_NT16(false, false);
  }
#undef LAUNCH_NT16
}

int log2_exact(int v) {
  int l = 0;
  while ((1 << l) < v) ++l;
  return ((1 << l) == v) ? l : -1;
}

}  // namespace

// 256x256-tile glds kernel for big plain bf16 shapes (gemm_plain.hip)
bool gemm_nt_plain256(const torch::Tensor& A, const torch::Tensor& B,
                      torch::Tensor& C,
                      const c10::optional<torch::Tensor>& bias, bool relu);

// C[b,M,N] = A[b,M,K] @ B[b,N,K]^T (+bias, +relu).  2-D inputs = batch 1.
torch::Tensor bmm_nt_bf16(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias, bool relu) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  const bool batched = A.dim() == 3;
  long long bsz = batched ? A.size(0) : 1;
  int M = (int)A.size(batched ? 1 : 0), K = (int)A.size(batched ? 2 : 1);
  int N = (int)B.size(batched ? 1 : 0);
  TORCH_CHECK((int)B.size(batched ? 2 : 1) == K, "K mismatch");
  TORCH_CHECK(K % 8 == 0, "K must be padded to a multiple of 8 (host)");
  auto C = batched ? torch::empty({bsz, M, N}, A.options())
                   : torch::empty({M, N}, A.options());
  if (!batched && gemm_nt_plain256(A, B, C, bias, relu)) return C;
  auto& zp = zero_page(A.device(), A.scalar_type());
  dim3 grid((N + g16::BN - 1) / g16::BN, (M + g16::BM - 1) / g16::BM,
            (unsigned)bsz);
  g16::ConvMeta cm{};
  auto stream = c10::hip::getCurrentHIPStream();
  long long sA = batched ? (long long)M * K : 0;
  long long sB = batched ? (long long)N * K : 0;
  long long sC = batched ? (long long)M * N : 0;
  if (A.scalar_type() == torch::kBFloat16)
    launch_nt16<bf16, g16::MODE_PLAIN>(A, B, C, bias, relu, zp, grid, M, N, K,
                                       sA, sB, sC, cm);
  else
    launch_nt16<_Float16, g16::MODE_PLAIN>(A, B, C, bias, relu, zp, grid, M,
                                           N, K, sA, sB, sC, cm);
  return C;
}

// conv_halo.hip: direct-from-LDS-halo 3x3 stride-1 fast path
bool conv2d_fwd_halo(torch::Tensor x, torch::Tensor w, torch::Tensor y,
                     torch::Tensor zp, int64_t stride, int64_t pad,
                     float* stats_ws, int ws_nblocks);
// glds implicit-GEMM conv fwd (conv_glds.hip): pow2 spatial, C%64==0
bool conv2d_fwd_glds(const torch::Tensor& x, const torch::Tensor& w,
                     torch::Tensor& y, const torch::Tensor& zp,
                     int64_t stride, int64_t pad, float* stats_ws,
                     int ws_nblocks);
bool conv2d_fwd_glds_ex(const torch::Tensor& x, const torch::Tensor& w,
                        torch::Tensor& y, const torch::Tensor& zp,
                        int64_t stride, int64_t pad, float* stats_ws,
                        int ws_nblocks, int lgSt);

// conv2d forward, NHWC x[N,H,W,C] * w[Kout,R,S,C] -> y[N,HO,WO,Kout]
// stats=true additionally returns the BN partial workspace ws
// ([2*nblocks][Kout], block-level sums/sumsqs of y) so the following
// BatchNorm needs no stats pass over y (bn_fwd_ws consumes it).
std::vector<torch::Tensor> conv2d_fwd_bf16_impl(
    torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias,
    int64_t stride, int64_t pad, bool relu, bool stats) {
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      Cin = (int)x.size(3);
  int Kout = (int)w.size(0), R = (int)w.size(1), S = (int)w.size(2);
  TORCH_CHECK((int)w.size(3) == Cin);
  int cl = log2_exact(Cin);
  TORCH_CHECK(cl >= 3, "conv fast path needs pow2 C >= 8 (stem uses im2col)");
  int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - S) / (int)stride + 1;
  int M = N * HO * WO, K = R * S * Cin;
  auto y = torch::empty({N, HO, WO, Kout}, x.options());
  auto& zp = zero_page(x.device(), x.scalar_type());
  dim3 grid((Kout + g16::BN - 1) / g16::BN, (M + g16::BM - 1) / g16::BM, 1);
  // 2 workspace rows per block (per wave-pair; see the epilogue), capped —
  // overflow blocks wrap around with atomic accumulation
  const int nblocks = std::min(
      8192,
      2 * (int)(((Kout <= 64 ? (Kout + 63) / 64 : grid.x)) * grid.y));
  g16::ConvMeta cm{H, W, cl, S, R, (int)stride, (int)pad, HO, WO};
  NtExtras ex{};
  torch::Tensor ws;
  if (stats) {
    ws = torch::zeros({2LL * nblocks, Kout},
                      x.options().dtype(torch::kFloat32));
    ex.stats_ws = ws.data_ptr<float>();
    ex.ws_nblocks = nblocks;
  }
  if (!relu && !bias.has_value() &&
      conv2d_fwd_glds(x, w, y, zp, stride, pad, ex.stats_ws,
                      ex.ws_nblocks)) {
    if (stats) return {y, ws};
    return {y};
  }
  if (!relu && !bias.has_value() &&
      conv2d_fwd_halo(x, w, y, zp, stride, pad, ex.stats_ws,
                      ex.ws_nblocks)) {
    if (stats) return {y, ws};
    return {y};
  }
  if (x.scalar_type() == torch::kBFloat16)
    launch_nt16<bf16, g16::MODE_CONV>(x, w, y, bias, relu, zp, grid, M, Kout,
                                      K, 0, 0, 0, cm, ex, /*wide*/ true);
  else
    launch_nt16<_Float16, g16::MODE_CONV>(x, w, y, bias, relu, zp, grid, M,
                                          Kout, K, 0, 0, 0, cm, ex,
                                          /*wide*/ true);
  if (stats) return {y, ws};
  return {y};
}

torch::Tensor conv2d_fwd_bf16(torch::Tensor x, torch::Tensor w,
                              c10::optional<torch::Tensor> bias, int64_t stride,
                              int64_t pad, bool relu) {
  return conv2d_fwd_bf16_impl(x, w, bias, stride, pad, relu, false)[0];
}

std::vector<torch::Tensor> conv2d_fwd_stats_bf16(
    torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias,
    int64_t stride, int64_t pad, bool relu) {
  return conv2d_fwd_bf16_impl(x, w, bias, stride, pad, relu, true);
}

// conv2d dgrad, direct: dx = conv(zero_stuffed(dy), wr, 1, R-1-pad) with the
// zero-stuffing folded into the im2col gather (ConvMeta.ss) — the stuffed
// tensor is never materialized.  dy[N,HO,WO,Kout], wr[Cin,R,S,Kout] (the
// 180°-rotated weight), stride/pad are the ORIGINAL forward conv's.
torch::Tensor conv2d_dgrad_bf16(torch::Tensor dy, torch::Tensor wr,
                                int64_t stride, int64_t pad, int64_t H,
                                int64_t W,
                                c10::optional<torch::Tensor> addend,
                                c10::optional<torch::Tensor> addend_mask) {
  int N = (int)dy.size(0), HOs = (int)dy.size(1), WOs = (int)dy.size(2),
      Kout = (int)dy.size(3);
  int Cin = (int)wr.size(0), R = (int)wr.size(1), S = (int)wr.size(2);
  TORCH_CHECK((int)wr.size(3) == Kout);
  int cl = log2_exact(Kout);
  TORCH_CHECK(cl >= 3, "dgrad fast path needs pow2 Kout >= 8");
  // logical (stuffed) image dims: (HO-1)*stride + 1 + output-padding
  const int opad_h = (int)((H + 2 * pad - R) % stride);
  const int opad_w = (int)((W + 2 * pad - S) % stride);
  const int HS = (HOs - 1) * (int)stride + 1 + opad_h;
  const int WS = (WOs - 1) * (int)stride + 1 + opad_w;
  const int dpad = R - 1 - (int)pad;
  TORCH_CHECK(dpad >= 0 && S - 1 - pad == dpad, "dgrad needs pad <= kernel-1");
  const int HOut = HS + 2 * dpad - R + 1, WOut = WS + 2 * dpad - S + 1;
  TORCH_CHECK(HOut == H && WOut == W, "dgrad shape mismatch");
  int M = N * HOut * WOut, K = R * S * Kout;
  auto dx = torch::empty({N, HOut, WOut, Cin}, dy.options());
  auto& zp = zero_page(dy.device(), dy.scalar_type());
  dim3 grid((Cin + g16::BN - 1) / g16::BN, (M + g16::BM - 1) / g16::BM, 1);
  g16::ConvMeta cm{HS,   WS,   cl,   S,  R, 1, dpad, HOut, WOut,
                   (int)stride, HOs, WOs};
  // dense (stride-1) dgrad is exactly a 3x3 stride-1 pad-1 conv of dy with
  // the rotated weight: take the LDS-halo fast path when it applies
  // glds path: stride-1 dgrad is a plain conv of dy with the rotated
  // weight; pow2-stride dgrad uses the stuffed-coordinate gather (valid
  // only on the stride grid, indexed by the un-stuffed position)
  static const char* e_gs2 = getenv("PDT_CONV_GLDS_S2");
  const bool glds_s2 = !(e_gs2 && e_gs2[0] == '0');
  if (!addend.has_value() && ((int)stride & ((int)stride - 1)) == 0 &&
      ((int)stride == 1 || glds_s2)) {
    const int lgSt = log2_exact((int)stride);
    if (lgSt >= 0 &&
        conv2d_fwd_glds_ex(dy, wr, dx, zp, 1, dpad, nullptr, 0, lgSt))
      return dx;
  }
  if ((int)stride == 1 && !addend.has_value() &&
      conv2d_fwd_halo(dy, wr, dx, zp, 1, dpad, nullptr, 0))
    return dx;
  NtExtras ex{};
  if (addend.has_value()) {
    TORCH_CHECK(addend->is_contiguous() && addend->numel() == dx.numel() &&
                addend->scalar_type() == dy.scalar_type());
    ex.addend = addend->data_ptr();
    if (addend_mask.has_value()) {
      TORCH_CHECK(addend_mask->is_contiguous() &&
                  addend_mask->numel() == dx.numel() &&
                  addend_mask->scalar_type() == dy.scalar_type());
      ex.addend_mask = addend_mask->data_ptr();
    }
  }
  if (dy.scalar_type() == torch::kBFloat16)
    launch_nt16<bf16, g16::MODE_CONV>(dy, wr, dx, {}, false, zp, grid, M, Cin,
                                      K, 0, 0, 0, cm, ex);
  else
    launch_nt16<_Float16, g16::MODE_CONV>(dy, wr, dx, {}, false, zp, grid, M,
                                          Cin, K, 0, 0, 0, cm, ex);
  return dx;
}

// C[I,J] = A[...,M,I]^T @ B[...,M,J] summed over batch? No — per batch.
// Returns f32.  2-D inputs only here; batched variant loops z on grid.
torch::Tensor bmm_tn_bf16(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  const bool batched = A.dim() == 3;
  TORCH_CHECK(!batched || B.dim() == 3);
  long long bsz = batched ? A.size(0) : 1;
  int M = (int)A.size(batched ? 1 : 0), I = (int)A.size(batched ? 2 : 1);
  int J = (int)B.size(batched ? 2 : 1);
  TORCH_CHECK((int)B.size(batched ? 1 : 0) == (batched ? M : M));
  auto C = batched
               ? torch::zeros({bsz, I, J},
                              A.options().dtype(torch::kFloat32))
               : torch::zeros({I, J}, A.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  // split-M for parallelism: aim for >= 512 blocks
  int n_chunks = (M + 31) / 32;
  int tiles = ((J + 63) / 64) * ((I + 63) / 64);
  // ~768 total blocks: enough to fill 256 CUs ~3 deep while keeping the
  // f32 atomic writeback fan-in (z * tile bytes) small.
  int z = std::max(1, std::min(n_chunks, (768 + tiles - 1) / std::max(1, tiles)));
  g16::ConvMeta cm{};
  auto& zp = zero_page(A.device(), A.scalar_type());
  // one launch for the whole batch: grid.z = bsz * split (ViT attention
  // backward has ~1.5K batches; a per-batch host loop dominated its step)
  if (bsz > 1) z = std::max(1, std::min(n_chunks, (int)(2048 / bsz) + 1));
  const bool wide = I >= 128 && J >= 128;  // 128x128 tiles for big GEMMs
  // 256x256 tiles on 8 waves for the really big shapes: halves the per-
  // operand HBM re-reads (the 128x128 traffic roofline measured ~450-510 TF
  // on the ViT wgrad shapes — the same band rocBLAS hits).
  // PDT_TN_TILE=256 forces the 256 route (A/B measurements); default keeps
  // the measured-faster 128 tiles.
  static const char* tn_tile = getenv("PDT_TN_TILE");
  const bool xwide = (tn_tile && tn_tile[0] == '2') && I >= 256 && J >= 256 &&
                     M >= 4096;
  if (wide) {
    tiles = ((J + 127) / 128) * ((I + 127) / 128);
    // measured z sweep (ViT wgrad shapes, 1xMI355X): ~1024 blocks (= 2 full
    // occupancy rounds) beats the old ~512 target by 18-27% — more
    // independent m-sweeps per CU hide the staging latency; beyond ~3x the
    // atomic fan-in starts to cost.
    z = std::max(1,
                 std::min(n_chunks, (1024 + tiles - 1) / std::max(1, tiles)));
    if (bsz > 1) z = std::max(1, std::min(n_chunks, (int)(2048 / bsz) + 1));
    static const char* e_z = getenv("PDT_TN_Z");  // split-M sweep knob
    if (e_z && bsz == 1) z = std::max(1, std::min(n_chunks, atoi(e_z)));
  }
  if (xwide) {
    // one 512-thread WG per CU: target ~256 resident blocks (one full round)
    tiles = ((J + 255) / 256) * ((I + 255) / 256);
    z = std::max(1, std::min(n_chunks, (256 + tiles - 1) / std::max(1, tiles)));
    if (bsz > 1) z = std::max(1, std::min(n_chunks, (int)(1024 / bsz) + 1));
  }
  auto run = [&](auto tag) {
    using t16 = decltype(tag);
    if (xwide) {
      dim3 grid((J + 255) / 256, (I + 255) / 256, (unsigned)(bsz * z));
      hipLaunchKernelGGL(
          (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_PLAIN, 8, 8, 2, 4, 3>),
          grid, dim3(512), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J,
          /*ldc=*/J, cm, (long long)M * I, (long long)M * J,
          (long long)I * J, z);
      return;
    }
    if (wide) {
      dim3 grid((J + 127) / 128, (I + 127) / 128, (unsigned)(bsz * z));
      // PDT_TN_KERN=old falls back to the [32m][16ch]-image kernel (A/B)
      static const char* e_kern = getenv("PDT_TN_KERN");
      if (e_kern && e_kern[0] == 'o') {
        hipLaunchKernelGGL(
            (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_PLAIN, 4, 4>), grid,
            dim3(g16::THREADS), 0, stream,
            reinterpret_cast<const t16*>(A.data_ptr()),
            reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
            reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J,
            /*ldc=*/J, cm, (long long)M * I, (long long)M * J,
            (long long)I * J, z);
        return;
      }
      hipLaunchKernelGGL(
          (g16::gemm_tn_plain_kernel<t16, 4, 4, 2, 2>), grid,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J,
          /*ldc=*/J, (long long)M * I, (long long)M * J, (long long)I * J, z,
          nullptr);
      return;
    }
    dim3 grid((J + 63) / 64, (I + 63) / 64, (unsigned)(bsz * z));
    hipLaunchKernelGGL(
        (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_PLAIN>), grid,
        dim3(g16::THREADS), 0, stream,
        reinterpret_cast<const t16*>(A.data_ptr()),
        reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
        reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J,
        /*ldc=*/J, cm, (long long)M * I, (long long)M * J, (long long)I * J,
        z);
  };
  if (A.scalar_type() == torch::kBFloat16) run(bf16{});
  else run(_Float16{});
  return C;
}

// TN GEMM + bias grad in one pass: returns {C=f32[I,J], db=f32[I]} where
// db = column sums of A (the dy operand) — the kernel already stages every
// dy tile, so the separate col_sum pass over dy disappears (linear layers'
// backward).  2-D only.
std::vector<torch::Tensor> bmm_tn_bias_bf16(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2);
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  int M = (int)A.size(0), I = (int)A.size(1);
  int J = (int)B.size(1);
  TORCH_CHECK((int)B.size(0) == M);
  auto C = torch::zeros({I, J}, A.options().dtype(torch::kFloat32));
  auto db = torch::zeros({I}, A.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  int n_chunks = (M + 31) / 32;
  g16::ConvMeta cm{};
  auto& zp = zero_page(A.device(), A.scalar_type());
  static const char* tn_tile2 = getenv("PDT_TN_TILE");
  // 256x256 tiles on the 128-B-row tn_plain kernel (PDT_TN_TILE=3):
  // measured SLOWER than 128 tiles (qkv 380 vs 414 TF, proj 219 vs 352 —
  // the drop to one 8-wave block/CU costs more than the halved re-read
  // traffic, which the per-XCD L2s largely absorb).  Kept as a documented
  // negative result; '2' = the older xwide TR route (also slower).
  const bool xwide = (tn_tile2 && tn_tile2[0] == '2') && I >= 256 &&
                     J >= 256 && M >= 4096;
  const bool big256 = (tn_tile2 && tn_tile2[0] == '3') && I >= 256 &&
                      J >= 256 && M >= 4096;
  const bool wide = I >= 128 && J >= 128;
  int tiles = (xwide || big256) ? ((J + 255) / 256) * ((I + 255) / 256)
              : wide ? ((J + 127) / 128) * ((I + 127) / 128)
                     : ((J + 63) / 64) * ((I + 63) / 64);
  int z = std::max(
      1, std::min(n_chunks,
                  ((xwide ? 256 : 1024) + tiles - 1) / std::max(1, tiles)));
  auto run = [&](auto tag) {
    using t16 = decltype(tag);
    if (big256) {
      dim3 grid((J + 255) / 256, (I + 255) / 256, (unsigned)z);
      hipLaunchKernelGGL(
          (g16::gemm_tn_plain_kernel<t16, 8, 8, 2, 4, 1>), grid,
          dim3(512), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J, J, 0, 0,
          0, z, db.data_ptr<float>());
      return;
    }
    if (xwide) {
      dim3 grid((J + 255) / 256, (I + 255) / 256, (unsigned)z);
      hipLaunchKernelGGL(
          (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_PLAIN, 8, 8, 2, 4, 3>),
          grid, dim3(512), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J, J, cm, 0, 0,
          0, z, db.data_ptr<float>());
      return;
    }
    if (wide) {
      dim3 grid((J + 127) / 128, (I + 127) / 128, (unsigned)z);
      hipLaunchKernelGGL(
          (g16::gemm_tn_plain_kernel<t16, 4, 4, 2, 2>), grid,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J, J, 0, 0,
          0, z, db.data_ptr<float>());
    } else {
      dim3 grid((J + 63) / 64, (I + 63) / 64, (unsigned)z);
      hipLaunchKernelGGL(
          (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_PLAIN>), grid,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(A.data_ptr()),
          reinterpret_cast<const t16*>(B.data_ptr()), C.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, I, J, J, cm, 0, 0,
          0, z, db.data_ptr<float>());
    }
  };
  if (A.scalar_type() == torch::kBFloat16) run(bf16{});
  else run(_Float16{});
  return {C, db};
}

namespace g16 {
// -------------------------------------- TN conv wgrad, segment-staged ----
// 3x3 stride-1 wgrad with pow2 HO/WO: instead of gathering 9 shifted x
// tiles per chunk, stage dy + THREE spatial x row-tiles with a 2-column
// halo (the s-shift becomes a +s*32B tr-read base offset into the same
// tile).  Cuts the per-chunk glds count 40 -> 28 and the x LDS/gather
// traffic 3x; all addressing is shift arithmetic (pow2 gate).
//
// x tile layout: G = max(1, 32/WO) spatial segments of L = min(32, WO)
// output columns each, stored as q = seg*(L+2) + col rows of [16 ch]
// (E = G*(L+2) <= 48 rows, image padded to 64 so each is exactly 2 glds).
// A fragment m-window of 4 always sits inside one segment (L >= 4, windows
// 4-aligned); the second tr read's row jump is L>=8 ? +4 : +6 (uniform).
// TSPLIT=3: the 9-tap accumulator set is split by kernel ROW across a
// 12-wave block (wave = rt * 4 + quadrant) — 48 acc VGPRs per wave instead
// of 144 lifts occupancy 2 -> 3 waves/SIMD, and each wave reads only its
// own rt's x tiles (per-CU tr-read traffic -40%).  dw writes were already
// atomicAdd, so the rt-partials need no extra reduction.
// RSHIFT (TSPLIT==1 only): rolling x-tile ring.  One chunk advances the
// output row by RSHIFT (= 32/WO), so chunk ch+1's kernel-row-r tile IS
// chunk ch's row r+RSHIFT tile — a (3+RSHIFT)-slot ring restages RSHIFT
// tiles per chunk instead of three (WO=32: 12 glds vs 28; WO=16: 20),
// except at image boundaries (full restage behind an extra barrier).
template <typename T16, int TSPLIT = 1, int RSHIFT = 0>
__global__ __launch_bounds__(TSPLIT == 3 ? 768 : THREADS,
                             TSPLIT == 3 ? 1 : 2) void gemm_wgrad_seg_kernel(
    const T16* __restrict__ dy, const T16* __restrict__ x,
    float* __restrict__ dw, const T16* __restrict__ zpad, int Mtot,
    int I /*Kout*/, int J /*Cin*/, long long ldc, ConvMeta cm, int lgWO,
    int lgHO, int zsplit) {
  constexpr bool RING = RSHIFT > 0;
  constexpr int SLOTS = 3 + RSHIFT;
  constexpr int BI = 64, BJ = 64, BMC = 32;
  constexpr int IMG_A = 32 * 16;   // dy images: [32 m][16 ch]
  constexpr int IMG_X = 64 * 16;   // x images: [<=48 q rows][16 ch], padded
  constexpr int TILE_A = 4 * IMG_A;       // 4 KiB
  constexpr int TILE_X = 4 * IMG_X;       // 8 KiB per kernel row r
  typedef short v4s __attribute__((ext_vector_type(4)));
  using vec16 = typename M16<T16>::vec;
  // RING layout: [2 dy bufs][4 x rt-slots] = 40 KB; else the classic
  // [2][dy + 3 x tiles] = 56 KB
  __shared__ __attribute__((aligned(16)))
      T16 lds[RING ? (2 * TILE_A + SLOTS * TILE_X)
                   : 2 * (TILE_A + 3 * TILE_X)];

  // XCD-contiguous remap (private per-XCD L2s — see gemm_wgrad_tr_kernel):
  // blocks sharing a zidx read the same dy/x chunks; keep them on one L2.
  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  {
    const int gx = gridDim.x, gy = gridDim.y;
    const long long nwg = (long long)gx * gy * gridDim.z;
    long long id = ((long long)bz * gy + by) * gx + bx;
    if (nwg >= 16) {
      const long long q = nwg >> 3, r = nwg & 7;
      const int xcd = (int)(id & 7);
      const long long idx = id >> 3;
      id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
    }
    bx = (int)(id % gx);
    const long long t = id / gx;
    by = (int)(t % gy);
    bz = (int)(t / gy);
  }
  const int Cin = J;
  const int i0 = by * BI;
  const int j0 = bx * BJ;
  const int n_chunks = (Mtot + BMC - 1) / BMC;
  const int per_z = (n_chunks + zsplit - 1) / zsplit;
  const int ch0 = bz * per_z;
  const int ch1 = min(n_chunks, ch0 + per_z);
  if (ch0 >= ch1) return;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int w4 = TSPLIT == 3 ? (wave & 3) : wave;
  const int rtw = TSPLIT == 3 ? (wave >> 2) : 0;  // this wave's kernel row
  const int wm = (w4 >> 1) * 32, wn = (w4 & 1) * 32;
  const int sm = lane >> 1;
  const int sh8 = (lane & 1) * 8;
  const int L = min(32, 1 << lgWO);  // seg length (pow2)
  const int lgL = lgWO < 5 ? lgWO : 5;
  const int segs = max(1, 32 >> lgL);
  const int E = segs * (L + 2);
  const int rcpL2 = 65536 / (L + 2) + 1;  // magic recip (q <= 63)

  f32x16 acc[TSPLIT == 3 ? 3 : 9] = {};

  // ---- stage chunk: dy (4 glds) + 3 x row-tiles (8 glds each) ----
  // RING helpers: dy buffers at lds[0..2*TILE_A), x slots after
  auto stage_dy_ring = [&](int buf, int ch) {
    if (!RING) return;
    const int m0 = ch * BMC;
    for (int u = wave; u < 4; u += 4) {
      const T16* src = zpad;
      const int gm = m0 + sm;
      const int ii = i0 + u * 16 + sh8;
      if (gm < Mtot && ii < I) src = dy + (long long)gm * I + ii;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)(
              lds + buf * TILE_A + u * IMG_A),
          16, 0, 0);
    }
  };
  auto stage_x_slot = [&](int slot, int ch, int rt) {
    if (!RING) return;
    const int m0 = ch * BMC;
    T16* dst0 = lds + 2 * TILE_A + slot * TILE_X;
    for (int u = wave & 3; u < 8; u += 4) {
      const int g2 = (u >> 2) & 1;
      const int ig = u & 3;
      const int q = g2 * 32 + sm;
      const T16* src = zpad;
      if (q < E) {
        const int seg = (q * rcpL2) >> 16;
        const int c = q - seg * (L + 2);
        const long long m_seg = (long long)m0 + (long long)seg * L;
        if (m_seg < Mtot) {
          const int wo0 = (int)(m_seg & ((1 << lgWO) - 1));
          const long long t = m_seg >> lgWO;
          const int ho = (int)(t & ((1 << lgHO) - 1));
          const int n = (int)(t >> lgHO);
          const int hi = ho - cm.pad + rt;
          const int wi = wo0 - cm.pad + c;
          const int jj = j0 + ig * 16 + sh8;
          if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W && jj < Cin)
            src = x + (((long long)n * cm.H + hi) * cm.W + wi) * Cin + jj;
        }
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)(
              dst0 + ig * IMG_X + g2 * (32 * 16)),
          16, 0, 0);
    }
  };
  auto stage = [&](int buf, int ch) {
    if (RING) return;
    const int m0 = ch * BMC;
    T16* base = lds + buf * (TILE_A + 3 * TILE_X);
    for (int u = wave; u < 4 + 24; u += (TSPLIT == 3 ? 12 : 4)) {
      const T16* src = zpad;
      T16* dst;
      if (u < 4) {  // dy image u
        const int gm = m0 + sm;
        const int ii = i0 + u * 16 + sh8;
        if (gm < Mtot && ii < I) src = dy + (long long)gm * I + ii;
        dst = base + u * IMG_A;
      } else {
        const int v = u - 4;          // 0..23
        const int rt = v / 8;         // kernel row 0..2
        const int g2 = (v / 4) & 1;   // which 32-row half of the image
        const int ig = v & 3;         // 16-ch group
        const int q = g2 * 32 + sm;   // lds row within the image
        dst = base + TILE_A + rt * TILE_X + ig * IMG_X + g2 * (32 * 16);
        if (q < E) {
          const int seg = (q * rcpL2) >> 16;
          const int c = q - seg * (L + 2);
          const long long m_seg = (long long)m0 + (long long)seg * L;
          if (m_seg < Mtot) {
            const int wo0 = (int)(m_seg & ((1 << lgWO) - 1));
            const long long t = m_seg >> lgWO;
            const int ho = (int)(t & ((1 << lgHO) - 1));
            const int n = (int)(t >> lgHO);
            const int hi = ho - cm.pad + rt;
            const int wi = wo0 - cm.pad + c;
            const int jj = j0 + ig * 16 + sh8;
            if (hi >= 0 && hi < cm.H && wi >= 0 && wi < cm.W && jj < Cin)
              src = x + (((long long)n * cm.H + hi) * cm.W + wi) * Cin + jj;
          }
        }
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  const int ks = lane >> 5;
  const unsigned tr_lane_off =
      (unsigned)((ks * 8 + ((lane & 15) >> 2)) * 32 + (lane & 3) * 8);
  const int img_sel = (lane >> 4) & 1;
  // x-tile q rows for this lane's two m-windows (kh = 0/1), plus the
  // uniform row jump of the second tr read of each window pair
  const int jump = (L >= 8 ? 4 : 6) * 32;  // bytes = rows*32B
  int qw[2];
#pragma unroll
  for (int kh = 0; kh < 2; ++kh) {
    const int mw = kh * 16 + ks * 8;
    qw[kh] = (mw >> lgL) * (L + 2) + (mw & (L - 1));
  }
#define LDS_BYTE(p)                                           \
  ((unsigned)(unsigned long long)(__attribute__((            \
      address_space(3))) const T16*)(p))

  int r0 = 0;  // RING rotation base
  if (RING) {
    stage_dy_ring(0, ch0);
    stage_x_slot(0, ch0, 0);
    stage_x_slot(1, ch0, 1);
    stage_x_slot(2, ch0, 2);
  } else {
    stage(0, ch0);
  }
  __syncthreads();

  for (int ch = ch0; ch < ch1; ++ch) {
    const int buf = (ch - ch0) & 1;
    const bool more = ch + 1 < ch1;
    // RING: does the NEXT chunk start a new image (ho wrap)?  Then the
    // shift-by-one identity breaks and all three tiles restage.
    const bool wrap =
        RING && ((int)((((long long)ch * BMC + BMC - 1) >> lgWO) &
                       ((1 << lgHO) - 1)) == (1 << lgHO) - 1);
    if (more) {
      if (RING) {
        stage_dy_ring(buf ^ 1, ch + 1);
        if (!wrap) {
#pragma unroll
          for (int rr = 3 - RSHIFT; rr < 3; ++rr)
            stage_x_slot((r0 + rr + RSHIFT) % SLOTS, ch + 1, rr);
        }
      } else {
        stage(buf ^ 1, ch + 1);
      }
    }
    const T16* base = RING ? lds + buf * TILE_A
                           : lds + buf * (TILE_A + 3 * TILE_X);

    // A fragments (dy): 4 tr reads issued, drained by the FIRST counted
    // wait of the B pipeline below (not a full lgkmcnt(0) drain)
    vec16 af[2];
    {
      const unsigned a0 =
          LDS_BYTE(base + ((wm >> 4) + img_sel) * IMG_A) + tr_lane_off;
      v4s l0, h0, l1, h1;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %4 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %4 offset:128\n\t"
          "ds_read_b64_tr_b16 %2, %4 offset:512\n\t"
          "ds_read_b64_tr_b16 %3, %4 offset:640\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(l0), "=&v"(h0), "=&v"(l1), "=&v"(h1)
          : "v"(a0), "v"(a0));
      reinterpret_cast<v4s*>(&af[0])[0] = l0;
      reinterpret_cast<v4s*>(&af[0])[1] = h0;
      reinterpret_cast<v4s*>(&af[1])[0] = l1;
      reinterpret_cast<v4s*>(&af[1])[1] = h1;
    }
    // B-pair software pipeline: 18 (2x tr-read, MFMA) pairs; pair i+1's
    // reads are issued BEFORE pair i's MFMA and the wait is a counted
    // lgkmcnt(2) (pair i done, pair i+1 pending) — the old full
    // lgkmcnt(0) before every MFMA serialized LDS-read -> MFMA and kept
    // the matrix pipe half-fed (round-1 PMC stopping point).
    {
      const unsigned lane_b =
          (unsigned)(((lane & 15) >> 2) * 32 + (lane & 3) * 8);
      // 3-slot rotation: the ds_read that REWRITES a slot is issued two
      // MFMAs after the MFMA that sourced it (a 2-slot ping-pong put the
      // async LDS write inside the in-flight MFMA's source-read window —
      // WAR hazard, caught by the cold-launch stress test)
      v4s pb[3][2];
      constexpr int NPAIR = TSPLIT == 3 ? 6 : 18;
      auto issueB = [&](int slot, int idx) {
        const int rt = TSPLIT == 3 ? rtw : idx / 6;
        const int rem = idx % 6, s2 = rem >> 1, kh = rem & 1;
        const T16* timg =
            RING ? lds + 2 * TILE_A + ((r0 + rt) % SLOTS) * TILE_X +
                       (img_sel + (wn >> 4)) * IMG_X
                 : base + TILE_A + rt * TILE_X +
                       (img_sel + (wn >> 4)) * IMG_X;
        const unsigned b0 =
            LDS_BYTE(timg) + (unsigned)((qw[kh] + s2) * 32) + lane_b;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %2 offset:0\n\t"
            "ds_read_b64_tr_b16 %1, %3 offset:0"
            : "=&v"(pb[slot][0]), "=&v"(pb[slot][1])
            : "v"(b0), "v"(b0 + (unsigned)jump));
      };
      issueB(0, 0);
#pragma unroll
      for (int idx = 0; idx < NPAIR; ++idx) {
        const int slot = idx % 3;
        // the wait carries the pair's registers as "+v" operands: the
        // compiler must order the copies/MFMA that consume them AFTER the
        // wait (a bare asm waitcnt does not stop it hoisting the v_movs
        // above the wait — stale-register NaNs at some shapes)
        if (idx + 1 < NPAIR) {
          issueB((idx + 1) % 3, idx + 1);
          asm volatile("s_waitcnt lgkmcnt(2)"
                       : "+v"(pb[slot][0]), "+v"(pb[slot][1]));
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)"
                       : "+v"(pb[slot][0]), "+v"(pb[slot][1]));
        }
        const int rt = TSPLIT == 3 ? 0 : idx / 6;
        const int rem = idx % 6, s2 = rem >> 1, kh = rem & 1;
        vec16 bf;
        reinterpret_cast<v4s*>(&bf)[0] = pb[slot][0];
        reinterpret_cast<v4s*>(&bf)[1] = pb[slot][1];
        acc[rt * 3 + s2] = M16<T16>::mma32(af[kh], bf, acc[rt * 3 + s2]);
      }
    }
    if (more) __syncthreads();
    if (RING && more) {
      if (wrap) {
        // new image: restage all three rt tiles (slots now all free) and
        // reset the rotation — one extra barrier every HO chunks
        stage_x_slot(0, ch + 1, 0);
        stage_x_slot(1, ch + 1, 1);
        stage_x_slot(2, ch + 1, 2);
        r0 = 0;
        __syncthreads();
      } else {
        r0 = (r0 + RSHIFT) % SLOTS;
      }
    }
  }

  // ---- writeback (32x32x16 C/D layout; TSPLIT: only this wave's taps) ----
#pragma unroll
  for (int t3 = 0; t3 < (TSPLIT == 3 ? 3 : 9); ++t3) {
    const int tap = TSPLIT == 3 ? rtw * 3 + t3 : t3;
    const long long coff = (long long)tap * Cin;
    const int col = j0 + wn + (lane & 31);
    if (col >= Cin) continue;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int row = i0 + wm + (reg & 3) + 8 * (reg >> 2) + 4 * ks;
      if (row >= I) continue;
      atomicAdd(&dw[(long long)row * ldc + coff + col], acc[t3][reg]);
    }
  }
}
}  // namespace g16

// conv wgrad: dw[Kout,R,S,C] (f32) from dy[N,HO,WO,Kout], x[N,H,W,C]
torch::Tensor conv2d_wgrad_bf16(torch::Tensor dy, torch::Tensor x,
                                int64_t stride, int64_t pad, int64_t R,
                                int64_t S) {
  int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
      Cin = (int)x.size(3);
  int HO = (int)dy.size(1), WO = (int)dy.size(2), Kout = (int)dy.size(3);
  int cl = log2_exact(Cin);
  TORCH_CHECK(cl >= 3, "conv wgrad fast path needs pow2 C (stem uses im2col)");
  int M = N * HO * WO;
  auto dw = torch::zeros({(long long)Kout, R, S, (long long)Cin},
                         x.options().dtype(torch::kFloat32));
  auto& zp = zero_page(x.device(), x.scalar_type());
  auto stream = c10::hip::getCurrentHIPStream();
  g16::ConvMeta cm{H, W, cl, (int)S, (int)R, (int)stride, (int)pad, HO, WO};
  int n_chunks = (M + 31) / 32;
  int tiles = ((Cin + 63) / 64) * ((Kout + 63) / 64);
  // block-count target: more split-M blocks hide the staging latency (the
  // plain-TN z sweep measured +18-27% at ~1024 vs ~512); PDT_CONV_WGRAD_B
  // overrides for sweeps.
  // e2e bench sweep at the flagship batch sizes AFTER the wide-N fix:
  // 512 edges 1024 on r18 (103.5k vs 103.0k wide-on; 106.1k vs 105.8k
  // wide-off) — the earlier 1024-wins result was contaminated by the
  // unwritten-half-of-C bug in the fused conv forwards.
  static const char* e_cwb = getenv("PDT_CONV_WGRAD_B");
  const int btarget = e_cwb ? atoi(e_cwb) : 512;
  int z = std::max(1,
                   std::min(n_chunks, (btarget + tiles - 1) / std::max(1, tiles)));
  const long long ldc = (long long)R * S * Cin;
  dim3 grid((Cin + 63) / 64, (Kout + 63) / 64, z);
  auto pow2l = [](int v) {
    int l = 0;
    while ((1 << l) < v) ++l;
    return ((1 << l) == v && v >= 4) ? l : -1;
  };
  const int lgWO = pow2l(WO), lgHO = pow2l(HO);
  auto run = [&](auto tag) {
    using t16 = decltype(tag);
    if (R == 3 && S == 3 && stride == 1 && lgWO >= 2 && lgHO >= 2) {
      // segment-staged fast path (stride-1, pow2 spatial): 28 glds/chunk
      // instead of 40, x gathered once per kernel ROW not per tap.
      // Measured: the 12-wave rt-split variant (92 VGPR, 3 waves/SIMD,
      // -40% per-CU tr traffic) LOSES to the 4-wave form (19.5 vs 18.4 ms
      // summed wgrad; bench 113.8k vs 115.7k) — one 12-wave block
      // locksteps every wave at the per-chunk barrier, while two 4-wave
      // blocks interleave around each other's glds drains.  Kept behind
      // PDT_WGRAD_TSPLIT=3 as a documented negative result.
      static const char* e_ts = getenv("PDT_WGRAD_TSPLIT");
      // rolling-ring x reuse (WO==32): default ON — L1 wgrad 1.30 ->
      // 0.94 ms (+38%), r18 bench 114.9k -> 118.8k same-box.
      // PDT_WGRAD_RING=0 reverts.
      static const char* e_rg = getenv("PDT_WGRAD_RING");
      const bool ring =
          !(e_rg && e_rg[0] == '0') && lgWO >= 4 && lgHO >= 1;
      if (ring && lgWO == 5) {
        hipLaunchKernelGGL((g16::gemm_wgrad_seg_kernel<t16, 1, 1>), grid,
                           dim3(g16::THREADS), 0, stream,
                           reinterpret_cast<const t16*>(dy.data_ptr()),
                           reinterpret_cast<const t16*>(x.data_ptr()),
                           dw.data_ptr<float>(),
                           reinterpret_cast<const t16*>(zp.data_ptr()), M,
                           Kout, Cin, ldc, cm, lgWO, lgHO, (int)grid.z);
      } else if (ring && lgWO == 4) {
        hipLaunchKernelGGL((g16::gemm_wgrad_seg_kernel<t16, 1, 2>), grid,
                           dim3(g16::THREADS), 0, stream,
                           reinterpret_cast<const t16*>(dy.data_ptr()),
                           reinterpret_cast<const t16*>(x.data_ptr()),
                           dw.data_ptr<float>(),
                           reinterpret_cast<const t16*>(zp.data_ptr()), M,
                           Kout, Cin, ldc, cm, lgWO, lgHO, (int)grid.z);
      } else if (e_ts && e_ts[0] == '3')
        hipLaunchKernelGGL((g16::gemm_wgrad_seg_kernel<t16, 3>), grid,
                           dim3(768), 0, stream,
                           reinterpret_cast<const t16*>(dy.data_ptr()),
                           reinterpret_cast<const t16*>(x.data_ptr()),
                           dw.data_ptr<float>(),
                           reinterpret_cast<const t16*>(zp.data_ptr()), M,
                           Kout, Cin, ldc, cm, lgWO, lgHO, (int)grid.z);
      else
        hipLaunchKernelGGL((g16::gemm_wgrad_seg_kernel<t16>), grid,
                           dim3(g16::THREADS), 0, stream,
                           reinterpret_cast<const t16*>(dy.data_ptr()),
                           reinterpret_cast<const t16*>(x.data_ptr()),
                           dw.data_ptr<float>(),
                           reinterpret_cast<const t16*>(zp.data_ptr()), M,
                           Kout, Cin, ldc, cm, lgWO, lgHO, (int)grid.z);
    } else if (R == 3 && S == 3) {
      hipLaunchKernelGGL((g16::gemm_wgrad_tr_kernel<t16, 9>), grid,
                         dim3(g16::THREADS), 0, stream,
                         reinterpret_cast<const t16*>(dy.data_ptr()),
                         reinterpret_cast<const t16*>(x.data_ptr()),
                         dw.data_ptr<float>(),
                         reinterpret_cast<const t16*>(zp.data_ptr()), M, Kout,
                         Cin, ldc, cm, 0, 0, 0, (int)grid.z);
    } else if (R == 1 && S == 1 && Kout >= 128 && Cin >= 128) {
      // wide 128x128 tiles: the bottleneck 1x1s (up to 512x2048 dw) re-read
      // dy grid.x times; halving grid.x halves that traffic
      int tiles2 = ((Cin + 127) / 128) * ((Kout + 127) / 128);
      int z2 = std::max(
          1,
          std::min(n_chunks, (btarget + tiles2 - 1) / std::max(1, tiles2)));
      dim3 grid2((Cin + 127) / 128, (Kout + 127) / 128, (unsigned)z2);
      hipLaunchKernelGGL(
          (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_CONV, 4, 4>), grid2,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(dy.data_ptr()),
          reinterpret_cast<const t16*>(x.data_ptr()), dw.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, Kout, Cin, ldc, cm,
          0, 0, 0, z2);
    } else if (R == 1 && S == 1) {
      hipLaunchKernelGGL((g16::gemm_wgrad_tr_kernel<t16, 1>), grid,
                         dim3(g16::THREADS), 0, stream,
                         reinterpret_cast<const t16*>(dy.data_ptr()),
                         reinterpret_cast<const t16*>(x.data_ptr()),
                         dw.data_ptr<float>(),
                         reinterpret_cast<const t16*>(zp.data_ptr()), M, Kout,
                         Cin, ldc, cm, 0, 0, 0, (int)grid.z);
    } else {  // generic R x S: taps-on-J, ONE launch (dw[I][R*S*Cin] as a
      // single TN GEMM; col j = tap*Cin + c gathers x at tap's shift).
      // The per-tap TN alternative re-read all of dy R*S times — the 7x7
      // ResNet-50 stem spent 13.4 ms/step (28% of the step) there.
      const int Jl = (int)ldc;  // = R*S*Cin
      int tiles2 = ((Jl + 127) / 128) * ((Kout + 63) / 64);
      int z2 = std::max(
          1,
          std::min(n_chunks, (btarget + tiles2 - 1) / std::max(1, tiles2)));
      dim3 grid2((Jl + 127) / 128, (Kout + 63) / 64, (unsigned)z2);
      hipLaunchKernelGGL(
          (g16::gemm_wgrad_tr_kernel<t16, 1, g16::MODE_CONVJ, 2, 4>), grid2,
          dim3(g16::THREADS), 0, stream,
          reinterpret_cast<const t16*>(dy.data_ptr()),
          reinterpret_cast<const t16*>(x.data_ptr()), dw.data_ptr<float>(),
          reinterpret_cast<const t16*>(zp.data_ptr()), M, Kout, Jl, ldc, cm,
          0, 0, 0, z2);
    }
  };
  if (x.scalar_type() == torch::kBFloat16) run(bf16{});
  else run(_Float16{});
  return dw;
}
