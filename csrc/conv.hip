// NHWC implicit-GEMM convolutions on MFMA (bf16 in, fp32 accumulate).
//
// Covers the reference model zoo's conv shapes: bias-free
// 3x3/1x1 ResNet convs at 64-512 channels on 32->4 px (resnet.hpp:3-9) and
// biased k3/k5 MNIST convs (event.cpp:51-83, nnet.hpp). No im2col
// materialization: the GEMM A-operand addresses x through the im2col map
// while staging tiles into LDS.
//
// GEMM views (all row-major, K-innermost "NT" tiles like gemm.hip):
//  fwd   y[np, k]  = im2col[np, rsc]  @ w[k, rsc]^T          np=(n,ho,wo)
//  dgrad dx[np, c] = col(dy)[np, rsk] @ wT[c, rsk]^T         np=(n,h,w)
//  wgrad dw[k, rsc] = dy[np, k]^T @ im2col[np, rsc]          (TN, split-NP
//        across blocks, fp32 atomic accumulate)
//
// Address-math discipline (the first profile showed these kernels
// VALU-bound on div/mod chains): each thread's output-pixel decomposition
// is hoisted OUT of the k-loop, and the reduction index (r,s,c)/(r,s,k)
// advances by an incremental cursor (add + wrap) instead of div/mod —
// FAST template path, used whenever the innermost dim is a multiple of 8
// (every ResNet conv). The generic slow path remains for C=3 stems and
// the small CNN channel counts.

#include "common.h"

namespace eg {

namespace conv {

constexpr int CBM = 128, CBN = 64, CBK = 64;
constexpr int CLDK = CBK + 8;

struct Geom {
  int N, H, W, C;     // input
  int K, R, S;        // filter
  int Ho, Wo;         // output
  int stride, pad;
};

__device__ __forceinline__ bf16x8 lds8(const __bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

__device__ __forceinline__ void zero8(__bf16* dst) {
#pragma unroll
  for (int j = 0; j < 8; ++j) dst[j] = (__bf16)0.f;
}

// XCD-aware tile remap (bijective; guide §6). The dispatcher places linear
// block b on XCD b % 8, so consecutive blockIdx.x tiles — whose im2col
// windows overlap in x — land on DIFFERENT 4 MiB L2s and re-read HBM/L3.
// Remap so each XCD owns a CONTIGUOUS M-tile range across all N-tiles:
// its L2 then serves the R*S-fold A re-reads.
__device__ __forceinline__ long xcd_tile_remap() {
  const long T = (long)gridDim.x * gridDim.y;
  const long orig = (long)blockIdx.x + (long)gridDim.x * blockIdx.y;
  const long q = T >> 3, r = T & 7;
  const long xcd = orig & 7, idx = orig >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// 3D variant for the wgrad grid (k-tiles × rsc-tiles × np-slices): all the
// (k, rsc) tiles of one np-slice read the SAME x/dy rows (the rsc tiles
// re-read x R*S-fold), so each XCD gets contiguous z-major chunks.
__device__ __forceinline__ long xcd_tile_remap3() {
  const long T = (long)gridDim.x * gridDim.y * gridDim.z;
  const long orig = (long)blockIdx.x +
                    (long)gridDim.x * (blockIdx.y +
                                       (long)gridDim.y * blockIdx.z);
  const long q = T >> 3, r = T & 7;
  const long xcd = orig & 7, idx = orig >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// ---- generic (slow) stagers: per-element guarded loads -------------------

__device__ __forceinline__ void stage_im2col_slow(const bf16* __restrict__ x,
                                                  const Geom& g, bool mvalid,
                                                  int n, int ho, int wo,
                                                  int red, __bf16* dst) {
  int RSC = g.R * g.S * g.C;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int rr = red + j;
    float v = 0.f;
    if (mvalid && rr < RSC) {
      int c = rr % g.C;
      int rs2 = rr / g.C;
      int r2 = rs2 / g.S, s2 = rs2 % g.S;
      int h2 = ho * g.stride - g.pad + r2;
      int w2 = wo * g.stride - g.pad + s2;
      if ((unsigned)h2 < (unsigned)g.H && (unsigned)w2 < (unsigned)g.W)
        v = b2f(x[(((long)n * g.H + h2) * g.W + w2) * g.C + c]);
    }
    dst[j] = (__bf16)v;
  }
}

__device__ __forceinline__ void stage_dgrad_slow(const bf16* __restrict__ dy,
                                                 const Geom& g, bool mvalid,
                                                 int n, int h, int w, int red,
                                                 __bf16* dst) {
  int RSK = g.R * g.S * g.K;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int rr = red + j;
    float v = 0.f;
    if (mvalid && rr < RSK) {
      int kk = rr % g.K;
      int rs = rr / g.K;
      int r = rs / g.S, s = rs % g.S;
      int hq = h + g.pad - r, wq = w + g.pad - s;
      if (hq >= 0 && wq >= 0 && hq % g.stride == 0 && wq % g.stride == 0) {
        int ho = hq / g.stride, wo = wq / g.stride;
        if (ho < g.Ho && wo < g.Wo)
          v = b2f(dy[(((long)n * g.Ho + ho) * g.Wo + wo) * g.K + kk]);
      }
    }
    dst[j] = (__bf16)v;
  }
}

// guarded vec8 row load of a [rows, cols] row-major bf16 matrix
__device__ __forceinline__ void row8(const bf16* g, long rows, long cols,
                                     long r, long c, __bf16* dst) {
  if (r < rows && c + 8 <= cols && ((r * cols + c) & 7) == 0) {
    *reinterpret_cast<s16x8*>(dst) =
        *reinterpret_cast<const s16x8*>(g + r * cols + c);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dst[j] = (__bf16)((r < rows && c + j < cols)
                            ? b2f(g[r * cols + c + j]) : 0.f);
  }
}

// ---- fwd & dgrad main loop ----------------------------------------------
// TBM x 64 block tile (TBM 128 for chip-filling shapes, 64 when the grid
// would drop under ~1.5 blocks/CU), BK=64, TBM*4 threads = TBM/16 waves of
// 32x32 wave-tiles. Software-pipelined register staging (guide G15/T14):
// the NEXT tile's global loads are issued right after this tile's LDS
// write, so HBM latency hides under the MFMA phase.

// backward-stats fusion pack (MODE 1 + BST): while writing dx — which IS
// the downstream-in-backward BN's incoming dy — accumulate that BN's
// sum(dy') and sum(dy' * xhat) into its grad views. y1 = the BN's output
// (this conv's forward input, used for the relu gating), x1 = the BN's
// input, mean/invstd its saved batch stats. Mirrors the MODE-0 STATS
// forward fusion; the BN's backward then skips its standalone stats pass.
struct BPack {
  const bf16* y1;
  const bf16* x1;
  const float* mean;
  const float* invstd;
  float* sum_dy;
  float* sum_dyx;
};

template <int MODE, bool FAST, int TBM, bool STATS = false, bool BST = false>
__global__ __launch_bounds__(TBM * 4) void conv_mm_kernel(
    const bf16* __restrict__ Asrc, const bf16* __restrict__ B,
    const float* __restrict__ bias, bf16* __restrict__ out, Geom g,
    long M, long N, long RED, int has_bias,
    float* __restrict__ stats_ws = nullptr, BPack bp = BPack{}) {
  constexpr int THREADS = TBM * 4;
  constexpr int BROWS = 512 / THREADS;  // B-staging rows per thread
  // ONE shared allocation (guide §5.4 trap 4a: a second __shared__ object
  // makes hipcc over-synchronize the k-loop — measured +14 waitcnts and an
  // extra barrier per step with separate stats arrays). Stats accumulators
  // live in a float-aliased tail of the same array.
  // double-buffered LDS for the TBM=64 (deep, issue-bound) shapes: one
  // barrier per k-tile instead of two. TBM=128 keeps a single buffer —
  // doubling its 27.6 KB LDS halves occupancy and measured net-slower.
  constexpr bool DB = (TBM == 64);
  constexpr bool ACC = STATS || BST;   // either stats epilogue form
  constexpr int BUFSZ = TBM * CLDK + CBN * CLDK;
  __shared__ __bf16 smem[(DB ? 2 : 1) * BUFSZ + (ACC ? 2 * CBN * 2 : 0)];
  float* s_sum = reinterpret_cast<float*>(smem + (DB ? 2 : 1) * BUFSZ);
  float* s_sq = s_sum + CBN;
  if (ACC && threadIdx.x < CBN) {
    s_sum[threadIdx.x] = 0.f;
    s_sq[threadIdx.x] = 0.f;
  }

  const long wg = xcd_tile_remap();  // m-major: contiguous M per XCD
  const long m0 = (wg / gridDim.y) * TBM;
  const long n0 = (wg % gridDim.y) * CBN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int lra = t >> 2;            // A staging row 0..TBM-1
  const int lka = (t & 3) * 8;       // A staging col {0,8,16,24}
  const int lrb = t >> 3;            // B staging row
  const int lkb = (t & 7) * 8;       // B staging col {0..56}

  // hoisted per-thread A-row pixel decomposition
  const long m = m0 + lra;
  const bool mvalid = m < M;
  int pn = 0, ph = 0, pw = 0;  // MODE0: (n, ho, wo); MODE1: (n, h, w)
  if (mvalid) {
    const int HW = MODE == 0 ? g.Ho * g.Wo : g.H * g.W;
    const int WW = MODE == 0 ? g.Wo : g.W;
    pn = (int)(m / HW);
    int rem = (int)(m % HW);
    ph = rem / WW;
    pw = rem % WW;
  }
  // FAST staging state (PMC: recomputed address chains made these kernels
  // VALU-bound at 14-18 VALU instructions per MFMA). Per thread:
  //  - a validity BITMASK over the R*S filter taps (computed once; the
  //    per-tap h/w range checks collapse to one shift+and);
  //  - an incremental source offset with CONSTANT wrap corrections
  //    (MODE 0: (s,c) is contiguous in NHWC x, so within a filter row the
  //    offset just advances by CBK; crossing to the next r adds
  //    (W-S)*C. MODE 1 (stride 1): s-wrap -2K, r-wrap K*(S-Wo-2)).
  // inner = c (MODE0) or k (MODE1); requires R*S <= 32 (host-gated).
  const int INNER = MODE == 0 ? g.C : g.K;
  unsigned long long vmask = 0;
  long aoff[2];
  int cur_i[2], rsn[2], scnt[2];
  long swrap_fix = 0, rwrap_fix = 0;
  if (FAST) {
    if (MODE == 0) {
      const int hb = ph * g.stride - g.pad;   // h = hb + r
      const int wb = pw * g.stride - g.pad;   // w = wb + s
      for (int r = 0; r < g.R; ++r)
        for (int s2 = 0; s2 < g.S; ++s2)
          if ((unsigned)(hb + r) < (unsigned)g.H &&
              (unsigned)(wb + s2) < (unsigned)g.W)
            vmask |= 1ull << (r * g.S + s2);
      swrap_fix = 0;
      rwrap_fix = (long)(g.W - g.S) * g.C;
      for (int p = 0; p < 2; ++p) {
        const int red = lka + 32 * p;
        cur_i[p] = red % INNER;
        rsn[p] = red / INNER;
        const int r = rsn[p] / g.S, s2 = rsn[p] % g.S;
        scnt[p] = s2;
        aoff[p] = (((long)pn * g.H + hb + r) * g.W + wb + s2) * g.C +
                  cur_i[p];
      }
    } else {
      const int hob = ph + g.pad;             // ho = hob - r (stride 1)
      const int wob = pw + g.pad;
      for (int r = 0; r < g.R; ++r)
        for (int s2 = 0; s2 < g.S; ++s2)
          if ((unsigned)(hob - r) < (unsigned)g.Ho &&
              (unsigned)(wob - s2) < (unsigned)g.Wo)
            vmask |= 1ull << (r * g.S + s2);
      swrap_fix = -2L * g.K;
      rwrap_fix = (long)g.K * (g.S - 2 - g.Wo);
      for (int p = 0; p < 2; ++p) {
        const int red = lka + 32 * p;
        cur_i[p] = red % INNER;
        rsn[p] = red / INNER;
        const int r = rsn[p] / g.S, s2 = rsn[p] % g.S;
        scnt[p] = s2;
        aoff[p] = (((long)pn * g.Ho + hob - r) * g.Wo + (wob - s2)) * g.K +
                  cur_i[p];
      }
    }
  }

  // B rows are fixed per thread (plain [N, RED] weight matrix): hoist the
  // guards and advance a raw pointer by CBK per tile
  const bf16* bptr[BROWS];
  bool bok[BROWS];
#pragma unroll
  for (int q = 0; q < BROWS; ++q) {
    const long rrow = n0 + lrb + q * (THREADS / 8);
    bok[q] = rrow < N;
    bptr[q] = B + rrow * RED + lkb;
  }

  __bf16 ra[2][8], rb[BROWS][8];

  auto stage = [&](long k0) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      if (FAST) {
        const bool valid = mvalid && ((vmask >> rsn[p]) & 1ull);
        if (valid) {
          *reinterpret_cast<s16x8*>(ra[p]) =
              *reinterpret_cast<const s16x8*>(Asrc + aoff[p]);
        } else {
          zero8(ra[p]);
        }
        aoff[p] += CBK;
        cur_i[p] += CBK;
        while (cur_i[p] >= INNER) {
          cur_i[p] -= INNER;
          rsn[p] += 1;
          if (++scnt[p] == g.S) {
            scnt[p] = 0;
            aoff[p] += rwrap_fix;
          } else {
            aoff[p] += swrap_fix;
          }
        }
      } else {
        int red = (int)k0 + lka + 32 * p;
        if (MODE == 0) {
          stage_im2col_slow(Asrc, g, mvalid, pn, ph, pw, red, ra[p]);
        } else {
          stage_dgrad_slow(Asrc, g, mvalid, pn, ph, pw, red, ra[p]);
        }
      }
    }
    if (FAST) {
#pragma unroll
      for (int q = 0; q < BROWS; ++q) {
        if (bok[q] && k0 + lkb < RED) {  // RED % 8 == 0 in FAST mode
          *reinterpret_cast<s16x8*>(rb[q]) =
              *reinterpret_cast<const s16x8*>(bptr[q]);
        } else {
          zero8(rb[q]);
        }
        bptr[q] += CBK;
      }
    } else {
#pragma unroll
      for (int q = 0; q < BROWS; ++q) {
        row8(B, N, RED, n0 + lrb + q * (THREADS / 8), k0 + lkb, rb[q]);
      }
    }
  };

  auto write_lds = [&](int pb) {
    __bf16* wA = smem + pb * BUFSZ;
    __bf16* wB = wA + TBM * CLDK;
    *reinterpret_cast<bf16x8*>(&wA[lra * CLDK + lka]) =
        *reinterpret_cast<bf16x8*>(ra[0]);
    *reinterpret_cast<bf16x8*>(&wA[lra * CLDK + lka + 32]) =
        *reinterpret_cast<bf16x8*>(ra[1]);
#pragma unroll
    for (int q = 0; q < BROWS; ++q) {
      *reinterpret_cast<bf16x8*>(
          &wB[(lrb + q * (THREADS / 8)) * CLDK + lkb]) =
          *reinterpret_cast<bf16x8*>(rb[q]);
    }
  };

  f32x4 acc[2][2] = {};
  stage(0);
  if (DB) {
    write_lds(0);                     // no readers yet: no barrier needed
    if (CBK < RED) stage(CBK);
  }
  int pb = 0;

  for (long k0 = 0; k0 < RED; k0 += CBK) {
    __syncthreads();  // LDS safe: prior tile's reads (and writes) done
    if (DB) {
      if (k0 + CBK < RED) {
        write_lds(pb ^ 1);            // tile k0+CBK from staged regs
        if (k0 + 2 * CBK < RED) stage(k0 + 2 * CBK);  // loads overlap MFMA
      }
    } else {
      write_lds(0);
      if (k0 + CBK < RED) stage(k0 + CBK);
      __syncthreads();
    }
    const __bf16* sA = smem + pb * BUFSZ;
    const __bf16* sB = sA + TBM * CLDK;

    const int ml = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kf = ks * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fi = 0; fi < 2; ++fi) {
        bf16x8 af = lds8(&sA[(wr * 32 + fi * 16 + ml) * CLDK + kf]);
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          bf16x8 bfr = lds8(&sB[(wc * 32 + fj * 16 + ml) * CLDK + kf]);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    if (DB) pb ^= 1;
  }

  const int cn = lane & 15;
  const int cm = (lane >> 4) * 4;
#pragma unroll
  for (int fj = 0; fj < 2; ++fj) {
    long nn = n0 + wc * 32 + fj * 16 + cn;
    if (nn >= N) continue;
    float bv = has_bias ? bias[nn] : 0.f;
    float bm = 0.f, bis = 0.f;
    if (BST) {
      bm = bp.mean[nn];
      bis = bp.invstd[nn];
    }
    float psum = 0.f, psq = 0.f;
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long mm = m0 + wr * 32 + fi * 16 + cm + r;
        if (mm >= M) continue;
        float v = acc[fi][fj][r] + bv;
        out[mm * N + nn] = f2b(v);
        if (STATS) {
          psum += v;
          psq += v * v;
        }
        if (BST) {
          // bf16 round-trip so the sums equal the unfused path, which
          // reads the stored bf16 dx
          const long o = mm * N + nn;
          float gg = b2f(bp.y1[o]) > 0.f ? b2f(f2b(v)) : 0.f;
          float xh = (b2f(bp.x1[o]) - bm) * bis;
          psum += gg;
          psq += gg * xh;
        }
      }
    }
    if (ACC) {
      // lanes l, l+16, l+32, l+48 hold the same output channel: reduce
      // across the wave first so only 16 lanes touch the LDS accumulators
      psum += __shfl_down(psum, 32);
      psq += __shfl_down(psq, 32);
      psum += __shfl_down(psum, 16);
      psq += __shfl_down(psq, 16);
      if ((lane >> 4) == 0) {
        int nl = wc * 32 + fj * 16 + cn;
        atomicAdd(&s_sum[nl], psum);
        atomicAdd(&s_sq[nl], psq);
      }
    }
  }
  if (STATS) {
    __syncthreads();
    if (threadIdx.x < CBN && n0 + threadIdx.x < N) {
      if (s_sum[threadIdx.x] != 0.f || s_sq[threadIdx.x] != 0.f) {
        atomicAdd(&stats_ws[n0 + threadIdx.x], s_sum[threadIdx.x]);
        atomicAdd(&stats_ws[N + n0 + threadIdx.x], s_sq[threadIdx.x]);
      }
    }
  }
  if (BST) {
    __syncthreads();
    if (threadIdx.x < CBN && n0 + threadIdx.x < N) {
      if (s_sum[threadIdx.x] != 0.f)
        atomicAdd(&bp.sum_dy[n0 + threadIdx.x], s_sum[threadIdx.x]);
      if (s_sq[threadIdx.x] != 0.f)
        atomicAdd(&bp.sum_dyx[n0 + threadIdx.x], s_sq[threadIdx.x]);
    }
  }
}

// ---- stride-2 dgrad, parity-decomposed ----------------------------------
// A stride-2 dgrad through the generic MODE=1 path wastes 3/4 of its MFMA
// work: for a fixed dx pixel only the (r,s) with matching parity satisfy
// (h+pad-r) % 2 == 0, so 75% of staged fragments are masked zeros. This
// kernel runs one launch per parity class (ph,pw) in {0,1}^2 over the
// quarter-resolution pixel grid h=2*hh+ph, w=2*ww+pw with the reduced
// filter footprint r = r0+2*ri (ri < Rp), s = s0+2*si — total work is
// exactly 1/4 of the naive form. ho = hh + cr - ri with cr constant per
// class, so the cursor stays O(1) add+wrap.

// per-parity-class geometry, all four classes dispatched in ONE launch
// (separate per-class launches left the deep stages' quarter grids at
// <1 block/CU each)
struct PClass {
  int ph, pw, r0, s0, Rp, Sp, cr, cs, Hh, Wh;
  long M, RED;
};
struct PClasses {
  PClass c[4];
};

template <int TBM>
__global__ __launch_bounds__(TBM * 4) void conv_dgrad_p_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ wt,  // CRSK
    bf16* __restrict__ dx, Geom g, PClasses pcs) {
  constexpr int THREADS = TBM * 4;
  constexpr int BROWS = 512 / THREADS;
  constexpr bool DB = (TBM == 64);   // see conv_mm_kernel note
  constexpr int BUFSZ = TBM * CLDK + CBN * CLDK;
  __shared__ __bf16 smem[(DB ? 2 : 1) * BUFSZ];

  const PClass P = pcs.c[blockIdx.z];
  const int ph = P.ph, pw = P.pw, r0 = P.r0, s0 = P.s0;
  const int Rp = P.Rp, Sp = P.Sp, cr = P.cr, cs = P.cs;
  const int Hh = P.Hh, Wh = P.Wh;
  const long M = P.M, RED = P.RED;
  const long wg = xcd_tile_remap();  // 2D remap within this class slice
  const long m0 = (wg / gridDim.y) * TBM;
  const long n0 = (wg % gridDim.y) * CBN;   // dx channel tile (C dim)
  if (m0 >= M) return;                       // class smaller than grid.x
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int lra = t >> 2;
  const int lka = (t & 3) * 8;
  const int lrb = t >> 3;
  const int lkb = (t & 7) * 8;

  // A-row pixel decomposition over the quarter grid
  const long m = m0 + lra;
  const bool mvalid = m < M;
  int pn = 0, phh = 0, pww = 0;
  if (mvalid) {
    const int HW = Hh * Wh;
    pn = (int)(m / HW);
    int rem = (int)(m % HW);
    phh = rem / Wh;
    pww = rem % Wh;
  }
  // staging state: validity bitmask over the (ri, si) taps (<= 4 bits) +
  // incremental offsets with constant wrap fixes — the div/mod form
  // measured 38 VALU instructions per MFMA (runtime int division)
  unsigned vmask = 0;
  for (int ri = 0; ri < Rp; ++ri)
    for (int si = 0; si < Sp; ++si)
      if ((unsigned)(phh + cr - ri) < (unsigned)g.Ho &&
          (unsigned)(pww + cs - si) < (unsigned)g.Wo)
        vmask |= 1u << (ri * Sp + si);
  const long a_swrap = -2L * g.K;                       // si+1: wo -= 1
  const long a_rwrap = (long)g.K * (Sp - 2 - g.Wo);     // ri+1: ho -= 1
  long aoff[2];
  int acur_k[2], arsn[2], asct[2];
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    acur_k[p] = lka + 32 * p;   // < 64 <= K
    arsn[p] = 0;
    asct[p] = 0;
    aoff[p] = (((long)pn * g.Ho + phh + cr) * g.Wo + pww + cs) * g.K +
              acur_k[p];
  }
  // B columns: wt[c][r0+2*ri][s0+2*si][lkb + k]; per-row base pointers
  // hoisted, one shared column offset advanced with constant wrap fixes
  const long CRSK_row = (long)g.R * g.S * g.K;  // wt row stride (per c)
  // si+1 advances s by 2 (+2K) while k resets (-K): net +K
  const long b_sw = (long)g.K;
  // ri+1: r += 2 (+2*S*K), si resets (-(Sp-1)*2K), k resets (-K)
  const long b_rw = 2L * g.S * g.K - 2L * g.K * (Sp - 1) - (long)g.K;
  long bcol = ((long)r0 * g.S + s0) * g.K + lkb;
  int bcur_k = lkb, bsct = 0;
  const bf16* bbase[BROWS];
  bool bok[BROWS];
#pragma unroll
  for (int q = 0; q < BROWS; ++q) {
    const long c = n0 + lrb + q * (THREADS / 8);
    bok[q] = c < g.C;
    bbase[q] = wt + c * CRSK_row;
  }

  __bf16 ra[2][8], rb[BROWS][8];

  auto stage = [&](long k0) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const bool valid = mvalid && arsn[p] < 32 && ((vmask >> arsn[p]) & 1u);
      if (valid) {
        *reinterpret_cast<s16x8*>(ra[p]) =
            *reinterpret_cast<const s16x8*>(dy + aoff[p]);
      } else {
        zero8(ra[p]);
      }
      aoff[p] += CBK;
      acur_k[p] += CBK;
      while (acur_k[p] >= g.K) {
        acur_k[p] -= g.K;
        arsn[p] += 1;
        if (++asct[p] == Sp) {
          asct[p] = 0;
          aoff[p] += a_rwrap;
        } else {
          aoff[p] += a_swrap;
        }
      }
    }
    // B: vec8 along k at the shared strided column offset
    {
#pragma unroll
      for (int q = 0; q < BROWS; ++q) {
        if (bok[q] && k0 + lkb < RED) {
          *reinterpret_cast<s16x8*>(rb[q]) =
              *reinterpret_cast<const s16x8*>(bbase[q] + bcol);
        } else {
          zero8(rb[q]);
        }
      }
      bcol += CBK;
      bcur_k += CBK;
      while (bcur_k >= g.K) {
        bcur_k -= g.K;
        if (++bsct == Sp) {
          bsct = 0;
          bcol += b_rw;
        } else {
          bcol += b_sw;
        }
      }
    }
  };

  auto write_lds = [&](int pbuf) {
    __bf16* wA = smem + pbuf * BUFSZ;
    __bf16* wB = wA + TBM * CLDK;
    *reinterpret_cast<bf16x8*>(&wA[lra * CLDK + lka]) =
        *reinterpret_cast<bf16x8*>(ra[0]);
    *reinterpret_cast<bf16x8*>(&wA[lra * CLDK + lka + 32]) =
        *reinterpret_cast<bf16x8*>(ra[1]);
#pragma unroll
    for (int q = 0; q < BROWS; ++q) {
      *reinterpret_cast<bf16x8*>(
          &wB[(lrb + q * (THREADS / 8)) * CLDK + lkb]) =
          *reinterpret_cast<bf16x8*>(rb[q]);
    }
  };

  f32x4 acc[2][2] = {};
  stage(0);
  if (DB) {
    write_lds(0);
    if (CBK < RED) stage(CBK);
  }
  int pb = 0;

  for (long k0 = 0; k0 < RED; k0 += CBK) {
    __syncthreads();
    if (DB) {
      if (k0 + CBK < RED) {
        write_lds(pb ^ 1);
        if (k0 + 2 * CBK < RED) stage(k0 + 2 * CBK);
      }
    } else {
      write_lds(0);
      if (k0 + CBK < RED) stage(k0 + CBK);
      __syncthreads();
    }
    const __bf16* sA = smem + pb * BUFSZ;
    const __bf16* sB = sA + TBM * CLDK;

    const int ml = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kf = ks * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fi = 0; fi < 2; ++fi) {
        bf16x8 af = lds8(&sA[(wr * 32 + fi * 16 + ml) * CLDK + kf]);
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          bf16x8 bfr = lds8(&sB[(wc * 32 + fj * 16 + ml) * CLDK + kf]);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    if (DB) pb ^= 1;
  }

  // epilogue: scatter the quarter-grid rows back to full-resolution dx
  const int cn = lane & 15;
  const int cm = (lane >> 4) * 4;
  const int HW = Hh * Wh;
#pragma unroll
  for (int fj = 0; fj < 2; ++fj) {
    long nn = n0 + wc * 32 + fj * 16 + cn;
    if (nn >= g.C) continue;
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long mm = m0 + wr * 32 + fi * 16 + cm + r;
        if (mm >= M) continue;
        const int n = (int)(mm / HW);
        const int rem = (int)(mm % HW);
        const int h = 2 * (rem / Wh) + ph;
        const int w = 2 * (rem % Wh) + pw;
        dx[(((long)n * g.H + h) * g.W + w) * g.C + nn] = f2b(acc[fi][fj][r]);
      }
    }
  }
}

// ---- wgrad: dw[k, rsc] = sum_np dy[np,k] * im2col[np,rsc]  (TN) ----------
// 64x64 tile over (k, rsc), WBK=64 deep in np. Both operands need their
// MFMA fragments transposed relative to the global [np, inner] layout; the
// FAST kernel stages LINEAR 16-B writes into a [inner/16][np][16] image and
// reads fragments with ds_read_b64_tr_b16 (gfx950 LDS hardware transpose:
// each lane gets 4 bf16 strided 16 elements from its own address), so the
// per-element transposed ds_write storm of the naive version disappears.

constexpr int WBM = 64, WBN = 64, WBK = 64;
constexpr int WLDN = WBK + 8;  // padded np-stride (generic fallback kernel)

typedef __bf16 bf16x4t __attribute__((ext_vector_type(4)));
typedef bf16x4t __attribute__((address_space(3)))* lds_b64p;

// two tr16 reads -> one 8-elem fragment: lane l receives image elements
// [np0 + (l>>4)*8 + j][l&15] of a [np][16] bf16 image starting at `base`.
//
// ds_read_b64_tr_b16 semantics (measured with tr16_probe, all 4 modes):
// within a 16-lane group, lane i receives in reg j (j=0..3) the 16-bit
// element at ALIGN8(byte addr supplied by lane 4j + (i>>2)) + 2*(i&3).
// So to deliver img[np0+8g+j][i] to lane i of group g, the lane at group
// coords (a = (l>>2)&3, b = l&3) must SUPPLY elem (np0 + 8g + a)*16 + 4b.
__device__ __forceinline__ bf16x8 tr16_frag(const __bf16* base, int np0,
                                            int lane) {
  const int np = np0 + ((lane >> 4) << 3) + ((lane >> 2) & 3);
  const __bf16* p = base + np * 16 + ((np >> 3) * 8) + (lane & 3) * 4;
  auto p3 = (lds_b64p)(__bf16 __attribute__((address_space(3)))*)p;
  bf16x4t v0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3);
  bf16x4t v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3 + 16);
  bf16x8 r;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    r[j] = v0[j];
    r[4 + j] = v1[j];
  }
  return r;
}

// image block stride: +16 elems (32 B) padding between the four 16-inner
// blocks — without it the b128 staging writes of different kb-blocks land
// on the same 8 banks (measured 4-way, SQ_LDS_BANK_CONFLICT ~= MFMA time).
// per-np-group padding: +8 elems (16 B) after every 8 np rows — without
// it the two 16-lane tr-read groups of a wave land on identical banks.
__device__ __forceinline__ int np_img(int np) { return np * 16 + (np >> 3) * 8; }
constexpr int WIMG = WBK * 16 + (WBK / 8) * 8 + 16;

template <bool ATOMIC>
__global__ __launch_bounds__(256) void conv_wgrad_fast_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    float* __restrict__ dw, Geom g, long NP, long npslice) {
  // images: [inner16-block][np(64)][16] per operand (4 blocks of 64 k/rsc)
  // DOUBLE-buffered: one barrier per np-tile, and the next tile's global
  // loads issue before this tile's MFMA phase (the single-buffered form
  // ran 80% WAIT: load latency sat inside the tile, unoverlapped)
  __shared__ __bf16 sA[2 * 4 * WIMG];
  __shared__ __bf16 sB[2 * 4 * WIMG];

  const long wgid = xcd_tile_remap3();  // z-major chunks per XCD
  const long bx = wgid % gridDim.x;
  const long by = (wgid / gridDim.x) % gridDim.y;
  const long bz = wgid / ((long)gridDim.x * gridDim.y);
  const long k0c = bx * WBM;
  const long n0 = by * WBN;
  const long np0 = bz * npslice;
  const long np1 = min(np0 + npslice, NP);
  const long RED = (long)g.R * g.S * g.C;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int snp = t >> 3;          // 0..31 (stages rows snp, snp+32)
  const int scol = (t & 7) * 8;    // 0..56
  const int simg = (scol >> 4) * WIMG + (scol & 15);

  // fixed per-thread B column (r, s, c)
  int fb_r = 0, fb_s = 0, fb_c = 0;
  const bool bcol_ok = (n0 + scol) < RED;
  if (bcol_ok) {
    int rsc = (int)(n0 + scol);
    fb_c = rsc % g.C;
    int rs = rsc / g.C;
    fb_r = rs / g.S;
    fb_s = rs % g.S;
  }
  const bool a_ok = k0c + scol + 8 <= g.K;

  // incremental pixel cursors + INCREMENTAL addresses for rows snp, snp+32
  int pn[2], pho[2], pwo[2], fh[2], fw[2];
  long aoff[2], xoff[2];
  const int sC = g.stride * g.C;
  const long sWC = (long)g.stride * g.W * g.C;
  const long HWC = (long)g.H * g.W * g.C;
#pragma unroll
  for (int q = 0; q < 2; ++q) {
    long m = np0 + snp + 32 * q;
    long mm = m < NP ? m : 0;
    pn[q] = (int)(mm / ((long)g.Ho * g.Wo));
    int rem = (int)(mm % ((long)g.Ho * g.Wo));
    pho[q] = rem / g.Wo;
    pwo[q] = rem % g.Wo;
    fh[q] = pho[q] * g.stride - g.pad + fb_r;
    fw[q] = pwo[q] * g.stride - g.pad + fb_s;
    aoff[q] = mm * g.K + k0c + scol;
    xoff[q] = (((long)pn[q] * g.H + fh[q]) * g.W + fw[q]) * g.C + fb_c;
  }
  // O(1) cursor advance: np += WBK decomposes into per-shape constants
  // (the while-wrap form costs up to ~16 serial VALU iterations per step
  // for the deep stages' small Wo/Ho)
  const int dwo = WBK % g.Wo;
  const int dcar = WBK / g.Wo;
  const int dho = dcar % g.Ho;
  const int dn = dcar / g.Ho;
  const long dA = (long)WBK * g.K;

  __bf16 ra[2][8], rb[2][8];

  auto stage = [&](long p0) {
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      const long m = p0 + snp + 32 * q;
      const bool v = m < np1;
      if (v && a_ok) {
        *reinterpret_cast<s16x8*>(ra[q]) =
            *reinterpret_cast<const s16x8*>(dy + aoff[q]);
      } else {
        zero8(ra[q]);
      }
      if (v && bcol_ok && (unsigned)fh[q] < (unsigned)g.H &&
          (unsigned)fw[q] < (unsigned)g.W) {
        *reinterpret_cast<s16x8*>(rb[q]) =
            *reinterpret_cast<const s16x8*>(x + xoff[q]);
      } else {
        zero8(rb[q]);
      }
      // constant-delta advance (dwo/dho/dn): <= 2 conditional wraps; the
      // addresses track the cursors with adds only
      aoff[q] += dA;
      pwo[q] += dwo;
      fw[q] += dwo * g.stride;
      xoff[q] += (long)dwo * sC;
      int car = 0;
      if (pwo[q] >= g.Wo) {
        pwo[q] -= g.Wo;
        fw[q] -= g.Wo * g.stride;
        xoff[q] -= (long)g.Wo * sC;
        car = 1;
      }
      pho[q] += dho + car;
      fh[q] += (dho + car) * g.stride;
      xoff[q] += (dho + car) * sWC;
      int car2 = 0;
      if (pho[q] >= g.Ho) {
        pho[q] -= g.Ho;
        fh[q] -= g.Ho * g.stride;
        xoff[q] -= (long)g.Ho * sWC;
        car2 = 1;
      }
      xoff[q] += (dn + car2) * HWC;
      pn[q] += dn + car2;
    }
  };

  auto write_lds = [&](int pbuf) {
    __bf16* wA = sA + pbuf * 4 * WIMG;
    __bf16* wB = sB + pbuf * 4 * WIMG;
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      const int npl = snp + 32 * q;
      *reinterpret_cast<bf16x8*>(&wA[simg + np_img(npl)]) =
          *reinterpret_cast<bf16x8*>(ra[q]);
      *reinterpret_cast<bf16x8*>(&wB[simg + np_img(npl)]) =
          *reinterpret_cast<bf16x8*>(rb[q]);
    }
  };

  f32x4 acc[2][2] = {};
  stage(np0);
  write_lds(0);
  if (np0 + WBK < np1) stage(np0 + WBK);
  int pb = 0;

  for (long p0 = np0; p0 < np1; p0 += WBK) {
    __syncthreads();
    if (p0 + WBK < np1) {
      write_lds(pb ^ 1);
      if (p0 + 2 * WBK < np1) stage(p0 + 2 * WBK);
    }
    const __bf16* rA = sA + pb * 4 * WIMG;
    const __bf16* rB = sB + pb * 4 * WIMG;

#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int fi = 0; fi < 2; ++fi) {
        bf16x8 af =
            tr16_frag(&rA[(wr * 2 + fi) * WIMG], ks * 32, lane);
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          bf16x8 bfr =
              tr16_frag(&rB[(wc * 2 + fj) * WIMG], ks * 32, lane);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr, acc[fi][fj], 0, 0, 0);
        }
      }
    }
    pb ^= 1;
  }

  const int cn = lane & 15;
  const int cm = (lane >> 4) * 4;
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      long nn = n0 + wc * 32 + fj * 16 + cn;
      if (nn >= RED) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long kk = k0c + wr * 32 + fi * 16 + cm + r;
        if (kk >= g.K) continue;
        if (ATOMIC) {
          atomicAdd(&dw[kk * RED + nn], acc[fi][fj][r]);
        } else {
          dw[kk * RED + nn] = acc[fi][fj][r];
        }
      }
    }
  }
}

template <bool FAST>  // generic fallback (any C/K); WBK_GEN=32 transposed
__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    float* __restrict__ dw, Geom g, long NP, long npslice) {
  // sA[k(64)][np(32)], sB[rsc(64)][np(32)] — transposed staging
  __shared__ __bf16 sA[WBM * WLDN];
  __shared__ __bf16 sB[WBN * WLDN];

  const long wgid = xcd_tile_remap3();       // z-major chunks per XCD
  const long k0c = (wgid % gridDim.x) * WBM;           // out-channel tile
  const long n0 = ((wgid / gridDim.x) % gridDim.y) * WBN;  // rsc tile
  const long np0 = (wgid / ((long)gridDim.x * gridDim.y)) * npslice;
  const long np1 = min(np0 + npslice, NP);
  const long RED = (long)g.R * g.S * g.C;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  // staging map: 256 threads = 32 np-rows x 8 col-chunks of 8; each thread
  // stages np rows {snp, snp+32}.
  const int snp = t >> 3;          // 0..31
  const int scol = (t & 7) * 8;    // 0..56

  // hoisted B-column decomposition: (r, s, c) is FIXED per thread
  int fb_r = 0, fb_s = 0, fb_c = 0;
  bool bcol_ok = (n0 + scol) < RED;
  if (bcol_ok) {
    int rsc = (int)(n0 + scol);
    fb_c = rsc % g.C;
    int rs = rsc / g.C;
    fb_r = rs / g.S;
    fb_s = rs % g.S;
  }
  const bool b_vec_ok = FAST && bcol_ok && (fb_c + 8 <= g.C);

  // incremental pixel cursors for the two staged np rows
  int pn[2], pho[2], pwo[2];
  bool pv[2];
#pragma unroll
  for (int q = 0; q < 2; ++q) {
    long m = np0 + snp + 32 * q;
    pv[q] = m < np1;
    long mm = pv[q] ? m : 0;
    pn[q] = (int)(mm / ((long)g.Ho * g.Wo));
    int rem = (int)(mm % ((long)g.Ho * g.Wo));
    pho[q] = rem / g.Wo;
    pwo[q] = rem % g.Wo;
  }

  f32x4 acc[2][2] = {};

  for (long p0 = np0; p0 < np1; p0 += WBK) {
    __bf16 ra[2][8], rb[2][8];
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      const long m = p0 + snp + 32 * q;
      const bool v = m < np1;
      // A: dy[np, k] chunk (vec8 along k)
      if (!v) {
        zero8(ra[q]);
      } else if (FAST && k0c + scol + 8 <= g.K) {
        *reinterpret_cast<s16x8*>(ra[q]) = *reinterpret_cast<const s16x8*>(
            dy + m * g.K + k0c + scol);
      } else {
        row8(dy, NP, g.K, m, k0c + scol, ra[q]);
      }
      // B: im2col[np, rsc] chunk at the thread's fixed (r,s,c)
      if (!v || !bcol_ok) {
        zero8(rb[q]);
      } else if (FAST) {
        int h = pho[q] * g.stride - g.pad + fb_r;
        int w = pwo[q] * g.stride - g.pad + fb_s;
        bool valid = (unsigned)h < (unsigned)g.H &&
                     (unsigned)w < (unsigned)g.W && b_vec_ok;
        if (valid) {
          long off = (((long)pn[q] * g.H + h) * g.W + w) * g.C + fb_c;
          *reinterpret_cast<s16x8*>(rb[q]) =
              *reinterpret_cast<const s16x8*>(x + off);
        } else {
          zero8(rb[q]);
        }
      } else {
        // generic: columns may cross (r,s) boundaries -> per element
        stage_im2col_slow(x, g, true, pn[q], pho[q], pwo[q],
                          (int)(n0 + scol), rb[q]);
      }
      // advance pixel cursor by WBK rows
      if (pv[q]) {
        pwo[q] += WBK;
        while (pwo[q] >= g.Wo) {
          pwo[q] -= g.Wo;
          pho[q] += 1;
        }
        while (pho[q] >= g.Ho) {
          pho[q] -= g.Ho;
          pn[q] += 1;
        }
      }
    }
    __syncthreads();
#pragma unroll
    for (int q = 0; q < 2; ++q) {
      const int npl = snp + 32 * q;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sA[(scol + j) * WLDN + npl] = ra[q][j];
        sB[(scol + j) * WLDN + npl] = rb[q][j];
      }
    }
    __syncthreads();

    const int ml = lane & 15;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int kf = ks * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fi = 0; fi < 2; ++fi) {
        bf16x8 af = lds8(&sA[(wr * 32 + fi * 16 + ml) * WLDN + kf]);
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          bf16x8 bfr = lds8(&sB[(wc * 32 + fj * 16 + ml) * WLDN + kf]);
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bfr, acc[fi][fj], 0, 0, 0);
        }
      }
    }
  }

  const int cn = lane & 15;
  const int cm = (lane >> 4) * 4;
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      long nn = n0 + wc * 32 + fj * 16 + cn;
      if (nn >= RED) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long kk = k0c + wr * 32 + fi * 16 + cm + r;
        if (kk >= g.K) continue;
        atomicAdd(&dw[kk * RED + nn], acc[fi][fj][r]);
      }
    }
  }
}

// debug probe: LDS filled with element indices; dump what each lane's
// tr16_frag (as used by wgrad) actually receives.
__global__ void tr16_probe_kernel(float* out, int mode) {
  __shared__ __bf16 s[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x)
    s[i] = (__bf16)(float)i;
  __syncthreads();
  if (threadIdx.x < 64) {
    int lane = threadIdx.x;
    long elem = 0;
    if (mode == 0) elem = ((lane >> 4) << 3) * 16 + (lane & 15);
    if (mode == 1) elem = lane * 16;
    if (mode == 2) elem = 0;
    if (mode == 3) elem = lane * 4;
    const __bf16* p = &s[elem];
    auto p3 = (lds_b64p)(__bf16 __attribute__((address_space(3)))*)p;
    bf16x4t v0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p3);
#pragma unroll
    for (int j = 0; j < 4; ++j) out[lane * 4 + j] = (float)v0[j];
  }
}

}  // namespace conv

torch::Tensor tr16_probe(long mode) {
  auto out = torch::zeros({64, 4},
                          torch::dtype(torch::kFloat32).device(torch::kCUDA));
  conv::tr16_probe_kernel<<<1, 64, 0, cur_stream()>>>(out.data_ptr<float>(),
                                                      (int)mode);
  return out;
}

// zsplit target: each dw element takes `zsplit` fp32 atomic hits, so the
// block count trades atomic contention against fill/loop length
// (EG_WGRAD_TB overrides for tuning sweeps).
static long wgrad_target_blocks() {
  static long v = [] {
    const char* e = getenv("EG_WGRAD_TB");
    return e ? atol(e) : 2048L;
  }();
  return v;
}

static conv::Geom make_geom(const torch::Tensor& x, int K, int R, int S,
                            long stride, long pad) {
  conv::Geom g;
  g.N = (int)x.size(0); g.H = (int)x.size(1); g.W = (int)x.size(2);
  g.C = (int)x.size(3);
  g.K = K; g.R = R; g.S = S;
  g.stride = (int)stride; g.pad = (int)pad;
  g.Ho = (g.H + 2 * g.pad - R) / g.stride + 1;
  g.Wo = (g.W + 2 * g.pad - S) / g.stride + 1;
  return g;
}

// x [N,H,W,C] bf16; w [K,R,S,C] bf16; bias fp32[K] or empty -> y [N,Ho,Wo,K]
// collect_bn_stats: also accumulate per-out-channel sum/sumsq of y into the
// shared BN workspace (bn.hip bn_stats_ws), consumed by bn_fwd(have_stats).
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         long stride, long pad, bool collect_bn_stats) {
  CHECK_IN(x); CHECK_IN(w);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16);
  auto g = make_geom(x, (int)w.size(0), (int)w.size(1), (int)w.size(2),
                     stride, pad);
  TORCH_CHECK((int)w.size(3) == g.C, "conv channel mismatch");
  auto y = torch::empty({g.N, g.Ho, g.Wo, g.K}, x.options());
  long M = (long)g.N * g.Ho * g.Wo;
  long RED = (long)g.R * g.S * g.C;
  bool has_bias = bias.numel() > 0;
  bool fast = (g.C % 8 == 0) && (g.R * g.S <= 32);
  // shape-adaptive tile: 128-row tiles starve the 256-CU chip on the deep
  // stages (M drops to 4096); switch to 64-row tiles below ~384 blocks.
  long blocks128 = (long)ceil_div(M, 128) * ceil_div(g.K, conv::CBN);
  bool narrow = blocks128 < 384;
  long tbm = narrow ? 64 : 128;
  dim3 grid(ceil_div(M, tbm), ceil_div(g.K, conv::CBN));
  float* ws = nullptr;
  if (collect_bn_stats) {
    // the stats epilogue's wave-shuffle reduction assumes no masked
    // output channels in any wave (true for every BN'd conv: K % 64 == 0)
    TORCH_CHECK(g.K % 64 == 0,
                "collect_bn_stats requires out-channels % 64 == 0");
    ws = bn_stats_ws_ptr(g.K, x.options());
  }
  if (ws != nullptr) {
    auto* fn = fast ? (narrow ? conv::conv_mm_kernel<0, true, 64, true>
                              : conv::conv_mm_kernel<0, true, 128, true>)
                    : (narrow ? conv::conv_mm_kernel<0, false, 64, true>
                              : conv::conv_mm_kernel<0, false, 128, true>);
    fn<<<grid, (unsigned)(tbm * 4), 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
        has_bias ? bias.data_ptr<float>() : nullptr, (bf16*)y.data_ptr(), g,
        M, g.K, RED, has_bias ? 1 : 0, ws, conv::BPack{});
  } else {
    auto* fn = fast ? (narrow ? conv::conv_mm_kernel<0, true, 64>
                              : conv::conv_mm_kernel<0, true, 128>)
                    : (narrow ? conv::conv_mm_kernel<0, false, 64>
                              : conv::conv_mm_kernel<0, false, 128>);
    fn<<<grid, (unsigned)(tbm * 4), 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
        has_bias ? bias.data_ptr<float>() : nullptr, (bf16*)y.data_ptr(), g,
        M, g.K, RED, has_bias ? 1 : 0, nullptr, conv::BPack{});
  }
  return y;
}

// dy [N,Ho,Wo,K] bf16; wt [C,R,S,K] bf16 (w.permute(3,1,2,0)) -> dx [N,H,W,C]
// Optional bs_* pack: fuse the upstream BN's backward stats into the dx
// epilogue (see BPack). Only taken on the FAST stride-1 path; the caller
// gates eligibility.
torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor wt, long stride,
                           long pad, long H, long W,
                           c10::optional<torch::Tensor> bs_y1,
                           c10::optional<torch::Tensor> bs_x1,
                           c10::optional<torch::Tensor> bs_mean,
                           c10::optional<torch::Tensor> bs_invstd,
                           c10::optional<torch::Tensor> bs_dgamma,
                           c10::optional<torch::Tensor> bs_dbeta) {
  CHECK_IN(dy); CHECK_IN(wt);
  int C = (int)wt.size(0), R = (int)wt.size(1), S = (int)wt.size(2);
  int K = (int)wt.size(3);
  conv::Geom g;
  g.N = (int)dy.size(0); g.Ho = (int)dy.size(1); g.Wo = (int)dy.size(2);
  TORCH_CHECK((int)dy.size(3) == K);
  g.H = (int)H; g.W = (int)W; g.C = C; g.K = K; g.R = R; g.S = S;
  g.stride = (int)stride; g.pad = (int)pad;
  long M = (long)g.N * g.H * g.W;
  long RED = (long)R * S * K;
  // stride-2 parity decomposition: 4 quarter-grid launches at 1/4 the
  // total work (the naive MODE=1 path masks 75% of fragments to zero)
  if (g.stride == 2 && g.K % 64 == 0) {
    conv::PClasses pcs{};
    int nc = 0;
    bool any_empty = false;
    long maxM = 0;
    for (int ph = 0; ph < 2; ++ph) {
      const int r0 = (ph + g.pad) & 1;
      const int Rp = ((int)R - r0 + 1) / 2;
      for (int pw = 0; pw < 2; ++pw) {
        const int s0 = (pw + g.pad) & 1;
        const int Sp = ((int)S - s0 + 1) / 2;
        if (Rp <= 0 || Sp <= 0 || ph >= g.H || pw >= g.W) {
          any_empty = true;  // e.g. the 1x1 downsampler's odd classes
          continue;
        }
        conv::PClass& P = pcs.c[nc++];
        P.ph = ph; P.pw = pw; P.r0 = r0; P.s0 = s0;
        P.Rp = Rp; P.Sp = Sp;
        P.cr = (ph + g.pad - r0) >> 1;
        P.cs = (pw + g.pad - s0) >> 1;
        P.Hh = (g.H - ph + 1) >> 1;
        P.Wh = (g.W - pw + 1) >> 1;
        P.M = (long)g.N * P.Hh * P.Wh;
        P.RED = (long)Rp * Sp * K;
        maxM = std::max(maxM, P.M);
      }
    }
    auto dx = any_empty
        ? torch::zeros({g.N, g.H, g.W, g.C}, dy.options())
        : torch::empty({g.N, g.H, g.W, g.C}, dy.options());
    if (nc > 0) {
      long blocks128 =
          (long)ceil_div(maxM, 128) * ceil_div(C, conv::CBN) * nc;
      long tbm = blocks128 < 384 ? 64 : 128;
      dim3 grid(ceil_div(maxM, tbm), ceil_div(C, conv::CBN), (unsigned)nc);
      auto* fn = tbm == 64 ? conv::conv_dgrad_p_kernel<64>
                           : conv::conv_dgrad_p_kernel<128>;
      fn<<<grid, (unsigned)(tbm * 4), 0, cur_stream()>>>(
          (const bf16*)dy.data_ptr(), (const bf16*)wt.data_ptr(),
          (bf16*)dx.data_ptr(), g, pcs);
    }
    return dx;
  }
  auto dx = torch::empty({g.N, g.H, g.W, g.C}, dy.options());
  // FAST MODE1 handles stride 1 only (stride 2 goes to the parity path
  // when K % 64 == 0, else the generic slow path)
  bool fast = (g.K % 8 == 0) && (g.stride == 1) && (g.R * g.S <= 32);
  long blocks128 = (long)ceil_div(M, 128) * ceil_div(C, conv::CBN);
  bool narrow = blocks128 < 384;
  long tbm = narrow ? 64 : 128;
  dim3 grid(ceil_div(M, tbm), ceil_div(C, conv::CBN));
  const bool bst = fast && bs_y1.has_value();
  if (bst) {
    TORCH_CHECK(C % 64 == 0, "bwd-stats fusion requires C % 64 == 0");
    conv::BPack bp{(const bf16*)bs_y1->data_ptr(),
                   (const bf16*)bs_x1->data_ptr(),
                   bs_mean->data_ptr<float>(), bs_invstd->data_ptr<float>(),
                   bs_dbeta->data_ptr<float>(), bs_dgamma->data_ptr<float>()};
    auto* fn = narrow ? conv::conv_mm_kernel<1, true, 64, false, true>
                      : conv::conv_mm_kernel<1, true, 128, false, true>;
    fn<<<grid, (unsigned)(tbm * 4), 0, cur_stream()>>>(
        (const bf16*)dy.data_ptr(), (const bf16*)wt.data_ptr(), nullptr,
        (bf16*)dx.data_ptr(), g, M, C, RED, 0, nullptr, bp);
    return dx;
  }
  auto* fn = fast ? (narrow ? conv::conv_mm_kernel<1, true, 64>
                            : conv::conv_mm_kernel<1, true, 128>)
                  : (narrow ? conv::conv_mm_kernel<1, false, 64>
                            : conv::conv_mm_kernel<1, false, 128>);
  fn<<<grid, (unsigned)(tbm * 4), 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), (const bf16*)wt.data_ptr(), nullptr,
      (bf16*)dx.data_ptr(), g, M, C, RED, 0, nullptr, conv::BPack{});
  return dx;
}

// x [N,H,W,C] bf16; dy [N,Ho,Wo,K] bf16 -> dw fp32 [K,R,S,C]
torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor dy, long R, long S,
                           long stride, long pad) {
  CHECK_IN(x); CHECK_IN(dy);
  auto g = make_geom(x, (int)dy.size(3), (int)R, (int)S, stride, pad);
  TORCH_CHECK(g.Ho == (int)dy.size(1) && g.Wo == (int)dy.size(2),
              "wgrad geometry mismatch");
  long NP = (long)g.N * g.Ho * g.Wo;
  long RED = (long)R * S * g.C;
  // split the NP reduction across blocks for parallelism; fp32 atomics
  long target_blocks = wgrad_target_blocks();
  long tiles = (long)ceil_div(g.K, conv::WBM) * ceil_div(RED, conv::WBN);
  long zsplit =
      std::max(1L, std::min(512L, target_blocks / std::max(tiles, 1L)));
  long npslice = (NP + zsplit - 1) / zsplit;
  npslice = ((npslice + conv::WBK - 1) / conv::WBK) * conv::WBK;
  zsplit = (NP + npslice - 1) / npslice;
  bool fast = (g.K % 8 == 0) && (g.C % 8 == 0);
  // single NP slice per tile -> plain stores, no zero-init needed
  auto dw = zsplit == 1
      ? torch::empty({(long)g.K, RED}, x.options().dtype(torch::kFloat32))
      : torch::zeros({(long)g.K, RED}, x.options().dtype(torch::kFloat32));
  dim3 grid(ceil_div(g.K, conv::WBM), ceil_div(RED, conv::WBN),
            (unsigned)zsplit);
  if (fast && zsplit == 1) {
    conv::conv_wgrad_fast_kernel<false><<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(),
        dw.data_ptr<float>(), g, NP, npslice);
  } else if (fast) {
    conv::conv_wgrad_fast_kernel<true><<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(),
        dw.data_ptr<float>(), g, NP, npslice);
  } else {
    TORCH_CHECK(zsplit >= 1);
    if (zsplit == 1) dw.zero_();
    conv::conv_wgrad_kernel<false><<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(),
        dw.data_ptr<float>(), g, NP, npslice);
  }
  return dw.view({(long)g.K, R, S, (long)g.C});
}

// persistent KRSC wgrad scratch, shared by every conv: allocated zeroed
// once and RETURNED to zero by the consuming transform kernel (same
// recycling trick as the BN stats workspace), so the per-conv zero-fill
// launches disappear. Grows monotonically; stable after the first pass,
// so hipGraph capture sees a fixed address.
static torch::Tensor& wgrad_scratch(long n, const torch::TensorOptions& f32) {
  static auto* t = new torch::Tensor();
  if (!t->defined() || t->numel() < n) *t = torch::zeros({n}, f32);
  return *t;
}

namespace conv {

// grad_oihw[i] += scratch_krsc[perm(i)]; scratch reset to 0 after the read.
// dst-indexed (coalesced grad writes + scattered scratch reads, same shape
// as the old krsc_to_oihw transform that measured ~6 us per conv).
__global__ void krsc_accum_oihw_reset_kernel(float* __restrict__ scratch,
                                             float* __restrict__ grad, int K,
                                             int C, int R, int S) {
  long total = (long)K * C * R * S;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    // OIHW i -> (k, c, r, s) -> KRSC index
    int s = (int)(i % S);
    long rem = i / S;
    int r = (int)(rem % R);
    rem /= R;
    int c = (int)(rem % C);
    int k = (int)(rem / C);
    long j = ((long)(k * R + r) * S + s) * C + c;
    grad[i] += scratch[j];
    scratch[j] = 0.f;
  }
}

}  // namespace conv

// wgrad accumulated into the fp32 OIHW flat-grad view (pre-zeroed by the
// per-step zero_grad): the MFMA kernel keeps its fast coalesced KRSC
// epilogue (an OIHW epilogue measured 4x slower: the rsc->hw permutation
// scatters the fp32 atomics across cachelines), writing into the shared
// self-rezeroing scratch; one cheap transform then folds scratch into the
// grad view. No fresh allocation, no zero fill, no autograd
// accumulate-add. Also serves Linear (x [M,1,1,Cin], dy [M,1,1,Nout],
// R=S=1: OIHW == [Nout, Cin] row-major == the torch weight-grad layout).
void conv2d_wgrad_into(torch::Tensor x, torch::Tensor dy, torch::Tensor dw,
                       long R, long S, long stride, long pad) {
  CHECK_IN(x); CHECK_IN(dy); CHECK_IN(dw);
  TORCH_CHECK(dw.scalar_type() == torch::kFloat32);
  auto g = make_geom(x, (int)dy.size(3), (int)R, (int)S, stride, pad);
  TORCH_CHECK(g.Ho == (int)dy.size(1) && g.Wo == (int)dy.size(2),
              "wgrad geometry mismatch");
  long RED = (long)R * S * g.C;
  long total = (long)g.K * RED;
  TORCH_CHECK(dw.numel() == total);
  long NP = (long)g.N * g.Ho * g.Wo;
  long target_blocks = wgrad_target_blocks();
  long tiles = (long)ceil_div(g.K, conv::WBM) * ceil_div(RED, conv::WBN);
  long zsplit =
      std::max(1L, std::min(512L, target_blocks / std::max(tiles, 1L)));
  long npslice = (NP + zsplit - 1) / zsplit;
  npslice = ((npslice + conv::WBK - 1) / conv::WBK) * conv::WBK;
  zsplit = (NP + npslice - 1) / npslice;
  bool fast = (g.K % 8 == 0) && (g.C % 8 == 0);
  dim3 grid(ceil_div(g.K, conv::WBM), ceil_div(RED, conv::WBN),
            (unsigned)zsplit);
  // 1x1 (and Linear): KRSC == OIHW, so the epilogue atomics accumulate
  // straight into the pre-zeroed grad view — no scratch, no fold kernel
  float* dst = dw.data_ptr<float>();
  const bool direct = (R == 1 && S == 1);
  if (!direct) {
    auto& scratch = wgrad_scratch(total, x.options().dtype(torch::kFloat32));
    dst = scratch.data_ptr<float>();
  }
  // scratch is zero here (invariant); atomics accumulate, the fold resets
  if (fast) {
    conv::conv_wgrad_fast_kernel<true><<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(), dst, g, NP,
        npslice);
  } else {
    conv::conv_wgrad_kernel<false><<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(), dst, g, NP,
        npslice);
  }
  if (!direct) {
    int tgrid = (int)std::min<long>((total + 255) / 256, 2048L);
    conv::krsc_accum_oihw_reset_kernel<<<tgrid, 256, 0, cur_stream()>>>(
        dst, dw.data_ptr<float>(), g.K, g.C, (int)R, (int)S);
  }
}

}  // namespace eg
