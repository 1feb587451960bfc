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
//        (full-corr with rotated kernel: dy upsampled by stride)
//  wgrad dw[k, rsc] = dy[np, k]^T @ im2col[np, rsc]          (TN, split-NP
//        across blocks, fp32 atomic accumulate)
//
// Tiles: 64x64xBK32, 4 waves (2x2) of 2x2 v_mfma_f32_16x16x32_bf16
// fragments; LDS rows padded +8 bf16 against b128 bank conflicts.

#include "common.h"

namespace eg {

namespace conv {

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int LDK = BK + 8;

struct Geom {
  int N, H, W, C;     // input
  int K, R, S;        // filter
  int Ho, Wo;         // output
  int stride, pad;
};

__device__ __forceinline__ bf16x8 lds8(const __bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

// stage one vec8 of the im2col A-operand: row m (output pixel), red r0..r0+7.
__device__ __forceinline__ void stage_im2col8(const bf16* __restrict__ x,
                                              const Geom g, long m, int red,
                                              __bf16* dst) {
  long NP = (long)g.N * g.Ho * g.Wo;
  if (m >= NP) {
#pragma unroll
    for (int j = 0; j < 8; ++j) dst[j] = (__bf16)0.f;
    return;
  }
  int n = (int)(m / ((long)g.Ho * g.Wo));
  int rem = (int)(m % ((long)g.Ho * g.Wo));
  int ho = rem / g.Wo, wo = rem % g.Wo;
  int rsc = red;
  int c0 = rsc % g.C;
  int rs = rsc / g.C;
  int r = rs / g.S, s = rs % g.S;
  int h = ho * g.stride - g.pad + r;
  int w = wo * g.stride - g.pad + s;
  int RSC = g.R * g.S * g.C;
  bool fast = (c0 + 8 <= g.C) && (red + 8 <= RSC);
  if (fast) {
    bool valid = (unsigned)h < (unsigned)g.H && (unsigned)w < (unsigned)g.W;
    long off = (((long)n * g.H + h) * g.W + w) * g.C + c0;
    if (valid && (off & 7) == 0) {
      *reinterpret_cast<s16x8*>(dst) = *reinterpret_cast<const s16x8*>(x + off);
    } else if (valid) {
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = (__bf16)b2f(x[off + j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = (__bf16)0.f;
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int rr = red + j;
      float v = 0.f;
      if (rr < RSC) {
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
}

// stage one vec8 of the dgrad A-operand: row m = input pixel (n,h,w),
// red = (r, s, kk) over the OUTPUT-grad channels.
__device__ __forceinline__ void stage_dgrad8(const bf16* __restrict__ dy,
                                             const Geom g, long m, int red,
                                             __bf16* dst) {
  long NP = (long)g.N * g.H * g.W;  // over INPUT pixels
  int RSK = g.R * g.S * g.K;
  if (m >= NP) {
#pragma unroll
    for (int j = 0; j < 8; ++j) dst[j] = (__bf16)0.f;
    return;
  }
  int n = (int)(m / ((long)g.H * g.W));
  int rem = (int)(m % ((long)g.H * g.W));
  int h = rem / g.W, w = rem % g.W;
  int k0 = red % g.K;
  bool fast = (k0 + 8 <= g.K) && (red + 8 <= RSK);
  if (fast) {
    int rs = red / g.K;
    int r = rs / g.S, s = rs % g.S;
    int hq = h + g.pad - r, wq = w + g.pad - s;
    bool valid = hq >= 0 && wq >= 0 && hq % g.stride == 0 &&
                 wq % g.stride == 0;
    int ho = hq / g.stride, wo = wq / g.stride;
    valid = valid && ho < g.Ho && wo < g.Wo;
    long off = (((long)n * g.Ho + ho) * g.Wo + wo) * g.K + k0;
    if (valid && (off & 7) == 0) {
      *reinterpret_cast<s16x8*>(dst) = *reinterpret_cast<const s16x8*>(dy + off);
    } else if (valid) {
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = (__bf16)b2f(dy[off + j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[j] = (__bf16)0.f;
    }
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int rr = red + j;
      float v = 0.f;
      if (rr < RSK) {
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

// ---- fwd & dgrad share the main loop (template on the A-stager) ----------
// 128x64 block tile, BK=64, 8 waves (4x2) of 32x32, 512 threads.

constexpr int CBM = 128, CBN = 64, CBK = 64;
constexpr int CLDK = CBK + 8;

template <int MODE>  // 0 = fwd, 1 = dgrad
__global__ __launch_bounds__(512) void conv_mm_kernel(
    const bf16* __restrict__ Asrc, const bf16* __restrict__ B,
    const float* __restrict__ bias, bf16* __restrict__ out, Geom g,
    long M, long N, long RED, int has_bias) {
  __shared__ __bf16 sA[CBM * CLDK];
  __shared__ __bf16 sB[CBN * CLDK];

  const long m0 = (long)blockIdx.x * CBM;
  const long n0 = (long)blockIdx.y * CBN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;          // 8 waves: wr in 0..3, wc in 0..1
  const int wr = wave >> 1, wc = wave & 1;
  // A staging: 512 thr x 2 vec8 = 128x64; B staging: 512 thr x 1 vec8 = 64x64
  const int lra = t >> 2;            // 0..127
  const int lka = (t & 3) * 8;       // 0..24
  const int lrb = t >> 3;            // 0..63
  const int lkb = (t & 7) * 8;       // 0..56

  f32x4 acc[2][2] = {};

  for (long k0 = 0; k0 < RED; k0 += CBK) {
    __bf16 ra0[8], ra1[8], rb[8];
    if (MODE == 0) {
      stage_im2col8(Asrc, g, m0 + lra, (int)(k0 + lka), ra0);
      stage_im2col8(Asrc, g, m0 + lra, (int)(k0 + lka + 32), ra1);
    } else {
      stage_dgrad8(Asrc, g, m0 + lra, (int)(k0 + lka), ra0);
      stage_dgrad8(Asrc, g, m0 + lra, (int)(k0 + lka + 32), ra1);
    }
    row8(B, N, RED, n0 + lrb, k0 + lkb, rb);
    __syncthreads();
    *reinterpret_cast<bf16x8*>(&sA[lra * CLDK + lka]) =
        *reinterpret_cast<bf16x8*>(ra0);
    *reinterpret_cast<bf16x8*>(&sA[lra * CLDK + lka + 32]) =
        *reinterpret_cast<bf16x8*>(ra1);
    *reinterpret_cast<bf16x8*>(&sB[lrb * CLDK + lkb]) =
        *reinterpret_cast<bf16x8*>(rb);
    __syncthreads();

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
  }

  const int cn = lane & 15;
  const int cm = (lane >> 4) * 4;
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      long nn = n0 + wc * 32 + fj * 16 + cn;
      if (nn >= N) continue;
      float bv = has_bias ? bias[nn] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long mm = m0 + wr * 32 + fi * 16 + cm + r;
        if (mm >= M) continue;
        out[mm * N + nn] = f2b(acc[fi][fj][r] + bv);
      }
    }
  }
}

// ---- wgrad: dw[k, rsc] = sum_np dy[np,k] * im2col[np,rsc]  (TN) ----------

constexpr int WLDK = BK + 8;  // padded np-stride for transposed LDS tiles

__global__ __launch_bounds__(256) void conv_wgrad_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ dy,
    float* __restrict__ dw, Geom g, long NP, long npslice) {
  // sA[k(64)][np(32)], sB[rsc(64)][np(32)] — transposed staging
  __shared__ __bf16 sA[BM * WLDK];
  __shared__ __bf16 sB[BN * WLDK];

  const long k0c = (long)blockIdx.x * BM;   // out-channel tile
  const long n0 = (long)blockIdx.y * BN;    // rsc tile
  const long np0 = (long)blockIdx.z * npslice;
  const long np1 = min(np0 + npslice, NP);
  const long RED = g.R * g.S * g.C;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  // staging map: 256 threads = 32 np-rows x 8 col-chunks of 8
  const int snp = t >> 3;          // 0..31 np row
  const int scol = (t & 7) * 8;    // 0..56 col chunk

  f32x4 acc[2][2] = {};

  for (long p0 = np0; p0 < np1; p0 += BK) {
    __bf16 ra[8], rb[8];
    long m = p0 + snp;
    // A: dy[np, k] chunk (vec8 along k), transposed into sA[k][np]
    if (m < np1) row8(dy, NP, g.K, m, k0c + scol, ra);
    else {
#pragma unroll
      for (int j = 0; j < 8; ++j) ra[j] = (__bf16)0.f;
    }
    // B: im2col[np, rsc] chunk, transposed into sB[rsc][np]
    if (m < np1) stage_im2col8(x, g, m, (int)(n0 + scol), rb);
    else {
#pragma unroll
      for (int j = 0; j < 8; ++j) rb[j] = (__bf16)0.f;
    }
    __syncthreads();
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sA[(scol + j) * WLDK + snp] = ra[j];
      sB[(scol + j) * WLDK + snp] = rb[j];
    }
    __syncthreads();

    const int kf = (lane >> 4) * 8;
    const int ml = lane & 15;
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      bf16x8 af = lds8(&sA[(wr * 32 + fi * 16 + ml) * WLDK + kf]);
#pragma unroll
      for (int fj = 0; fj < 2; ++fj) {
        bf16x8 bfr = lds8(&sB[(wc * 32 + fj * 16 + ml) * WLDK + kf]);
        acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bfr, acc[fi][fj], 0, 0, 0);
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

}  // namespace conv

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
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         long stride, long pad) {
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
  dim3 grid(ceil_div(M, conv::CBM), ceil_div(g.K, conv::CBN));
  conv::conv_mm_kernel<0><<<grid, 512, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
      has_bias ? bias.data_ptr<float>() : nullptr, (bf16*)y.data_ptr(), g,
      M, g.K, RED, has_bias ? 1 : 0);
  return y;
}

// dy [N,Ho,Wo,K] bf16; wt [C,R,S,K] bf16 (w.permute(3,1,2,0)) -> dx [N,H,W,C]
torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor wt, long stride,
                           long pad, long H, long W) {
  CHECK_IN(dy); CHECK_IN(wt);
  int C = (int)wt.size(0), R = (int)wt.size(1), S = (int)wt.size(2);
  int K = (int)wt.size(3);
  conv::Geom g;
  g.N = (int)dy.size(0); g.Ho = (int)dy.size(1); g.Wo = (int)dy.size(2);
  TORCH_CHECK((int)dy.size(3) == K);
  g.H = (int)H; g.W = (int)W; g.C = C; g.K = K; g.R = R; g.S = S;
  g.stride = (int)stride; g.pad = (int)pad;
  auto dx = torch::empty({g.N, g.H, g.W, g.C}, dy.options());
  long M = (long)g.N * g.H * g.W;
  long RED = (long)R * S * K;
  dim3 grid(ceil_div(M, conv::CBM), ceil_div(C, conv::CBN));
  conv::conv_mm_kernel<1><<<grid, 512, 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), (const bf16*)wt.data_ptr(), nullptr,
      (bf16*)dx.data_ptr(), g, M, C, RED, 0);
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
  auto dw = torch::zeros({(long)g.K, RED}, x.options().dtype(torch::kFloat32));
  // split the NP reduction across blocks for parallelism; fp32 atomics
  long target_blocks = 1024;
  long tiles = (long)ceil_div(g.K, conv::BM) * ceil_div(RED, conv::BN);
  long zsplit =
      std::max(1L, std::min(512L, target_blocks / std::max(tiles, 1L)));
  long npslice = (NP + zsplit - 1) / zsplit;
  npslice = ((npslice + conv::BK - 1) / conv::BK) * conv::BK;
  zsplit = (NP + npslice - 1) / npslice;
  dim3 grid(ceil_div(g.K, conv::BM), ceil_div(RED, conv::BN),
            (unsigned)zsplit);
  conv::conv_wgrad_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), (const bf16*)dy.data_ptr(),
      dw.data_ptr<float>(), g, NP, npslice);
  return dw.view({(long)g.K, R, S, (long)g.C});
}

}  // namespace eg
