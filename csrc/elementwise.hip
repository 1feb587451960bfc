// Elementwise kernels: ReLU fwd/bwd, residual add+ReLU, dropout/dropout2d,
// channel sums (bias gradients). All bf16 activations, vectorized 8-wide
// (guide §6 G13: hipcc does not auto-vectorize bf16 loads).

#include "common.h"

namespace eg {

using u8 = unsigned char;

// ---- relu ----------------------------------------------------------------

// EW: 0 = relu fwd, 1 = relu bwd (a=dy, b=y), 2 = add_relu
template <int EW>
__device__ __forceinline__ float ew_op(float a, float b) {
  if (EW == 0) return a > 0.f ? a : 0.f;
  if (EW == 1) return b > 0.f ? a : 0.f;
  return (a + b) > 0.f ? a + b : 0.f;
}

template <int EW>
__global__ void ew_kernel(const bf16* __restrict__ a,
                          const bf16* __restrict__ b, bf16* __restrict__ y,
                          long n) {
  const long n8 = n / 8;
  const long stride = (long)gridDim.x * blockDim.x;
  const long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = t0; v < n8; v += stride) {
    s16x8 va = reinterpret_cast<const s16x8*>(a)[v];
    s16x8 vb;
    if (EW != 0) vb = reinterpret_cast<const s16x8*>(b)[v];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float fa = b2f(__ushort_as_bfloat16((unsigned short)va[j]));
      float fb = EW == 0 ? 0.f
                         : b2f(__ushort_as_bfloat16((unsigned short)vb[j]));
      va[j] = (short)__bfloat16_as_ushort(f2b(ew_op<EW>(fa, fb)));
    }
    reinterpret_cast<s16x8*>(y)[v] = va;
  }
  if (t0 == 0) {  // scalar tail (< 8 elems)
    for (long i = n8 * 8; i < n; ++i) {
      float fa = b2f(a[i]);
      float fb = EW == 0 ? 0.f : b2f(b[i]);
      y[i] = f2b(ew_op<EW>(fa, fb));
    }
  }
}

template <int EW>
static torch::Tensor ew_launch(torch::Tensor a, const torch::Tensor* b) {
  CHECK_IN(a);
  auto y = torch::empty_like(a);
  long n = a.numel();
  int grid = (int)std::min<long>((n / 8 + 255) / 256 + 1, 4096L);
  ew_kernel<EW><<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)a.data_ptr(), b ? (const bf16*)b->data_ptr() : nullptr,
      (bf16*)y.data_ptr(), n);
  return y;
}

torch::Tensor relu_fwd(torch::Tensor x) { return ew_launch<0>(x, nullptr); }

torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y) {
  CHECK_IN(y);
  return ew_launch<1>(dy, &y);
}

torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b) {
  CHECK_IN(b);
  return ew_launch<2>(a, &b);
}

// ---- dropout -------------------------------------------------------------
// element dropout: mask per element; channel dropout (Dropout2d on NHWC):
// mask per (n, c) applied across H*W.

__global__ void dropout_fwd_kernel(const bf16* __restrict__ x,
                                   bf16* __restrict__ y, u8* __restrict__ mask,
                                   long n, float p, float scale,
                                   unsigned seed) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    bool keep = hash_uniform(seed, (unsigned)i) >= p;
    mask[i] = keep;
    y[i] = keep ? f2b(b2f(x[i]) * scale) : f2b(0.f);
  }
}

__global__ void dropout2d_fwd_kernel(const bf16* __restrict__ x,
                                     bf16* __restrict__ y,
                                     u8* __restrict__ mask, long n, int hw,
                                     int c, float p, float scale,
                                     unsigned seed) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    long nidx = i / ((long)hw * c);
    int cidx = (int)(i % c);
    long mi = nidx * c + cidx;
    bool keep = hash_uniform(seed, (unsigned)mi) >= p;
    if (i % ((long)hw * c) < (long)c) mask[mi] = keep;  // write once per (n,c)
    y[i] = keep ? f2b(b2f(x[i]) * scale) : f2b(0.f);
  }
}

__global__ void dropout_bwd_kernel(const bf16* __restrict__ dy,
                                   const u8* __restrict__ mask,
                                   bf16* __restrict__ dx, long n, int hw,
                                   int c, float scale, int per_channel) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    long mi = per_channel ? (i / ((long)hw * c)) * c + (i % c) : i;
    dx[i] = mask[mi] ? f2b(b2f(dy[i]) * scale) : f2b(0.f);
  }
}

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p, long seed,
                                       bool per_channel) {
  CHECK_IN(x);
  auto y = torch::empty_like(x);
  long n = x.numel();
  float scale = 1.0f / (1.0f - (float)p);
  int grid = (int)std::min<long>((n + 255) / 256, 4096L);
  torch::Tensor mask;
  if (per_channel) {
    TORCH_CHECK(x.dim() == 4, "dropout2d expects NHWC");
    int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
    mask = torch::empty({(long)N * C}, x.options().dtype(torch::kUInt8));
    dropout2d_fwd_kernel<<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(),
        mask.data_ptr<u8>(), n, H * W, C, (float)p, scale, (unsigned)seed);
  } else {
    mask = torch::empty({n}, x.options().dtype(torch::kUInt8));
    dropout_fwd_kernel<<<grid, 256, 0, cur_stream()>>>(
        (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(),
        mask.data_ptr<u8>(), n, (float)p, scale, (unsigned)seed);
  }
  return {y, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p,
                          bool per_channel) {
  CHECK_IN(dy); CHECK_IN(mask);
  auto dx = torch::empty_like(dy);
  long n = dy.numel();
  float scale = 1.0f / (1.0f - (float)p);
  int grid = (int)std::min<long>((n + 255) / 256, 4096L);
  int hw = 1, c = 1;
  if (per_channel) {
    hw = dy.size(1) * dy.size(2);
    c = dy.size(3);
  }
  dropout_bwd_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), mask.data_ptr<u8>(), (bf16*)dx.data_ptr(),
      n, hw, c, scale, per_channel ? 1 : 0);
  return dx;
}

// ---- channel sum (bias grad): x [rows, C] bf16 -> out fp32 [C] -----------

__global__ void channel_sum_kernel(const bf16* __restrict__ x,
                                   float* __restrict__ out, long rows, int c) {
  // one block per channel-chunk: block handles 64 channels, grid.y strides rows
  int ch = blockIdx.x * 64 + (threadIdx.x % 64);
  int rlane = threadIdx.x / 64;  // 4 row-groups per block
  if (ch >= c) return;
  float acc = 0.f;
  for (long r = blockIdx.y * 4 + rlane; r < rows; r += (long)gridDim.y * 4) {
    acc += b2f(x[r * c + ch]);
  }
  atomicAdd(&out[ch], acc);
}

torch::Tensor channel_sum(torch::Tensor x) {
  CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  auto out = torch::zeros({c}, x.options().dtype(torch::kFloat32));
  dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
  channel_sum_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), out.data_ptr<float>(), rows, c);
  return out;
}

// accumulate the channel sum straight into a pre-zeroed flat-grad view
// (bias grads written by the producing kernel; no aten accumulate-add)
void channel_sum_into(torch::Tensor x, torch::Tensor out) {
  CHECK_IN(x);
  CHECK_IN(out);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
  channel_sum_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), out.data_ptr<float>(), rows, c);
}


// ---- fused conv-weight layout transforms (one kernel per transform) ------
// OIHW fp32 -> KRSC bf16 (fwd operand), KRSC bf16 -> CRSK bf16 (dgrad
// operand), KRSC fp32 -> OIHW fp32 (wgrad output). Each replaces a
// permute+contiguous+cast aten chain (2-3 launches) on the per-step path.

namespace wt {

__global__ void oihw_to_krsc_kernel(const float* __restrict__ w,
                                    bf16* __restrict__ out, int K, int C,
                                    int R, int S) {
  long total = (long)K * C * R * S;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    // out index: (((k*R + r)*S + s)*C + c)
    int c = (int)(i % C);
    long rem = i / C;
    int s = (int)(rem % S);
    rem /= S;
    int r = (int)(rem % R);
    int k = (int)(rem / R);
    out[i] = f2b(w[((long)(k * C + c) * R + r) * S + s]);
  }
}

__global__ void krsc_to_crsk_kernel(const bf16* __restrict__ wk,
                                    bf16* __restrict__ out, int K, int C,
                                    int R, int S) {
  long total = (long)K * C * R * S;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    // out index: (((c*R + r)*S + s)*K + k)
    int k = (int)(i % K);
    long rem = i / K;
    int s = (int)(rem % S);
    rem /= S;
    int r = (int)(rem % R);
    int c = (int)(rem / R);
    out[i] = wk[((long)(k * R + r) * S + s) * C + c];
  }
}

__global__ void krsc_to_oihw_kernel(const float* __restrict__ dwk,
                                    float* __restrict__ out, int K, int C,
                                    int R, int S) {
  long total = (long)K * C * R * S;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    // out (OIHW) index: ((k*C + c)*R + r)*S + s
    int s = (int)(i % S);
    long rem = i / S;
    int r = (int)(rem % R);
    rem /= R;
    int c = (int)(rem % C);
    int k = (int)(rem / C);
    out[i] = dwk[((long)(k * R + r) * S + s) * C + c];
  }
}

}  // namespace wt

static inline int wt_grid(long total) {
  return (int)std::min<long>((total + 255) / 256, 2048L);
}

torch::Tensor oihw_to_krsc(torch::Tensor w) {
  CHECK_IN(w);
  int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto out = torch::empty({K, R, S, C}, w.options().dtype(torch::kBFloat16));
  long total = (long)K * C * R * S;
  wt::oihw_to_krsc_kernel<<<wt_grid(total), 256, 0, cur_stream()>>>(
      w.data_ptr<float>(), (bf16*)out.data_ptr(), K, C, R, S);
  return out;
}

torch::Tensor krsc_to_crsk(torch::Tensor wk) {
  CHECK_IN(wk);
  int K = wk.size(0), R = wk.size(1), S = wk.size(2), C = wk.size(3);
  auto out = torch::empty({C, R, S, K}, wk.options());
  long total = (long)K * C * R * S;
  wt::krsc_to_crsk_kernel<<<wt_grid(total), 256, 0, cur_stream()>>>(
      (const bf16*)wk.data_ptr(), (bf16*)out.data_ptr(), K, C, R, S);
  return out;
}

torch::Tensor krsc_to_oihw(torch::Tensor dwk) {
  CHECK_IN(dwk);
  int K = dwk.size(0), R = dwk.size(1), S = dwk.size(2), C = dwk.size(3);
  auto out = torch::empty({K, C, R, S}, dwk.options());
  long total = (long)K * C * R * S;
  wt::krsc_to_oihw_kernel<<<wt_grid(total), 256, 0, cur_stream()>>>(
      dwk.data_ptr<float>(), out.data_ptr<float>(), K, C, R, S);
  return out;
}

// ---- persistent weight-shadow refresh ------------------------------------
// ONE launch regenerates every conv/linear weight's bf16 KRSC (fwd operand)
// and CRSK (dgrad operand) shadow from the updated fp32 flat params —
// replacing the per-conv oihw_to_krsc + krsc_to_crsk launches on every
// forward/backward. Runs eagerly right after the fused
// SGD step (the last param mutation of a pass), so the hipGraph-captured
// fwd+bwd reads shadows at fixed addresses with fresh values.
//
// Thread mapping: vec8 chunks over the PADDED shadow offset space
// (sh_start[j] is 8-aligned); dst-indexed so shadow writes are coalesced
// 16-B stores, the fp32 reads scatter (stride R*S or C*R*S) but stay
// within one filter's footprint and hit L2.

namespace wt {

__global__ void refresh_shadows_kernel(
    const float* __restrict__ param, const long* __restrict__ p_start,
    const long* __restrict__ sh_start, const int* __restrict__ Ks,
    const int* __restrict__ Cs, const int* __restrict__ Rs,
    const int* __restrict__ Ss, int nseg, long total8,
    bf16* __restrict__ krsc, bf16* __restrict__ crsk) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < 2 * total8;
       v += stride) {
    const bool is_crsk = v >= total8;
    const long e0 = (is_crsk ? v - total8 : v) * 8;  // padded shadow offset
    int lo = 0, hi = nseg - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (sh_start[mid] <= e0) lo = mid; else hi = mid - 1;
    }
    const int j = lo;
    const int K = Ks[j], C = Cs[j], R = Rs[j], S = Ss[j];
    const long numel = (long)K * C * R * S;
    const long loc = e0 - sh_start[j];
    if (loc >= numel) continue;  // pad gap
    const float* w = param + p_start[j];
    __bf16* dst = reinterpret_cast<__bf16*>(
        (is_crsk ? crsk : krsc) + e0);
    const int RS = R * S, CRS = C * RS;
    if (!is_crsk) {
      // krsc idx loc+t = ((k*R+r)*S+s)*C + c  (c fastest)
      const int c0 = (int)(loc % C);
      long rem = loc / C;
      const int s = (int)(rem % S);
      rem /= S;
      const int r = (int)(rem % R);
      const int k = (int)(rem / R);
      if (c0 + 8 <= C && loc + 8 <= numel) {
        const float* src = w + (long)k * CRS + (long)c0 * RS + r * S + s;
        bf16x8 o;
#pragma unroll
        for (int t = 0; t < 8; ++t) o[t] = (__bf16)src[(long)t * RS];
        *reinterpret_cast<bf16x8*>(dst) = o;
      } else {
        for (int t = 0; t < 8 && loc + t < numel; ++t) {
          long i = loc + t;
          int c = (int)(i % C);
          long rm = i / C;
          int ss = (int)(rm % S);
          rm /= S;
          int rr = (int)(rm % R);
          int kk = (int)(rm / R);
          dst[t] = (__bf16)w[(long)kk * CRS + (long)c * RS + rr * S + ss];
        }
      }
    } else {
      // crsk idx loc+t = ((c*R+r)*S+s)*K + k  (k fastest)
      const int k0 = (int)(loc % K);
      long rem = loc / K;
      const int s = (int)(rem % S);
      rem /= S;
      const int r = (int)(rem % R);
      const int c = (int)(rem / R);
      if (k0 + 8 <= K && loc + 8 <= numel) {
        const float* src = w + (long)k0 * CRS + (long)c * RS + r * S + s;
        bf16x8 o;
#pragma unroll
        for (int t = 0; t < 8; ++t) o[t] = (__bf16)src[(long)t * CRS];
        *reinterpret_cast<bf16x8*>(dst) = o;
      } else {
        for (int t = 0; t < 8 && loc + t < numel; ++t) {
          long i = loc + t;
          int kk = (int)(i % K);
          long rm = i / K;
          int ss = (int)(rm % S);
          rm /= S;
          int rr = (int)(rm % R);
          int cc = (int)(rm / R);
          dst[t] = (__bf16)w[(long)kk * CRS + (long)cc * RS + rr * S + ss];
        }
      }
    }
  }
}

}  // namespace wt

void refresh_conv_shadows(torch::Tensor param, torch::Tensor p_start,
                          torch::Tensor sh_start, torch::Tensor Ks,
                          torch::Tensor Cs, torch::Tensor Rs,
                          torch::Tensor Ss, torch::Tensor krsc,
                          torch::Tensor crsk) {
  CHECK_IN(param);
  CHECK_IN(p_start);
  int nseg = (int)p_start.numel();
  if (nseg == 0) return;
  long total8 = krsc.numel() / 8;
  TORCH_CHECK(krsc.numel() % 8 == 0 && crsk.numel() == krsc.numel());
  long work = 2 * total8;
  int grid = (int)std::min<long>((work + 255) / 256, 4096L);
  wt::refresh_shadows_kernel<<<grid, 256, 0, cur_stream()>>>(
      param.data_ptr<float>(), p_start.data_ptr<long>(),
      sh_start.data_ptr<long>(), Ks.data_ptr<int>(), Cs.data_ptr<int>(),
      Rs.data_ptr<int>(), Ss.data_ptr<int>(), nseg, total8,
      (bf16*)krsc.data_ptr(), (bf16*)crsk.data_ptr());
}

}  // namespace eg
