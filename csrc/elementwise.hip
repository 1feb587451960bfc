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

}  // namespace eg
