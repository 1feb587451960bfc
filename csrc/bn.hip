// NHWC BatchNorm2d (training + eval) with optional fused ReLU epilogue.
// Replaces torch::nn::BatchNorm2d in the ResNet blocks (resnet.hpp:18-19,
// 41-44). Stats are fp32 over the (N*H*W) rows of the [rows, C] view;
// running stats update matches torch semantics (biased var for
// normalization, unbiased for the running buffer).

#include "common.h"

#include <unordered_map>

namespace eg {

// shared persistent per-channel-count BN stats workspace (zeroed once;
// finalize_kernel resets it after each consume). Also written by the conv
// forward epilogue when collect_bn_stats is set (conv.hip).
static torch::Tensor& bn_ws_tensor(int c, const torch::TensorOptions& f32) {
  static auto* cache = new std::unordered_map<int, torch::Tensor>();
  auto it = cache->find(c);
  if (it == cache->end())
    it = cache->emplace(c, torch::zeros({2 * c}, f32)).first;
  return it->second;
}

float* bn_stats_ws_ptr(int c, torch::TensorOptions opts) {
  return bn_ws_tensor(c, opts.dtype(torch::kFloat32))
      .data_ptr<float>();
}

namespace bn {

// per-channel sum & sumsq of x (bf16 [rows, C]) -> fp32 [C] each.
// Vectorized form (C % 8 == 0, C <= 1024): each lane owns 8 consecutive
// channels via 16-B loads; a 256-thread block covers 256*8/C rows per step
// (guide §6 G13 — scalar bf16 channel reads ran at ~6% of HBM peak).
constexpr int BN_MAXC = 1024;

__global__ void stats_kernel_v(const bf16* __restrict__ x,
                               float* __restrict__ s, float* __restrict__ sq,
                               long rows, int c) {
  __shared__ float ls[BN_MAXC], lsq[BN_MAXC];
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    ls[i] = 0.f;
    lsq[i] = 0.f;
  }
  __syncthreads();
  const int lpr = c >> 3;                 // lanes per row
  const int c0 = (threadIdx.x % lpr) * 8;
  const int rpb = blockDim.x / lpr;       // rows per block-step
  float a[8] = {}, b[8] = {};
  for (long r = (long)blockIdx.x * rpb + threadIdx.x / lpr; r < rows;
       r += (long)gridDim.x * rpb) {
    s16x8 v = *reinterpret_cast<const s16x8*>(&x[r * c + c0]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = b2f(__ushort_as_bfloat16((unsigned short)v[j]));
      a[j] += f;
      b[j] += f * f;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&ls[c0 + j], a[j]);
    atomicAdd(&lsq[c0 + j], b[j]);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    atomicAdd(&s[i], ls[i]);
    atomicAdd(&sq[i], lsq[i]);
  }
}

// scalar fallback for C % 8 != 0
__global__ void stats_kernel(const bf16* __restrict__ x, float* __restrict__ s,
                             float* __restrict__ sq, long rows, int c) {
  int ch = blockIdx.x * 64 + (threadIdx.x & 63);
  int rlane = threadIdx.x >> 6;  // 4 row-groups
  if (ch >= c) return;
  float a = 0.f, b = 0.f;
  for (long r = blockIdx.y * 4 + rlane; r < rows; r += (long)gridDim.y * 4) {
    float v = b2f(x[r * c + ch]);
    a += v;
    b += v * v;
  }
  atomicAdd(&s[ch], a);
  atomicAdd(&sq[ch], b);
}

__global__ void finalize_kernel(float* __restrict__ s, float* __restrict__ sq,
                                float* __restrict__ mean,
                                float* __restrict__ invstd,
                                float* __restrict__ running_mean,
                                float* __restrict__ running_var, long rows,
                                int c, float momentum, float eps,
                                int update_running) {
  int ch = blockIdx.x * blockDim.x + threadIdx.x;
  if (ch >= c) return;
  float m = s[ch] / rows;
  float var = sq[ch] / rows - m * m;
  var = fmaxf(var, 0.f);
  mean[ch] = m;
  invstd[ch] = rsqrtf(var + eps);
  s[ch] = 0.f;   // recycle the persistent workspace (no per-call zeros)
  sq[ch] = 0.f;
  if (update_running) {
    float unbiased = rows > 1 ? var * rows / (rows - 1) : var;
    running_mean[ch] = (1.f - momentum) * running_mean[ch] + momentum * m;
    running_var[ch] = (1.f - momentum) * running_var[ch] + momentum * unbiased;
  }
}

__global__ void eval_stats_kernel(const float* __restrict__ running_mean,
                                  const float* __restrict__ running_var,
                                  float* __restrict__ mean,
                                  float* __restrict__ invstd, int c,
                                  float eps) {
  int ch = blockIdx.x * blockDim.x + threadIdx.x;
  if (ch >= c) return;
  mean[ch] = running_mean[ch];
  invstd[ch] = rsqrtf(running_var[ch] + eps);
}

__global__ void norm_kernel_v(const bf16* __restrict__ x,
                              bf16* __restrict__ y,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta, long n8, int c8,
                              int relu) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n8;
       v += stride) {
    int c0 = (int)(v % c8) * 8;
    s16x8 xv = reinterpret_cast<const s16x8*>(x)[v];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int ch = c0 + j;
      float f = (b2f(__ushort_as_bfloat16((unsigned short)xv[j])) - mean[ch])
                * invstd[ch] * gamma[ch] + beta[ch];
      if (relu && f < 0.f) f = 0.f;
      xv[j] = (short)__bfloat16_as_ushort(f2b(f));
    }
    reinterpret_cast<s16x8*>(y)[v] = xv;
  }
}

__global__ void norm_kernel(const bf16* __restrict__ x, bf16* __restrict__ y,
                            const float* __restrict__ mean,
                            const float* __restrict__ invstd,
                            const float* __restrict__ gamma,
                            const float* __restrict__ beta, long n, int c,
                            int relu) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int ch = (int)(i % c);
    float v = (b2f(x[i]) - mean[ch]) * invstd[ch] * gamma[ch] + beta[ch];
    if (relu && v < 0.f) v = 0.f;
    y[i] = f2b(v);
  }
}

// fused block-tail forward: out = relu(xhat*gamma + beta + residual) —
// the separate normalize kernel's y2 write and the add_relu's y2 read
// disappear (y2 is never needed in backward: the BN bwd gating uses the
// RELU output and its dx path uses x2)
__global__ void norm_add_relu_kernel_v(
    const bf16* __restrict__ x, const bf16* __restrict__ res,
    bf16* __restrict__ out, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, long n8, int c8) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < n8;
       v += stride) {
    int c0 = (int)(v % c8) * 8;
    s16x8 xv = reinterpret_cast<const s16x8*>(x)[v];
    s16x8 rv = reinterpret_cast<const s16x8*>(res)[v];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int ch = c0 + j;
      float f = (b2f(__ushort_as_bfloat16((unsigned short)xv[j])) - mean[ch])
                * invstd[ch] * gamma[ch] + beta[ch]
                + b2f(__ushort_as_bfloat16((unsigned short)rv[j]));
      if (f < 0.f) f = 0.f;
      xv[j] = (short)__bfloat16_as_ushort(f2b(f));
    }
    reinterpret_cast<s16x8*>(out)[v] = xv;
  }
}

// vectorized backward reductions (C % 8 == 0): sum(dy'), sum(dy'*xhat)
__global__ void bwd_stats_kernel_v(const bf16* __restrict__ dy,
                                   const bf16* __restrict__ x,
                                   const bf16* __restrict__ y,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd,
                                   float* __restrict__ sum_dy,
                                   float* __restrict__ sum_dyx, long rows,
                                   int c, int relu) {
  __shared__ float ls[BN_MAXC], lsq[BN_MAXC];
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    ls[i] = 0.f;
    lsq[i] = 0.f;
  }
  __syncthreads();
  const int lpr = c >> 3;
  const int c0 = (threadIdx.x % lpr) * 8;
  const int rpb = blockDim.x / lpr;
  float m[8], is[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    m[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
  }
  float a[8] = {}, b[8] = {};
  for (long r = (long)blockIdx.x * rpb + threadIdx.x / lpr; r < rows;
       r += (long)gridDim.x * rpb) {
    long base = r * c + c0;
    s16x8 vg = *reinterpret_cast<const s16x8*>(&dy[base]);
    s16x8 vx = *reinterpret_cast<const s16x8*>(&x[base]);
    s16x8 vy;
    if (relu) vy = *reinterpret_cast<const s16x8*>(&y[base]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = b2f(__ushort_as_bfloat16((unsigned short)vg[j]));
      if (relu &&
          b2f(__ushort_as_bfloat16((unsigned short)vy[j])) <= 0.f)
        g = 0.f;
      float xh =
          (b2f(__ushort_as_bfloat16((unsigned short)vx[j])) - m[j]) * is[j];
      a[j] += g;
      b[j] += g * xh;
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&ls[c0 + j], a[j]);
    atomicAdd(&lsq[c0 + j], b[j]);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    atomicAdd(&sum_dy[i], ls[i]);
    atomicAdd(&sum_dyx[i], lsq[i]);
  }
}

// backward reductions: sum(dy'), sum(dy' * xhat) per channel (fallback)
__global__ void bwd_stats_kernel(const bf16* __restrict__ dy,
                                 const bf16* __restrict__ x,
                                 const bf16* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 float* __restrict__ sum_dy,
                                 float* __restrict__ sum_dyx, long rows,
                                 int c, int relu) {
  int ch = blockIdx.x * 64 + (threadIdx.x & 63);
  int rlane = threadIdx.x >> 6;
  if (ch >= c) return;
  float a = 0.f, b = 0.f;
  float m = mean[ch], is = invstd[ch];
  for (long r = blockIdx.y * 4 + rlane; r < rows; r += (long)gridDim.y * 4) {
    long i = r * c + ch;
    float g = b2f(dy[i]);
    if (relu && b2f(y[i]) <= 0.f) g = 0.f;
    float xh = (b2f(x[i]) - m) * is;
    a += g;
    b += g * xh;
  }
  atomicAdd(&sum_dy[ch], a);
  atomicAdd(&sum_dyx[ch], b);
}

// fused add_relu backward + the FOLLOWING-in-backward BN's stats: writes
// da = dy * (y > 0) AND accumulates sum(da), sum(da * xhat) for the BN
// that PRODUCED the add_relu's first operand (its backward then skips the
// standalone bwd_stats pass over the same arrays). xb/mean/invstd are that
// BN's saved input/batch stats; the sums land in its pre-zeroed
// dgamma/dbeta flat-grad views (direct-grad convention).
__global__ void relu_bwd_bnstats_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ y,
    const bf16* __restrict__ xb, const float* __restrict__ mean,
    const float* __restrict__ invstd, bf16* __restrict__ da,
    float* __restrict__ sum_dy, float* __restrict__ sum_dyx, long rows,
    int c) {
  __shared__ float ls[BN_MAXC], lsq[BN_MAXC];
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    ls[i] = 0.f;
    lsq[i] = 0.f;
  }
  __syncthreads();
  const int lpr = c >> 3;
  const int c0 = (threadIdx.x % lpr) * 8;
  const int rpb = blockDim.x / lpr;
  float m[8], is[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    m[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
  }
  float a[8] = {}, b[8] = {};
  for (long r = (long)blockIdx.x * rpb + threadIdx.x / lpr; r < rows;
       r += (long)gridDim.x * rpb) {
    long base = r * c + c0;
    s16x8 vg = *reinterpret_cast<const s16x8*>(&dy[base]);
    s16x8 vy = *reinterpret_cast<const s16x8*>(&y[base]);
    s16x8 vx = *reinterpret_cast<const s16x8*>(&xb[base]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = b2f(__ushort_as_bfloat16((unsigned short)vg[j]));
      if (b2f(__ushort_as_bfloat16((unsigned short)vy[j])) <= 0.f) g = 0.f;
      vg[j] = (short)__bfloat16_as_ushort(f2b(g));
      float xh =
          (b2f(__ushort_as_bfloat16((unsigned short)vx[j])) - m[j]) * is[j];
      a[j] += g;
      b[j] += g * xh;
    }
    *reinterpret_cast<s16x8*>(&da[base]) = vg;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(&ls[c0 + j], a[j]);
    atomicAdd(&lsq[c0 + j], b[j]);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < c; i += blockDim.x) {
    atomicAdd(&sum_dy[i], ls[i]);
    atomicAdd(&sum_dyx[i], lsq[i]);
  }
}

__global__ void bwd_dx_kernel_v(const bf16* __restrict__ dy,
                                const bf16* __restrict__ x,
                                const bf16* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ sum_dy,
                                const float* __restrict__ sum_dyx,
                                bf16* __restrict__ dx, long rows, int c,
                                int relu, int training) {
  // fixed 8-channel chunk per thread; per-channel params hoisted out of the
  // rows loop (a %-per-vector variant of this kernel measured 2.7x slower)
  const int lpr = c >> 3;
  const int c0 = (threadIdx.x % lpr) * 8;
  const int rpb = blockDim.x / lpr;
  const float inv_n = 1.0f / (float)rows;
  float m[8], is[8], gm[8], a1[8], a2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int ch = c0 + j;
    m[j] = mean[ch];
    is[j] = invstd[ch];
    gm[j] = gamma[ch];
    a1[j] = sum_dy[ch] * inv_n;
    a2[j] = sum_dyx[ch] * inv_n;
  }
  for (long r = (long)blockIdx.x * rpb + threadIdx.x / lpr; r < rows;
       r += (long)gridDim.x * rpb) {
    long base = r * c + c0;
    s16x8 vg = *reinterpret_cast<const s16x8*>(&dy[base]);
    s16x8 vx = *reinterpret_cast<const s16x8*>(&x[base]);
    s16x8 vy;
    if (relu) vy = *reinterpret_cast<const s16x8*>(&y[base]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gg = b2f(__ushort_as_bfloat16((unsigned short)vg[j]));
      if (relu && b2f(__ushort_as_bfloat16((unsigned short)vy[j])) <= 0.f)
        gg = 0.f;
      float f;
      if (training) {
        float xh = (b2f(__ushort_as_bfloat16((unsigned short)vx[j]))
                    - m[j]) * is[j];
        f = gm[j] * is[j] * (gg - a1[j] - xh * a2[j]);
      } else {
        f = gm[j] * is[j] * gg;
      }
      vg[j] = (short)__bfloat16_as_ushort(f2b(f));
    }
    *reinterpret_cast<s16x8*>(&dx[base]) = vg;
  }
}

__global__ void bwd_dx_kernel(const bf16* __restrict__ dy,
                              const bf16* __restrict__ x,
                              const bf16* __restrict__ y,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const float* __restrict__ gamma,
                              const float* __restrict__ sum_dy,
                              const float* __restrict__ sum_dyx,
                              bf16* __restrict__ dx, long n, long rows, int c,
                              int relu, int training) {
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv_n = 1.0f / (float)rows;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int ch = (int)(i % c);
    float g = b2f(dy[i]);
    if (relu && b2f(y[i]) <= 0.f) g = 0.f;
    float is = invstd[ch];
    float v;
    if (training) {
      float xh = (b2f(x[i]) - mean[ch]) * is;
      v = gamma[ch] * is *
          (g - sum_dy[ch] * inv_n - xh * sum_dyx[ch] * inv_n);
    } else {
      v = gamma[ch] * is * g;
    }
    dx[i] = f2b(v);
  }
}

}  // namespace bn

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  torch::Tensor running_mean,
                                  torch::Tensor running_var, double momentum,
                                  double eps, bool training, bool relu,
                                  bool have_stats) {
  CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({c}, f32);
  auto invstd = torch::empty({c}, f32);
  auto stream = cur_stream();
  if (training) {
    // persistent per-C workspace, zeroed once and reset by finalize_kernel;
    // when have_stats the producing conv's epilogue already filled it.
    auto ssq = bn_ws_tensor(c, f32);
    auto s = ssq.narrow(0, 0, c);
    auto sq = ssq.narrow(0, c, c);
    if (have_stats) {
      // sums already accumulated by the conv epilogue
    } else if (c % 8 == 0 && c <= bn::BN_MAXC) {
      // block cap 256: measured sweet spot between per-thread TLP and the
      // per-channel global-atomic merge (128/512/1024/4096 all slower)
      int rpb = 256 / (c / 8);
      int grid = (int)std::min<long>((rows + rpb - 1) / rpb, 256L);
      bn::stats_kernel_v<<<grid, 256, 0, stream>>>(
          (const bf16*)x.data_ptr(), s.data_ptr<float>(),
          sq.data_ptr<float>(), rows, c);
    } else {
      dim3 grid((c + 63) / 64,
                (unsigned)std::min<long>((rows + 3) / 4, 256L));
      bn::stats_kernel<<<grid, 256, 0, stream>>>(
          (const bf16*)x.data_ptr(), s.data_ptr<float>(),
          sq.data_ptr<float>(), rows, c);
    }
    bn::finalize_kernel<<<ceil_div(c, 128), 128, 0, stream>>>(
        s.data_ptr<float>(), sq.data_ptr<float>(), mean.data_ptr<float>(),
        invstd.data_ptr<float>(), running_mean.data_ptr<float>(),
        running_var.data_ptr<float>(), rows, c, (float)momentum, (float)eps,
        1);
  } else {
    bn::eval_stats_kernel<<<ceil_div(c, 128), 128, 0, stream>>>(
        running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
        mean.data_ptr<float>(), invstd.data_ptr<float>(), c, (float)eps);
  }
  auto y = torch::empty_like(x);
  long n = x.numel();
  if (c % 8 == 0) {
    long n8 = n / 8;
    int grid1 = (int)std::min<long>((n8 + 255) / 256, 4096L);
    bn::norm_kernel_v<<<grid1, 256, 0, stream>>>(
        (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(),
        mean.data_ptr<float>(), invstd.data_ptr<float>(),
        gamma.data_ptr<float>(), beta.data_ptr<float>(), n8, c / 8,
        relu ? 1 : 0);
  } else {
    int grid1 = (int)std::min<long>((n + 255) / 256, 4096L);
    bn::norm_kernel<<<grid1, 256, 0, stream>>>(
        (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(),
        mean.data_ptr<float>(), invstd.data_ptr<float>(),
        gamma.data_ptr<float>(), beta.data_ptr<float>(), n, c, relu ? 1 : 0);
  }
  return {y, mean, invstd};
}

// batch stats + running-stats update WITHOUT the normalize pass: returns
// (mean, invstd). have_stats: the producing conv's epilogue already
// filled the shared workspace.
std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor x,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var,
                                             double momentum, double eps,
                                             bool have_stats) {
  CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({c}, f32);
  auto invstd = torch::empty({c}, f32);
  auto stream = cur_stream();
  auto ssq = bn_ws_tensor(c, f32);
  auto sum = ssq.narrow(0, 0, c);
  auto sq = ssq.narrow(0, c, c);
  if (have_stats) {
    // already accumulated by the conv epilogue
  } else if (c % 8 == 0 && c <= bn::BN_MAXC) {
    int rpb = 256 / (c / 8);
    int grid = (int)std::min<long>((rows + rpb - 1) / rpb, 256L);
    bn::stats_kernel_v<<<grid, 256, 0, stream>>>(
        (const bf16*)x.data_ptr(), sum.data_ptr<float>(),
        sq.data_ptr<float>(), rows, c);
  } else {
    dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
    bn::stats_kernel<<<grid, 256, 0, stream>>>(
        (const bf16*)x.data_ptr(), sum.data_ptr<float>(),
        sq.data_ptr<float>(), rows, c);
  }
  bn::finalize_kernel<<<ceil_div(c, 128), 128, 0, stream>>>(
      sum.data_ptr<float>(), sq.data_ptr<float>(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), running_mean.data_ptr<float>(),
      running_var.data_ptr<float>(), rows, c, (float)momentum, (float)eps,
      1);
  return {mean, invstd};
}

torch::Tensor bn_norm_add_relu(torch::Tensor x, torch::Tensor res,
                               torch::Tensor mean, torch::Tensor invstd,
                               torch::Tensor gamma, torch::Tensor beta) {
  CHECK_IN(x); CHECK_IN(res);
  int c = (int)x.size(-1);
  TORCH_CHECK(c % 8 == 0);
  auto out = torch::empty_like(x);
  long n8 = x.numel() / 8;
  int grid = (int)std::min<long>((n8 + 255) / 256, 4096L);
  bn::norm_add_relu_kernel_v<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), (const bf16*)res.data_ptr(),
      (bf16*)out.data_ptr(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), gamma.data_ptr<float>(),
      beta.data_ptr<float>(), n8, c / 8);
  return out;
}

// da = dy * (y > 0), plus the producing BN's backward stats accumulated
// into its dgamma/dbeta grad views (see relu_bwd_bnstats_kernel)
torch::Tensor relu_bwd_bnstats(torch::Tensor dy, torch::Tensor y,
                               torch::Tensor xb, torch::Tensor mean,
                               torch::Tensor invstd, torch::Tensor dgamma,
                               torch::Tensor dbeta) {
  CHECK_IN(dy); CHECK_IN(y); CHECK_IN(xb);
  CHECK_IN(dgamma); CHECK_IN(dbeta);
  int c = (int)xb.size(-1);
  long rows = xb.numel() / c;
  TORCH_CHECK(c % 8 == 0 && c <= bn::BN_MAXC);
  auto da = torch::empty_like(dy);
  // 512-thread blocks: this kernel also carries the elementwise da write,
  // so it wants more in-flight rows per block than the pure reductions
  // (1024 blocks x 256 thr measured slower: 4x the global atomic merges)
  int rpb = 512 / (c / 8);
  int grid = (int)std::min<long>((rows + rpb - 1) / rpb, 256L);
  bn::relu_bwd_bnstats_kernel<<<grid, 512, 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
      (const bf16*)xb.data_ptr(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), (bf16*)da.data_ptr(),
      dbeta.data_ptr<float>(), dgamma.data_ptr<float>(), rows, c);
  return da;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, torch::Tensor y,
                                  bool relu, bool training,
                                  c10::optional<torch::Tensor> dgamma_out,
                                  c10::optional<torch::Tensor> dbeta_out,
                                  bool stats_ready) {
  CHECK_IN(dy); CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  long n = x.numel();
  auto f32 = x.options().dtype(torch::kFloat32);
  // dgamma == sum(dy'*xhat), dbeta == sum(dy'): when the flat-grad views
  // are supplied (pre-zeroed by the step's zero_grad), the reduction
  // accumulates the gradients in place — no allocation, no aten
  // accumulate-add afterwards.
  torch::Tensor sum_dy, sum_dyx;
  const bool direct = dgamma_out.has_value() && dgamma_out->numel() == c;
  if (direct) {
    CHECK_IN(*dgamma_out); CHECK_IN(*dbeta_out);
    sum_dyx = *dgamma_out;
    sum_dy = *dbeta_out;
  } else {
    auto sums = torch::zeros({2 * c}, f32);
    sum_dy = sums.narrow(0, 0, c);
    sum_dyx = sums.narrow(0, c, c);
  }
  auto stream = cur_stream();
  if (stats_ready) {
    // sums already accumulated into the supplied grad views by the
    // consumer's fused backward kernel (relu_bwd_bnstats)
    TORCH_CHECK(direct, "stats_ready requires dgamma/dbeta views");
  } else if (c % 8 == 0 && c <= bn::BN_MAXC) {
    int rpb = 256 / (c / 8);
    int grid = (int)std::min<long>((rows + rpb - 1) / rpb, 256L);
    bn::bwd_stats_kernel_v<<<grid, 256, 0, stream>>>(
        (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
        (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
        invstd.data_ptr<float>(), sum_dy.data_ptr<float>(),
        sum_dyx.data_ptr<float>(), rows, c, relu ? 1 : 0);
  } else {
    dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
    bn::bwd_stats_kernel<<<grid, 256, 0, stream>>>(
        (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
        (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
        invstd.data_ptr<float>(), sum_dy.data_ptr<float>(),
        sum_dyx.data_ptr<float>(), rows, c, relu ? 1 : 0);
  }
  auto dx = torch::empty_like(dy);
  if (c % 8 == 0) {
    int rpb = 256 / (c / 8);
    int grid1 = (int)std::min<long>((rows + rpb - 1) / rpb, 256L);
    bn::bwd_dx_kernel_v<<<grid1, 256, 0, stream>>>(
        (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
        (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
        invstd.data_ptr<float>(), gamma.data_ptr<float>(),
        sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
        (bf16*)dx.data_ptr(), rows, c, relu ? 1 : 0, training ? 1 : 0);
  } else {
    int grid1 = (int)std::min<long>((n + 255) / 256, 4096L);
    bn::bwd_dx_kernel<<<grid1, 256, 0, stream>>>(
        (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
        (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
        invstd.data_ptr<float>(), gamma.data_ptr<float>(),
        sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
        (bf16*)dx.data_ptr(), n, rows, c, relu ? 1 : 0, training ? 1 : 0);
  }
  // dgamma = sum_dyx, dbeta = sum_dy (already per-channel fp32)
  return {dx, sum_dyx, sum_dy};
}

}  // namespace eg
