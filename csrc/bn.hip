// NHWC BatchNorm2d (training + eval) with optional fused ReLU epilogue.
// Replaces torch::nn::BatchNorm2d in the ResNet blocks (resnet.hpp:18-19,
// 41-44). Stats are fp32 over the (N*H*W) rows of the [rows, C] view;
// running stats update matches torch semantics (biased var for
// normalization, unbiased for the running buffer).

#include "common.h"

namespace eg {

namespace bn {

// per-channel sum & sumsq of x (bf16 [rows, C]) -> fp32 [C] each
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

// backward reductions: sum(dy'), sum(dy' * xhat) per channel
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
                                  double eps, bool training, bool relu) {
  CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  auto f32 = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({c}, f32);
  auto invstd = torch::empty({c}, f32);
  auto stream = cur_stream();
  if (training) {
    auto s = torch::zeros({c}, f32);
    auto sq = torch::zeros({c}, f32);
    dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
    bn::stats_kernel<<<grid, 256, 0, stream>>>(
        (const bf16*)x.data_ptr(), s.data_ptr<float>(), sq.data_ptr<float>(),
        rows, c);
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
  int grid1 = (int)std::min<long>((n + 255) / 256, 4096L);
  bn::norm_kernel<<<grid1, 256, 0, stream>>>(
      (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), gamma.data_ptr<float>(),
      beta.data_ptr<float>(), n, c, relu ? 1 : 0);
  return {y, mean, invstd};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, torch::Tensor y,
                                  bool relu, bool training) {
  CHECK_IN(dy); CHECK_IN(x);
  int c = (int)x.size(-1);
  long rows = x.numel() / c;
  long n = x.numel();
  auto f32 = x.options().dtype(torch::kFloat32);
  auto sum_dy = torch::zeros({c}, f32);
  auto sum_dyx = torch::zeros({c}, f32);
  auto stream = cur_stream();
  dim3 grid((c + 63) / 64, (unsigned)std::min<long>((rows + 3) / 4, 256L));
  bn::bwd_stats_kernel<<<grid, 256, 0, stream>>>(
      (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
      (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), sum_dy.data_ptr<float>(),
      sum_dyx.data_ptr<float>(), rows, c, relu ? 1 : 0);
  auto dx = torch::empty_like(dy);
  int grid1 = (int)std::min<long>((n + 255) / 256, 4096L);
  bn::bwd_dx_kernel<<<grid1, 256, 0, stream>>>(
      (const bf16*)dy.data_ptr(), (const bf16*)x.data_ptr(),
      (const bf16*)y.data_ptr(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(), gamma.data_ptr<float>(),
      sum_dy.data_ptr<float>(), sum_dyx.data_ptr<float>(),
      (bf16*)dx.data_ptr(), n, rows, c, relu ? 1 : 0, training ? 1 : 0);
  // dgamma = sum_dyx, dbeta = sum_dy (already per-channel fp32)
  return {dx, sum_dyx, sum_dy};
}

}  // namespace eg
