// NHWC pooling: max_pool2d(2,2) with saved argmax (event.cpp:68-70) and
// avg_pool2d(k,k) (resnet.hpp:152). Floor semantics like torch defaults.

#include "common.h"

namespace eg {

using u8 = unsigned char;

__global__ void maxpool_fwd_kernel(const bf16* __restrict__ x,
                                   bf16* __restrict__ y, u8* __restrict__ idx,
                                   int N, int H, int W, int C, int Ho,
                                   int Wo) {
  long total = (long)N * Ho * Wo * C;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long rem = i / C;
    int wo = (int)(rem % Wo);
    rem /= Wo;
    int ho = (int)(rem % Ho);
    int n = (int)(rem / Ho);
    int h0 = ho * 2, w0 = wo * 2;
    float best = -3.4e38f;
    int bj = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int h = h0 + (j >> 1), w = w0 + (j & 1);
      float v = b2f(x[(((long)n * H + h) * W + w) * C + c]);
      if (v > best) { best = v; bj = j; }
    }
    y[i] = f2b(best);
    idx[i] = (u8)bj;
  }
}

__global__ void maxpool_bwd_kernel(const bf16* __restrict__ dy,
                                   const u8* __restrict__ idx,
                                   bf16* __restrict__ dx, int N, int H, int W,
                                   int C, int Ho, int Wo) {
  long total = (long)N * Ho * Wo * C;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long rem = i / C;
    int wo = (int)(rem % Wo);
    rem /= Wo;
    int ho = (int)(rem % Ho);
    int n = (int)(rem / Ho);
    int j = idx[i];
    int h = ho * 2 + (j >> 1), w = wo * 2 + (j & 1);
    dx[(((long)n * H + h) * W + w) * C + c] = dy[i];
  }
}

std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x) {
  CHECK_IN(x);
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int Ho = H / 2, Wo = W / 2;
  auto y = torch::empty({N, Ho, Wo, C}, x.options());
  auto idx = torch::empty({(long)N * Ho * Wo * C},
                          x.options().dtype(torch::kUInt8));
  long total = (long)N * Ho * Wo * C;
  int grid = (int)std::min<long>((total + 255) / 256, 4096L);
  maxpool_fwd_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(), idx.data_ptr<u8>(), N,
      H, W, C, Ho, Wo);
  return {y, idx};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor dy, torch::Tensor idx, long H,
                             long W) {
  CHECK_IN(dy); CHECK_IN(idx);
  int N = dy.size(0), Ho = dy.size(1), Wo = dy.size(2), C = dy.size(3);
  auto dx = torch::zeros({(long)N, H, W, (long)C}, dy.options());
  long total = (long)N * Ho * Wo * C;
  int grid = (int)std::min<long>((total + 255) / 256, 4096L);
  maxpool_bwd_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), idx.data_ptr<u8>(), (bf16*)dx.data_ptr(),
      N, (int)H, (int)W, C, Ho, Wo);
  return dx;
}

__global__ void avgpool_fwd_kernel(const bf16* __restrict__ x,
                                   bf16* __restrict__ y, int N, int H, int W,
                                   int C, int Ho, int Wo, int k) {
  long total = (long)N * Ho * Wo * C;
  const long stride = (long)gridDim.x * blockDim.x;
  float inv = 1.0f / (k * k);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long rem = i / C;
    int wo = (int)(rem % Wo);
    rem /= Wo;
    int ho = (int)(rem % Ho);
    int n = (int)(rem / Ho);
    float acc = 0.f;
    for (int dh = 0; dh < k; ++dh)
      for (int dw = 0; dw < k; ++dw)
        acc += b2f(x[(((long)n * H + ho * k + dh) * W + wo * k + dw) * C + c]);
    y[i] = f2b(acc * inv);
  }
}

__global__ void avgpool_bwd_kernel(const bf16* __restrict__ dy,
                                   bf16* __restrict__ dx, int N, int H, int W,
                                   int C, int Ho, int Wo, int k) {
  long total = (long)N * H * W * C;
  const long stride = (long)gridDim.x * blockDim.x;
  float inv = 1.0f / (k * k);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    long rem = i / C;
    int w = (int)(rem % W);
    rem /= W;
    int h = (int)(rem % H);
    int n = (int)(rem / H);
    int ho = h / k, wo = w / k;
    float v = 0.f;
    if (ho < Ho && wo < Wo)
      v = b2f(dy[(((long)n * Ho + ho) * Wo + wo) * C + c]) * inv;
    dx[i] = f2b(v);
  }
}

torch::Tensor avgpool_fwd(torch::Tensor x, long k) {
  CHECK_IN(x);
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int Ho = H / (int)k, Wo = W / (int)k;
  auto y = torch::empty({N, Ho, Wo, C}, x.options());
  long total = (long)N * Ho * Wo * C;
  int grid = (int)std::min<long>((total + 255) / 256, 4096L);
  avgpool_fwd_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)x.data_ptr(), (bf16*)y.data_ptr(), N, H, W, C, Ho, Wo,
      (int)k);
  return y;
}

torch::Tensor avgpool_bwd(torch::Tensor dy, long k, long H, long W) {
  CHECK_IN(dy);
  int N = dy.size(0), Ho = dy.size(1), Wo = dy.size(2), C = dy.size(3);
  auto dx = torch::empty({(long)N, H, W, (long)C}, dy.options());
  long total = (long)N * H * W * C;
  int grid = (int)std::min<long>((total + 255) / 256, 4096L);
  avgpool_bwd_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)dy.data_ptr(), (bf16*)dx.data_ptr(), N, (int)H, (int)W, C,
      Ho, Wo, (int)k);
  return dx;
}

}  // namespace eg
