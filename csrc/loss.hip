// Fused log_softmax + mean NLL loss over [B, C] logits (C small, e.g. 10).
// Matches torch::nll_loss(torch::log_softmax(x,1), t) (cent.cpp:119,
// event.cpp:291). One wave per 64 rows; fp32 math.

#include "common.h"

namespace eg {

__global__ void lsm_nll_fwd_kernel(const bf16* __restrict__ logits,
                                   const long* __restrict__ target,
                                   float* __restrict__ logp,
                                   float* __restrict__ loss_sum, int B,
                                   int C) {
  int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  float mx = -3.4e38f;
  for (int j = 0; j < C; ++j) mx = fmaxf(mx, b2f(logits[row * C + j]));
  float denom = 0.f;
  for (int j = 0; j < C; ++j) denom += __expf(b2f(logits[row * C + j]) - mx);
  float lse = mx + __logf(denom);
  for (int j = 0; j < C; ++j)
    logp[row * C + j] = b2f(logits[row * C + j]) - lse;
  atomicAdd(loss_sum, -logp[row * C + target[row]]);
}

__global__ void lsm_nll_bwd_kernel(const float* __restrict__ logp,
                                   const long* __restrict__ target,
                                   float* __restrict__ dlogits, int B, int C,
                                   const float* __restrict__ dloss) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= B * C) return;
  // dloss read on device (graph-capture safe: no host sync)
  float gscale = dloss[0] / (float)B;
  int row = i / C, j = i % C;
  float soft = __expf(logp[i]);
  float g = (soft - (j == target[row] ? 1.f : 0.f)) * gscale;
  dlogits[i] = g;
}

std::vector<torch::Tensor> logsoftmax_nll_fwd(torch::Tensor logits,
                                              torch::Tensor target) {
  CHECK_IN(logits); CHECK_IN(target);
  int B = (int)logits.size(0), C = (int)logits.size(1);
  auto f32 = logits.options().dtype(torch::kFloat32);
  auto logp = torch::empty({B, C}, f32);
  auto loss = torch::zeros({}, f32);
  lsm_nll_fwd_kernel<<<ceil_div(B, 256), 256, 0, cur_stream()>>>(
      (const bf16*)logits.data_ptr(), target.data_ptr<long>(),
      logp.data_ptr<float>(), loss.data_ptr<float>(), B, C);
  loss.div_((double)B);
  return {loss, logp};
}

torch::Tensor logsoftmax_nll_bwd(torch::Tensor logp, torch::Tensor target,
                                 torch::Tensor dloss) {
  CHECK_IN(logp); CHECK_IN(target); CHECK_DEV(dloss);
  int B = (int)logp.size(0), C = (int)logp.size(1);
  auto d = torch::empty_like(logp);
  lsm_nll_bwd_kernel<<<ceil_div((long)B * C, 256), 256, 0, cur_stream()>>>(
      logp.data_ptr<float>(), target.data_ptr<long>(), d.data_ptr<float>(),
      B, C, dloss.data_ptr<float>());
  return d;
}

}  // namespace eg
