// Top-k of |theta - theta_prev| per parameter tensor (spevent mode,
// dcifar10/spevent/spevent.cpp:344-363), as an 8-bit-radix histogram select
// on the float bit pattern (non-negative floats order like their bits).
// Selected values are gathered from theta and theta_prev is updated at the
// selected indices in the same pass (spevent.cpp:407-413). Indices are
// int32 on the wire (bitcast — fixing the reference's float-conversion
// precision hazard).

#include "common.h"

namespace eg {

namespace topk {

struct State {
  unsigned prefix;     // selected high bits so far
  unsigned prefix_mask;  // which high bits are fixed
  int k_remain;        // k among elements matching prefix
  int out_count;       // compact-phase cursor (strictly-greater elems)
  int tie_count;       // compact-phase cursor for prefix-equal elems
};

__device__ __forceinline__ unsigned keyof(float a, float b) {
  return __float_as_uint(fabsf(a - b));
}

__global__ void hist_kernel(const float* __restrict__ x,
                            const float* __restrict__ prev, long n,
                            const State* __restrict__ st, int shift,
                            int* __restrict__ hist) {
  __shared__ int h[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) h[i] = 0;
  __syncthreads();
  unsigned prefix = st->prefix, mask = st->prefix_mask;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned key = keyof(x[i], prev[i]);
    if ((key & mask) == prefix) atomicAdd(&h[(key >> shift) & 0xFF], 1);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    if (h[i]) atomicAdd(&hist[i], h[i]);
}

// single thread: walk the histogram from the top bin down, fix 8 more bits
__global__ void scan_kernel(State* st, int* hist, int shift) {
  int k = st->k_remain;
  int bin = 255;
  for (; bin >= 0; --bin) {
    if (hist[bin] >= k) break;
    k -= hist[bin];
  }
  if (bin < 0) bin = 0;  // defensive; cannot happen when k <= n
  st->prefix |= ((unsigned)bin) << shift;
  st->prefix_mask |= 0xFFu << shift;
  st->k_remain = k;
  for (int i = 0; i < 256; ++i) hist[i] = 0;
}

// after 4 rounds prefix_mask == 0xFFFFFFFF: threshold key = prefix.
// Elements with key > prefix are all selected; prefix-equal elements fill
// the remaining slots (ties broken arbitrarily).
__global__ void compact_kernel(const float* __restrict__ x,
                               float* __restrict__ prev, long n, State* st,
                               int k, float* __restrict__ vals,
                               int* __restrict__ idx) {
  unsigned thr = st->prefix;
  int greater_total = k - st->k_remain;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned key = keyof(x[i], prev[i]);
    int slot = -1;
    if (key > thr) {
      slot = atomicAdd(&st->out_count, 1);
    } else if (key == thr) {
      int t = atomicAdd(&st->tie_count, 1);
      if (t < st->k_remain) slot = greater_total + t;
    }
    if (slot >= 0 && slot < k) {
      float v = x[i];
      vals[slot] = v;
      idx[slot] = (int)i;
      prev[i] = v;  // update last-sent value at the sent index
    }
  }
}

}  // namespace topk

// x, prev: fp32 views of ONE parameter segment (same length). Returns
// (vals fp32[k], idx int32[k]) and updates prev[idx] = x[idx] in place.
std::vector<torch::Tensor> topk_absdiff(torch::Tensor x, torch::Tensor prev,
                                        long k) {
  CHECK_DEV(x); CHECK_DEV(prev);
  long n = x.numel();
  TORCH_CHECK(k >= 1 && k <= n, "bad k");
  auto opts_i = x.options().dtype(torch::kInt32);
  auto vals = torch::empty({k}, x.options());
  auto idx = torch::empty({k}, opts_i);
  auto hist = torch::zeros({256}, opts_i);
  auto st = torch::zeros({(long)sizeof(topk::State) / 4}, opts_i);
  // init state: k_remain = k (prefix/mask/counters zero)
  {
    auto st_cpu = torch::zeros_like(st, st.options().device(torch::kCPU));
    st_cpu[2] = (int)k;
    st.copy_(st_cpu, /*non_blocking=*/true);
  }
  auto stream = cur_stream();
  int grid = (int)std::min<long>((n + 255) / 256, 1024L);
  auto* stp = reinterpret_cast<topk::State*>(st.data_ptr<int>());
  for (int shift = 24; shift >= 0; shift -= 8) {
    topk::hist_kernel<<<grid, 256, 0, stream>>>(
        x.data_ptr<float>(), prev.data_ptr<float>(), n, stp, shift,
        hist.data_ptr<int>());
    topk::scan_kernel<<<1, 1, 0, stream>>>(stp, hist.data_ptr<int>(), shift);
  }
  topk::compact_kernel<<<grid, 256, 0, stream>>>(
      x.data_ptr<float>(), prev.data_ptr<float>(), n, stp, (int)k,
      vals.data_ptr<float>(), idx.data_ptr<int>());
  return {vals, idx};
}

// scatter received (vals, idx) into a replica segment (spevent receive side)
__global__ void scatter_update_kernel(float* __restrict__ seg,
                                      const float* __restrict__ vals,
                                      const int* __restrict__ idx, int k) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < k) seg[idx[i]] = vals[i];
}

void scatter_update(torch::Tensor seg, torch::Tensor vals, torch::Tensor idx) {
  CHECK_DEV(seg);
  int k = (int)vals.numel();
  if (k == 0) return;
  scatter_update_kernel<<<ceil_div(k, 256), 256, 0, cur_stream()>>>(
      seg.data_ptr<float>(), vals.data_ptr<float>(), idx.data_ptr<int>(), k);
}

}  // namespace eg
