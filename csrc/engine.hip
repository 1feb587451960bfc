// Engine kernels: fused SGD step + per-tensor L2 norms, trigger controller,
// (p+l+r)/3 averaging, payload gather/scatter.
//
// These implement, MI355X-natively, the reference's communication-adjacent
// hot loops: per-tensor torch::norm (event.cpp:325), SGD
// step (optimizer.step), neighbor average p.add_(l).add_(r).div_(3)
// (event.cpp:469-471), flatten/pack to wire format (decent.cpp:183-189),
// and the trigger + adaptive-threshold controller (event.cpp:324-392),
// which runs device-resident here.
//
// Design: ONE flat fp32 buffer per role; per-tensor segments start at
// 64-float-aligned offsets (flat.py), so all elementwise work is float4.
// The SGD kernel accumulates per-segment sum-of-squares of the UPDATED
// params through an LDS per-segment table (per-thread run-accumulation ->
// LDS atomics -> one global atomic per block per segment), fusing the
// trigger's norm input into the step as the BASELINE north star requires.

#include "common.h"

namespace eg {

constexpr int MAX_SEG = 512;  // max named-parameter tensors (ref max: 86)

// binary search: largest i with starts[i] <= idx;  returns -1 if idx < starts[0]
__device__ __forceinline__ int find_seg(const long* starts, int sz, long idx) {
  int lo = 0, hi = sz - 1;
  if (idx < starts[0]) return -1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (starts[mid] <= idx) lo = mid; else hi = mid - 1;
  }
  return lo;
}

// --------------------------------------------------------------------------
// fused SGD + per-segment sqnorm
// --------------------------------------------------------------------------

template <bool DO_STEP>
__global__ void sgd_step_norm_kernel(
    float* __restrict__ param, const float* __restrict__ grad,
    float* __restrict__ mom, const long* __restrict__ g_starts,
    const long* __restrict__ g_numels, int sz, long total4,
    float lr, float momentum, float wd, float* __restrict__ norms_sq) {
  __shared__ long s_starts[MAX_SEG];
  __shared__ long s_ends[MAX_SEG];
  __shared__ float s_norm[MAX_SEG];
  for (int i = threadIdx.x; i < sz; i += blockDim.x) {
    s_starts[i] = g_starts[i];
    s_ends[i] = g_starts[i] + g_numels[i];
    s_norm[i] = 0.f;
  }
  __syncthreads();

  int run_seg = -2;        // current accumulation segment (-2: none)
  float run_acc = 0.f;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < total4;
       v += stride) {
    const long base = v * 4;
    float4 p = reinterpret_cast<float4*>(param)[v];
    float4 g = reinterpret_cast<const float4*>(grad)[v];
    int seg = find_seg(s_starts, sz, base);
    // a float4 never straddles two segments (64-float-aligned starts), but
    // may straddle a segment's end into its pad gap.
    long end = (seg >= 0) ? s_ends[seg] : 0;
    float4 m;
    if (DO_STEP) m = reinterpret_cast<float4*>(mom)[v];
    float* pp = reinterpret_cast<float*>(&p);
    float* gg = reinterpret_cast<float*>(&g);
    float* mm = reinterpret_cast<float*>(&m);
    if (seg != run_seg) {
      if (run_seg >= 0) atomicAdd(&s_norm[run_seg], run_acc);
      run_seg = seg;
      run_acc = 0.f;
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bool in = seg >= 0 && (base + j) < end;
      if (DO_STEP) {
        float gj = gg[j] + wd * pp[j];
        float mj = momentum * mm[j] + gj;
        mm[j] = mj;
        pp[j] -= lr * mj;
      }
      if (in) run_acc += pp[j] * pp[j];
    }
    if (DO_STEP) {
      reinterpret_cast<float4*>(param)[v] = p;
      reinterpret_cast<float4*>(mom)[v] = m;
    }
  }
  if (run_seg >= 0) atomicAdd(&s_norm[run_seg], run_acc);
  __syncthreads();
  for (int i = threadIdx.x; i < sz; i += blockDim.x) {
    if (s_norm[i] != 0.f) atomicAdd(&norms_sq[i], s_norm[i]);
  }
}

static torch::Tensor sgd_or_norm(torch::Tensor param, torch::Tensor grad,
                                 torch::Tensor mom, torch::Tensor starts,
                                 torch::Tensor numels, double lr,
                                 double momentum, double wd, bool do_step) {
  CHECK_IN(param);
  CHECK_IN(starts);
  CHECK_IN(numels);
  int sz = (int)starts.numel();
  TORCH_CHECK(sz <= MAX_SEG, "too many parameter tensors");
  TORCH_CHECK(param.numel() % 4 == 0, "flat buffer must be float4-aligned");
  auto norms = torch::zeros({sz}, param.options());
  long total4 = param.numel() / 4;
  int block = 256;
  int grid = std::min<long>((total4 + block - 1) / block, 2048L);
  if (do_step) {
    CHECK_IN(grad);
    CHECK_IN(mom);
    sgd_step_norm_kernel<true><<<grid, block, 0, cur_stream()>>>(
        param.data_ptr<float>(), grad.data_ptr<float>(),
        mom.data_ptr<float>(), starts.data_ptr<long>(),
        numels.data_ptr<long>(), sz, total4, (float)lr, (float)momentum,
        (float)wd, norms.data_ptr<float>());
  } else {
    sgd_step_norm_kernel<false><<<grid, block, 0, cur_stream()>>>(
        param.data_ptr<float>(), nullptr, nullptr, starts.data_ptr<long>(),
        numels.data_ptr<long>(), sz, total4, 0.f, 0.f, 0.f,
        norms.data_ptr<float>());
  }
  return norms;
}

torch::Tensor sgd_step_norm(torch::Tensor param, torch::Tensor grad,
                            torch::Tensor mom, torch::Tensor starts,
                            torch::Tensor numels, double lr, double momentum,
                            double wd) {
  return sgd_or_norm(param, grad, mom, starts, numels, lr, momentum, wd, true);
}

torch::Tensor seg_sqnorms(torch::Tensor buf, torch::Tensor starts,
                          torch::Tensor numels) {
  return sgd_or_norm(buf, buf, buf, starts, numels, 0, 0, 0, false);
}

// --------------------------------------------------------------------------
// neighbor averaging: p = (p + l + r) / 3   (event.cpp:469-471)
// --------------------------------------------------------------------------

__global__ void avg3_kernel(float* __restrict__ p, const float* __restrict__ l,
                            const float* __restrict__ r, long total4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < total4;
       v += stride) {
    float4 a = reinterpret_cast<float4*>(p)[v];
    float4 b = reinterpret_cast<const float4*>(l)[v];
    float4 c = reinterpret_cast<const float4*>(r)[v];
    const float k = 1.0f / 3.0f;
    a.x = (a.x + b.x + c.x) * k;
    a.y = (a.y + b.y + c.y) * k;
    a.z = (a.z + b.z + c.z) * k;
    a.w = (a.w + b.w + c.w) * k;
    reinterpret_cast<float4*>(p)[v] = a;
  }
}

void avg3(torch::Tensor p, torch::Tensor l, torch::Tensor r) {
  CHECK_IN(p); CHECK_IN(l); CHECK_IN(r);
  long total4 = p.numel() / 4;
  int block = 256;
  int grid = std::min<long>((total4 + block - 1) / block, 4096L);
  avg3_kernel<<<grid, block, 0, cur_stream()>>>(
      p.data_ptr<float>(), l.data_ptr<float>(), r.data_ptr<float>(), total4);
}

// --------------------------------------------------------------------------
// payload gather/scatter (wire format: tight-packed fired segments)
// --------------------------------------------------------------------------

__global__ void gather_segments_kernel(const float* __restrict__ flat,
                                       const long* __restrict__ src_starts,
                                       const long* __restrict__ dst_offs,
                                       int nf, float* __restrict__ out,
                                       long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int f = find_seg(dst_offs, nf, i);
    out[i] = flat[src_starts[f] + (i - dst_offs[f])];
  }
}

torch::Tensor gather_segments(torch::Tensor flat, torch::Tensor src_starts,
                              torch::Tensor dst_offs, long total) {
  CHECK_IN(flat); CHECK_IN(src_starts); CHECK_IN(dst_offs);
  int nf = (int)src_starts.numel();
  auto out = torch::empty({total}, flat.options());
  if (total == 0) return out;
  int block = 256;
  int grid = std::min<long>((total + block - 1) / block, 2048L);
  gather_segments_kernel<<<grid, block, 0, cur_stream()>>>(
      flat.data_ptr<float>(), src_starts.data_ptr<long>(),
      dst_offs.data_ptr<long>(), nf, out.data_ptr<float>(), total);
  return out;
}

__global__ void scatter_segments_kernel(const float* __restrict__ payload,
                                        const long* __restrict__ dst_starts,
                                        const long* __restrict__ src_offs,
                                        int nf, float* __restrict__ inbox,
                                        long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int f = find_seg(src_offs, nf, i);
    inbox[dst_starts[f] + (i - src_offs[f])] = payload[i];
  }
}

void scatter_segments(torch::Tensor payload, torch::Tensor dst_starts,
                      torch::Tensor src_offs, torch::Tensor inbox) {
  CHECK_IN(payload); CHECK_IN(dst_starts); CHECK_IN(src_offs); CHECK_IN(inbox);
  long total = payload.numel();
  if (total == 0) return;
  int nf = (int)dst_starts.numel();
  int block = 256;
  int grid = std::min<long>((total + block - 1) / block, 2048L);
  scatter_segments_kernel<<<grid, block, 0, cur_stream()>>>(
      payload.data_ptr<float>(), dst_starts.data_ptr<long>(),
      src_offs.data_ptr<long>(), nf, inbox.data_ptr<float>(), total);
}

// --------------------------------------------------------------------------
// device-resident trigger + adaptive-threshold controller
// (event.cpp:324-392; state mirrors parallel/controller.py exactly)
// --------------------------------------------------------------------------

__global__ void trigger_update_kernel(
    const float* __restrict__ norms_sq, float* __restrict__ thres,
    float* __restrict__ last_sent_norm, float* __restrict__ last_sent_iter,
    float* __restrict__ slopes, unsigned char* __restrict__ mask,
    int* __restrict__ num_events, int sz, int hist, float pass_num,
    int adaptive, float horizon, float constant, int warmup,
    int always_fire) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= sz) return;
  float norm = sqrtf(norms_sq[i]);
  float value_diff = fabsf(norm - last_sent_norm[i]);
  float iter_diff = fmaxf(pass_num - last_sent_iter[i], 1.0f);
  bool fire;
  if (always_fire) {
    fire = true;
  } else {
    thres[i] = adaptive ? thres[i] * horizon : constant;
    fire = (value_diff >= thres[i]) || (pass_num < (float)warmup);
  }
  if (fire) {
    float slope_avg = 0.f;
    for (int j = 0; j < hist - 1; ++j) {
      float s = slopes[i * hist + j + 1];
      slopes[i * hist + j] = s;
      slope_avg += s;
    }
    float ns = value_diff / iter_diff;
    slopes[i * hist + hist - 1] = ns;
    slope_avg = (slope_avg + ns) / (float)hist;
    if (adaptive && !always_fire) thres[i] = slope_avg;
    last_sent_norm[i] = norm;
    last_sent_iter[i] = pass_num;
    atomicAdd(num_events, 2);
  }
  mask[i] = fire ? 1 : 0;
}

// Pure fire decision — no state mutation. Bit-identical to the decision
// trigger_update takes (same float expressions), so it can run one pass
// EARLY (at the end of the previous optimizer step) to post the mask
// exchange off the host critical path; trigger_update commits later.
__global__ void trigger_decide_kernel(
    const float* __restrict__ norms_sq, const float* __restrict__ thres,
    const float* __restrict__ last_sent_norm, unsigned char* __restrict__ mask,
    int sz, float pass_num, int adaptive, float horizon, float constant,
    int warmup) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= sz) return;
  float norm = sqrtf(norms_sq[i]);
  float value_diff = fabsf(norm - last_sent_norm[i]);
  float th = adaptive ? thres[i] * horizon : constant;
  mask[i] = ((value_diff >= th) || (pass_num < (float)warmup)) ? 1 : 0;
}

torch::Tensor trigger_decide(torch::Tensor norms_sq, torch::Tensor thres,
                             torch::Tensor last_sent_norm, long pass_num,
                             bool adaptive, double horizon, double constant,
                             long warmup) {
  CHECK_IN(norms_sq);
  int sz = (int)norms_sq.numel();
  auto mask = torch::empty({sz}, norms_sq.options().dtype(torch::kUInt8));
  trigger_decide_kernel<<<ceil_div(sz, 128), 128, 0, cur_stream()>>>(
      norms_sq.data_ptr<float>(), thres.data_ptr<float>(),
      last_sent_norm.data_ptr<float>(), mask.data_ptr<unsigned char>(), sz,
      (float)pass_num, adaptive ? 1 : 0, (float)horizon, (float)constant,
      (int)warmup);
  return mask;
}

torch::Tensor trigger_update(torch::Tensor norms_sq, torch::Tensor thres,
                             torch::Tensor last_sent_norm,
                             torch::Tensor last_sent_iter,
                             torch::Tensor slopes, torch::Tensor num_events,
                             long pass_num, bool adaptive, double horizon,
                             double constant, long warmup, bool always_fire) {
  CHECK_IN(norms_sq);
  int sz = (int)norms_sq.numel();
  int hist = (int)(slopes.numel() / sz);
  auto mask = torch::empty({sz}, norms_sq.options().dtype(torch::kUInt8));
  trigger_update_kernel<<<ceil_div(sz, 128), 128, 0, cur_stream()>>>(
      norms_sq.data_ptr<float>(), thres.data_ptr<float>(),
      last_sent_norm.data_ptr<float>(), last_sent_iter.data_ptr<float>(),
      slopes.data_ptr<float>(), mask.data_ptr<unsigned char>(),
      num_events.data_ptr<int>(), sz, hist, (float)pass_num,
      adaptive ? 1 : 0, (float)horizon, (float)constant, (int)warmup,
      always_fire ? 1 : 0);
  return mask;
}

}  // namespace eg
