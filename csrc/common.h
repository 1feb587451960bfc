// Common helpers for the eventgrad_amd gfx950 HIP kernels.
// MI355X-native: wave64, MFMA bf16 (v_mfma_f32_16x16x32_bf16), NHWC layouts.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_IN(x) CHECK_DEV(x); CHECK_CONTIG(x)

namespace eg {

using bf16 = __hip_bfloat16;

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float b2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2b(float v) { return __float2bfloat16(v); }

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// defined in bn.hip; used by conv.hip's fused-BN-stats epilogue
float* bn_stats_ws_ptr(int c, torch::TensorOptions opts);

// simple per-element hash RNG (wang hash), uniform in [0,1)
__device__ __forceinline__ float hash_uniform(unsigned seed, unsigned idx) {
  unsigned h = seed ^ (idx * 0x9E3779B9u);
  h = (h ^ 61u) ^ (h >> 16);
  h *= 9u;
  h = h ^ (h >> 4);
  h *= 0x27d4eb2du;
  h = h ^ (h >> 15);
  return (h & 0x00FFFFFFu) * (1.0f / 16777216.0f);
}

}  // namespace eg
