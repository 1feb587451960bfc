// bf16 MFMA GEMM (NT form): C[M,N] = A[M,K] * B[N,K]^T (+bias), fp32 acc.
//
// Used for the Linear layers (cent.cpp MLP, event.cpp CNN fc heads,
// resnet.hpp:154 fc). Both operands are row-major with K innermost, so
// global loads are contiguous 16-byte vectors along K for A and B alike
// ("NT" is the natural layout: torch Linear weight is already [N, K]).
//
// Geometry: 64x64 block tile, BK=32, 4 waves (2x2), each wave a 32x32
// sub-tile = 2x2 fragments of v_mfma_f32_16x16x32_bf16. LDS tiles are
// row-padded by 8 bf16 (16 B) to kill ds_read_b128 bank conflicts
// (guide §6 G4). Register-staged (load->ds_write), single-buffered.
//
// Fragment mapping for mfma_f32_16x16x32_bf16 (cdna guide §3):
//   A (16Mx32K): lane l holds A[m = l&15][k = (l>>4)*8 + j], j=0..7
//   B (32Kx16N): lane l holds B[k = (l>>4)*8 + j][n = l&15]
//   C (16x16 f32x4): lane l holds C[m = (l>>4)*4 + r][n = l&15], r=0..3
// Verified empirically by tests/test_gpu_numerics.py (asymmetric operands).

#include "common.h"

namespace eg {

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int LDK = BK + 8;  // padded LDS row stride (bf16 elems)

__device__ __forceinline__ bf16x8 lds_read8(const __bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

// guarded vec8 global load of row-major [rows, cols] at (r, c..c+7)
__device__ __forceinline__ void load_row8(const bf16* g, long rows, long cols,
                                          long r, long c, __bf16* dst) {
  if (r < rows && c + 8 <= cols && ((r * cols + c) & 7) == 0) {
    *reinterpret_cast<s16x8*>(dst) =
        *reinterpret_cast<const s16x8*>(g + r * cols + c);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = (r < rows && c + j < cols) ? b2f(g[r * cols + c + j]) : 0.f;
      dst[j] = (__bf16)v;
    }
  }
}

__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C, long M, long N,
    long K, int c_bf16, int has_bias) {
  __shared__ __bf16 sA[BM * LDK];
  __shared__ __bf16 sB[BN * LDK];

  const long m0 = (long)blockIdx.x * BM;
  const long n0 = (long)blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;          // 4 waves: (wr, wc) in 2x2
  const int wr = wave >> 1, wc = wave & 1;

  f32x4 acc[2][2] = {};

  // staging assignment: 256 threads, each loads one row-chunk of 8
  const int lr = t >> 2;            // 0..63  tile row
  const int lk = (t & 3) * 8;       // 0,8,16,24 within BK

  for (long k0 = 0; k0 < K; k0 += BK) {
    __bf16 ra[8], rb[8];
    load_row8(A, M, K, m0 + lr, k0 + lk, ra);
    load_row8(B, N, K, n0 + lr, k0 + lk, rb);
    __syncthreads();
    *reinterpret_cast<bf16x8*>(&sA[lr * LDK + lk]) =
        *reinterpret_cast<bf16x8*>(ra);
    *reinterpret_cast<bf16x8*>(&sB[lr * LDK + lk]) =
        *reinterpret_cast<bf16x8*>(rb);
    __syncthreads();

    const int kf = (lane >> 4) * 8;   // fragment k-offset
    const int ml = lane & 15;
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      bf16x8 af = lds_read8(&sA[(wr * 32 + fi * 16 + ml) * LDK + kf]);
#pragma unroll
      for (int fj = 0; fj < 2; ++fj) {
        bf16x8 bf = lds_read8(&sB[(wc * 32 + fj * 16 + ml) * LDK + kf]);
        acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, acc[fi][fj], 0, 0, 0);
      }
    }
  }

  // epilogue: C[m0 + wr*32 + fi*16 + (l>>4)*4 + r][n0 + wc*32 + fj*16 + (l&15)]
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
        float v = acc[fi][fj][r] + bv;
        if (c_bf16) {
          reinterpret_cast<bf16*>(C)[mm * N + nn] = f2b(v);
        } else {
          reinterpret_cast<float*>(C)[mm * N + nn] = v;
        }
      }
    }
  }
}

// A [M,K] bf16, B [N,K] bf16 (i.e. torch Linear weight layout), bias fp32[N]
// or empty; returns [M,N] (bf16 if out_bf16 else fp32).
torch::Tensor gemm_bias(torch::Tensor A, torch::Tensor B, torch::Tensor bias,
                        bool out_bf16) {
  CHECK_IN(A); CHECK_IN(B);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16, "A must be bf16");
  TORCH_CHECK(B.scalar_type() == torch::kBFloat16, "B must be bf16");
  long M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "GEMM K mismatch");
  auto out = torch::empty({M, N}, A.options().dtype(
      out_bf16 ? torch::kBFloat16 : torch::kFloat32));
  bool has_bias = bias.numel() > 0;
  dim3 grid(ceil_div(M, BM), ceil_div(N, BN));
  gemm_nt_kernel<<<grid, 256, 0, cur_stream()>>>(
      (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),
      has_bias ? bias.data_ptr<float>() : nullptr, out.data_ptr(), M, N, K,
      out_bf16 ? 1 : 0, has_bias ? 1 : 0);
  return out;
}

}  // namespace eg
