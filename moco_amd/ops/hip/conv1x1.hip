// 1x1 stride-1 convolution as an MFMA NT-GEMM (NHWC bf16).
//
// In channels_last a 1x1/s1 conv IS a GEMM with the contraction index
// contiguous in BOTH operands:
//   fwd:   out[M, Co] = x[M, Ci] @ W[Co, Ci]^T      (M = N*H*W)
//   dgrad: dx [M, Ci] = g[M, Co] @ Wt[Ci, Co]^T     (Wt = W transposed once)
// so ONE kernel shape serves both: C[M, N] = A[M, K] @ Bt[N, K] with
// bf16 A/B, fp32 accumulate, bf16 C.  (wgrad is a TN GEMM over huge M and
// stays on hipBLASLt.)
//
// Geometry: block = 4 waves; wave w owns rows [m0 + 16w, m0 + 16w + 16) x
// all BN=64 block columns as 4 16x16 v_mfma_f32_16x16x32_bf16 fragments.
// A-fragments load 16 B of K-contiguous data per lane straight from HBM;
// B (the weight tile) is shared by every M-block and stays L2-resident.
// The M-grid is huge (>= 196 blocks at the smallest ResNet-50 shape,
// ~12k at the largest), so occupancy comes from the grid, not the tile.
#include "common.h"
#include "launchers.h"

namespace {

__device__ __forceinline__ bf16x8 ld8(const bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

__global__ __launch_bounds__(256) void gemm_nt_bf16_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ Bt,
    bf16* __restrict__ C, int64_t M, int N, int K) {
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int64_t m0 = ((int64_t)blockIdx.x * 4 + wave) * 16;
  const int n0 = blockIdx.y * 64;
  if (m0 >= M) return;

  const int64_t arow = m0 + (lane & 15) < M ? m0 + (lane & 15) : M - 1;
  const int koff = (lane >> 4) * 8;
  const int nfr = (N - n0) >= 64 ? 4 : (N - n0 + 15) / 16;

  f32x4 acc[4] = {};
  for (int kk = 0; kk < K; kk += 32) {
    const bf16x8 a = ld8(A + arow * K + kk + koff);
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f >= nfr) break;
      const int brow = n0 + f * 16 + (lane & 15);  // < N (N % 16 == 0)
      const bf16x8 b = ld8(Bt + (int64_t)brow * K + kk + koff);
      acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[f], 0, 0, 0);
    }
  }

#pragma unroll
  for (int f = 0; f < 4; ++f) {
    if (f >= nfr) break;
    const int col = n0 + f * 16 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = m0 + (lane >> 4) * 4 + r;
      if (row < M) C[row * N + col] = f2bf(acc[f][r]);
    }
  }
}

}  // namespace

extern "C" int launch_gemm_nt_bf16(void* stream, const void* A,
                                   const void* Bt, void* C, int64_t M,
                                   int64_t N, int64_t K) {
  if (K % 32 != 0 || N % 16 != 0) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)((M + 63) / 64), (unsigned)((N + 63) / 64));
  gemm_nt_bf16_kernel<<<grid, 256, 0, (hipStream_t)stream>>>(
      (const bf16*)A, (const bf16*)Bt, (bf16*)C, M, (int)N, (int)K);
  HIP_CHECK_LAST();
  return 0;
}
