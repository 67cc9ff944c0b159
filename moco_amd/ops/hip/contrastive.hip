// Fused contrastive logits for MoCo:
//   fwd: out[n, 0]   = invT * (q[n] . k[n])
//        out[n, 1+j] = invT * (q[n] . queue[:, j])          (N x (1+K) fp32)
//   bwd: dq[n, c]    = invT * (g[n,0]*k[n,c] + sum_j g[n,1+j]*queue[c,j])
// replacing the two einsums + cat + scale at reference builder.py:146-155.
//
// MFMA mapping (v_mfma_f32_16x16x32_bf16, one 16x16 D-tile per fragment):
//   a-frag: lane l supplies A[l&15][(l>>4)*8 + t], t=0..7  (k-contiguous)
//   b-frag: lane l supplies B[(l>>4)*8 + t][l&15]
//   d-frag: lane l, reg r holds D[(l>>4)*4 + r][l&15]
// Both GEMMs are laid out NT (contraction index contiguous in BOTH operands):
//   fwd contracts over C: q [N,C] and queue_kc [K,C] are C-contiguous;
//   bwd contracts over K: g [N,1+K] and queue_ck [C,K] are K-contiguous
// — which is exactly why the queue keeps two working layouts.
#include "common.h"
#include "launchers.h"

namespace {

__device__ __forceinline__ bf16x8 load_bf16x8(const bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}

__device__ __forceinline__ bf16x8 load_f32_as_bf16x8(const float* p) {
  const float4 lo = *reinterpret_cast<const float4*>(p);
  const float4 hi = *reinterpret_cast<const float4*>(p + 4);
  bf16x8 r;
  r[0] = (__bf16)lo.x; r[1] = (__bf16)lo.y;
  r[2] = (__bf16)lo.z; r[3] = (__bf16)lo.w;
  r[4] = (__bf16)hi.x; r[5] = (__bf16)hi.y;
  r[6] = (__bf16)hi.z; r[7] = (__bf16)hi.w;
  return r;
}

// 4-byte-aligned-only variant for the logits-grad rows: their stride is
// 1 + K (odd), so 16-byte vector loads would be misaligned for most rows
// (dword-tolerant loads are a gfx950 behavior, not an ISA guarantee)
__device__ __forceinline__ bf16x8 load_f32_as_bf16x8_u(const float* p) {
  bf16x8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r[i] = (__bf16)p[i];
  return r;
}

// ---------------------------------------------------------------------------
// forward: l_neg GEMM.  Block = 4 waves; BM=16 rows x BN=256 cols
// (wave w covers cols [w*64, w*64+64) as 4 16-col fragments).
// grid = (K/256 or K/16 granules, ceil(N/16)).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void contrastive_fwd_neg_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ queue_kc,
    float* __restrict__ out, int N, int C, int64_t K, float invT) {
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int n0 = blockIdx.y * 16;
  const int64_t jbase = (int64_t)blockIdx.x * 256 + wave * 64;
  if (jbase >= K) return;

  const int arow = min(n0 + (lane & 15), N - 1);
  const int koff = (lane >> 4) * 8;

  f32x4 acc[4] = {};
  const int ksteps = C / 32;
  for (int kk = 0; kk < ksteps; ++kk) {
    const bf16x8 a = load_bf16x8(q + (int64_t)arow * C + kk * 32 + koff);
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int64_t j0 = jbase + f * 16;
      if (j0 >= K) break;
      const int64_t brow = j0 + (lane & 15);  // j0+16 <= K (K % 16 == 0)
      const bf16x8 b =
          load_bf16x8(queue_kc + brow * C + kk * 32 + koff);
      acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[f], 0, 0, 0);
    }
  }

  const int64_t ldo = K + 1;
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    const int64_t j0 = jbase + f * 16;
    if (j0 >= K) break;
    const int64_t col = 1 + j0 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = n0 + (lane >> 4) * 4 + r;
      if (row < N) out[(int64_t)row * ldo + col] = acc[f][r] * invT;
    }
  }
}

// forward: l_pos column (one wave per row)
__global__ __launch_bounds__(256) void contrastive_fwd_pos_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    float* __restrict__ out, int N, int C, int64_t K, float invT) {
  const int lane = threadIdx.x % WAVE;
  const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
  if (row >= N) return;
  float dot = 0.f;
  for (int c = lane; c < C; c += WAVE)
    dot += bf2f(q[(int64_t)row * C + c]) * bf2f(k[(int64_t)row * C + c]);
  dot = wave_reduce_sum(dot);
  if (lane == 0) out[(int64_t)row * (K + 1)] = dot * invT;
}

// ---------------------------------------------------------------------------
// backward: dq_neg = g[:,1:] @ queue_ck^T  (contract over K, split-K with
// fp32 atomics).  Block = 4 waves; wave w owns the 16x16 tile at
// cols c0 = blockIdx.y*64 + w*16; rows n0 = blockIdx.x*16;
// k-slice blockIdx.z covers [z*ks, (z+1)*ks).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void contrastive_bwd_neg_kernel(
    const float* __restrict__ g, const float* __restrict__ queue_ck,
    float* __restrict__ dq, int N, int C, int64_t K, int64_t kslice,
    float invT) {
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int n0 = blockIdx.x * 16;
  const int c0 = blockIdx.y * 64 + wave * 16;
  if (c0 >= C) return;
  const int64_t jlo = (int64_t)blockIdx.z * kslice;
  const int64_t jhi = min(jlo + kslice, K);

  const int arow = min(n0 + (lane & 15), N - 1);
  const int brow = c0 + (lane & 15);  // < C (C % 16 == 0)
  const int koff = (lane >> 4) * 8;
  const int64_t ldg = K + 1;

  f32x4 acc = {};
  for (int64_t j0 = jlo; j0 < jhi; j0 += 32) {
    const bf16x8 a =
        load_f32_as_bf16x8_u(g + (int64_t)arow * ldg + 1 + j0 + koff);
    const bf16x8 b =
        load_f32_as_bf16x8(queue_ck + (int64_t)brow * K + j0 + koff);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = n0 + (lane >> 4) * 4 + r;
    const int col = c0 + (lane & 15);
    if (row < N) atomicAdd(&dq[(int64_t)row * C + col], acc[r] * invT);
  }
}

// backward: dq += g[:,0:1] * k * invT  (runs after the GEMM kernel on the
// same stream — plain read-modify-write)
__global__ __launch_bounds__(256) void contrastive_bwd_pos_kernel(
    const float* __restrict__ g, const bf16* __restrict__ k,
    float* __restrict__ dq, int N, int C, int64_t K, float invT) {
  const int lane = threadIdx.x % WAVE;
  const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
  if (row >= N) return;
  const float g0 = g[(int64_t)row * (K + 1)] * invT;
  for (int c = lane; c < C; c += WAVE)
    dq[(int64_t)row * C + c] += g0 * bf2f(k[(int64_t)row * C + c]);
}

}  // namespace

extern "C" int launch_contrastive_fwd(void* stream, const void* q,
                                      const void* k, const void* queue_kc,
                                      float* out, int64_t N, int64_t C,
                                      int64_t K, float invT) {
  if (C % 32 != 0 || C > 512 || K % 16 != 0)
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  {
    dim3 grid((K + 255) / 256, (N + 15) / 16);
    contrastive_fwd_neg_kernel<<<grid, 256, 0, s>>>(
        (const bf16*)q, (const bf16*)queue_kc, out, (int)N, (int)C, K, invT);
    HIP_CHECK_LAST();
  }
  {
    dim3 grid((N + 3) / 4);
    contrastive_fwd_pos_kernel<<<grid, 256, 0, s>>>(
        (const bf16*)q, (const bf16*)k, out, (int)N, (int)C, K, invT);
    HIP_CHECK_LAST();
  }
  return 0;
}

extern "C" int launch_contrastive_bwd(void* stream, const float* g,
                                      const void* k, const float* queue_ck,
                                      float* dq, int64_t N, int64_t C,
                                      int64_t K, float invT) {
  if (C % 16 != 0 || K % 32 != 0) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  // pick a power-of-two split over K so total blocks land near ~1024
  const int64_t nb_base = ((N + 15) / 16) * ((C + 63) / 64);
  int64_t splits = 1;
  while (splits * 2 * nb_base <= 1024 && (K % (64 * splits * 2)) == 0 &&
         K / (splits * 2) >= 512)
    splits *= 2;
  const int64_t kslice = K / splits;
  {
    dim3 grid((N + 15) / 16, (C + 63) / 64, splits);
    contrastive_bwd_neg_kernel<<<grid, 256, 0, s>>>(
        g, queue_ck, dq, (int)N, (int)C, K, kslice, invT);
    HIP_CHECK_LAST();
  }
  {
    dim3 grid((N + 3) / 4);
    contrastive_bwd_pos_kernel<<<grid, 256, 0, s>>>(
        g, (const bf16*)k, dq, (int)N, (int)C, K, invT);
    HIP_CHECK_LAST();
  }
  return 0;
}
