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

// ---------------------------------------------------------------------------
// LDS-staged variant (guide §5 ladder step-3 structure): 128x128 tile,
// BK=64, double-buffered LDS filled by global_load_lds dwordx4, XOR-swizzled
// images (T2, rule 21: linear LDS dest + pre-swizzled SOURCE + swizzled
// read), 4 waves in a 2x2 wave grid, 32 v_mfma_f32_16x16x32_bf16 per K-tile.
// Requires M % 128 == 0, N % 128 == 0, K % 64 == 0 (all ResNet-50 1x1/s1
// shapes at batch 256 qualify); the launcher falls back to the naive kernel
// otherwise.
// ---------------------------------------------------------------------------
namespace {

constexpr int BM = 128, BK = 64;
constexpr int STAT_SLOTS = 256;  // slot-striped channel reductions
// tile images: A [128 rows][64 k] bf16 = 16 KiB; B [BNT rows][64 k]

__device__ __forceinline__ unsigned swz(unsigned row, unsigned kbyte) {
  return kbyte ^ ((row & 7) << 4);
}

// stage a [ROWS][64] bf16 tile (ROWS*128 bytes) into LDS via glds.
template <int ROWS>
__device__ __forceinline__ void stage_tile(const bf16* __restrict__ src,
                                           int64_t ldK, char* smem,
                                           int wave, int lane) {
#pragma unroll
  for (int c = 0; c < ROWS / 32; ++c) {
    const int piece = c * 256 + wave * 64 + lane;
    const int row = piece / 8;
    const int kb = (piece % 8) * 16;          // byte offset within the row
    const int src_elem = swz(row, kb) / 2;    // pre-swizzled source column
    // per-wave uniform LDS base; the DMA adds lane*16 itself
    char* lds_base = smem + (int64_t)(c * 256 + wave * 64) * 16;
    // C-style casts lower to addrspacecast (reinterpret_cast is rejected)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            src + (int64_t)row * ldK + src_elem),
        (__attribute__((address_space(3))) void*)lds_base, 16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8 lds_frag(const char* smem, int row,
                                           int kbyte) {
  return *reinterpret_cast<const bf16x8*>(smem + row * (BK * 2) +
                                          swz(row, kbyte));
}

// BNT = 128: 2x2 wave grid, 64x64 per wave (MR=NR=4)
// BNT =  64: 4x1 wave grid, 32x64 per wave (MR=2, NR=4)
// DBUF: double-buffered staging (off for single-K-tile problems)
// APPLY (conv+BN fusion): transform every A element with
//   x' = relu?(x * scale[k] + shift[k])  — i.e. the BatchNorm apply of the
//   PREVIOUS layer folded into this GEMM's fragment path (scale = rstd*gamma,
//   shift = beta - mean*rstd*gamma, precomputed per channel on the host).
// STATS: accumulate per-output-channel (sum, sumsq) of the fp32 accumulator
//   tile into stats_sum/stats_sumsq (pre-zeroed, fp32 atomics) — the BN
//   statistics of THIS layer's output for free.
// a_out (runtime, with APPLY): materialize the transformed A operand (the
//   consumer conv's backward wgrad needs it); only the blockIdx.y==0 / wc==0
//   waves write, so each element is written exactly once.
// c_add (runtime): epilogue C += c_add — fuses the residual-path gradient
//   accumulation into the dgrad GEMM's store tail.
template <int BNT, bool DBUF, bool APPLY, bool STATS>
__global__ __launch_bounds__(256, 2) void gemm_nt_lds_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ Bt,
    bf16* __restrict__ C, int64_t M, int N, int K,
    const float* __restrict__ a_scale, const float* __restrict__ a_shift,
    bool a_relu, float* __restrict__ stats_sum,
    float* __restrict__ stats_sumsq, bf16* __restrict__ a_out,
    const bf16* __restrict__ c_add) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int MR = BNT == 128 ? 4 : 2;
  constexpr int ABYTES = BM * BK * 2;    // 16 KiB
  constexpr int BBYTES = BNT * BK * 2;   // 16 or 8 KiB
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int wr = BNT == 128 ? (wave >> 1) : wave;
  const int wc = BNT == 128 ? (wave & 1) : 0;
  const int wrows = BNT == 128 ? 64 : 32;
  const int64_t m0 = (int64_t)blockIdx.x * BM;
  const int n0 = blockIdx.y * BNT;

  f32x4 acc[MR][4] = {};

  stage_tile<BM>(A + m0 * K, K, smem, wave, lane);
  stage_tile<BNT>(Bt + (int64_t)n0 * K, K, smem + ABYTES, wave, lane);
  __syncthreads();  // hipcc emits the vmcnt(0) drain for the in-flight DMA

  int cur = 0;
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    if (DBUF && t + 1 < ntiles) {
      char* nbuf = smem + (cur ^ 1) * (ABYTES + BBYTES);
      stage_tile<BM>(A + m0 * K + (t + 1) * BK, K, nbuf, wave, lane);
      stage_tile<BNT>(Bt + (int64_t)n0 * K + (t + 1) * BK, K, nbuf + ABYTES,
                      wave, lane);
    }
    const char* abuf = smem + cur * (ABYTES + BBYTES);
    const char* bbuf = abuf + ABYTES;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[MR], bf[4];
      const int kbyte = ks * 64 + (lane >> 4) * 16;
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
        af[mr] = lds_frag(abuf, wr * wrows + mr * 16 + (lane & 15), kbyte);
      if (APPLY) {
        const int c0 = t * BK + ks * 32 + (lane >> 4) * 8;
        const float4 sc0 = *reinterpret_cast<const float4*>(a_scale + c0);
        const float4 sc1 = *reinterpret_cast<const float4*>(a_scale + c0 + 4);
        const float4 sh0 = *reinterpret_cast<const float4*>(a_shift + c0);
        const float4 sh1 = *reinterpret_cast<const float4*>(a_shift + c0 + 4);
        const float scs[8] = {sc0.x, sc0.y, sc0.z, sc0.w,
                              sc1.x, sc1.y, sc1.z, sc1.w};
        const float shs[8] = {sh0.x, sh0.y, sh0.z, sh0.w,
                              sh1.x, sh1.y, sh1.z, sh1.w};
#pragma unroll
        for (int mr = 0; mr < MR; ++mr)
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float v = fmaf((float)af[mr][j], scs[j], shs[j]);
            if (a_relu) v = fmaxf(v, 0.f);
            af[mr][j] = (__bf16)v;
          }
        if (a_out != nullptr && blockIdx.y == 0 && wc == 0) {
          // each transformed A element written once: rows wr*wrows + mr*16 +
          // (lane&15), cols c0..c0+8 (this lane's fragment span)
#pragma unroll
          for (int mr = 0; mr < MR; ++mr) {
            const int64_t row = m0 + wr * wrows + mr * 16 + (lane & 15);
            *reinterpret_cast<bf16x8*>(a_out + row * K + c0) = af[mr];
          }
        }
      }
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        bf[nr] = lds_frag(bbuf, wc * 64 + nr * 16 + (lane & 15), kbyte);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
#pragma unroll
        for (int nr = 0; nr < 4; ++nr)
          acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mr], bf[nr], acc[mr][nr], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (DBUF || t + 1 < ntiles) __syncthreads();
    if (DBUF) cur ^= 1;
  }

  if (STATS) {
    // per-output-channel partial (sum, sumsq) of the fp32 accumulators.
    // Column c is held by lanes {l, l+16, l+32, l+48}: butterfly those
    // four, then one atomic pair per (lane<16, nr).  The destination is
    // SLOT-STRIPED [STAT_SLOTS, N] (slot = blockIdx.x % STAT_SLOTS): with a
    // flat [N] buffer every M-block of a big grid serializes on the same
    // few cache lines (measured 4.8 ms on an M=802816, N=64 dispatch);
    // striping cuts per-line contention ~STAT_SLOTS-fold and bn_finalize
    // reduces the slots.
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      float ssum = 0.f, ssq = 0.f;
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float v = acc[mr][nr][r];
          ssum += v;
          ssq += v * v;
        }
      ssum += __shfl_xor(ssum, 16, WAVE);
      ssq += __shfl_xor(ssq, 16, WAVE);
      ssum += __shfl_xor(ssum, 32, WAVE);
      ssq += __shfl_xor(ssq, 32, WAVE);
      if ((lane >> 4) == 0) {
        const int64_t slot = (int64_t)(blockIdx.x % STAT_SLOTS) * N;
        const int c = n0 + wc * 64 + nr * 16 + lane;
        atomicAdd(&stats_sum[slot + c], ssum);
        atomicAdd(&stats_sumsq[slot + c], ssq);
      }
    }
  }

  // Widened epilogue (guide T21 spirit): the fragment layout leaves each
  // lane with 2-byte column-scattered values; staging the wave's tile
  // through LDS turns the store tail into 16-byte coalesced rows.
  __syncthreads();  // k-loop LDS reads are done; smem is reusable
  bf16* ct = reinterpret_cast<bf16*>(smem) + wave * wrows * 64;
#pragma unroll
  for (int mr = 0; mr < MR; ++mr)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      const int col = nr * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int trow = mr * 16 + (lane >> 4) * 4 + r;
        ct[trow * 64 + col] = f2bf(acc[mr][nr][r]);
      }
    }
  __syncthreads();
  const int npieces = wrows * 64 / 8 / WAVE;  // 16B pieces per lane
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    if (i >= npieces) break;
    const int piece = i * WAVE + lane;
    const int trow = piece / 8;
    const int tcol = (piece % 8) * 8;
    const int64_t row = m0 + wr * wrows + trow;
    const int64_t off = row * N + n0 + wc * 64 + tcol;
    bf16x8 v = *reinterpret_cast<const bf16x8*>(ct + trow * 64 + tcol);
    if (c_add != nullptr) {
      const bf16x8 ad = *reinterpret_cast<const bf16x8*>(c_add + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (__bf16)((float)v[j] + (float)ad[j]);
    }
    *reinterpret_cast<bf16x8*>(C + off) = v;
  }
}

}  // namespace

extern "C" int launch_gemm_nt_lds_bf16(void* stream, const void* A,
                                       const void* Bt, void* C, int64_t M,
                                       int64_t N, int64_t K,
                                       const void* c_add) {
  const bool n128 = (N % 128 == 0);
  const bool n64 = (N % 64 == 0);
  if (M % 128 != 0 || K % 64 != 0 || !(n128 || n64)) {
    if (c_add != nullptr) return (int)hipErrorInvalidValue;
    return launch_gemm_nt_bf16(stream, A, Bt, C, M, N, K);
  }
  hipStream_t s = (hipStream_t)stream;
  const int BNT = n128 ? 128 : 64;
  const bool dbuf = (K / 64) > 1;
  const int smem = (16384 + BNT * 128) * (dbuf ? 2 : 1);
  dim3 grid((unsigned)(M / 128), (unsigned)(N / BNT));
#define GO(BNTv, DB)                                                       \
  gemm_nt_lds_kernel<BNTv, DB, false, false><<<grid, 256, smem, s>>>(      \
      (const bf16*)A, (const bf16*)Bt, (bf16*)C, M, (int)N, (int)K,        \
      nullptr, nullptr, false, nullptr, nullptr, nullptr,                  \
      (const bf16*)c_add)
  if (BNT == 128 && dbuf) GO(128, true);
  else if (BNT == 128) GO(128, false);
  else if (dbuf) GO(64, true);
  else GO(64, false);
#undef GO
  HIP_CHECK_LAST();
  return 0;
}

// Round-2 prototype entry: GEMM with the BN apply of the previous layer
// fused into the A-fragment path and/or BN statistics of the output fused
// into the epilogue.  stats_* must be pre-zeroed [N] fp32 when requested.
extern "C" int launch_gemm_nt_fused_bf16(void* stream, const void* A,
                                         const void* Bt, void* C, int64_t M,
                                         int64_t N, int64_t K,
                                         const float* a_scale,
                                         const float* a_shift, bool a_relu,
                                         float* stats_sum,
                                         float* stats_sumsq, void* a_out,
                                         const void* c_add) {
  if (M % 128 != 0 || K % 64 != 0 || (N % 64 != 0))
    return (int)hipErrorInvalidValue;
  if (a_out != nullptr && a_scale == nullptr) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int BNT = (N % 128 == 0) ? 128 : 64;
  const bool dbuf = (K / 64) > 1;
  const int smem = (16384 + BNT * 128) * (dbuf ? 2 : 1);
  const bool apply = a_scale != nullptr;
  const bool stats = stats_sum != nullptr;
  dim3 grid((unsigned)(M / 128), (unsigned)(N / BNT));
#define GO(BNTv, DB, AP, ST)                                               \
  gemm_nt_lds_kernel<BNTv, DB, AP, ST><<<grid, 256, smem, s>>>(            \
      (const bf16*)A, (const bf16*)Bt, (bf16*)C, M, (int)N, (int)K,        \
      a_scale, a_shift, a_relu, stats_sum, stats_sumsq, (bf16*)a_out,      \
      (const bf16*)c_add)
#define GO2(BNTv, DB)                                                      \
  do {                                                                     \
    if (apply && stats) GO(BNTv, DB, true, true);                          \
    else if (apply) GO(BNTv, DB, true, false);                             \
    else if (stats) GO(BNTv, DB, false, true);                             \
    else GO(BNTv, DB, false, false);                                       \
  } while (0)
  if (BNT == 128 && dbuf) GO2(128, true);
  else if (BNT == 128) GO2(128, false);
  else if (dbuf) GO2(64, true);
  else GO2(64, false);
#undef GO2
#undef GO
  HIP_CHECK_LAST();
  return 0;
}

// ---------------------------------------------------------------------------
// Round-2 prototype: dgrad GEMM with the NEXT stage's BN backward folded
// into the epilogue.  For the chain y = relu(bn(z) + res), the consumer
// conv's dgrad GEMM produces dy; this variant post-processes each output
// tile in-register/LDS:
//     g'      = dy * (y > 0)                      (the relu mask)
//     dbeta  += sum g'          per channel       (fp32 atomics)
//     dgamma += sum g' * zhat,  zhat = (z-mean)*rstd
// and stores g' — replacing the separate 3-read bn_bwd_reduce pass.
// Channel bookkeeping rides the widened LDS-transpose epilogue: a lane's
// 16-byte piece always covers the same 8 output channels ((lane%8)*8), so
// per-channel partials accumulate in 8 registers and reduce across the 8
// same-class lanes with three butterflies.
// ---------------------------------------------------------------------------
namespace {

// c_add (nullable): epilogue C += c_add BEFORE the relu mask — the
// residual-path gradient joins the dgrad sum, then the whole sum is masked
// (y = relu(bn(z) + res): upstream grads add first, then the mask).
template <int BNT>
__global__ __launch_bounds__(256, 2) void gemm_nt_dgrad_epi_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ Bt,
    bf16* __restrict__ C, int64_t M, int N, int K,
    const bf16* __restrict__ y, const bf16* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dbeta, float* __restrict__ dgamma,
    const bf16* __restrict__ c_add) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int MR = BNT == 128 ? 4 : 2;
  constexpr int ABYTES = BM * BK * 2;
  constexpr int BBYTES = BNT * BK * 2;
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int wr = BNT == 128 ? (wave >> 1) : wave;
  const int wc = BNT == 128 ? (wave & 1) : 0;
  const int wrows = BNT == 128 ? 64 : 32;
  const int64_t m0 = (int64_t)blockIdx.x * BM;
  const int n0 = blockIdx.y * BNT;

  f32x4 acc[MR][4] = {};
  stage_tile<BM>(A + m0 * K, K, smem, wave, lane);
  stage_tile<BNT>(Bt + (int64_t)n0 * K, K, smem + ABYTES, wave, lane);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / BK;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      char* nbuf = smem + (cur ^ 1) * (ABYTES + BBYTES);
      stage_tile<BM>(A + m0 * K + (t + 1) * BK, K, nbuf, wave, lane);
      stage_tile<BNT>(Bt + (int64_t)n0 * K + (t + 1) * BK, K, nbuf + ABYTES,
                      wave, lane);
    }
    const char* abuf = smem + cur * (ABYTES + BBYTES);
    const char* bbuf = abuf + ABYTES;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[MR], bf[4];
      const int kbyte = ks * 64 + (lane >> 4) * 16;
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
        af[mr] = lds_frag(abuf, wr * wrows + mr * 16 + (lane & 15), kbyte);
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        bf[nr] = lds_frag(bbuf, wc * 64 + nr * 16 + (lane & 15), kbyte);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
#pragma unroll
        for (int nr = 0; nr < 4; ++nr)
          acc[mr][nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mr], bf[nr], acc[mr][nr], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // transpose through LDS, then mask + reduce + store row-contiguously
  __syncthreads();
  bf16* ct = reinterpret_cast<bf16*>(smem) + wave * wrows * 64;
#pragma unroll
  for (int mr = 0; mr < MR; ++mr)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      const int col = nr * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int trow = mr * 16 + (lane >> 4) * 4 + r;
        ct[trow * 64 + col] = f2bf(acc[mr][nr][r]);
      }
    }
  __syncthreads();

  // per-lane channel class: piece = i*64 + lane, 64 % 8 == 0 => this lane
  // always touches channels c8 = (lane % 8) * 8 .. +8
  const int c8 = n0 + wc * 64 + (lane % 8) * 8;
  float sdb[8] = {}, sdg[8] = {};
  float mn[8], rs[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mn[j] = mean[c8 + j];
    rs[j] = rstd[c8 + j];
  }
  const int npieces = wrows * 64 / 8 / WAVE;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    if (i >= npieces) break;
    const int piece = i * WAVE + lane;
    const int trow = piece / 8;
    const int tcol = (piece % 8) * 8;
    const int64_t row = m0 + wr * wrows + trow;
    const int64_t off = row * N + n0 + wc * 64 + tcol;
    const bf16x8 gv = *reinterpret_cast<const bf16x8*>(ct + trow * 64 + tcol);
    const bf16x8 yv = *reinterpret_cast<const bf16x8*>(y + off);
    const bf16x8 zv = *reinterpret_cast<const bf16x8*>(z + off);
    bf16x8 av = {};
    if (c_add != nullptr)
      av = *reinterpret_cast<const bf16x8*>(c_add + off);
    bf16x8 gp;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = (float)gv[j];
      if (c_add != nullptr) g += (float)av[j];
      if ((float)yv[j] <= 0.f) g = 0.f;
      gp[j] = (__bf16)g;
      sdb[j] += g;
      sdg[j] += g * ((float)zv[j] - mn[j]) * rs[j];
    }
    *reinterpret_cast<bf16x8*>(C + off) = gp;
  }
  // reduce across the 8 lanes of each channel class (xor 8/16/32 preserve
  // lane % 8), then one atomic pair per channel from lanes 0..7 into the
  // slot-striped [STAT_SLOTS, N] partials (see the STATS epilogue note)
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    for (int m = 8; m <= 32; m <<= 1) {
      sdb[j] += __shfl_xor(sdb[j], m, WAVE);
      sdg[j] += __shfl_xor(sdg[j], m, WAVE);
    }
  }
  if (lane < 8) {
    const int64_t slot = (int64_t)(blockIdx.x % STAT_SLOTS) * N;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      atomicAdd(&dbeta[slot + c8 + j], sdb[j]);
      atomicAdd(&dgamma[slot + c8 + j], sdg[j]);
    }
  }
}

}  // namespace

extern "C" int launch_gemm_nt_dgrad_epi(void* stream, const void* A,
                                        const void* Bt, void* C, int64_t M,
                                        int64_t N, int64_t K, const void* y,
                                        const void* z, const float* mean,
                                        const float* rstd, float* dbeta,
                                        float* dgamma, const void* c_add) {
  if (M % 128 != 0 || K % 64 != 0 || N % 64 != 0)
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int BNT = (N % 128 == 0) ? 128 : 64;
  const int smem = (16384 + BNT * 128) * 2;
  dim3 grid((unsigned)(M / 128), (unsigned)(N / BNT));
  if (BNT == 128)
    gemm_nt_dgrad_epi_kernel<128><<<grid, 256, smem, s>>>(
        (const bf16*)A, (const bf16*)Bt, (bf16*)C, M, (int)N, (int)K,
        (const bf16*)y, (const bf16*)z, mean, rstd, dbeta, dgamma,
        (const bf16*)c_add);
  else
    gemm_nt_dgrad_epi_kernel<64><<<grid, 256, smem, s>>>(
        (const bf16*)A, (const bf16*)Bt, (bf16*)C, M, (int)N, (int)K,
        (const bf16*)y, (const bf16*)z, mean, rstd, dbeta, dgamma,
        (const bf16*)c_add);
  HIP_CHECK_LAST();
  return 0;
}

// ---------------------------------------------------------------------------
// fp8 (OCP e4m3) GEMM on the MX-scaled MFMA — BASELINE config-4 path.
// gfx950's NON-scaled fp8 MFMA runs at the bf16 rate; the 2x fp8 rate is
// only reachable through v_mfma_scale_f32_16x16x128_f8f6f4 (guide §3/§4).
// Per-TENSOR software scaling: operands are pre-scaled into e4m3 range at
// quantization, the hardware per-block E8M0 scales are set to 1.0
// (byte 127), and the epilogue multiplies the fp32 accumulators by
// sa*sb.  A BK=128-fp8 K-tile is byte-identical to the bf16 BK=64 tile
// (128 B/row), so the stage_tile/XOR-swizzle machinery is reused as-is.
// ---------------------------------------------------------------------------
namespace {

typedef char f8x32 __attribute__((ext_vector_type(32)));

__device__ __forceinline__ f8x32 lds_frag32(const char* smem, int row,
                                            int kbyte) {
  union {
    f8x32 v;
    bf16x8 h[2];
  } u;
  u.h[0] = *reinterpret_cast<const bf16x8*>(smem + row * 128 +
                                            (kbyte ^ ((row & 7) << 4)));
  u.h[1] = *reinterpret_cast<const bf16x8*>(
      smem + row * 128 + ((kbyte + 16) ^ ((row & 7) << 4)));
  return u.v;
}

template <int BNT>
__global__ __launch_bounds__(256, 2) void gemm_nt_fp8_kernel(
    const char* __restrict__ A, const char* __restrict__ Bt,
    bf16* __restrict__ C, int64_t M, int N, int K, float out_scale,
    float* __restrict__ stats_sum, float* __restrict__ stats_sumsq) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int MR = BNT == 128 ? 4 : 2;
  constexpr int ABYTES = BM * 128;   // 128 rows x 128 B
  constexpr int BBYTES = BNT * 128;
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int wr = BNT == 128 ? (wave >> 1) : wave;
  const int wc = BNT == 128 ? (wave & 1) : 0;
  const int wrows = BNT == 128 ? 64 : 32;
  const int64_t m0 = (int64_t)blockIdx.x * BM;
  const int n0 = blockIdx.y * BNT;

  f32x4 acc[MR][4] = {};
  // stage via the bf16 path: row stride K bytes = K/2 "bf16 columns"
  stage_tile<BM>((const bf16*)(A + m0 * K), K / 2, smem, wave, lane);
  stage_tile<BNT>((const bf16*)(Bt + (int64_t)n0 * K), K / 2, smem + ABYTES,
                  wave, lane);
  __syncthreads();

  int cur = 0;
  const int ntiles = K / 128;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      char* nbuf = smem + (cur ^ 1) * (ABYTES + BBYTES);
      stage_tile<BM>((const bf16*)(A + m0 * K + (t + 1) * 128), K / 2, nbuf,
                     wave, lane);
      stage_tile<BNT>((const bf16*)(Bt + (int64_t)n0 * K + (t + 1) * 128),
                      K / 2, nbuf + ABYTES, wave, lane);
    }
    const char* abuf = smem + cur * (ABYTES + BBYTES);
    const char* bbuf = abuf + ABYTES;
    const int kbyte = (lane >> 4) * 32;  // lane's 32-byte K-block
    f8x32 af[MR], bf[4];
#pragma unroll
    for (int mr = 0; mr < MR; ++mr)
      af[mr] = lds_frag32(abuf, wr * wrows + mr * 16 + (lane & 15), kbyte);
#pragma unroll
    for (int nr = 0; nr < 4; ++nr)
      bf[nr] = lds_frag32(bbuf, wc * 64 + nr * 16 + (lane & 15), kbyte);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mr = 0; mr < MR; ++mr)
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        acc[mr][nr] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            af[mr], bf[nr], acc[mr][nr], /*cbsz=*/0, /*blgp=*/0,
            /*opselA=*/0, /*scaleA=*/127, /*opselB=*/0, /*scaleB=*/127);
    __builtin_amdgcn_s_setprio(0);
    if (t + 1 < ntiles) __syncthreads();
    cur ^= 1;
  }

  // epilogue: per-tensor dequant scale, optional bn-stats accumulation,
  // widened LDS-transpose store (same structure as the bf16 kernel)
#pragma unroll
  for (int mr = 0; mr < MR; ++mr)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[mr][nr][r] *= out_scale;

  if (stats_sum != nullptr) {
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      float ssum = 0.f, ssq = 0.f;
#pragma unroll
      for (int mr = 0; mr < MR; ++mr)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float v = acc[mr][nr][r];
          ssum += v;
          ssq += v * v;
        }
      ssum += __shfl_xor(ssum, 16, WAVE);
      ssq += __shfl_xor(ssq, 16, WAVE);
      ssum += __shfl_xor(ssum, 32, WAVE);
      ssq += __shfl_xor(ssq, 32, WAVE);
      if ((lane >> 4) == 0) {
        const int64_t slot = (int64_t)(blockIdx.x % STAT_SLOTS) * N;
        const int c = n0 + wc * 64 + nr * 16 + lane;
        atomicAdd(&stats_sum[slot + c], ssum);
        atomicAdd(&stats_sumsq[slot + c], ssq);
      }
    }
  }

  __syncthreads();
  bf16* ct = reinterpret_cast<bf16*>(smem) + wave * wrows * 64;
#pragma unroll
  for (int mr = 0; mr < MR; ++mr)
#pragma unroll
    for (int nr = 0; nr < 4; ++nr) {
      const int col = nr * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int trow = mr * 16 + (lane >> 4) * 4 + r;
        ct[trow * 64 + col] = f2bf(acc[mr][nr][r]);
      }
    }
  __syncthreads();
  const int npieces = wrows * 64 / 8 / WAVE;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    if (i >= npieces) break;
    const int piece = i * WAVE + lane;
    const int trow = piece / 8;
    const int tcol = (piece % 8) * 8;
    const int64_t row = m0 + wr * wrows + trow;
    *reinterpret_cast<bf16x8*>(C + row * N + n0 + wc * 64 + tcol) =
        *reinterpret_cast<const bf16x8*>(ct + trow * 64 + tcol);
  }
}

}  // namespace

extern "C" int launch_gemm_nt_fp8(void* stream, const void* A,
                                  const void* Bt, void* C, int64_t M,
                                  int64_t N, int64_t K, float out_scale,
                                  float* stats_sum, float* stats_sumsq) {
  if (M % 128 != 0 || K % 128 != 0 || N % 64 != 0)
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int BNT = (N % 128 == 0) ? 128 : 64;
  const bool dbuf = (K / 128) > 1;
  const int smem = (BM * 128 + BNT * 128) * (dbuf ? 2 : 1);
  dim3 grid((unsigned)(M / 128), (unsigned)(N / BNT));
  if (BNT == 128)
    gemm_nt_fp8_kernel<128><<<grid, 256, smem, s>>>(
        (const char*)A, (const char*)Bt, (bf16*)C, M, (int)N, (int)K,
        out_scale, stats_sum, stats_sumsq);
  else
    gemm_nt_fp8_kernel<64><<<grid, 256, smem, s>>>(
        (const char*)A, (const char*)Bt, (bf16*)C, M, (int)N, (int)K,
        out_scale, stats_sum, stats_sumsq);
  HIP_CHECK_LAST();
  return 0;
}
