// 3x3 convolution (pad 1, stride 1 or 2) as an implicit MFMA GEMM,
// NHWC bf16 — round-2 prototype (compile-verified + oracle-tested;
// integration gated behind MOCO_AMD_EXPERIMENTAL).
//
//   out[M, Co] = sum_{tap, ci} x[n, ho*s+ky-1, wo*s+kx-1, ci] * W[co, tap, ci]
//   M = N*Ho*Wo,  K' = 9*Ci,  tap = ky*3+kx
//
// Same LDS structure as conv1x1.hip's gemm_nt_lds_kernel (128-row tiles,
// BK=64 K-slices, glds staging with XOR-swizzled source addresses, 16-byte
// LDS-transpose store epilogue).  Two deltas:
//
// 1. A staging uses IM2COL ADDRESSING: each 16-byte piece maps its row to
//    (n, ho, wo) and its K-slice to (tap, ci).  Padding needs no masking:
//    out-of-bounds pieces redirect their glds SOURCE to a per-channel
//    NEUTRAL page sized [Ci + 8].  Without the BN-apply fusion it holds
//    zeros; with it, neutral[ci] = -shift[ci]/scale[ci], so the fused
//    transform maps padding to exactly 0 (and relu(0) == 0).  Requires
//    Ci % 64 == 0 so one 64-wide K-slice never crosses a tap.
//
// 2. Channels-last conv weights (Co, Ci, 3, 3) are stored [Co][ky][kx][Ci]
//    = exactly the [Co][K'] row-major B^T the NT-GEMM wants, k-contiguous.
//
// The BN-apply prologue hook indexes its per-channel params by ci = k' % Ci
// (every tap reads the same transformed activation); the BN-stats epilogue
// is identical to the 1x1 kernel's.
#include "common.h"
#include "launchers.h"

namespace {

constexpr int BM3 = 128, BK3 = 64;

__device__ __forceinline__ unsigned swz3(unsigned row, unsigned kbyte) {
  return kbyte ^ ((row & 7) << 4);
}

// stage a [128 rows][64 k'] A tile with im2col addressing.
// kt: K-slice index (k' base = kt*64); geometry args are wave-uniform.
__device__ __forceinline__ void stage_im2col(
    const bf16* __restrict__ x, const bf16* __restrict__ zero_page,
    char* smem, int wave, int lane, int64_t m0, int kt, int N, int H, int W,
    int Ci, int Ho, int Wo, int stride) {
  const int kbase = kt * BK3;          // k' base of this slice
  const int tap = kbase / Ci;          // constant across the slice (Ci%64==0)
  const int ky = tap / 3, kx = tap % 3;
  const int ci_base = kbase - tap * Ci;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int piece = c * 256 + wave * 64 + lane;
    const int row = piece / 8;
    const int kb = (piece % 8) * 16;                 // byte offset in-slice
    const int ci = ci_base + swz3(row, kb) / 2;      // pre-swizzled source
    // row -> (n, ho, wo)
    const int64_t m = m0 + row;
    const int wo = (int)(m % Wo);
    const int64_t t = m / Wo;
    const int ho = (int)(t % Ho);
    const int n = (int)(t / Ho);
    const int iy = ho * stride + ky - 1;
    const int ix = wo * stride + kx - 1;
    const bool valid = (iy >= 0) & (iy < H) & (ix >= 0) & (ix < W);
    const bf16* src = valid
        ? x + (((int64_t)n * H + iy) * W + ix) * Ci + ci
        : zero_page + ci;  // neutral page: transform(neutral[ci]) == 0
    char* lds_base = smem + (int64_t)(c * 256 + wave * 64) * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)lds_base, 16, 0, 0);
  }
}

// stage a [ROWS][64] weight slice (identical to conv1x1's stage_tile)
template <int ROWS>
__device__ __forceinline__ void stage_w(const bf16* __restrict__ src,
                                        int64_t ldK, char* smem, int wave,
                                        int lane) {
#pragma unroll
  for (int c = 0; c < ROWS / 32; ++c) {
    const int piece = c * 256 + wave * 64 + lane;
    const int row = piece / 8;
    const int kb = (piece % 8) * 16;
    const int src_elem = swz3(row, kb) / 2;
    char* lds_base = smem + (int64_t)(c * 256 + wave * 64) * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(
            src + (int64_t)row * ldK + src_elem),
        (__attribute__((address_space(3))) void*)lds_base, 16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8 lds_frag3(const char* smem, int row,
                                            int kbyte) {
  return *reinterpret_cast<const bf16x8*>(smem + row * (BK3 * 2) +
                                          swz3(row, kbyte));
}

template <int BNT, bool APPLY, bool STATS>
__global__ __launch_bounds__(256, 2) void conv3x3_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ Wt,
    const bf16* __restrict__ zero_page, bf16* __restrict__ C, int N, int H,
    int W, int Ci, int Co, int Ho, int Wo, int stride,
    const float* __restrict__ a_scale, const float* __restrict__ a_shift,
    bool a_relu, float* __restrict__ stats_sum,
    float* __restrict__ stats_sumsq) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int MR = BNT == 128 ? 4 : 2;
  constexpr int ABYTES = BM3 * BK3 * 2;
  constexpr int BBYTES = BNT * BK3 * 2;
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int wr = BNT == 128 ? (wave >> 1) : wave;
  const int wc = BNT == 128 ? (wave & 1) : 0;
  const int wrows = BNT == 128 ? 64 : 32;
  const int64_t m0 = (int64_t)blockIdx.x * BM3;
  const int n0 = blockIdx.y * BNT;
  const int Kp = 9 * Ci;

  f32x4 acc[MR][4] = {};

  stage_im2col(x, zero_page, smem, wave, lane, m0, 0, N, H, W, Ci, Ho, Wo,
               stride);
  stage_w<BNT>(Wt + (int64_t)n0 * Kp, Kp, smem + ABYTES, wave, lane);
  __syncthreads();

  int cur = 0;
  const int ntiles = Kp / BK3;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      char* nbuf = smem + (cur ^ 1) * (ABYTES + BBYTES);
      stage_im2col(x, zero_page, nbuf, wave, lane, m0, t + 1, N, H, W, Ci,
                   Ho, Wo, stride);
      stage_w<BNT>(Wt + (int64_t)n0 * Kp + (t + 1) * BK3, Kp, nbuf + ABYTES,
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
        af[mr] = lds_frag3(abuf, wr * wrows + mr * 16 + (lane & 15), kbyte);
      if (APPLY) {
        // params indexed by input channel ci = k' % Ci (tap-invariant)
        const int c0 = (t * BK3 + ks * 32 + (lane >> 4) * 8) % Ci;
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
            // padding correctness: af holds neutral[ci] = -shift/scale for
            // out-of-bounds pieces, so v == 0 there (and relu keeps it 0)
            float v = fmaf((float)af[mr][j], scs[j], shs[j]);
            if (a_relu) v = fmaxf(v, 0.f);
            af[mr][j] = (__bf16)v;
          }
      }
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        bf[nr] = lds_frag3(bbuf, wc * 64 + nr * 16 + (lane & 15), kbyte);
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

  if (STATS) {
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
        const int c = n0 + wc * 64 + nr * 16 + lane;
        atomicAdd(&stats_sum[c], ssum);
        atomicAdd(&stats_sumsq[c], ssq);
      }
    }
  }

  // widened LDS-transpose store epilogue (same as conv1x1)
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
    *reinterpret_cast<bf16x8*>(C + row * Co + n0 + wc * 64 + tcol) =
        *reinterpret_cast<const bf16x8*>(ct + trow * 64 + tcol);
  }
}

}  // namespace

extern "C" int launch_conv3x3_fwd(void* stream, const void* x, const void* w,
                                  const void* zero_page, void* out, int64_t N,
                                  int64_t H, int64_t W, int64_t Ci,
                                  int64_t Co, int64_t stride,
                                  const float* a_scale, const float* a_shift,
                                  bool a_relu, float* stats_sum,
                                  float* stats_sumsq) {
  const int64_t Ho = (H + 2 - 3) / stride + 1;
  const int64_t Wo = (W + 2 - 3) / stride + 1;
  const int64_t M = N * Ho * Wo;
  if (Ci % 64 != 0 || M % 128 != 0 || Co % 64 != 0 ||
      (stride != 1 && stride != 2))
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int BNT = (Co % 128 == 0) ? 128 : 64;
  const int smem = (16384 + BNT * 128) * 2;  // always double-buffered
  const bool apply = a_scale != nullptr;
  const bool stats = stats_sum != nullptr;
  dim3 grid((unsigned)(M / 128), (unsigned)(Co / BNT));
#define GO(BNTv, AP, ST)                                                   \
  conv3x3_kernel<BNTv, AP, ST><<<grid, 256, smem, s>>>(                    \
      (const bf16*)x, (const bf16*)w, (const bf16*)zero_page, (bf16*)out,  \
      (int)N, (int)H, (int)W, (int)Ci, (int)Co, (int)Ho, (int)Wo,          \
      (int)stride, a_scale, a_shift, a_relu, stats_sum, stats_sumsq)
#define GO2(BNTv)                                                          \
  do {                                                                     \
    if (apply && stats) GO(BNTv, true, true);                              \
    else if (apply) GO(BNTv, true, false);                                 \
    else if (stats) GO(BNTv, false, true);                                 \
    else GO(BNTv, false, false);                                           \
  } while (0)
  if (BNT == 128) GO2(128);
  else GO2(64);
#undef GO2
#undef GO
  HIP_CHECK_LAST();
  return 0;
}

// ---------------------------------------------------------------------------
// Strided 1x1 conv (the bottleneck downsample path): the same implicit-GEMM
// structure with a trivial row map — output row (n, ho, wo) reads input row
// (n, ho*s, wo*s), always in bounds, no padding page needed.
// ---------------------------------------------------------------------------
namespace {

__device__ __forceinline__ void stage_strided_rows(
    const bf16* __restrict__ x, char* smem, int wave, int lane, int64_t m0,
    int kt, int H, int W, int Ci, int Ho, int Wo, int stride) {
  const int ci_b = kt * BK3;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int piece = c * 256 + wave * 64 + lane;
    const int row = piece / 8;
    const int kb = (piece % 8) * 16;
    const int ci = ci_b + swz3(row, kb) / 2;
    const int64_t m = m0 + row;
    const int wo = (int)(m % Wo);
    const int64_t t = m / Wo;
    const int ho = (int)(t % Ho);
    const int n = (int)(t / Ho);
    const bf16* src =
        x + (((int64_t)n * H + ho * stride) * W + wo * stride) * Ci + ci;
    char* lds_base = smem + (int64_t)(c * 256 + wave * 64) * 16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)lds_base, 16, 0, 0);
  }
}

template <int BNT>
__global__ __launch_bounds__(256, 2) void conv1x1s_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ Wt,
    bf16* __restrict__ C, int H, int W, int Ci, int Co, int Ho, int Wo,
    int stride) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int MR = BNT == 128 ? 4 : 2;
  constexpr int ABYTES = BM3 * BK3 * 2;
  constexpr int BBYTES = BNT * BK3 * 2;
  const int lane = threadIdx.x % WAVE;
  const int wave = threadIdx.x / WAVE;
  const int wr = BNT == 128 ? (wave >> 1) : wave;
  const int wc = BNT == 128 ? (wave & 1) : 0;
  const int wrows = BNT == 128 ? 64 : 32;
  const int64_t m0 = (int64_t)blockIdx.x * BM3;
  const int n0 = blockIdx.y * BNT;

  f32x4 acc[MR][4] = {};
  stage_strided_rows(x, smem, wave, lane, m0, 0, H, W, Ci, Ho, Wo, stride);
  stage_w<BNT>(Wt + (int64_t)n0 * Ci, Ci, smem + ABYTES, wave, lane);
  __syncthreads();

  int cur = 0;
  const int ntiles = Ci / BK3;
  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      char* nbuf = smem + (cur ^ 1) * (ABYTES + BBYTES);
      stage_strided_rows(x, nbuf, wave, lane, m0, t + 1, H, W, Ci, Ho, Wo,
                         stride);
      stage_w<BNT>(Wt + (int64_t)n0 * Ci + (t + 1) * BK3, Ci, nbuf + ABYTES,
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
        af[mr] = lds_frag3(abuf, wr * wrows + mr * 16 + (lane & 15), kbyte);
#pragma unroll
      for (int nr = 0; nr < 4; ++nr)
        bf[nr] = lds_frag3(bbuf, wc * 64 + nr * 16 + (lane & 15), kbyte);
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
    *reinterpret_cast<bf16x8*>(C + row * Co + n0 + wc * 64 + tcol) =
        *reinterpret_cast<const bf16x8*>(ct + trow * 64 + tcol);
  }
}

}  // namespace

extern "C" int launch_conv1x1_strided(void* stream, const void* x,
                                      const void* w, void* out, int64_t N,
                                      int64_t H, int64_t W, int64_t Ci,
                                      int64_t Co, int64_t stride) {
  const int64_t Ho = (H - 1) / stride + 1, Wo = (W - 1) / stride + 1;
  const int64_t M = N * Ho * Wo;
  if (Ci % 64 != 0 || M % 128 != 0 || Co % 64 != 0)
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int BNT = (Co % 128 == 0) ? 128 : 64;
  const int smem = (16384 + BNT * 128) * 2;
  dim3 grid((unsigned)(M / 128), (unsigned)(Co / BNT));
  if (BNT == 128)
    conv1x1s_kernel<128><<<grid, 256, smem, s>>>(
        (const bf16*)x, (const bf16*)w, (bf16*)out, (int)H, (int)W, (int)Ci,
        (int)Co, (int)Ho, (int)Wo, (int)stride);
  else
    conv1x1s_kernel<64><<<grid, 256, smem, s>>>(
        (const bf16*)x, (const bf16*)w, (bf16*)out, (int)H, (int)W, (int)Ci,
        (int)Co, (int)Ho, (int)Wo, (int)stride);
  HIP_CHECK_LAST();
  return 0;
}
