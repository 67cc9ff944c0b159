// Fused BatchNorm (+ReLU, +residual-add) for NHWC bf16 activations.
//
// Motivation (profiles/r01_step_breakdown.md): MIOpen's spatial BN kernels +
// the separate residual-add / ReLU elementwise kernels were 32 ms of a 58 ms
// ResNet-50 step on MI355X.  These kernels do the whole BN(+add)(+relu)
// block in the minimum number of memory passes, every pass reading/writing
// bf16 as 16-byte vectors (guide G13: scalar bf16 loads are 2-2.5x slower):
//   train fwd:  stats (1 read)  ->  apply (1 read [+res], 1 write)
//   bwd:        reduce (3 reads)->  dx (3 reads, 1-2 writes)
// Statistics accumulate in fp32; activations stay bf16 NHWC end to end.
//
// Geometry: x is [R, C], R = N*H*W, C fastest (channels_last), C a multiple
// of 64 up to 2048.  A 256-thread block covers Q = C/8 "channel octets" x
// rpi = 256/Q rows per iteration; per-channel partials cross-reduce through
// LDS and land in a [BN_PARTIAL, C] fp32 workspace, finalized by a second
// tiny kernel that also updates the running stats.
#include "common.h"
#include "launchers.h"

namespace {

constexpr int BN_PARTIAL = 1024;  // 4 blocks/CU: memory-bound reduction
                                  // needs ~16 waves/CU to hide HBM latency

typedef unsigned short ushort8v __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float us2f(unsigned short u) {
  union {
    unsigned int i;
    float f;
  } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f2us(float f) {
  union {
    float f;
    unsigned int i;
  } v;
  v.f = f;
  const unsigned int lsb = (v.i >> 16) & 1;  // round-to-nearest-even
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

// cross-rowgroup reduce of one float[8] accumulator set through LDS.
// layout: ls[tid][j]; rg==0 threads sum over their column's rpi groups.
__device__ __forceinline__ void lds_reduce8(float (&a)[8], float* ls, int Q,
                                            int rpi, int q8, int rg) {
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 8; ++j) ls[threadIdx.x * 8 + j] = a[j];
  __syncthreads();
  if (rg == 0) {
    for (int g = 1; g < rpi; ++g)
#pragma unroll
      for (int j = 0; j < 8; ++j) a[j] += ls[(q8 + g * Q) * 8 + j];
  }
}

// ---- stats pass 1: partial sums -------------------------------------------
__global__ __launch_bounds__(256) void bn_stats_partial_kernel(
    const unsigned short* __restrict__ x, float* __restrict__ part_s,
    float* __restrict__ part_q, int64_t R, int C) {
  const int Q = C / 8;
  const int rpi = 256 / Q;
  const int q8 = threadIdx.x % Q;
  const int rg = threadIdx.x / Q;
  float s[8] = {}, q[8] = {};
  for (int64_t row = (int64_t)blockIdx.x * rpi + rg; row < R;
       row += (int64_t)gridDim.x * rpi) {
    const ushort8v v =
        *reinterpret_cast<const ushort8v*>(x + row * C + q8 * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = us2f(v[j]);
      s[j] += f;
      q[j] += f * f;
    }
  }
  __shared__ float ls[256 * 8];
  lds_reduce8(s, ls, Q, rpi, q8, rg);
  if (rg == 0)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      part_s[(int64_t)blockIdx.x * C + q8 * 8 + j] = s[j];
  lds_reduce8(q, ls, Q, rpi, q8, rg);
  if (rg == 0)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      part_q[(int64_t)blockIdx.x * C + q8 * 8 + j] = q[j];
}

// ---- stats pass 2: finalize mean/rstd, update running stats ---------------
// ZERO_SRC: write zeros back to the partial slabs after reading — the
// conv-GEMM stats path reuses one pre-zeroed workspace per shape, so the
// finalize restores the invariant for the next dispatch (no separate
// fill kernels).
template <bool ZERO_SRC>
__global__ __launch_bounds__(256) void bn_stats_final_kernel(
    const float* __restrict__ part_s, const float* __restrict__ part_q,
    float* __restrict__ mean, float* __restrict__ rstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int64_t R, int C, int P, float eps, float momentum) {
  // 32 partial-groups x 8 channels per block: enough independent loads in
  // flight to cover the P-deep strided reads even for small C
  const int c = blockIdx.x * 8 + (threadIdx.x & 7);
  const int pg = threadIdx.x >> 3;
  float s = 0.f, q = 0.f;
  if (c < C)
    for (int p = pg; p < P; p += 32) {
      s += part_s[(int64_t)p * C + c];
      q += part_q[(int64_t)p * C + c];
      if (ZERO_SRC) {
        const_cast<float*>(part_s)[(int64_t)p * C + c] = 0.f;
        const_cast<float*>(part_q)[(int64_t)p * C + c] = 0.f;
      }
    }
  __shared__ float lss[256], lqq[256];
  lss[threadIdx.x] = s;
  lqq[threadIdx.x] = q;
  __syncthreads();
  if (pg != 0 || c >= C) return;
  for (int g = 1; g < 32; ++g) {
    s += lss[(threadIdx.x & 7) + g * 8];
    q += lqq[(threadIdx.x & 7) + g * 8];
  }
  const float m = s / (float)R;
  const float var = fmaxf(q / (float)R - m * m, 0.f);
  mean[c] = m;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
    const float unbiased = R > 1 ? var * (float)R / (float)(R - 1) : var;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---- forward apply --------------------------------------------------------
template <bool RELU, bool RES>
__global__ __launch_bounds__(256) void bn_apply_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ res,
    unsigned short* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, int64_t total8, int Q) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total8;
       i += (int64_t)gridDim.x * 256) {
    const int c8 = (int)(i % Q) * 8;
    const ushort8v xv = *reinterpret_cast<const ushort8v*>(x + i * 8);
    ushort8v rv = {};
    if (RES) rv = *reinterpret_cast<const ushort8v*>(res + i * 8);
    ushort8v out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = (us2f(xv[j]) - mean[c8 + j]) * rstd[c8 + j] * gamma[c8 + j] +
                beta[c8 + j];
      if (RES) v += us2f(rv[j]);
      if (RELU) v = fmaxf(v, 0.f);
      out[j] = f2us(v);
    }
    *reinterpret_cast<ushort8v*>(y + i * 8) = out;
  }
}

// ---- backward reduce: dbeta = sum g', dgamma = sum g' * xhat --------------
template <bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_reduce_kernel(
    const unsigned short* __restrict__ g, const unsigned short* __restrict__ y,
    const unsigned short* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ part_db,
    float* __restrict__ part_dg, int64_t R, int C) {
  const int Q = C / 8;
  const int rpi = 256 / Q;
  const int q8 = threadIdx.x % Q;
  const int rg = threadIdx.x / Q;
  float m[8], rs[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    m[j] = mean[q8 * 8 + j];
    rs[j] = rstd[q8 * 8 + j];
  }
  float db[8] = {}, dg[8] = {};
  for (int64_t row = (int64_t)blockIdx.x * rpi + rg; row < R;
       row += (int64_t)gridDim.x * rpi) {
    const int64_t base = row * C + q8 * 8;
    const ushort8v gv = *reinterpret_cast<const ushort8v*>(g + base);
    ushort8v yv = {};
    if (RELU) yv = *reinterpret_cast<const ushort8v*>(y + base);
    const ushort8v xv = *reinterpret_cast<const ushort8v*>(x + base);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gp = us2f(gv[j]);
      if (RELU && us2f(yv[j]) <= 0.f) gp = 0.f;
      db[j] += gp;
      dg[j] += gp * (us2f(xv[j]) - m[j]) * rs[j];
    }
  }
  __shared__ float ls[256 * 8];
  lds_reduce8(db, ls, Q, rpi, q8, rg);
  if (rg == 0)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      part_db[(int64_t)blockIdx.x * C + q8 * 8 + j] = db[j];
  lds_reduce8(dg, ls, Q, rpi, q8, rg);
  if (rg == 0)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      part_dg[(int64_t)blockIdx.x * C + q8 * 8 + j] = dg[j];
}

__global__ __launch_bounds__(256) void bn_bwd_final_kernel(
    const float* __restrict__ part_db, const float* __restrict__ part_dg,
    float* __restrict__ dbeta, float* __restrict__ dgamma, int C, int P) {
  const int c = blockIdx.x * 8 + (threadIdx.x & 7);
  const int pg = threadIdx.x >> 3;
  float db = 0.f, dg = 0.f;
  if (c < C)
    for (int p = pg; p < P; p += 32) {
      db += part_db[(int64_t)p * C + c];
      dg += part_dg[(int64_t)p * C + c];
    }
  __shared__ float lss[256], lqq[256];
  lss[threadIdx.x] = db;
  lqq[threadIdx.x] = dg;
  __syncthreads();
  if (pg != 0 || c >= C) return;
  for (int g = 1; g < 32; ++g) {
    db += lss[(threadIdx.x & 7) + g * 8];
    dg += lqq[(threadIdx.x & 7) + g * 8];
  }
  dbeta[c] = db;
  dgamma[c] = dg;
}

// ---- backward dx (+dres) --------------------------------------------------
// train: dx = gamma*rstd * (g' - dbeta/R - xhat * dgamma/R)
// eval:  dx = gamma*rstd * g'
template <bool RELU, bool RES, bool TRAIN>
__global__ __launch_bounds__(256) void bn_bwd_dx_kernel(
    const unsigned short* __restrict__ g, const unsigned short* __restrict__ y,
    const unsigned short* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ dbeta, const float* __restrict__ dgamma,
    unsigned short* __restrict__ dx, unsigned short* __restrict__ dres,
    int64_t total8, int Q, float invR) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total8;
       i += (int64_t)gridDim.x * 256) {
    const int c8 = (int)(i % Q) * 8;
    const ushort8v gv = *reinterpret_cast<const ushort8v*>(g + i * 8);
    ushort8v yv = {};
    if (RELU) yv = *reinterpret_cast<const ushort8v*>(y + i * 8);
    ushort8v xv = {};
    if (TRAIN) xv = *reinterpret_cast<const ushort8v*>(x + i * 8);
    ushort8v dxv, drv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c8 + j;
      float gp = us2f(gv[j]);
      if (RELU && us2f(yv[j]) <= 0.f) gp = 0.f;
      if (RES) drv[j] = f2us(gp);
      float v;
      if (TRAIN) {
        const float xhat = (us2f(xv[j]) - mean[c]) * rstd[c];
        v = gamma[c] * rstd[c] *
            (gp - dbeta[c] * invR - xhat * dgamma[c] * invR);
      } else {
        v = gamma[c] * rstd[c] * gp;
      }
      dxv[j] = f2us(v);
    }
    *reinterpret_cast<ushort8v*>(dx + i * 8) = dxv;
    if (RES) *reinterpret_cast<ushort8v*>(dres + i * 8) = drv;
  }
}

// small feature maps need few partial blocks (the finalize cost scales
// with P); big ones want the chip full.  >=4 row-iterations per block.
int bn_partial_blocks(int64_t R, int64_t C) {
  const int64_t rpi = 256 / (C / 8);
  int64_t p = R / (rpi * 4);
  if (p > BN_PARTIAL) p = BN_PARTIAL;
  if (p < 1) p = 1;
  return (int)p;
}

int elementwise_grid(int64_t total8) {
  int64_t b = (total8 + 255) / 256;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

bool bn_c_ok(int64_t C) {
  // Q = C/8 must divide 256: C in {64,128,...,2048} multiples of 64 with
  // power-of-two octet counts
  if (C % 8 != 0) return false;
  const int64_t Q = C / 8;
  return Q >= 1 && Q <= 256 && (256 % Q) == 0;
}

}  // namespace

extern "C" int launch_bn_stats(void* stream, const void* x, float* part_s,
                               float* part_q, float* mean, float* rstd,
                               float* running_mean, float* running_var,
                               int64_t R, int64_t C, float eps,
                               float momentum) {
  if (!bn_c_ok(C)) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int P = bn_partial_blocks(R, C);
  bn_stats_partial_kernel<<<P, 256, 0, s>>>(
      (const unsigned short*)x, part_s, part_q, R, (int)C);
  HIP_CHECK_LAST();
  bn_stats_final_kernel<false><<<dim3((unsigned)((C + 7) / 8)), 256, 0, s>>>(
      part_s, part_q, mean, rstd, running_mean, running_var, R, (int)C,
      P, eps, momentum);
  HIP_CHECK_LAST();
  return 0;
}

// Finalize mean/rstd (+ running stats) from the conv-GEMM epilogue's
// slot-striped [P, C] partials.  Same math as the stats pass 2 above —
// reuse its kernel (32 slot-groups x 8 channels per block through LDS;
// a naive one-thread-per-channel slot loop measured 75 us per call).
extern "C" int launch_bn_finalize(void* stream, const float* ssum,
                                  const float* ssq, float* mean, float* rstd,
                                  float* running_mean, float* running_var,
                                  int64_t R, int64_t C, int64_t P, float eps,
                                  float momentum, bool zero_src) {
  hipStream_t s = (hipStream_t)stream;
  if (zero_src)
    bn_stats_final_kernel<true>
        <<<dim3((unsigned)((C + 7) / 8)), 256, 0, s>>>(
            ssum, ssq, mean, rstd, running_mean, running_var, R, (int)C,
            (int)P, eps, momentum);
  else
    bn_stats_final_kernel<false>
        <<<dim3((unsigned)((C + 7) / 8)), 256, 0, s>>>(
            ssum, ssq, mean, rstd, running_mean, running_var, R, (int)C,
            (int)P, eps, momentum);
  HIP_CHECK_LAST();
  return 0;
}

// reduce the dgrad-epilogue's two [P, N] slot slabs to [N] totals and
// re-zero them (same workspace-reuse contract as bn_finalize)
__global__ __launch_bounds__(256) void slot_reduce2_kernel(
    float* __restrict__ s1, float* __restrict__ s2, float* __restrict__ o1,
    float* __restrict__ o2, int C, int P) {
  const int c = blockIdx.x * 8 + (threadIdx.x & 7);
  const int pg = threadIdx.x >> 3;
  float a = 0.f, b = 0.f;
  if (c < C)
    for (int p = pg; p < P; p += 32) {
      a += s1[(int64_t)p * C + c];
      b += s2[(int64_t)p * C + c];
      s1[(int64_t)p * C + c] = 0.f;
      s2[(int64_t)p * C + c] = 0.f;
    }
  __shared__ float lsa[256], lsb[256];
  lsa[threadIdx.x] = a;
  lsb[threadIdx.x] = b;
  __syncthreads();
  if (pg != 0 || c >= C) return;
  for (int g = 1; g < 32; ++g) {
    a += lsa[(threadIdx.x & 7) + g * 8];
    b += lsb[(threadIdx.x & 7) + g * 8];
  }
  o1[c] = a;
  o2[c] = b;
}

extern "C" int launch_slot_reduce2(void* stream, float* s1, float* s2,
                                   float* o1, float* o2, int64_t C,
                                   int64_t P) {
  hipStream_t s = (hipStream_t)stream;
  slot_reduce2_kernel<<<dim3((unsigned)((C + 7) / 8)), 256, 0, s>>>(
      s1, s2, o1, o2, (int)C, (int)P);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_bn_apply(void* stream, const void* x, const void* res,
                               void* y, const float* mean, const float* rstd,
                               const float* gamma, const float* beta,
                               int64_t R, int64_t C, bool relu) {
  if (!bn_c_ok(C)) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int64_t total8 = R * C / 8;
  const int grid = elementwise_grid(total8);
  const int Q = (int)(C / 8);
#define APPLY(RELU, RES)                                                   \
  bn_apply_kernel<RELU, RES><<<grid, 256, 0, s>>>(                         \
      (const unsigned short*)x, (const unsigned short*)res,                \
      (unsigned short*)y, mean, rstd, gamma, beta, total8, Q)
  if (relu && res) APPLY(true, true);
  else if (relu) APPLY(true, false);
  else if (res) APPLY(false, true);
  else APPLY(false, false);
#undef APPLY
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_bn_bwd_reduce(void* stream, const void* g, const void* y,
                                    const void* x, const float* mean,
                                    const float* rstd, float* part_db,
                                    float* part_dg, float* dbeta,
                                    float* dgamma, int64_t R, int64_t C,
                                    bool relu) {
  if (!bn_c_ok(C)) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int P = bn_partial_blocks(R, C);
  if (relu)
    bn_bwd_reduce_kernel<true><<<P, 256, 0, s>>>(
        (const unsigned short*)g, (const unsigned short*)y,
        (const unsigned short*)x, mean, rstd, part_db, part_dg, R, (int)C);
  else
    bn_bwd_reduce_kernel<false><<<P, 256, 0, s>>>(
        (const unsigned short*)g, (const unsigned short*)y,
        (const unsigned short*)x, mean, rstd, part_db, part_dg, R, (int)C);
  HIP_CHECK_LAST();
  bn_bwd_final_kernel<<<dim3((unsigned)((C + 7) / 8)), 256, 0, s>>>(
      part_db, part_dg, dbeta, dgamma, (int)C, P);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_bn_bwd_dx(void* stream, const void* g, const void* y,
                                const void* x, const float* mean,
                                const float* rstd, const float* gamma,
                                const float* dbeta, const float* dgamma,
                                void* dx, void* dres, int64_t R, int64_t C,
                                bool relu, bool train) {
  if (!bn_c_ok(C)) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int64_t total8 = R * C / 8;
  const int grid = elementwise_grid(total8);
  const int Q = (int)(C / 8);
  const float invR = 1.f / (float)R;
  const bool res = dres != nullptr;
#define DX(RELU, RES, TRAIN)                                               \
  bn_bwd_dx_kernel<RELU, RES, TRAIN><<<grid, 256, 0, s>>>(                 \
      (const unsigned short*)g, (const unsigned short*)y,                  \
      (const unsigned short*)x, mean, rstd, gamma, dbeta, dgamma,          \
      (unsigned short*)dx, (unsigned short*)dres, total8, Q, invR)
  if (train) {
    if (relu && res) DX(true, true, true);
    else if (relu) DX(true, false, true);
    else if (res) DX(false, true, true);
    else DX(false, false, true);
  } else {
    if (relu && res) DX(true, true, false);
    else if (relu) DX(true, false, false);
    else if (res) DX(false, true, false);
    else DX(false, false, false);
  }
#undef DX
  HIP_CHECK_LAST();
  return 0;
}
