// Fused BatchNorm (+ReLU, +residual-add) for NHWC bf16 activations.
//
// Motivation (profiles/r01_step_breakdown.md): MIOpen's spatial BN kernels +
// the separate residual-add / ReLU elementwise kernels were 32 ms of a 58 ms
// ResNet-50 step on MI355X.  These kernels do the whole BN(+add)(+relu)
// block in the minimum number of memory passes:
//   train fwd:  stats (1 read)  ->  apply (1 read [+res], 1 write)
//   bwd:        reduce (3 reads)->  dx (3 reads, 1-2 writes)
// All statistics accumulate in fp32; activations stay bf16 NHWC end to end
// (the MI355X-native dtype policy — MIOpen's path casts through fp32).
//
// Layout: x is [R, C] with R = N*H*W and C the fastest (channels_last).
// Stats geometry: C <= 256 -> a 256-thread block covers (256/C) rows per
// iteration with an LDS cross-group reduce; C > 256 (multiple of 256) ->
// each thread owns C/256 channels, no LDS needed.  Partials land in a
// [P, C] fp32 workspace, P = BN_PARTIAL blocks; a second tiny kernel
// finalizes mean/rstd and updates the running stats in the same launch.
#include "common.h"
#include "launchers.h"

namespace {

constexpr int BN_PARTIAL = 304;  // ~1.2 blocks/CU on 256 CUs
constexpr int MAX_L = 8;         // C <= 2048

typedef unsigned short ushort4v __attribute__((ext_vector_type(4)));

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
  // round-to-nearest-even bf16
  unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

// ---- stats pass 1: partial sums -------------------------------------------
__global__ __launch_bounds__(256) void bn_stats_partial_kernel(
    const unsigned short* __restrict__ x, float* __restrict__ part_s,
    float* __restrict__ part_q, int64_t R, int C) {
  if (C <= 256) {
    const int c = threadIdx.x % C;
    const int rg = threadIdx.x / C;
    const int rpi = 256 / C;
    float s = 0.f, q = 0.f;
    for (int64_t row = (int64_t)blockIdx.x * rpi + rg; row < R;
         row += (int64_t)gridDim.x * rpi) {
      const float v = us2f(x[row * C + c]);
      s += v;
      q += v * v;
    }
    __shared__ float ls[256], lq[256];
    ls[threadIdx.x] = s;
    lq[threadIdx.x] = q;
    __syncthreads();
    if (rg == 0) {
      for (int g = 1; g < rpi; ++g) {
        s += ls[c + g * C];
        q += lq[c + g * C];
      }
      part_s[(int64_t)blockIdx.x * C + c] = s;
      part_q[(int64_t)blockIdx.x * C + c] = q;
    }
  } else {
    const int L = C / 256;
    float s[MAX_L] = {}, q[MAX_L] = {};
    for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
      const unsigned short* xr = x + row * C;
#pragma unroll
      for (int k = 0; k < MAX_L; ++k) {
        if (k >= L) break;
        const float v = us2f(xr[threadIdx.x + k * 256]);
        s[k] += v;
        q[k] += v * v;
      }
    }
#pragma unroll
    for (int k = 0; k < MAX_L; ++k) {
      if (k >= L) break;
      part_s[(int64_t)blockIdx.x * C + threadIdx.x + k * 256] = s[k];
      part_q[(int64_t)blockIdx.x * C + threadIdx.x + k * 256] = q[k];
    }
  }
}

// ---- stats pass 2: finalize mean/rstd, update running stats ---------------
__global__ __launch_bounds__(256) void bn_stats_final_kernel(
    const float* __restrict__ part_s, const float* __restrict__ part_q,
    float* __restrict__ mean, float* __restrict__ rstd,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    int64_t R, int C, int P, float eps, float momentum) {
  const int c = blockIdx.x * 256 + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, q = 0.f;
  for (int p = 0; p < P; ++p) {
    s += part_s[(int64_t)p * C + c];
    q += part_q[(int64_t)p * C + c];
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
    const float* __restrict__ beta, int64_t total4, int C4) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total4;
       i += (int64_t)gridDim.x * 256) {
    const int c4 = (int)(i % C4) * 4;
    const ushort4v xv = *reinterpret_cast<const ushort4v*>(x + i * 4);
    ushort4v rv = {};
    if (RES) rv = *reinterpret_cast<const ushort4v*>(res + i * 4);
    const float4 mn = *reinterpret_cast<const float4*>(mean + c4);
    const float4 rs = *reinterpret_cast<const float4*>(rstd + c4);
    const float4 ga = *reinterpret_cast<const float4*>(gamma + c4);
    const float4 be = *reinterpret_cast<const float4*>(beta + c4);
    ushort4v out;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float mj = j == 0 ? mn.x : j == 1 ? mn.y : j == 2 ? mn.z : mn.w;
      const float rj = j == 0 ? rs.x : j == 1 ? rs.y : j == 2 ? rs.z : rs.w;
      const float gj = j == 0 ? ga.x : j == 1 ? ga.y : j == 2 ? ga.z : ga.w;
      const float bj = j == 0 ? be.x : j == 1 ? be.y : j == 2 ? be.z : be.w;
      float v = (us2f(xv[j]) - mj) * rj * gj + bj;
      if (RES) v += us2f(rv[j]);
      if (RELU) v = fmaxf(v, 0.f);
      out[j] = f2us(v);
    }
    *reinterpret_cast<ushort4v*>(y + i * 4) = out;
  }
}

// ---- backward reduce: dbeta = sum g', dgamma = sum g' * xhat --------------
template <bool RELU>
__global__ __launch_bounds__(256) void bn_bwd_reduce_kernel(
    const unsigned short* __restrict__ g, const unsigned short* __restrict__ y,
    const unsigned short* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ part_db,
    float* __restrict__ part_dg, int64_t R, int C) {
  if (C <= 256) {
    const int c = threadIdx.x % C;
    const int rg = threadIdx.x / C;
    const int rpi = 256 / C;
    const float m = mean[c], rs = rstd[c];
    float db = 0.f, dg = 0.f;
    for (int64_t row = (int64_t)blockIdx.x * rpi + rg; row < R;
         row += (int64_t)gridDim.x * rpi) {
      const int64_t idx = row * C + c;
      float gv = us2f(g[idx]);
      if (RELU && us2f(y[idx]) <= 0.f) gv = 0.f;
      db += gv;
      dg += gv * (us2f(x[idx]) - m) * rs;
    }
    __shared__ float ls[256], lq[256];
    ls[threadIdx.x] = db;
    lq[threadIdx.x] = dg;
    __syncthreads();
    if (rg == 0) {
      for (int gg = 1; gg < rpi; ++gg) {
        db += ls[c + gg * C];
        dg += lq[c + gg * C];
      }
      part_db[(int64_t)blockIdx.x * C + c] = db;
      part_dg[(int64_t)blockIdx.x * C + c] = dg;
    }
  } else {
    const int L = C / 256;
    float db[MAX_L] = {}, dg[MAX_L] = {};
    float m[MAX_L], rs[MAX_L];
#pragma unroll
    for (int k = 0; k < MAX_L; ++k) {
      if (k >= L) break;
      m[k] = mean[threadIdx.x + k * 256];
      rs[k] = rstd[threadIdx.x + k * 256];
    }
    for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
      const int64_t base = row * C;
#pragma unroll
      for (int k = 0; k < MAX_L; ++k) {
        if (k >= L) break;
        const int64_t idx = base + threadIdx.x + k * 256;
        float gv = us2f(g[idx]);
        if (RELU && us2f(y[idx]) <= 0.f) gv = 0.f;
        db[k] += gv;
        dg[k] += gv * (us2f(x[idx]) - m[k]) * rs[k];
      }
    }
#pragma unroll
    for (int k = 0; k < MAX_L; ++k) {
      if (k >= L) break;
      part_db[(int64_t)blockIdx.x * C + threadIdx.x + k * 256] = db[k];
      part_dg[(int64_t)blockIdx.x * C + threadIdx.x + k * 256] = dg[k];
    }
  }
}

__global__ __launch_bounds__(256) void bn_bwd_final_kernel(
    const float* __restrict__ part_db, const float* __restrict__ part_dg,
    float* __restrict__ dbeta, float* __restrict__ dgamma, int C, int P) {
  const int c = blockIdx.x * 256 + threadIdx.x;
  if (c >= C) return;
  float db = 0.f, dg = 0.f;
  for (int p = 0; p < P; ++p) {
    db += part_db[(int64_t)p * C + c];
    dg += part_dg[(int64_t)p * C + c];
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
    int64_t total4, int C4, float invR) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total4;
       i += (int64_t)gridDim.x * 256) {
    const int c4 = (int)(i % C4) * 4;
    const ushort4v gv4 = *reinterpret_cast<const ushort4v*>(g + i * 4);
    ushort4v yv4 = {};
    if (RELU) yv4 = *reinterpret_cast<const ushort4v*>(y + i * 4);
    ushort4v xv4 = {};
    if (TRAIN) xv4 = *reinterpret_cast<const ushort4v*>(x + i * 4);
    const float4 mn = *reinterpret_cast<const float4*>(mean + c4);
    const float4 rs = *reinterpret_cast<const float4*>(rstd + c4);
    const float4 ga = *reinterpret_cast<const float4*>(gamma + c4);
    float4 db = {}, dgm = {};
    if (TRAIN) {
      db = *reinterpret_cast<const float4*>(dbeta + c4);
      dgm = *reinterpret_cast<const float4*>(dgamma + c4);
    }
    ushort4v dxv, drv;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float mj = j == 0 ? mn.x : j == 1 ? mn.y : j == 2 ? mn.z : mn.w;
      const float rj = j == 0 ? rs.x : j == 1 ? rs.y : j == 2 ? rs.z : rs.w;
      const float gj = j == 0 ? ga.x : j == 1 ? ga.y : j == 2 ? ga.z : ga.w;
      float gp = us2f(gv4[j]);
      if (RELU && us2f(yv4[j]) <= 0.f) gp = 0.f;
      if (RES) drv[j] = f2us(gp);
      float v;
      if (TRAIN) {
        const float dbj = j == 0 ? db.x : j == 1 ? db.y : j == 2 ? db.z : db.w;
        const float dgj =
            j == 0 ? dgm.x : j == 1 ? dgm.y : j == 2 ? dgm.z : dgm.w;
        const float xhat = (us2f(xv4[j]) - mj) * rj;
        v = gj * rj * (gp - dbj * invR - xhat * dgj * invR);
      } else {
        v = gj * rj * gp;
      }
      dxv[j] = f2us(v);
    }
    *reinterpret_cast<ushort4v*>(dx + i * 4) = dxv;
    if (RES) *reinterpret_cast<ushort4v*>(dres + i * 4) = drv;
  }
}

int elementwise_grid(int64_t total4) {
  int64_t b = (total4 + 255) / 256;
  if (b > 4096) b = 4096;
  if (b < 1) b = 1;
  return (int)b;
}

}  // namespace

extern "C" int launch_bn_stats(void* stream, const void* x, float* part_s,
                               float* part_q, float* mean, float* rstd,
                               float* running_mean, float* running_var,
                               int64_t R, int64_t C, float eps,
                               float momentum) {
  if (!(C <= 256 ? (256 % C == 0) : (C % 256 == 0 && C <= 2048)))
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  bn_stats_partial_kernel<<<BN_PARTIAL, 256, 0, s>>>(
      (const unsigned short*)x, part_s, part_q, R, (int)C);
  HIP_CHECK_LAST();
  bn_stats_final_kernel<<<dim3((unsigned)((C + 255) / 256)), 256, 0, s>>>(
      part_s, part_q, mean, rstd, running_mean, running_var, R, (int)C,
      BN_PARTIAL, eps, momentum);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_bn_apply(void* stream, const void* x, const void* res,
                               void* y, const float* mean, const float* rstd,
                               const float* gamma, const float* beta,
                               int64_t R, int64_t C, bool relu) {
  if (C % 4 != 0) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int64_t total4 = R * C / 4;
  const int grid = elementwise_grid(total4);
  const int C4 = (int)(C / 4);
#define APPLY(RELU, RES)                                                   \
  bn_apply_kernel<RELU, RES><<<grid, 256, 0, s>>>(                         \
      (const unsigned short*)x, (const unsigned short*)res,                \
      (unsigned short*)y, mean, rstd, gamma, beta, total4, C4)
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
  if (!(C <= 256 ? (256 % C == 0) : (C % 256 == 0 && C <= 2048)))
    return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  if (relu)
    bn_bwd_reduce_kernel<true><<<BN_PARTIAL, 256, 0, s>>>(
        (const unsigned short*)g, (const unsigned short*)y,
        (const unsigned short*)x, mean, rstd, part_db, part_dg, R, (int)C);
  else
    bn_bwd_reduce_kernel<false><<<BN_PARTIAL, 256, 0, s>>>(
        (const unsigned short*)g, (const unsigned short*)y,
        (const unsigned short*)x, mean, rstd, part_db, part_dg, R, (int)C);
  HIP_CHECK_LAST();
  bn_bwd_final_kernel<<<dim3((unsigned)((C + 255) / 256)), 256, 0, s>>>(
      part_db, part_dg, dbeta, dgamma, (int)C, BN_PARTIAL);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_bn_bwd_dx(void* stream, const void* g, const void* y,
                                const void* x, const float* mean,
                                const float* rstd, const float* gamma,
                                const float* dbeta, const float* dgamma,
                                void* dx, void* dres, int64_t R, int64_t C,
                                bool relu, bool train) {
  if (C % 4 != 0) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int64_t total4 = R * C / 4;
  const int grid = elementwise_grid(total4);
  const int C4 = (int)(C / 4);
  const float invR = 1.f / (float)R;
  const bool res = dres != nullptr;
#define DX(RELU, RES, TRAIN)                                               \
  bn_bwd_dx_kernel<RELU, RES, TRAIN><<<grid, 256, 0, s>>>(                 \
      (const unsigned short*)g, (const unsigned short*)y,                  \
      (const unsigned short*)x, mean, rstd, gamma, dbeta, dgamma,          \
      (unsigned short*)dx, (unsigned short*)dres, total4, C4, invR)
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
