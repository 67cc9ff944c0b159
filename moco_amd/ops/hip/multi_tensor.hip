// Fused multi-tensor kernels (fp32 master weights):
//   - EMA key-encoder update:  dst = m*dst + (1-m)*src
//     (one launch replaces the reference's 161-iteration Python loop,
//      builder.py:51-56)
//   - SGD + momentum + weight decay (torch.optim.SGD semantics,
//     dampening=0, nesterov=False — reference main_moco.py:205-208)
// One 256-thread workgroup per 64Ki-element chunk; float4 main loop.
#include "common.h"
#include "launchers.h"

namespace {

__global__ __launch_bounds__(256) void mt_ema_kernel(
    const int64_t* __restrict__ pd, const int64_t* __restrict__ ps,
    const int32_t* __restrict__ sizes, float m) {
  const int chunk = blockIdx.x;
  float* dst = reinterpret_cast<float*>(pd[chunk]);
  const float* src = reinterpret_cast<const float*>(ps[chunk]);
  const int n = sizes[chunk];
  const float om = 1.f - m;
  const int n4 = n & ~3;
  for (int i = threadIdx.x * 4; i < n4; i += 256 * 4) {
    const float4 d = *reinterpret_cast<const float4*>(dst + i);
    const float4 s = *reinterpret_cast<const float4*>(src + i);
    float4 r;
    r.x = m * d.x + om * s.x;
    r.y = m * d.y + om * s.y;
    r.z = m * d.z + om * s.z;
    r.w = m * d.w + om * s.w;
    *reinterpret_cast<float4*>(dst + i) = r;
  }
  for (int i = n4 + threadIdx.x; i < n; i += 256)
    dst[i] = m * dst[i] + om * src[i];
}

__global__ __launch_bounds__(256) void mt_sgd_kernel(
    const int64_t* __restrict__ pp, const int64_t* __restrict__ pg,
    const int64_t* __restrict__ pb, const int32_t* __restrict__ sizes,
    float lr, float momentum, float wd) {
  const int chunk = blockIdx.x;
  float* p = reinterpret_cast<float*>(pp[chunk]);
  const float* g = reinterpret_cast<const float*>(pg[chunk]);
  float* b = reinterpret_cast<float*>(pb[chunk]);
  const int n = sizes[chunk];
  const int n4 = n & ~3;
  for (int i = threadIdx.x * 4; i < n4; i += 256 * 4) {
    float4 pv = *reinterpret_cast<const float4*>(p + i);
    const float4 gv = *reinterpret_cast<const float4*>(g + i);
    float4 bv = *reinterpret_cast<const float4*>(b + i);
    bv.x = momentum * bv.x + fmaf(wd, pv.x, gv.x);
    bv.y = momentum * bv.y + fmaf(wd, pv.y, gv.y);
    bv.z = momentum * bv.z + fmaf(wd, pv.z, gv.z);
    bv.w = momentum * bv.w + fmaf(wd, pv.w, gv.w);
    pv.x -= lr * bv.x;
    pv.y -= lr * bv.y;
    pv.z -= lr * bv.z;
    pv.w -= lr * bv.w;
    *reinterpret_cast<float4*>(b + i) = bv;
    *reinterpret_cast<float4*>(p + i) = pv;
  }
  for (int i = n4 + threadIdx.x; i < n; i += 256) {
    const float bv = momentum * b[i] + fmaf(wd, p[i], g[i]);
    b[i] = bv;
    p[i] -= lr * bv;
  }
}

__global__ __launch_bounds__(256) void ema_flat_kernel(
    float* __restrict__ dst, const float* __restrict__ src, int64_t n,
    float m) {
  const float om = 1.f - m;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256)
    dst[i] = m * dst[i] + om * src[i];
}

}  // namespace

extern "C" int launch_multi_tensor_ema(void* stream, const int64_t* ptrs_dst,
                                       const int64_t* ptrs_src,
                                       const int32_t* sizes, int64_t nchunks,
                                       float m) {
  if (nchunks == 0) return 0;
  mt_ema_kernel<<<dim3((unsigned)nchunks), 256, 0, (hipStream_t)stream>>>(
      ptrs_dst, ptrs_src, sizes, m);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_multi_tensor_sgd(void* stream, const int64_t* ptrs_p,
                                       const int64_t* ptrs_g,
                                       const int64_t* ptrs_b,
                                       const int32_t* sizes, int64_t nchunks,
                                       float lr, float momentum, float wd) {
  if (nchunks == 0) return 0;
  mt_sgd_kernel<<<dim3((unsigned)nchunks), 256, 0, (hipStream_t)stream>>>(
      ptrs_p, ptrs_g, ptrs_b, sizes, lr, momentum, wd);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_ema_flat(void* stream, float* dst, const float* src,
                               int64_t n, float m) {
  int64_t blocks = (n + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  ema_flat_kernel<<<dim3((unsigned)blocks), 256, 0,
                    (hipStream_t)stream>>>(dst, src, n, m);
  HIP_CHECK_LAST();
  return 0;
}
