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

// ---- weight mirrors --------------------------------------------------------
// One launch refreshes every conv weight's bf16 working copy from the fp32
// master (the fused-block GEMMs read bf16 mirrors; per-weight .to(bfloat16)
// calls were ~115 launches/step).  Chunk table like the EMA/SGD kernels.
namespace {
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
__global__ __launch_bounds__(256) void multi_cast_bf16_kernel(
    const int64_t* __restrict__ src_ptrs, const int64_t* __restrict__ dst_ptrs,
    const int32_t* __restrict__ sizes) {
  const float* __restrict__ src =
      reinterpret_cast<const float*>(src_ptrs[blockIdx.x]);
  bf16* __restrict__ dst = reinterpret_cast<bf16*>(dst_ptrs[blockIdx.x]);
  const int n = sizes[blockIdx.x];
  for (int i = threadIdx.x * 4; i < n; i += 256 * 4) {
    if (i + 4 <= n) {
      const float4 v = *reinterpret_cast<const float4*>(src + i);
      bf16x4 o = {f2bf(v.x), f2bf(v.y), f2bf(v.z), f2bf(v.w)};
      *reinterpret_cast<bf16x4*>(dst + i) = o;
    } else {
      for (int j = i; j < n; ++j) dst[j] = f2bf(src[j]);
    }
  }
}

// One launch transposes every 1x1-conv bf16 mirror [Co, Ci] -> [Ci, Co]
// (the dgrad GEMMs want the contraction contiguous).  One workgroup per
// 64x64 tile; meta = {co0, ci0, Co, Ci} per tile; Co, Ci % 64 == 0.
__global__ __launch_bounds__(256) void multi_transpose_bf16_kernel(
    const int64_t* __restrict__ src_ptrs, const int64_t* __restrict__ dst_ptrs,
    const int32_t* __restrict__ meta) {
  __shared__ bf16 tile[64][72];  // 72: pad away bank conflicts
  const bf16* __restrict__ src =
      reinterpret_cast<const bf16*>(src_ptrs[blockIdx.x]);
  bf16* __restrict__ dst = reinterpret_cast<bf16*>(dst_ptrs[blockIdx.x]);
  const int co0 = meta[blockIdx.x * 4 + 0];
  const int ci0 = meta[blockIdx.x * 4 + 1];
  const int Co = meta[blockIdx.x * 4 + 2];
  const int Ci = meta[blockIdx.x * 4 + 3];
  // load 64 rows x 64 cols: thread t covers row t/4, cols (t%4)*16..+16
  const int lr = threadIdx.x / 4;
  const int lc = (threadIdx.x % 4) * 16;
#pragma unroll
  for (int j = 0; j < 16; j += 8)
    *reinterpret_cast<bf16x8*>(&tile[lr][lc + j]) =
        *reinterpret_cast<const bf16x8*>(src + (int64_t)(co0 + lr) * Ci +
                                         ci0 + lc + j);
  __syncthreads();
  bf16x8 o[2];
#pragma unroll
  for (int p = 0; p < 2; ++p)
#pragma unroll
    for (int j = 0; j < 8; ++j) o[p][j] = tile[lc + p * 8 + j][lr];
#pragma unroll
  for (int p = 0; p < 2; ++p)
    *reinterpret_cast<bf16x8*>(dst + (int64_t)(ci0 + lr) * Co + co0 + lc +
                               p * 8) = o[p];
}
}  // namespace

extern "C" int launch_multi_cast_bf16(void* stream, const int64_t* src_ptrs,
                                      const int64_t* dst_ptrs,
                                      const int32_t* sizes, int64_t nchunks) {
  hipStream_t s = (hipStream_t)stream;
  multi_cast_bf16_kernel<<<dim3((unsigned)nchunks), 256, 0, s>>>(
      src_ptrs, dst_ptrs, sizes);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_multi_transpose_bf16(void* stream,
                                           const int64_t* src_ptrs,
                                           const int64_t* dst_ptrs,
                                           const int32_t* meta,
                                           int64_t ntiles) {
  hipStream_t s = (hipStream_t)stream;
  multi_transpose_bf16_kernel<<<dim3((unsigned)ntiles), 256, 0, s>>>(
      src_ptrs, dst_ptrs, meta);
  HIP_CHECK_LAST();
  return 0;
}
