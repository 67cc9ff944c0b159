// Row-wise L2 normalization, forward + backward.
// Replaces nn.functional.normalize(dim=1) at reference builder.py:131,138.
// One wave per row (C <= 512 elements => <= 8 per lane), block = 4 waves.
#include "common.h"
#include "launchers.h"

namespace {

constexpr int ROWS_PER_BLOCK = 4;
constexpr float EPS = 1e-12f;

template <typename T>
__device__ __forceinline__ float load_f(const T* p);
template <>
__device__ __forceinline__ float load_f<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float load_f<bf16>(const bf16* p) {
  return bf2f(*p);
}

template <typename T>
__device__ __forceinline__ void store_f(T* p, float v);
template <>
__device__ __forceinline__ void store_f<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void store_f<bf16>(bf16* p, float v) {
  *p = f2bf(v);
}

template <typename T>
__global__ void l2norm_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  float* __restrict__ inv_norm, int64_t N,
                                  int C) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int64_t row = (int64_t)blockIdx.x * ROWS_PER_BLOCK + wave;
  if (row >= N) return;
  const T* xr = x + row * C;
  float vals[8];
  float ss = 0.f;
  const int per_lane = (C + WAVE - 1) / WAVE;
  for (int i = 0; i < per_lane; ++i) {
    const int c = lane + i * WAVE;
    float v = (c < C) ? load_f(xr + c) : 0.f;
    vals[i] = v;
    ss += v * v;
  }
  ss = wave_reduce_sum(ss);
  const float inv = rsqrtf(fmaxf(ss, EPS * EPS));
  if (lane == 0) inv_norm[row] = inv;
  T* yr = y + row * C;
  for (int i = 0; i < per_lane; ++i) {
    const int c = lane + i * WAVE;
    if (c < C) store_f(yr + c, vals[i] * inv);
  }
}

// gx = (g - y * (g . y)) * inv_norm
template <typename T>
__global__ void l2norm_bwd_kernel(const T* __restrict__ g,
                                  const T* __restrict__ y,
                                  const float* __restrict__ inv_norm,
                                  T* __restrict__ gx, int64_t N, int C) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int64_t row = (int64_t)blockIdx.x * ROWS_PER_BLOCK + wave;
  if (row >= N) return;
  const T* gr = g + row * C;
  const T* yr = y + row * C;
  float gv[8], yv[8];
  float dot = 0.f;
  const int per_lane = (C + WAVE - 1) / WAVE;
  for (int i = 0; i < per_lane; ++i) {
    const int c = lane + i * WAVE;
    const float gvi = (c < C) ? load_f(gr + c) : 0.f;
    const float yvi = (c < C) ? load_f(yr + c) : 0.f;
    gv[i] = gvi;
    yv[i] = yvi;
    dot += gvi * yvi;
  }
  dot = wave_reduce_sum(dot);
  const float inv = inv_norm[row];
  T* gxr = gx + row * C;
  for (int i = 0; i < per_lane; ++i) {
    const int c = lane + i * WAVE;
    if (c < C) store_f(gxr + c, (gv[i] - yv[i] * dot) * inv);
  }
}

}  // namespace

extern "C" int launch_l2norm_fwd(void* stream, const void* x, void* y,
                                 float* inv_norm, int64_t N, int64_t C,
                                 bool is_bf16) {
  if (C > 8 * WAVE) return (int)hipErrorInvalidValue;
  dim3 grid((N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  dim3 block(ROWS_PER_BLOCK * WAVE);
  hipStream_t s = (hipStream_t)stream;
  if (is_bf16)
    l2norm_fwd_kernel<bf16><<<grid, block, 0, s>>>(
        (const bf16*)x, (bf16*)y, inv_norm, N, (int)C);
  else
    l2norm_fwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)x, (float*)y, inv_norm, N, (int)C);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_l2norm_bwd(void* stream, const void* g, const void* y,
                                 const float* inv_norm, void* gx, int64_t N,
                                 int64_t C, bool is_bf16) {
  if (C > 8 * WAVE) return (int)hipErrorInvalidValue;
  dim3 grid((N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK);
  dim3 block(ROWS_PER_BLOCK * WAVE);
  hipStream_t s = (hipStream_t)stream;
  if (is_bf16)
    l2norm_bwd_kernel<bf16><<<grid, block, 0, s>>>(
        (const bf16*)g, (const bf16*)y, inv_norm, (bf16*)gx, N, (int)C);
  else
    l2norm_bwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)g, (const float*)y, inv_norm, (float*)gx, N, (int)C);
  HIP_CHECK_LAST();
  return 0;
}
