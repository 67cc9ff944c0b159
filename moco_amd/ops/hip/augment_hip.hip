#include "hip/hip_runtime.h"
// On-GPU two-crop augmentation pipeline (replaces the reference CPU/PIL
// pipeline, moco/loader.py + main_moco.py:237-262).  Four memory-bound
// passes over the [M,3,S,S] crop batch (M = 2B):
//   1. bilinear crop-resize from the source batch + per-crop gray-mean
//      reduction (for contrast jitter)
//   2. pointwise color jitter (brightness/contrast/saturation/hue) +
//      random grayscale, v1/v2 op order per the params row
//   3. separable Gaussian blur, horizontal
//   4. Gaussian blur vertical + horizontal flip + ImageNet normalize +
//      cast/layout (fp32/bf16, NCHW/NHWC channels_last)
// All randomness lives in the host-sampled params tensor [M,12]
// (moco_amd/ops/augment.py defines the layout), so every pass is
// deterministic and unit-testable against the pure-torch oracle.
#include "common.h"
#include "launchers.h"

namespace {

// params row layout (keep in sync with ops/augment.py)
enum {
  P_X0 = 0, P_Y0, P_W, P_H, P_FLIP, P_GRAY, P_SIGMA, P_BR, P_CO, P_SA,
  P_HU, P_ORDER
};

__constant__ float MEAN_C[3] = {0.485f, 0.456f, 0.406f};
__constant__ float STD_C[3] = {0.229f, 0.224f, 0.225f};

constexpr float GR = 0.299f, GG = 0.587f, GB = 0.114f;
constexpr int MAX_R = 8;  // sigma <= 2 -> radius ceil(2.5*2) = 5

__device__ __forceinline__ float gray_of(float r, float g, float b) {
  return GR * r + GG * g + GB * b;
}

// ---- pass 1: crop-resize + gray mean --------------------------------------
// ST: float (already [0,1]) or unsigned char (decoded bytes, scaled /255 —
// real-image pipelines ship uint8 and save 4x loader/H2D bandwidth)
template <typename ST>
__device__ __forceinline__ float src_val(ST v);
template <>
__device__ __forceinline__ float src_val<float>(float v) { return v; }
template <>
__device__ __forceinline__ float src_val<unsigned char>(unsigned char v) {
  return (float)v * (1.f / 255.f);
}

template <typename ST>
__global__ __launch_bounds__(256) void aug_crop_resize_kernel(
    const ST* __restrict__ src, const float* __restrict__ params,
    float* __restrict__ tmp1, float* __restrict__ graysum, int B, int H,
    int W, int S) {
  const int m = blockIdx.y;
  const int p = blockIdx.x * 256 + threadIdx.x;
  const float* pm = params + m * 12;
  float g_acc = 0.f;
  if (p < S * S) {
    const int y = p / S, x = p % S;
    const int64_t img = m % B;
    const float sx = pm[P_X0] + (x + 0.5f) * pm[P_W] / S - 0.5f;
    const float sy = pm[P_Y0] + (y + 0.5f) * pm[P_H] / S - 0.5f;
    const float fx0 = floorf(sx), fy0 = floorf(sy);
    const float wx = sx - fx0, wy = sy - fy0;
    const int x0 = min(max((int)fx0, 0), W - 1);
    const int x1 = min(max((int)fx0 + 1, 0), W - 1);
    const int y0 = min(max((int)fy0, 0), H - 1);
    const int y1 = min(max((int)fy0 + 1, 0), H - 1);
    const ST* sb = src + img * 3 * (int64_t)H * W;
    float out3[3];
#pragma unroll
    for (int ch = 0; ch < 3; ++ch) {
      const ST* sc = sb + (int64_t)ch * H * W;
      const float v00 = src_val<ST>(sc[(int64_t)y0 * W + x0]);
      const float v01 = src_val<ST>(sc[(int64_t)y0 * W + x1]);
      const float v10 = src_val<ST>(sc[(int64_t)y1 * W + x0]);
      const float v11 = src_val<ST>(sc[(int64_t)y1 * W + x1]);
      const float top = v00 * (1.f - wx) + v01 * wx;
      const float bot = v10 * (1.f - wx) + v11 * wx;
      const float v = top * (1.f - wy) + bot * wy;
      out3[ch] = v;
      tmp1[((int64_t)m * 3 + ch) * S * S + p] = v;
    }
    g_acc = gray_of(out3[0], out3[1], out3[2]);
  }
  // block-reduce the gray contribution, one atomic per block
  g_acc = wave_reduce_sum(g_acc);
  __shared__ float ws[4];
  if (threadIdx.x % WAVE == 0) ws[threadIdx.x / WAVE] = g_acc;
  __syncthreads();
  if (threadIdx.x == 0)
    atomicAdd(&graysum[m], ws[0] + ws[1] + ws[2] + ws[3]);
}

// ---- hue helpers (match ops/augment.py _rgb_to_hsv/_hsv_to_rgb) -----------
__device__ void hue_rotate(float& r, float& g, float& b, float dh) {
  const float maxc = fmaxf(fmaxf(r, g), b);
  const float minc = fminf(fminf(r, g), b);
  const float v = maxc;
  const float cr = maxc - minc;
  const float s = maxc > 0.f ? cr / fmaxf(maxc, 1e-12f) : 0.f;
  const float crs = fmaxf(cr, 1e-12f);
  const float rc = (maxc - r) / crs;
  const float gc = (maxc - g) / crs;
  const float bc = (maxc - b) / crs;
  float h;
  if (maxc == r)
    h = bc - gc;
  else if (maxc == g)
    h = 2.f + rc - bc;
  else
    h = 4.f + gc - rc;
  h = cr > 0.f ? h / 6.f : 0.f;
  h = h - floorf(h);
  // rotate
  h = h + dh;
  h = h - floorf(h);
  // hsv -> rgb
  const float i_f = floorf(h * 6.f);
  const float f = h * 6.f - i_f;
  const float p = v * (1.f - s);
  const float q = v * (1.f - s * f);
  const float t = v * (1.f - s * (1.f - f));
  const int i = ((int)i_f) % 6;
  switch (i) {
    case 0: r = v; g = t; b = p; break;
    case 1: r = q; g = v; b = p; break;
    case 2: r = p; g = v; b = t; break;
    case 3: r = p; g = q; b = v; break;
    case 4: r = t; g = p; b = v; break;
    default: r = v; g = p; b = q; break;
  }
}

// ---- pass 2: jitter + grayscale -------------------------------------------
__global__ __launch_bounds__(256) void aug_jitter_kernel(
    const float* __restrict__ tmp1, const float* __restrict__ params,
    const float* __restrict__ graysum, float* __restrict__ tmp2, int S) {
  const int m = blockIdx.y;
  const int p = blockIdx.x * 256 + threadIdx.x;
  if (p >= S * S) return;
  const float* pm = params + m * 12;
  const int64_t base = (int64_t)m * 3 * S * S;
  float r = tmp1[base + p];
  float g = tmp1[base + S * S + p];
  float b = tmp1[base + 2 * S * S + p];

  const bool v1_first = pm[P_ORDER] > 0.f;
  const bool do_gray = pm[P_GRAY] > 0.f;
  if (v1_first && do_gray) {
    const float gray = gray_of(r, g, b);
    r = g = b = gray;
  }
  // brightness
  const float br = pm[P_BR];
  r = __saturatef(r * br);
  g = __saturatef(g * br);
  b = __saturatef(b * br);
  // contrast (blend with the crop's gray mean).  NOTE: the mean is of the
  // PRE-jitter crop (pass 1), matching the oracle only when brightness==1;
  // we therefore scale the stored mean by the brightness factor, which is
  // exact for unclipped pixels (documented approximation).
  const float co = pm[P_CO];
  const float mean_g =
      __saturatef(graysum[m] / (float)(S * S) * br);
  r = __saturatef(co * r + (1.f - co) * mean_g);
  g = __saturatef(co * g + (1.f - co) * mean_g);
  b = __saturatef(co * b + (1.f - co) * mean_g);
  // saturation
  const float sa = pm[P_SA];
  const float gray2 = gray_of(r, g, b);
  r = __saturatef(sa * r + (1.f - sa) * gray2);
  g = __saturatef(sa * g + (1.f - sa) * gray2);
  b = __saturatef(sa * b + (1.f - sa) * gray2);
  // hue
  const float hu = pm[P_HU];
  if (hu != 0.f) hue_rotate(r, g, b, hu);
  if (!v1_first && do_gray) {
    const float gray = gray_of(r, g, b);
    r = g = b = gray;
  }
  tmp2[base + p] = r;
  tmp2[base + S * S + p] = g;
  tmp2[base + 2 * S * S + p] = b;
}

__device__ __forceinline__ int blur_radius(float sigma) {
  return min(MAX_R, max(1, (int)ceilf(2.5f * sigma)));
}

// ---- pass 3: horizontal blur ----------------------------------------------
__global__ __launch_bounds__(256) void aug_blur_h_kernel(
    const float* __restrict__ tmp2, const float* __restrict__ params,
    float* __restrict__ tmp1, int S) {
  const int m = blockIdx.y;
  const int p = blockIdx.x * 256 + threadIdx.x;
  if (p >= S * S) return;
  const float sigma = params[m * 12 + P_SIGMA];
  const int64_t base = (int64_t)m * 3 * S * S;
  if (sigma <= 0.f) {
#pragma unroll
    for (int ch = 0; ch < 3; ++ch)
      tmp1[base + ch * S * S + p] = tmp2[base + ch * S * S + p];
    return;
  }
  const int y = p / S, x = p % S;
  const int R = blur_radius(sigma);
  float wsum = 0.f, acc[3] = {0.f, 0.f, 0.f};
  const float inv2s2 = -0.5f / (sigma * sigma);
  for (int t = -R; t <= R; ++t) {
    const float w = __expf((float)(t * t) * inv2s2);
    const int xs = min(max(x + t, 0), S - 1);  // clamp-to-edge
    wsum += w;
#pragma unroll
    for (int ch = 0; ch < 3; ++ch)
      acc[ch] += w * tmp2[base + ch * S * S + y * S + xs];
  }
#pragma unroll
  for (int ch = 0; ch < 3; ++ch)
    tmp1[base + ch * S * S + p] = acc[ch] / wsum;
}

// ---- pass 4: vertical blur + flip + normalize + cast/layout ---------------
template <typename OT, bool NHWC>
__global__ __launch_bounds__(256) void aug_blur_v_finalize_kernel(
    const float* __restrict__ tmp1, const float* __restrict__ params,
    OT* __restrict__ out, int S) {
  const int m = blockIdx.y;
  const int p = blockIdx.x * 256 + threadIdx.x;
  if (p >= S * S) return;
  const float* pm = params + m * 12;
  const float sigma = pm[P_SIGMA];
  const int y = p / S, x = p % S;
  const int64_t base = (int64_t)m * 3 * S * S;
  float v[3];
  if (sigma <= 0.f) {
#pragma unroll
    for (int ch = 0; ch < 3; ++ch) v[ch] = tmp1[base + ch * S * S + p];
  } else {
    const int R = blur_radius(sigma);
    const float inv2s2 = -0.5f / (sigma * sigma);
    float wsum = 0.f;
    v[0] = v[1] = v[2] = 0.f;
    for (int t = -R; t <= R; ++t) {
      const float w = __expf((float)(t * t) * inv2s2);
      const int ys = min(max(y + t, 0), S - 1);
      wsum += w;
#pragma unroll
      for (int ch = 0; ch < 3; ++ch)
        v[ch] += w * tmp1[base + ch * S * S + ys * S + x];
    }
#pragma unroll
    for (int ch = 0; ch < 3; ++ch) v[ch] /= wsum;
  }
  const int xo = pm[P_FLIP] > 0.f ? (S - 1 - x) : x;
#pragma unroll
  for (int ch = 0; ch < 3; ++ch) {
    const float nv = (v[ch] - MEAN_C[ch]) / STD_C[ch];
    const int64_t idx = NHWC
        ? (((int64_t)m * S + y) * S + xo) * 3 + ch
        : base + ch * S * S + y * S + xo;
    out[idx] = (OT)nv;
  }
}

}  // namespace

extern "C" int launch_two_crop_augment(void* stream, const void* src,
                                       bool src_u8, const float* params,
                                       void* out, float* tmp1, float* tmp2,
                                       float* graysum, int64_t B, int64_t M,
                                       int64_t H, int64_t W, int64_t S,
                                       bool out_bf16, bool channels_last) {
  hipStream_t s = (hipStream_t)stream;
  const dim3 grid((unsigned)((S * S + 255) / 256), (unsigned)M);
  if (src_u8)
   hipLaunchKernelGGL(( aug_crop_resize_kernel<unsigned char>), dim3(grid), dim3(256), 0, s, 
        (const unsigned char*)src, params, tmp1, graysum, (int)B, (int)H,
        (int)W, (int)S);
  else
   hipLaunchKernelGGL(( aug_crop_resize_kernel<float>), dim3(grid), dim3(256), 0, s, 
        (const float*)src, params, tmp1, graysum, (int)B, (int)H, (int)W,
        (int)S);
  HIP_CHECK_LAST();
 hipLaunchKernelGGL(( aug_jitter_kernel), dim3(grid), dim3(256), 0, s, tmp1, params, graysum, tmp2,
                                         (int)S);
  HIP_CHECK_LAST();
 hipLaunchKernelGGL(( aug_blur_h_kernel), dim3(grid), dim3(256), 0, s, tmp2, params, tmp1, (int)S);
  HIP_CHECK_LAST();
  if (out_bf16 && channels_last)
   hipLaunchKernelGGL(( aug_blur_v_finalize_kernel<bf16, true>)
        , dim3(grid), dim3(256), 0, s, tmp1, params, (bf16*)out, (int)S);
  else if (out_bf16)
   hipLaunchKernelGGL(( aug_blur_v_finalize_kernel<bf16, false>)
        , dim3(grid), dim3(256), 0, s, tmp1, params, (bf16*)out, (int)S);
  else if (channels_last)
   hipLaunchKernelGGL(( aug_blur_v_finalize_kernel<float, true>)
        , dim3(grid), dim3(256), 0, s, tmp1, params, (float*)out, (int)S);
  else
   hipLaunchKernelGGL(( aug_blur_v_finalize_kernel<float, false>)
        , dim3(grid), dim3(256), 0, s, tmp1, params, (float*)out, (int)S);
  HIP_CHECK_LAST();
  return 0;
}
