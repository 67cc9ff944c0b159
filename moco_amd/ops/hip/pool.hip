// MaxPool2d 3x3 stride-2 pad-1 (the ResNet stem pool), NHWC bf16.
// MIOpen's pooling kernels measured 1.2 ms/step on the ResNet-50 step;
// these are plain memory-bound kernels with 16-byte channel-octet loads.
// fwd saves a per-output 4-bit tap id; bwd is a GATHER over the <=4 output
// windows covering each input element (no atomics).
#include "common.h"
#include "launchers.h"

namespace {

typedef unsigned short ushort8v __attribute__((ext_vector_type(8)));
typedef unsigned char uchar8v __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float us2f_(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

// fwd: one thread per (n, ho, wo, channel-octet)
__global__ __launch_bounds__(256) void maxpool_fwd_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    unsigned char* __restrict__ idx, int N, int C, int H, int W, int Ho,
    int Wo) {
  const int Q = C / 8;
  const int64_t total = (int64_t)N * Ho * Wo * Q;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * 256) {
    const int q = (int)(i % Q);
    int64_t t = i / Q;
    const int wo = (int)(t % Wo);
    t /= Wo;
    const int ho = (int)(t % Ho);
    const int n = (int)(t / Ho);
    float best[8];
    int bidx[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      best[j] = -3.0e38f;
      bidx[j] = 0;
    }
    const int h0 = ho * 2 - 1, w0 = wo * 2 - 1;
#pragma unroll
    for (int dh = 0; dh < 3; ++dh) {
      const int h = h0 + dh;
      if (h < 0 || h >= H) continue;
#pragma unroll
      for (int dw = 0; dw < 3; ++dw) {
        const int w = w0 + dw;
        if (w < 0 || w >= W) continue;
        const ushort8v v = *reinterpret_cast<const ushort8v*>(
            x + (((int64_t)n * H + h) * W + w) * C + q * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = us2f_(v[j]);
          if (f > best[j]) {
            best[j] = f;
            bidx[j] = dh * 3 + dw;
          }
        }
      }
    }
    ushort8v out;
    uchar8v oidx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { float f; unsigned int u; } cv;
      cv.f = best[j];
      const unsigned int lsb = (cv.u >> 16) & 1;
      cv.u += 0x7fffu + lsb;
      out[j] = (unsigned short)(cv.u >> 16);
      oidx[j] = (unsigned char)bidx[j];
    }
    *reinterpret_cast<ushort8v*>(y + i * 8) = out;
    *reinterpret_cast<uchar8v*>(idx + i * 8) = oidx;
  }
}

// bwd: one thread per (n, h, w, channel-octet); gathers from the <=4
// output windows (ho, wo) whose 3x3 taps cover (h, w)
__global__ __launch_bounds__(256) void maxpool_bwd_kernel(
    const unsigned short* __restrict__ g, const unsigned char* __restrict__ idx,
    unsigned short* __restrict__ dx, int N, int C, int H, int W, int Ho,
    int Wo) {
  const int Q = C / 8;
  const int64_t total = (int64_t)N * H * W * Q;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * 256) {
    const int q = (int)(i % Q);
    int64_t t = i / Q;
    const int w = (int)(t % W);
    t /= W;
    const int h = (int)(t % H);
    const int n = (int)(t / H);
    float acc[8] = {};
    // window ho covers h iff ho*2-1 <= h <= ho*2+1, i.e. ho in
    // [h/2, (h+1)/2] (integer division); same for wo
    for (int ho = h / 2; ho <= min(Ho - 1, (h + 1) / 2); ++ho) {
      const int dh = h - (ho * 2 - 1);
      for (int wo = w / 2; wo <= min(Wo - 1, (w + 1) / 2); ++wo) {
        const int dw = w - (wo * 2 - 1);
        const int64_t o = (((int64_t)n * Ho + ho) * Wo + wo) * C + q * 8;
        const uchar8v iv = *reinterpret_cast<const uchar8v*>(idx + o);
        const unsigned char tap = (unsigned char)(dh * 3 + dw);
        bool any = false;
#pragma unroll
        for (int j = 0; j < 8; ++j) any |= (iv[j] == tap);
        if (!any) continue;
        const ushort8v gv = *reinterpret_cast<const ushort8v*>(g + o);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (iv[j] == tap) acc[j] += us2f_(gv[j]);
      }
    }
    ushort8v out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { float f; unsigned int u; } cv;
      cv.f = acc[j];
      const unsigned int lsb = (cv.u >> 16) & 1;
      cv.u += 0x7fffu + lsb;
      out[j] = (unsigned short)(cv.u >> 16);
    }
    *reinterpret_cast<ushort8v*>(dx + i * 8) = out;
  }
}

int pool_grid(int64_t total) {
  int64_t b = (total + 255) / 256;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

}  // namespace

extern "C" int launch_maxpool_fwd(void* stream, const void* x, void* y,
                                  void* idx, int64_t N, int64_t C, int64_t H,
                                  int64_t W, int64_t Ho, int64_t Wo) {
  if (C % 8 != 0) return (int)hipErrorInvalidValue;
  const int64_t total = N * Ho * Wo * (C / 8);
  maxpool_fwd_kernel<<<pool_grid(total), 256, 0, (hipStream_t)stream>>>(
      (const unsigned short*)x, (unsigned short*)y, (unsigned char*)idx,
      (int)N, (int)C, (int)H, (int)W, (int)Ho, (int)Wo);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_maxpool_bwd(void* stream, const void* g,
                                  const void* idx, void* dx, int64_t N,
                                  int64_t C, int64_t H, int64_t W, int64_t Ho,
                                  int64_t Wo) {
  if (C % 8 != 0) return (int)hipErrorInvalidValue;
  const int64_t total = N * H * W * (C / 8);
  maxpool_bwd_kernel<<<pool_grid(total), 256, 0, (hipStream_t)stream>>>(
      (const unsigned short*)g, (const unsigned char*)idx,
      (unsigned short*)dx, (int)N, (int)C, (int)H, (int)W, (int)Ho, (int)Wo);
  HIP_CHECK_LAST();
  return 0;
}

// ---------------------------------------------------------------------------
// Head adaptive-avg-pool to 1x1 (the ResNet global pool), NHWC bf16.
// fwd: one block per image; each thread averages its channel octets over
// the HW positions.  bwd: broadcast g/HW (elementwise).
// ---------------------------------------------------------------------------
namespace {

__device__ __forceinline__ unsigned short f2us_(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  const unsigned int lsb = (v.i >> 16) & 1;
  v.i += 0x7fffu + lsb;
  return (unsigned short)(v.i >> 16);
}

__global__ __launch_bounds__(256) void avgpool_fwd_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    int C, int HW) {
  const int n = blockIdx.x;
  const int Q = C / 8;
  const float inv = 1.0f / (float)HW;
  for (int q = threadIdx.x; q < Q; q += 256) {
    float acc[8] = {};
    const unsigned short* base = x + ((int64_t)n * HW) * C + q * 8;
    for (int p = 0; p < HW; ++p) {
      const ushort8v v =
          *reinterpret_cast<const ushort8v*>(base + (int64_t)p * C);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += us2f_(v[j]);
    }
    ushort8v out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = f2us_(acc[j] * inv);
    *reinterpret_cast<ushort8v*>(y + (int64_t)n * C + q * 8) = out;
  }
}

__global__ __launch_bounds__(256) void avgpool_bwd_kernel(
    const unsigned short* __restrict__ g, unsigned short* __restrict__ dx,
    int C, int HW, int64_t total8) {
  const float inv = 1.0f / (float)HW;
  const int Q = C / 8;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < total8;
       i += (int64_t)gridDim.x * 256) {
    const int64_t q = i % Q;          // channel octet
    const int64_t n = i / (Q * (int64_t)HW);
    const ushort8v gv =
        *reinterpret_cast<const ushort8v*>(g + (n * Q + q) * 8);
    ushort8v out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = f2us_(us2f_(gv[j]) * inv);
    *reinterpret_cast<ushort8v*>(dx + i * 8) = out;
  }
}

}  // namespace

extern "C" int launch_avgpool_fwd(void* stream, const void* x, void* y,
                                  int64_t N, int64_t C, int64_t HW) {
  if (C % 8 != 0) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  avgpool_fwd_kernel<<<dim3((unsigned)N), 256, 0, s>>>(
      (const unsigned short*)x, (unsigned short*)y, (int)C, (int)HW);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_avgpool_bwd(void* stream, const void* g, void* dx,
                                  int64_t N, int64_t C, int64_t HW) {
  if (C % 8 != 0) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int64_t total8 = N * HW * C / 8;
  int64_t b = (total8 + 255) / 256;
  if (b > 2048) b = 2048;
  avgpool_bwd_kernel<<<dim3((unsigned)b), 256, 0, s>>>(
      (const unsigned short*)g, (unsigned short*)dx, (int)C, (int)HW,
      total8);
  HIP_CHECK_LAST();
  return 0;
}
