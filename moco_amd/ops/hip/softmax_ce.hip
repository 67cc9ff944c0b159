// Softmax cross-entropy specialized for target class 0 (the MoCo InfoNCE:
// reference main_moco.py:202,310 with labels always zero, builder.py:158).
// fp32 logits [N, M], M = 1 + K (up to ~131k).
//
// fwd: one 256-thread block per row, single online (max, sum) pass with
//      float4 loads; saves per-row max and log-sum-exp for the backward.
// bwd: elementwise dlogits = (softmax - onehot0) * scale.
#include "common.h"
#include "launchers.h"

namespace {

__global__ __launch_bounds__(256) void ce0_fwd_kernel(
    const float* __restrict__ logits, float* __restrict__ loss_rows,
    float* __restrict__ row_max, float* __restrict__ row_lse, int64_t M) {
  const int64_t row = blockIdx.x;
  const float* lr = logits + row * M;
  const int tid = threadIdx.x;

  // online max/sum over this thread's strided slice (vectorized by 4).
  // -FLT_MAX, not -inf: threads with an empty slice must combine as
  // exp(m - nm) = exp(finite), never (-inf) - (-inf) = nan.
  float m = -3.0e38f;
  float s = 0.f;
  // peel a misaligned head: the row stride M = 1 + K is odd, so lr is
  // only 4-byte aligned for row > 0 — 16-byte vector loads must start at
  // an aligned element (dword-tolerant loads are a gfx950
  // behavior, not an ISA guarantee)
  const int head0 = (int)((4 - ((((unsigned long long)(size_t)lr) >> 2)
                               & 3)) & 3);
  const int64_t head = head0 < M ? head0 : M;
  for (int64_t i = tid; i < head; i += 256) {
    const float v = lr[i];
    if (v > m) {
      s *= __expf(m - v);
      m = v;
    }
    s += __expf(v - m);
  }
  const int64_t M4 = head + ((M - head) & ~3LL);
  for (int64_t i = head + (int64_t)tid * 4; i < M4; i += 256 * 4) {
    const float4 v = *reinterpret_cast<const float4*>(lr + i);
    const float vm = fmaxf(fmaxf(v.x, v.y), fmaxf(v.z, v.w));
    if (vm > m) {
      s *= __expf(m - vm);
      m = vm;
    }
    s += __expf(v.x - m) + __expf(v.y - m) + __expf(v.z - m) +
         __expf(v.w - m);
  }
  for (int64_t i = M4 + tid; i < M; i += 256) {
    const float v = lr[i];
    if (v > m) {
      s *= __expf(m - v);
      m = v;
    }
    s += __expf(v - m);
  }

  // combine across the block: wave butterfly, then cross-wave via LDS
  __shared__ float sm[4], ss[4];
  for (int off = 32; off > 0; off >>= 1) {
    const float om = __shfl_xor(m, off, WAVE);
    const float os = __shfl_xor(s, off, WAVE);
    const float nm = fmaxf(m, om);
    s = s * __expf(m - nm) + os * __expf(om - nm);
    m = nm;
  }
  const int wave = tid / WAVE;
  if (tid % WAVE == 0) {
    sm[wave] = m;
    ss[wave] = s;
  }
  __syncthreads();
  if (tid == 0) {
    float fm = sm[0], fs = ss[0];
    for (int w = 1; w < 4; ++w) {
      const float nm = fmaxf(fm, sm[w]);
      fs = fs * __expf(fm - nm) + ss[w] * __expf(sm[w] - nm);
      fm = nm;
    }
    const float lse = __logf(fs);
    row_max[row] = fm;
    row_lse[row] = lse;
    loss_rows[row] = fm + lse - lr[0];
  }
}

// gscale (nullable): extra scale read FROM DEVICE (the upstream grad of
// the mean loss under hipGraph capture, where a host read would sync)
__global__ __launch_bounds__(256) void ce0_bwd_kernel(
    const float* __restrict__ logits, const float* __restrict__ row_max,
    const float* __restrict__ row_lse, float* __restrict__ dlogits,
    float scale, const float* __restrict__ gscale, int64_t N, int64_t M) {
  const int64_t idx = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 4;
  const int64_t total = N * M;
  if (idx >= total) return;
  if (gscale != nullptr) scale *= *gscale;
  const int64_t row = idx / M;
  const int64_t col = idx - row * M;
  const float z = row_max[row] + row_lse[row];
  if (col + 4 <= M) {
    const float4 v = *reinterpret_cast<const float4*>(logits + idx);
    float4 d;
    d.x = (__expf(v.x - z) - (col == 0 ? 1.f : 0.f)) * scale;
    d.y = __expf(v.y - z) * scale;
    d.z = __expf(v.z - z) * scale;
    d.w = __expf(v.w - z) * scale;
    *reinterpret_cast<float4*>(dlogits + idx) = d;
  } else {
    for (int64_t i = idx; i < min(idx + 4, total); ++i) {
      const int64_t r = i / M, c = i - r * M;
      const float zz = row_max[r] + row_lse[r];
      dlogits[i] =
          (__expf(logits[i] - zz) - (c == 0 ? 1.f : 0.f)) * scale;
    }
  }
}

}  // namespace

extern "C" int launch_softmax_ce0_fwd(void* stream, const float* logits,
                                      float* loss_rows, float* row_max,
                                      float* row_lse, int64_t N, int64_t M) {
  hipStream_t s = (hipStream_t)stream;
  ce0_fwd_kernel<<<dim3((unsigned)N), 256, 0, s>>>(logits, loss_rows, row_max,
                                                   row_lse, M);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_softmax_ce0_bwd(void* stream, const float* logits,
                                      const float* row_max,
                                      const float* row_lse, float* dlogits,
                                      float scale, const float* gscale,
                                      int64_t N, int64_t M) {
  hipStream_t s = (hipStream_t)stream;
  const int64_t total = (N * M + 3) / 4;
  const int64_t blocks = (total + 255) / 256;
  ce0_bwd_kernel<<<dim3((unsigned)blocks), 256, 0, s>>>(
      logits, row_max, row_lse, dlogits, scale, gscale, N, M);
  HIP_CHECK_LAST();
  return 0;
}
