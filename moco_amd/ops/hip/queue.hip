// Ring-buffer enqueue (reference _dequeue_and_enqueue, builder.py:59-74):
// write the gathered key batch at column block [ptr, ptr+B) of the canonical
// fp32 queue [C, K] AND row block [ptr, ptr+B) of the bf16 GEMM working copy
// queue_kc [K, C], in one launch.
#include "common.h"
#include "launchers.h"

namespace {

template <typename KT>
__global__ __launch_bounds__(256) void enqueue_kernel(
    float* __restrict__ queue, bf16* __restrict__ queue_kc,
    const KT* __restrict__ keys, int C, int64_t K, int64_t ptr) {
  const int b = blockIdx.x;  // key row
  const KT* krow = keys + (int64_t)b * C;
  for (int c = threadIdx.x; c < C; c += 256) {
    const float v = (float)krow[c];
    queue[(int64_t)c * K + ptr + b] = v;            // column write, [C,K]
    if (queue_kc) queue_kc[(ptr + b) * C + c] = f2bf(v);  // row write, [K,C]
  }
}

// Device-pointer variant for hipGraph replay: the ring position lives in
// the queue_ptr buffer and is read/advanced ON DEVICE, so a captured
// enqueue rotates correctly across replays (a host-side ptr would be
// frozen into the graph).
template <typename KT>
__global__ __launch_bounds__(256) void enqueue_dev_kernel(
    float* __restrict__ queue, bf16* __restrict__ queue_kc,
    const KT* __restrict__ keys, int C, int64_t K,
    const int64_t* __restrict__ ptr_dev) {
  const int64_t ptr = *ptr_dev;
  const int b = blockIdx.x;
  const KT* krow = keys + (int64_t)b * C;
  for (int c = threadIdx.x; c < C; c += 256) {
    const float v = (float)krow[c];
    queue[(int64_t)c * K + ptr + b] = v;
    if (queue_kc) queue_kc[(ptr + b) * C + c] = f2bf(v);
  }
}

__global__ void ptr_advance_kernel(int64_t* ptr_dev, int64_t B, int64_t K) {
  *ptr_dev = (*ptr_dev + B) % K;
}

}  // namespace

extern "C" int launch_queue_enqueue_dev(void* stream, float* queue,
                                        void* queue_kc, const void* keys,
                                        bool keys_bf16, int64_t C, int64_t K,
                                        int64_t B, int64_t* ptr_dev) {
  hipStream_t s = (hipStream_t)stream;
  if (keys_bf16)
    enqueue_dev_kernel<bf16><<<dim3((unsigned)B), 256, 0, s>>>(
        queue, (bf16*)queue_kc, (const bf16*)keys, (int)C, K, ptr_dev);
  else
    enqueue_dev_kernel<float><<<dim3((unsigned)B), 256, 0, s>>>(
        queue, (bf16*)queue_kc, (const float*)keys, (int)C, K, ptr_dev);
  HIP_CHECK_LAST();
  ptr_advance_kernel<<<1, 1, 0, s>>>(ptr_dev, B, K);
  HIP_CHECK_LAST();
  return 0;
}

extern "C" int launch_queue_enqueue(void* stream, float* queue,
                                    void* queue_kc, const void* keys,
                                    bool keys_bf16, int64_t C, int64_t K,
                                    int64_t B, int64_t ptr) {
  hipStream_t s = (hipStream_t)stream;
  if (keys_bf16)
    enqueue_kernel<bf16><<<dim3((unsigned)B), 256, 0, s>>>(
        queue, (bf16*)queue_kc, (const bf16*)keys, (int)C, K, ptr);
  else
    enqueue_kernel<float><<<dim3((unsigned)B), 256, 0, s>>>(
        queue, (bf16*)queue_kc, (const float*)keys, (int)C, K, ptr);
  HIP_CHECK_LAST();
  return 0;
}
