// Flat-parameter EMA update: mean <- (1-decay)*x + decay*mean.
// One float4-vectorized pass over the contiguous flat buffer (2 reads +
// 1 write; HBM-bound). Replaces the reference's parameters_to_vector pack +
// ATen blend (/root/reference/main.py:158-161,255).
#include "common.h"

// decay arrives either as an immediate (decay_dev == nullptr) or in a
// 1-float device scalar — the latter keeps the kernel hipGraph-capturable
// with a per-step cosine-ramped decay (the replay wrapper rewrites the
// scalar outside the graph).
__global__ void flat_ema_update_kernel(float* __restrict__ mean,
                                       const float* __restrict__ x,
                                       float decay,
                                       const float* __restrict__ decay_dev,
                                       int64_t n4,
                                       int64_t n_tail, int64_t tail_base) {
  if (decay_dev != nullptr) decay = *decay_dev;
  const float w = 1.0f - decay;
  float4* m4 = reinterpret_cast<float4*>(mean);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n4; i += stride) {
    float4 m = m4[i];
    float4 v = x4[i];
    m.x = fmaf(w, v.x - m.x, m.x);
    m.y = fmaf(w, v.y - m.y, m.y);
    m.z = fmaf(w, v.z - m.z, m.z);
    m.w = fmaf(w, v.w - m.w, m.w);
    m4[i] = m;
  }
  // scalar tail (flat buffer length not necessarily a multiple of 4)
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_tail; i += stride) {
    const int64_t j = tail_base + i;
    mean[j] = fmaf(w, x[j] - mean[j], mean[j]);
  }
}

void launch_flat_ema_update(float* mean, const float* x, float decay,
                            const float* decay_dev, int64_t n,
                            hipStream_t stream) {
  const int64_t n4 = n / 4;
  const int64_t tail_base = n4 * 4;
  const int64_t n_tail = n - tail_base;
  const int block = 256;
  const int grid = grid_1d(n4 > 0 ? n4 : n_tail, block);
  hipLaunchKernelGGL(flat_ema_update_kernel, dim3(grid), dim3(block), 0,
                     stream, mean, x, decay, decay_dev, n4, n_tail,
                     tail_base);
}
