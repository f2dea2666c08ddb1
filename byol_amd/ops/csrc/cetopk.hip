// Fused cross-entropy + top-k accuracy (the probe loss and
// top-1/top-5 metrics the reference computes with ~6 separate ATen kernels,
// /root/reference/main.py:596-598).
//
// Forward: one block per row — stable logsumexp, per-row CE loss, top-5
// membership of the label, atomically accumulated into
// out[0..2] = {loss_sum, correct@1, correct@5}; saves per-row (max, lse)
// for the backward.
// Backward: dlogits = (softmax - onehot(label)) * go / M.
#include "common.h"

__global__ void ce_topk_fwd_kernel(const float* __restrict__ logits,
                                   const int64_t* __restrict__ labels,
                                   float* __restrict__ out,   // [3]
                                   float* __restrict__ row_stats,  // [M,2]
                                   int m, int n) {
  const int row = blockIdx.x;
  if (row >= m) return;
  const float* lrow = logits + (int64_t)row * n;
  const int64_t label = labels[row];

  // pass 1: max
  float vmax = -INFINITY;
  for (int j = threadIdx.x; j < n; j += blockDim.x)
    vmax = fmaxf(vmax, lrow[j]);
  __shared__ float sred[4];
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  #pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
    vmax = fmaxf(vmax, __shfl_down(vmax, off, WAVE_SIZE));
  if (lane == 0) sred[wid] = vmax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float v = sred[0];
    for (int w = 1; w < (int)(blockDim.x / WAVE_SIZE); ++w) v = fmaxf(v, sred[w]);
    sred[0] = v;
  }
  __syncthreads();
  vmax = sred[0];

  // pass 2: sumexp + per-thread top-5 + label logit
  float sumexp = 0.f;
  float t5[5] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY, -INFINITY};
  float label_val = -INFINITY;
  for (int j = threadIdx.x; j < n; j += blockDim.x) {
    const float v = lrow[j];
    sumexp += __expf(v - vmax);
    if (j == (int)label) label_val = v;
    // insert into the local top-5 (descending)
    if (v > t5[4]) {
      float x = v;
      #pragma unroll
      for (int k = 0; k < 5; ++k) {
        if (x > t5[k]) { const float tmp = t5[k]; t5[k] = x; x = tmp; }
      }
    }
  }
  __shared__ float ssum[4];
  float se = wave_reduce_sum(sumexp);
  if (lane == 0) ssum[wid] = se;
  // block top-5 via LDS: each thread dumps its 5, thread 0 merges
  __shared__ float stop[256 * 5 / WAVE_SIZE * WAVE_SIZE];  // 1280 floats
  #pragma unroll
  for (int k = 0; k < 5; ++k) stop[threadIdx.x * 5 + k] = t5[k];
  // label value reduction (exactly one thread saw it)
  __shared__ float slabel;
  if (threadIdx.x == 0) slabel = -INFINITY;
  __syncthreads();
  if (label_val > -INFINITY) slabel = label_val;
  __syncthreads();

  if (threadIdx.x == 0) {
    float total = 0.f;
    for (int w = 0; w < (int)(blockDim.x / WAVE_SIZE); ++w)
      total += ssum[w];
    const float lse = logf(total) + vmax;
    const float lv = slabel;
    const float loss = lse - lv;
    row_stats[row * 2 + 0] = vmax;
    row_stats[row * 2 + 1] = lse;
    // rank of the label: count strictly-greater among the block's merged
    // top-5 candidates is not enough; count strictly greater overall via
    // the candidate pool (any value greater than lv must appear in some
    // thread's local top-5 unless that thread saw >5 greater values; for
    // rank<=5 checks the pool is sufficient).
    int greater = 0;
    for (int t = 0; t < (int)(blockDim.x * 5); ++t)
      if (stop[t] > lv) ++greater;
    if (greater > 5) greater = 5;  // only need <1 and <5 decisions
    atomicAdd(&out[0], loss);
    atomicAdd(&out[1], greater < 1 ? 1.f : 0.f);
    atomicAdd(&out[2], greater < 5 ? 1.f : 0.f);
  }
}

__global__ void ce_bwd_kernel(const float* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ row_stats,
                              const float* __restrict__ grad_out,
                              float* __restrict__ dlogits,
                              float inv_m, int64_t total, int n) {
  const float go = grad_out[0] * inv_m;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int64_t row = i / n;
    const int j = (int)(i % n);
    const float lse = row_stats[row * 2 + 1];
    const float p = __expf(logits[i] - lse);
    const float onehot = (j == (int)labels[row]) ? 1.f : 0.f;
    dlogits[i] = go * (p - onehot);
  }
}

void launch_ce_topk_fwd(const float* logits, const int64_t* labels,
                        float* out, float* row_stats, int m, int n,
                        hipStream_t stream) {
  hipLaunchKernelGGL(ce_topk_fwd_kernel, dim3(m), dim3(256), 0, stream,
                     logits, labels, out, row_stats, m, n);
}

void launch_ce_bwd(const float* logits, const int64_t* labels,
                   const float* row_stats, const float* grad_out,
                   float* dlogits, int m, int n, hipStream_t stream) {
  const int64_t total = (int64_t)m * n;
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid_1d(total, 256)), dim3(256), 0,
                     stream, logits, labels, row_stats, grad_out, dlogits,
                     1.0f / (float)m, total, n);
}
