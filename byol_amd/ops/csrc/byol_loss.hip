// Fused BYOL loss (reference semantics: /root/reference/objective.py:6-25).
//
//   loss = mean_i[ -2*dot(p1_i,z2_i)/(|p1|_F |z2|_F)
//                  -2*dot(p2_i,z1_i)/(|p2|_F |z1|_F) ]
//
// Forward: ONE reduction kernel accumulates the six global sums
//   stats = [S_a, |p1|^2, |z2|^2, S_b, |p2|^2, |z1|^2]
// (S = sum of row dots) + a tiny finisher computing the scalar loss on
// device (no host sync). The reference pays 4 norm reductions + row-dot +
// scale + mean (~8 ATen launches).
//
// Backward w.r.t. p1 (z2 detached; symmetric for p2):
//   d/dp1 = (-2*go/B) * [ z2/(N1*N2) - S_a * p1 / (N1^3 * N2) ]
#include "common.h"

__global__ void byol_loss_stats_kernel(const float* __restrict__ p1,
                                       const float* __restrict__ p2,
                                       const float* __restrict__ z1,
                                       const float* __restrict__ z2,
                                       float* __restrict__ stats,
                                       int64_t n) {
  __shared__ float scratch[6][4];  // up to 4 waves/block
  float acc[6] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    const float a = p1[i], b = z2[i], c = p2[i], d = z1[i];
    acc[0] = fmaf(a, b, acc[0]);
    acc[1] = fmaf(a, a, acc[1]);
    acc[2] = fmaf(b, b, acc[2]);
    acc[3] = fmaf(c, d, acc[3]);
    acc[4] = fmaf(c, c, acc[4]);
    acc[5] = fmaf(d, d, acc[5]);
  }
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  #pragma unroll
  for (int k = 0; k < 6; ++k) {
    float v = wave_reduce_sum(acc[k]);
    if (lane == 0) scratch[k][wid] = v;
  }
  __syncthreads();
  if (wid == 0) {
    const int nwaves = blockDim.x / WAVE_SIZE;
    #pragma unroll
    for (int k = 0; k < 6; ++k) {
      float v = (lane < nwaves) ? scratch[k][lane] : 0.f;
      v = wave_reduce_sum(v);
      if (lane == 0) atomicAdd(&stats[k], v);
    }
  }
}

__global__ void byol_loss_finish_kernel(const float* __restrict__ stats,
                                        float* __restrict__ loss,
                                        float inv_batch) {
  const float sa = stats[0];
  const float n1 = sqrtf(stats[1]), n2 = sqrtf(stats[2]);
  const float sb = stats[3];
  const float n3 = sqrtf(stats[4]), n4 = sqrtf(stats[5]);
  loss[0] = -2.0f * inv_batch * (sa / (n1 * n2) + sb / (n3 * n4));
}

__global__ void byol_loss_backward_kernel(const float* __restrict__ p1,
                                          const float* __restrict__ p2,
                                          const float* __restrict__ z1,
                                          const float* __restrict__ z2,
                                          const float* __restrict__ stats,
                                          const float* __restrict__ grad_out,
                                          float* __restrict__ g1,
                                          float* __restrict__ g2,
                                          float inv_batch, int64_t n) {
  const float go = grad_out[0] * -2.0f * inv_batch;
  const float sa = stats[0];
  const float n1 = sqrtf(stats[1]), n2 = sqrtf(stats[2]);
  const float sb = stats[3];
  const float n3 = sqrtf(stats[4]), n4 = sqrtf(stats[5]);
  const float ia = 1.0f / (n1 * n2);
  const float ca = sa / (n1 * n1 * n1 * n2);
  const float ib = 1.0f / (n3 * n4);
  const float cb = sb / (n3 * n3 * n3 * n4);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    g1[i] = go * (z2[i] * ia - p1[i] * ca);
    g2[i] = go * (z1[i] * ib - p2[i] * cb);
  }
}

void launch_byol_loss_forward(const float* p1, const float* p2,
                              const float* z1, const float* z2,
                              float* stats, float* loss, int64_t n,
                              int64_t batch, hipStream_t stream) {
  const int block = 256;
  const int grid = grid_1d(n, block, 4096);
  hipLaunchKernelGGL(byol_loss_stats_kernel, dim3(grid), dim3(block), 0,
                     stream, p1, p2, z1, z2, stats, n);
  hipLaunchKernelGGL(byol_loss_finish_kernel, dim3(1), dim3(1), 0, stream,
                     stats, loss, 1.0f / (float)batch);
}

void launch_byol_loss_backward(const float* p1, const float* p2,
                               const float* z1, const float* z2,
                               const float* stats, const float* grad_out,
                               float* g1, float* g2, int64_t n,
                               int64_t batch, hipStream_t stream) {
  const int block = 256;
  const int grid = grid_1d(n, block);
  hipLaunchKernelGGL(byol_loss_backward_kernel, dim3(grid), dim3(block), 0,
                     stream, p1, p2, z1, z2, stats, grad_out, g1, g2,
                     1.0f / (float)batch, n);
}
