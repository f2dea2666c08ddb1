// Fused multi-tensor LARS + SGD-momentum step over the flat parameter space.
// Three launches replace the reference's ~6 ATen kernels x ~160 tensors
// (/root/reference/optimizers/lars.py:84-127 + torch SGD):
//   1. lars_norms_kernel  — per-segment sum(p^2) and sum((g+wd*p)^2),
//      block-per-chunk (chunk table precomputed host-side) with atomic
//      accumulation into per-segment slots;
//   2. lars_ratio_kernel  — adaptive LR per segment:
//      trust * ||p|| / (||g_eff|| + eps), 1.0 when skipped/degenerate;
//   3. lars_update_kernel — g_eff = (g + wd*p) * alr[seg];
//      m = momentum*m + g_eff (first step: m = g_eff); p -= lr*m.
#include "common.h"

__global__ void lars_norms_kernel(const float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ norm_acc,
                                  const int64_t* __restrict__ seg_off,
                                  const int64_t* __restrict__ seg_len,
                                  const float* __restrict__ seg_wd,
                                  const int* __restrict__ chunk_seg,
                                  const int64_t* __restrict__ chunk_base,
                                  int64_t chunk, int nchunks) {
  __shared__ float scratch[4];
  const int ci = blockIdx.x;
  if (ci >= nchunks) return;
  const int s = chunk_seg[ci];
  const int64_t off = seg_off[s];
  const int64_t base = chunk_base[ci];
  const int64_t len = seg_len[s];
  const int64_t end = (base + chunk < len) ? base + chunk : len;
  const float wd = seg_wd[s];
  float acc_p = 0.f, acc_g = 0.f;
  for (int64_t i = base + threadIdx.x; i < end; i += blockDim.x) {
    const float pv = p[off + i];
    const float gv = fmaf(wd, pv, g[off + i]);
    acc_p = fmaf(pv, pv, acc_p);
    acc_g = fmaf(gv, gv, acc_g);
  }
  const float tot_p = block_reduce_sum(acc_p, scratch);
  const float tot_g = block_reduce_sum(acc_g, scratch);
  if (threadIdx.x == 0) {
    atomicAdd(&norm_acc[2 * s], tot_p);
    atomicAdd(&norm_acc[2 * s + 1], tot_g);
  }
}

__global__ void lars_ratio_kernel(const float* __restrict__ norm_acc,
                                  const int* __restrict__ seg_adapt,
                                  float* __restrict__ alr,
                                  float trust, float eps, int nseg) {
  for (int s = blockIdx.x * blockDim.x + threadIdx.x; s < nseg;
       s += gridDim.x * blockDim.x) {
    float out = 1.0f;
    if (seg_adapt[s]) {
      const float pn = sqrtf(norm_acc[2 * s]);
      const float gn = sqrtf(norm_acc[2 * s + 1]);
      if (pn > 0.f && gn > 0.f) out = trust * pn / (gn + eps);
    }
    alr[s] = out;
  }
}

__global__ void lars_update_kernel(float* __restrict__ p,
                                   const float* __restrict__ g,
                                   float* __restrict__ m,
                                   const float* __restrict__ alr,
                                   const int64_t* __restrict__ seg_off,
                                   const int64_t* __restrict__ seg_len,
                                   const float* __restrict__ seg_wd,
                                   const int* __restrict__ chunk_seg,
                                   const int64_t* __restrict__ chunk_base,
                                   int64_t chunk, int nchunks,
                                   float lr, const float* __restrict__ lr_dev,
                                   float momentum, int m_init) {
  // lr optionally read from a device scalar so the step is hipGraph-
  // capturable across scheduler changes (wrapper rewrites the scalar)
  if (lr_dev != nullptr) lr = *lr_dev;
  const int ci = blockIdx.x;
  if (ci >= nchunks) return;
  const int s = chunk_seg[ci];
  const int64_t off = seg_off[s];
  const int64_t base = chunk_base[ci];
  const int64_t len = seg_len[s];
  const int64_t end = (base + chunk < len) ? base + chunk : len;
  const float wd = seg_wd[s];
  const float a = alr[s];
  for (int64_t i = base + threadIdx.x; i < end; i += blockDim.x) {
    const int64_t j = off + i;
    const float pv = p[j];
    const float geff = fmaf(wd, pv, g[j]) * a;
    const float mv = m_init ? fmaf(momentum, m[j], geff) : geff;
    m[j] = mv;
    p[j] = fmaf(-lr, mv, pv);
  }
}

void launch_lars_momentum_step(float* p, const float* g, float* m,
                               float* norm_acc, float* alr,
                               const int64_t* seg_off,
                               const int64_t* seg_len,
                               const float* seg_wd, const int* seg_adapt,
                               const int* chunk_seg,
                               const int64_t* chunk_base,
                               int nseg, int nchunks, int64_t chunk,
                               float trust, float eps, float lr,
                               const float* lr_dev,
                               float momentum, int m_init,
                               hipStream_t stream) {
  (void)hipMemsetAsync(norm_acc, 0, sizeof(float) * 2 * nseg, stream);
  hipLaunchKernelGGL(lars_norms_kernel, dim3(nchunks), dim3(256), 0, stream,
                     p, g, norm_acc, seg_off, seg_len, seg_wd, chunk_seg,
                     chunk_base, chunk, nchunks);
  hipLaunchKernelGGL(lars_ratio_kernel, dim3((nseg + 255) / 256), dim3(256),
                     0, stream, norm_acc, seg_adapt, alr, trust, eps, nseg);
  hipLaunchKernelGGL(lars_update_kernel, dim3(nchunks), dim3(256), 0, stream,
                     p, g, m, alr, seg_off, seg_len, seg_wd, chunk_seg,
                     chunk_base, chunk, nchunks, lr, lr_dev, momentum,
                     m_init);
}
