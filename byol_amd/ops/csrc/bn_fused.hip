// Fused BatchNorm(+residual)(+ReLU) for NHWC fp32 on gfx950.
//
// Replaces the reference's per-BN kernel parade (MIOpen SpatialMeanVariance +
// SpatialNorm + ATen ReLU + ATen residual add; backward dScaleDBias + dX +
// ReLU-bwd + add-bwd — SURVEY.md K3/K6) with:
//   fwd:  bn_stats (one pass, per-channel sum/sumsq, atomics)
//         bn_finalize (tiny: mean/invstd + running-stat update)
//         bn_apply (normalize + affine + residual-add + ReLU, one pass)
//   bwd:  bn_bwd_reduce (dy_eff = relu-masked dy; per-channel sum_dy,
//         sum_dy_xhat -> also dbias/dweight)
//         bn_bwd_apply (dx and optional dresidual, one pass)
//
// Layout contract: x is [M rows][C channels] with C contiguous — i.e. torch
// channels_last 4-D tensors (NHWC) or plain-contiguous 2-D [B, C] tensors
// (the projector/predictor BN1d).  C must be a multiple of 4 (float4 path).
//
// SyncBN: the host wrapper all-reduces the packed stats buffer between
// bn_stats and bn_finalize (and between bn_bwd_reduce and bn_bwd_apply),
// giving cross-replica batch statistics with ONE small RCCL message per
// layer.
#include "common.h"

// how many float4 channel-quads one thread owns at most: supports C<=8192
#define MAX_Q 8

struct Quad {
  float4 s;
  float4 ss;
};

// ---------------------------------------------------------------------------
// stats: acc[0:C] = sum, acc[C:2C] = sum of squares
// ---------------------------------------------------------------------------
template <int KQ>
__global__ void bn_stats_kernel(const float* __restrict__ x,
                                float* __restrict__ acc,
                                int64_t m, int c, int slot_mask) {
  // acc is [nslots][2C]; blocks hash into slots so per-address atomic
  // contention is num_blocks/nslots instead of num_blocks
  float* const acc_slot = acc + (int64_t)(blockIdx.x & slot_mask) * 2 * c;
  const int c4 = c >> 2;
  const int nthread = blockDim.x;
  // RPB row-lanes of CW channel-threads each (CW = min(c4, nthread))
  const int cw = c4 < nthread ? c4 : nthread;
  const int rpb = nthread / cw;           // rows processed in parallel
  const int tc = threadIdx.x % cw;        // channel-thread index
  const int tr = threadIdx.x / cw;        // row lane
  const bool active = tr < rpb;

  // KQ is a compile-time bound so the accumulators stay in VGPRs
  float4 s[KQ], ss[KQ];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    s[k] = make_float4(0.f, 0.f, 0.f, 0.f);
    ss[k] = make_float4(0.f, 0.f, 0.f, 0.f);
  }

  const float4* x4 = reinterpret_cast<const float4*>(x);
  // 4-row unroll: 4 independent 16-B loads in flight per thread (a single
  // outstanding load leaves the kernel latency-bound at ~1.3 TB/s)
  const int64_t row_stride = (int64_t)rpb * 4 * gridDim.x;
  if (active) {
    for (int64_t row = (int64_t)blockIdx.x * rpb * 4 + tr; row < m;
         row += row_stride) {
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int64_t r = row + (int64_t)j * rpb;
        if (r < m) {
          const int64_t base = r * c4;
          #pragma unroll
          for (int k = 0; k < KQ; ++k) {
            const int q = tc + k * cw;
            if (q < c4) {
              const float4 v = x4[base + q];
              s[k].x += v.x; s[k].y += v.y; s[k].z += v.z; s[k].w += v.w;
              ss[k] = make_float4(fmaf(v.x, v.x, ss[k].x),
                                  fmaf(v.y, v.y, ss[k].y),
                                  fmaf(v.z, v.z, ss[k].z),
                                  fmaf(v.w, v.w, ss[k].w));
            }
          }
        }
      }
    }
  }

  // reduce across row lanes (same tc) through LDS, then atomics
  __shared__ Quad scratch[256];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    scratch[threadIdx.x].s = s[k];
    scratch[threadIdx.x].ss = ss[k];
    __syncthreads();
    if (tr == 0) {
      float4 ts = s[k], tss = ss[k];
      for (int r = 1; r < rpb; ++r) {
        const Quad& o = scratch[tc + r * cw];
        ts.x += o.s.x; ts.y += o.s.y; ts.z += o.s.z; ts.w += o.s.w;
        tss.x += o.ss.x; tss.y += o.ss.y; tss.z += o.ss.z; tss.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int ch = q * 4;
        atomicAdd(&acc_slot[ch + 0], ts.x);
        atomicAdd(&acc_slot[ch + 1], ts.y);
        atomicAdd(&acc_slot[ch + 2], ts.z);
        atomicAdd(&acc_slot[ch + 3], ts.w);
        atomicAdd(&acc_slot[c + ch + 0], tss.x);
        atomicAdd(&acc_slot[c + ch + 1], tss.y);
        atomicAdd(&acc_slot[c + ch + 2], tss.z);
        atomicAdd(&acc_slot[c + ch + 3], tss.w);
      }
    }
    __syncthreads();
  }
}

// out[0:2C] = sum over slots of in[slot][0:2C]
__global__ void bn_reduce_slots_kernel(const float* __restrict__ in,
                                       float* __restrict__ out,
                                       int n2c, int nslots) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n2c;
       i += gridDim.x * blockDim.x) {
    float v = 0.f;
    for (int sl = 0; sl < nslots; ++sl) v += in[(int64_t)sl * n2c + i];
    out[i] = v;
  }
}

// mean/invstd from (possibly all-reduced) acc; update running stats
__global__ void bn_finalize_kernel(const float* __restrict__ acc,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float count, float eps, float momentum,
                                   int c, int update_running) {
  for (int ch = blockIdx.x * blockDim.x + threadIdx.x; ch < c;
       ch += gridDim.x * blockDim.x) {
    const float mu = acc[ch] / count;
    float var = acc[c + ch] / count - mu * mu;
    var = var < 0.f ? 0.f : var;
    mean[ch] = mu;
    invstd[ch] = rsqrtf(var + eps);
    if (update_running) {
      const float unbiased = var * (count / fmaxf(count - 1.f, 1.f));
      running_mean[ch] = fmaf(momentum, mu - running_mean[ch],
                              running_mean[ch]);
      running_var[ch] = fmaf(momentum, unbiased - running_var[ch],
                             running_var[ch]);
    }
  }
}

// y = relu?(w*(x-mean)*invstd + b [+ residual])
__global__ void bn_apply_kernel(const float* __restrict__ x,
                                const float* __restrict__ residual,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ weight,
                                const float* __restrict__ bias,
                                float* __restrict__ y,
                                int64_t m, int c, int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* r4 = reinterpret_cast<const float4*>(residual);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* b4 = reinterpret_cast<const float4*>(bias);
  float4* y4 = reinterpret_cast<float4*>(y);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
    const float4 v = x4[i];
    const float4 mu = mean4[q], is = inv4[q], w = w4[q], b = b4[q];
    float4 o;
    o.x = fmaf((v.x - mu.x) * is.x, w.x, b.x);
    o.y = fmaf((v.y - mu.y) * is.y, w.y, b.y);
    o.z = fmaf((v.z - mu.z) * is.z, w.z, b.z);
    o.w = fmaf((v.w - mu.w) * is.w, w.w, b.w);
    if (residual != nullptr) {
      const float4 r = r4[i];
      o.x += r.x; o.y += r.y; o.z += r.z; o.w += r.w;
    }
    if (relu) {
      o.x = fmaxf(o.x, 0.f); o.y = fmaxf(o.y, 0.f);
      o.z = fmaxf(o.z, 0.f); o.w = fmaxf(o.w, 0.f);
    }
    y4[i] = o;
  }
}

// ---------------------------------------------------------------------------
// backward
// red[0:C] = sum(dy_eff), red[C:2C] = sum(dy_eff * xhat)
// dy_eff = relu? (y > 0 ? dy : 0) : dy       (xhat from x, mean, invstd)
// ---------------------------------------------------------------------------
template <int KQ>
__global__ void bn_bwd_reduce_kernel(const float* __restrict__ dy,
                                     const float* __restrict__ y,
                                     const float* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ red,
                                     int64_t m, int c, int relu,
                                     int slot_mask) {
  float* const red_slot = red + (int64_t)(blockIdx.x & slot_mask) * 2 * c;
  const int c4 = c >> 2;
  const int nthread = blockDim.x;
  const int cw = c4 < nthread ? c4 : nthread;
  const int rpb = nthread / cw;
  const int tc = threadIdx.x % cw;
  const int tr = threadIdx.x / cw;
  const bool active = tr < rpb;

  float4 s1[KQ], s2[KQ];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    s1[k] = make_float4(0.f, 0.f, 0.f, 0.f);
    s2[k] = make_float4(0.f, 0.f, 0.f, 0.f);
  }
  const float4* dy4 = reinterpret_cast<const float4*>(dy);
  const float4* y4 = reinterpret_cast<const float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  // 2-row unroll: 6 independent loads in flight (3 streams x 2 rows)
  const int64_t row_stride = (int64_t)rpb * 2 * gridDim.x;
  if (active) {
    for (int64_t row = (int64_t)blockIdx.x * rpb * 2 + tr; row < m;
         row += row_stride) {
      #pragma unroll
      for (int j = 0; j < 2; ++j) {
      const int64_t r = row + (int64_t)j * rpb;
      if (r >= m) continue;
      const int64_t base = r * c4;
      #pragma unroll
      for (int k = 0; k < KQ; ++k) {
        const int q = tc + k * cw;
        if (q < c4) {
          float4 g = dy4[base + q];
          if (relu) {
            const float4 yy = y4[base + q];
            g.x = yy.x > 0.f ? g.x : 0.f;
            g.y = yy.y > 0.f ? g.y : 0.f;
            g.z = yy.z > 0.f ? g.z : 0.f;
            g.w = yy.w > 0.f ? g.w : 0.f;
          }
          const float4 v = x4[base + q];
          const float4 mu = mean4[q], is = inv4[q];
          s1[k].x += g.x; s1[k].y += g.y; s1[k].z += g.z; s1[k].w += g.w;
          s2[k].x = fmaf(g.x, (v.x - mu.x) * is.x, s2[k].x);
          s2[k].y = fmaf(g.y, (v.y - mu.y) * is.y, s2[k].y);
          s2[k].z = fmaf(g.z, (v.z - mu.z) * is.z, s2[k].z);
          s2[k].w = fmaf(g.w, (v.w - mu.w) * is.w, s2[k].w);
        }
      }
      }
    }
  }

  __shared__ Quad scratch[256];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    scratch[threadIdx.x].s = s1[k];
    scratch[threadIdx.x].ss = s2[k];
    __syncthreads();
    if (tr == 0) {
      float4 t1 = s1[k], t2 = s2[k];
      for (int r = 1; r < rpb; ++r) {
        const Quad& o = scratch[tc + r * cw];
        t1.x += o.s.x; t1.y += o.s.y; t1.z += o.s.z; t1.w += o.s.w;
        t2.x += o.ss.x; t2.y += o.ss.y; t2.z += o.ss.z; t2.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int ch = q * 4;
        atomicAdd(&red_slot[ch + 0], t1.x);
        atomicAdd(&red_slot[ch + 1], t1.y);
        atomicAdd(&red_slot[ch + 2], t1.z);
        atomicAdd(&red_slot[ch + 3], t1.w);
        atomicAdd(&red_slot[c + ch + 0], t2.x);
        atomicAdd(&red_slot[c + ch + 1], t2.y);
        atomicAdd(&red_slot[c + ch + 2], t2.z);
        atomicAdd(&red_slot[c + ch + 3], t2.w);
      }
    }
    __syncthreads();
  }
}

// dx = (w*invstd) * (dy_eff - red1/M - xhat * red2/M); dresidual = dy_eff
__global__ void bn_bwd_apply_kernel(const float* __restrict__ dy,
                                    const float* __restrict__ y,
                                    const float* __restrict__ x,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ weight,
                                    const float* __restrict__ red,
                                    float* __restrict__ dx,
                                    float* __restrict__ dresidual,
                                    float inv_count, int64_t m, int c,
                                    int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* dy4 = reinterpret_cast<const float4*>(dy);
  const float4* y4 = reinterpret_cast<const float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* r1 = reinterpret_cast<const float4*>(red);
  const float4* r2 = reinterpret_cast<const float4*>(red + c);
  float4* dx4 = reinterpret_cast<float4*>(dx);
  float4* dr4 = reinterpret_cast<float4*>(dresidual);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
    float4 g = dy4[i];
    if (relu) {
      const float4 yy = y4[i];
      g.x = yy.x > 0.f ? g.x : 0.f;
      g.y = yy.y > 0.f ? g.y : 0.f;
      g.z = yy.z > 0.f ? g.z : 0.f;
      g.w = yy.w > 0.f ? g.w : 0.f;
    }
    if (dresidual != nullptr) dr4[i] = g;
    const float4 v = x4[i];
    const float4 mu = mean4[q], is = inv4[q], w = w4[q];
    const float4 m1 = r1[q], m2 = r2[q];
    float4 o;
    o.x = w.x * is.x * (g.x - m1.x * inv_count
                        - (v.x - mu.x) * is.x * m2.x * inv_count);
    o.y = w.y * is.y * (g.y - m1.y * inv_count
                        - (v.y - mu.y) * is.y * m2.y * inv_count);
    o.z = w.z * is.z * (g.z - m1.z * inv_count
                        - (v.z - mu.z) * is.z * m2.z * inv_count);
    o.w = w.w * is.w * (g.w - m1.w * inv_count
                        - (v.w - mu.w) * is.w * m2.w * inv_count);
    dx4[i] = o;
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
static int stats_grid(int64_t m, int c) {
  // enough blocks to fill 256 CUs x 8 XCDs comfortably, capped
  const int c4 = c >> 2;
  const int cw = c4 < 256 ? c4 : 256;
  const int rpb = 256 / cw;
  int64_t g = (m + rpb - 1) / rpb;
  if (g > 4096) g = 4096;
  if (g < 1) g = 1;
  return (int)g;
}

static int kq_for(int c) {
  const int c4 = c >> 2;
  const int cw = c4 < 256 ? c4 : 256;
  const int kq = (c4 + cw - 1) / cw;
  if (kq <= 1) return 1;
  if (kq <= 2) return 2;
  if (kq <= 4) return 4;
  return 8;
}

// (defined at the end of this file)
void launch_bn_stats_v2(const float* x, float* acc, int64_t m, int c,
                        int slot_mask, int grid, hipStream_t stream);
void launch_bn_bwd_reduce_v2(const float* dy, const float* y, const float* x,
                             const float* mean, const float* invstd,
                             float* red, int64_t m, int c, int relu,
                             int slot_mask, int grid, hipStream_t stream);

// v2 (8-row unroll + shfl tail) won the sweep on every shape (r2 call 4:
// C=2048 stats 0.240 -> 0.082 ms, C=1024 1.8x, +8% small C) — the default
// launchers use it; grid from the measured knee (1024, 2048 for stem-size
// inputs).  BYOL_BN_V1=1 restores the v1 kernels for A/Bs.
static bool bn_v1_forced() {
  static bool f = [] {
    const char* v = getenv("BYOL_BN_V1");
    return v != nullptr && v[0] == '1';
  }();
  return f;
}

static int stats_grid_v2(int64_t m) { return m > 4000000 ? 2048 : 1024; }

void launch_bn_stats(const float* x, float* acc, int64_t m, int c,
                     int slot_mask, hipStream_t stream) {
  if (!bn_v1_forced()) {
    launch_bn_stats_v2(x, acc, m, c, slot_mask, stats_grid_v2(m), stream);
    return;
  }
  const dim3 g(stats_grid(m, c)), b(256);
  switch (kq_for(c)) {
    case 1: hipLaunchKernelGGL(bn_stats_kernel<1>, g, b, 0, stream, x, acc,
                               m, c, slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_stats_kernel<2>, g, b, 0, stream, x, acc,
                               m, c, slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_stats_kernel<4>, g, b, 0, stream, x, acc,
                               m, c, slot_mask); break;
    default: hipLaunchKernelGGL(bn_stats_kernel<8>, g, b, 0, stream, x, acc,
                                m, c, slot_mask); break;
  }
}

void launch_bn_reduce_slots(const float* in, float* out, int n2c, int nslots,
                            hipStream_t stream) {
  hipLaunchKernelGGL(bn_reduce_slots_kernel, dim3((n2c + 255) / 256),
                     dim3(256), 0, stream, in, out, n2c, nslots);
}

void launch_bn_finalize(const float* acc, float* mean, float* invstd,
                        float* running_mean, float* running_var, float count,
                        float eps, float momentum, int c, int update_running,
                        hipStream_t stream) {
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((c + 255) / 256), dim3(256), 0,
                     stream, acc, mean, invstd, running_mean, running_var,
                     count, eps, momentum, c, update_running);
}

void launch_bn_apply(const float* x, const float* residual, const float* mean,
                     const float* invstd, const float* weight,
                     const float* bias, float* y, int64_t m, int c, int relu,
                     hipStream_t stream) {
  const int64_t n4 = m * (c >> 2);
  hipLaunchKernelGGL(bn_apply_kernel, dim3(grid_1d(n4, 256)), dim3(256), 0,
                     stream, x, residual, mean, invstd, weight, bias, y, m, c,
                     relu);
}

void launch_bn_bwd_reduce(const float* dy, const float* y, const float* x,
                          const float* mean, const float* invstd, float* red,
                          int64_t m, int c, int relu, int slot_mask,
                          hipStream_t stream) {
  if (!bn_v1_forced()) {
    launch_bn_bwd_reduce_v2(dy, y, x, mean, invstd, red, m, c, relu,
                            slot_mask, stats_grid_v2(m), stream);
    return;
  }
  const dim3 g(stats_grid(m, c)), b(256);
  switch (kq_for(c)) {
    case 1: hipLaunchKernelGGL(bn_bwd_reduce_kernel<1>, g, b, 0, stream, dy,
                               y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_bwd_reduce_kernel<2>, g, b, 0, stream, dy,
                               y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_bwd_reduce_kernel<4>, g, b, 0, stream, dy,
                               y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    default: hipLaunchKernelGGL(bn_bwd_reduce_kernel<8>, g, b, 0, stream, dy,
                                y, x, mean, invstd, red, m, c, relu,
                                slot_mask); break;
  }
}

void launch_bn_bwd_apply(const float* dy, const float* y, const float* x,
                         const float* mean, const float* invstd,
                         const float* weight, const float* red, float* dx,
                         float* dresidual, float inv_count, int64_t m, int c,
                         int relu, hipStream_t stream) {
  const int64_t n4 = m * (c >> 2);
  hipLaunchKernelGGL(bn_bwd_apply_kernel, dim3(grid_1d(n4, 256)), dim3(256),
                     0, stream, dy, y, x, mean, invstd, weight, red, dx,
                     dresidual, inv_count, m, c, relu);
}

// ---------------------------------------------------------------------------
// stats v2: 8-row unroll + in-wave shuffle tail for small channel widths.
// Experiment vehicle for the round-2 BN bandwidth track (v1 measured
// ~3.5 TB/s vs bn_apply's 4.9 — suspicion: not enough loads in flight and
// a serial per-block reduce tail at small C).  Grid is a launcher argument
// so the microbench can sweep it.
// ---------------------------------------------------------------------------
template <int KQ>
__global__ void bn_stats_v2_kernel(const float* __restrict__ x,
                                   float* __restrict__ acc,
                                   int64_t m, int c, int slot_mask) {
  float* const acc_slot = acc + (int64_t)(blockIdx.x & slot_mask) * 2 * c;
  const int c4 = c >> 2;
  const int nthread = blockDim.x;
  const int cw = c4 < nthread ? c4 : nthread;
  const int rpb = nthread / cw;
  const int tc = threadIdx.x % cw;
  const int tr = threadIdx.x / cw;
  const bool active = tr < rpb;

  float4 s[KQ], ss[KQ];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    s[k] = make_float4(0.f, 0.f, 0.f, 0.f);
    ss[k] = make_float4(0.f, 0.f, 0.f, 0.f);
  }

  const float4* x4 = reinterpret_cast<const float4*>(x);
  // 8-row unroll: 8 independent 16-B loads in flight per thread
  const int64_t row_stride = (int64_t)rpb * 8 * gridDim.x;
  if (active) {
    for (int64_t row = (int64_t)blockIdx.x * rpb * 8 + tr; row < m;
         row += row_stride) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int64_t r = row + (int64_t)j * rpb;
        if (r < m) {
          const int64_t base = r * c4;
          #pragma unroll
          for (int k = 0; k < KQ; ++k) {
            const int q = tc + k * cw;
            if (q < c4) {
              const float4 v = x4[base + q];
              s[k].x += v.x; s[k].y += v.y; s[k].z += v.z; s[k].w += v.w;
              ss[k] = make_float4(fmaf(v.x, v.x, ss[k].x),
                                  fmaf(v.y, v.y, ss[k].y),
                                  fmaf(v.z, v.z, ss[k].z),
                                  fmaf(v.w, v.w, ss[k].w));
            }
          }
        }
      }
    }
  }

  // tail: reduce across row lanes.  For cw <= 32 (rows share a wave) fold
  // in-wave first with shfl_down — the v1 serial loop is up to 16 scalar
  // float4 adds on 1/16 of the threads.
  __shared__ Quad scratch[256];
  // shfl tree needs power-of-two cw; otherwise degrade to the v1 serial
  // tail (rows_in_wave = 1 disables the in-wave fold)
  const bool cw_pow2 = (cw & (cw - 1)) == 0;
  const int rows_in_wave =
      (cw_pow2 && cw < WAVE_SIZE) ? (WAVE_SIZE / cw) : 1;
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    float4 ts = s[k], tss = ss[k];
    for (int off = cw * (rows_in_wave >> 1); off >= cw && off > 0;
         off >>= 1) {
      ts.x += __shfl_down(ts.x, off, WAVE_SIZE);
      ts.y += __shfl_down(ts.y, off, WAVE_SIZE);
      ts.z += __shfl_down(ts.z, off, WAVE_SIZE);
      ts.w += __shfl_down(ts.w, off, WAVE_SIZE);
      tss.x += __shfl_down(tss.x, off, WAVE_SIZE);
      tss.y += __shfl_down(tss.y, off, WAVE_SIZE);
      tss.z += __shfl_down(tss.z, off, WAVE_SIZE);
      tss.w += __shfl_down(tss.w, off, WAVE_SIZE);
    }
    // after the in-wave fold, row-lane tr % rows_in_wave == 0 holds the
    // wave-partial; cross-wave partials go through LDS
    scratch[threadIdx.x].s = ts;
    scratch[threadIdx.x].ss = tss;
    __syncthreads();
    if (tr == 0) {
      float4 fs = ts, fss = tss;
      for (int r = rows_in_wave; r < rpb; r += rows_in_wave) {
        const Quad& o = scratch[tc + r * cw];
        fs.x += o.s.x; fs.y += o.s.y; fs.z += o.s.z; fs.w += o.s.w;
        fss.x += o.ss.x; fss.y += o.ss.y; fss.z += o.ss.z;
        fss.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int ch = q * 4;
        atomicAdd(&acc_slot[ch + 0], fs.x);
        atomicAdd(&acc_slot[ch + 1], fs.y);
        atomicAdd(&acc_slot[ch + 2], fs.z);
        atomicAdd(&acc_slot[ch + 3], fs.w);
        atomicAdd(&acc_slot[c + ch + 0], fss.x);
        atomicAdd(&acc_slot[c + ch + 1], fss.y);
        atomicAdd(&acc_slot[c + ch + 2], fss.z);
        atomicAdd(&acc_slot[c + ch + 3], fss.w);
      }
    }
    __syncthreads();
  }
}

void launch_bn_stats_v2(const float* x, float* acc, int64_t m, int c,
                        int slot_mask, int grid, hipStream_t stream) {
  const dim3 g(grid), b(256);
  switch (kq_for(c)) {
    case 1: hipLaunchKernelGGL(bn_stats_v2_kernel<1>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_stats_v2_kernel<2>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_stats_v2_kernel<4>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    default: hipLaunchKernelGGL(bn_stats_v2_kernel<8>, g, b, 0, stream, x,
                                acc, m, c, slot_mask); break;
  }
}

// bwd-reduce v2: 4-row unroll (12 independent loads in flight across the
// dy/y/x streams) + the same shfl tail as stats v2.
template <int KQ>
__global__ void bn_bwd_reduce_v2_kernel(const float* __restrict__ dy,
                                        const float* __restrict__ y,
                                        const float* __restrict__ x,
                                        const float* __restrict__ mean,
                                        const float* __restrict__ invstd,
                                        float* __restrict__ red,
                                        int64_t m, int c, int relu,
                                        int slot_mask) {
  float* const red_slot = red + (int64_t)(blockIdx.x & slot_mask) * 2 * c;
  const int c4 = c >> 2;
  const int nthread = blockDim.x;
  const int cw = c4 < nthread ? c4 : nthread;
  const int rpb = nthread / cw;
  const int tc = threadIdx.x % cw;
  const int tr = threadIdx.x / cw;
  const bool active = tr < rpb;

  float4 s1[KQ], s2[KQ];
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    s1[k] = make_float4(0.f, 0.f, 0.f, 0.f);
    s2[k] = make_float4(0.f, 0.f, 0.f, 0.f);
  }
  const float4* dy4 = reinterpret_cast<const float4*>(dy);
  const float4* y4 = reinterpret_cast<const float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const int64_t row_stride = (int64_t)rpb * 4 * gridDim.x;
  if (active) {
    for (int64_t row = (int64_t)blockIdx.x * rpb * 4 + tr; row < m;
         row += row_stride) {
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int64_t r = row + (int64_t)j * rpb;
        if (r >= m) continue;
        const int64_t base = r * c4;
        #pragma unroll
        for (int k = 0; k < KQ; ++k) {
          const int q = tc + k * cw;
          if (q < c4) {
            float4 g = dy4[base + q];
            if (relu) {
              const float4 yy = y4[base + q];
              g.x = yy.x > 0.f ? g.x : 0.f;
              g.y = yy.y > 0.f ? g.y : 0.f;
              g.z = yy.z > 0.f ? g.z : 0.f;
              g.w = yy.w > 0.f ? g.w : 0.f;
            }
            const float4 v = x4[base + q];
            const float4 mu = mean4[q], is = inv4[q];
            s1[k].x += g.x; s1[k].y += g.y;
            s1[k].z += g.z; s1[k].w += g.w;
            s2[k].x = fmaf(g.x, (v.x - mu.x) * is.x, s2[k].x);
            s2[k].y = fmaf(g.y, (v.y - mu.y) * is.y, s2[k].y);
            s2[k].z = fmaf(g.z, (v.z - mu.z) * is.z, s2[k].z);
            s2[k].w = fmaf(g.w, (v.w - mu.w) * is.w, s2[k].w);
          }
        }
      }
    }
  }

  __shared__ Quad scratch[256];
  const bool cw_pow2 = (cw & (cw - 1)) == 0;
  const int rows_in_wave =
      (cw_pow2 && cw < WAVE_SIZE) ? (WAVE_SIZE / cw) : 1;
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    float4 t1 = s1[k], t2 = s2[k];
    for (int off = cw * (rows_in_wave >> 1); off >= cw && off > 0;
         off >>= 1) {
      t1.x += __shfl_down(t1.x, off, WAVE_SIZE);
      t1.y += __shfl_down(t1.y, off, WAVE_SIZE);
      t1.z += __shfl_down(t1.z, off, WAVE_SIZE);
      t1.w += __shfl_down(t1.w, off, WAVE_SIZE);
      t2.x += __shfl_down(t2.x, off, WAVE_SIZE);
      t2.y += __shfl_down(t2.y, off, WAVE_SIZE);
      t2.z += __shfl_down(t2.z, off, WAVE_SIZE);
      t2.w += __shfl_down(t2.w, off, WAVE_SIZE);
    }
    scratch[threadIdx.x].s = t1;
    scratch[threadIdx.x].ss = t2;
    __syncthreads();
    if (tr == 0) {
      float4 f1 = t1, f2 = t2;
      for (int r = rows_in_wave; r < rpb; r += rows_in_wave) {
        const Quad& o = scratch[tc + r * cw];
        f1.x += o.s.x; f1.y += o.s.y; f1.z += o.s.z; f1.w += o.s.w;
        f2.x += o.ss.x; f2.y += o.ss.y; f2.z += o.ss.z; f2.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int ch = q * 4;
        atomicAdd(&red_slot[ch + 0], f1.x);
        atomicAdd(&red_slot[ch + 1], f1.y);
        atomicAdd(&red_slot[ch + 2], f1.z);
        atomicAdd(&red_slot[ch + 3], f1.w);
        atomicAdd(&red_slot[c + ch + 0], f2.x);
        atomicAdd(&red_slot[c + ch + 1], f2.y);
        atomicAdd(&red_slot[c + ch + 2], f2.z);
        atomicAdd(&red_slot[c + ch + 3], f2.w);
      }
    }
    __syncthreads();
  }
}

void launch_bn_bwd_reduce_v2(const float* dy, const float* y, const float* x,
                             const float* mean, const float* invstd,
                             float* red, int64_t m, int c, int relu,
                             int slot_mask, int grid, hipStream_t stream) {
  const dim3 g(grid), b(256);
  switch (kq_for(c)) {
    case 1: hipLaunchKernelGGL(bn_bwd_reduce_v2_kernel<1>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_bwd_reduce_v2_kernel<2>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_bwd_reduce_v2_kernel<4>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    default: hipLaunchKernelGGL(bn_bwd_reduce_v2_kernel<8>, g, b, 0, stream,
                                dy, y, x, mean, invstd, red, m, c, relu,
                                slot_mask); break;
  }
}

// ---------------------------------------------------------------------------
// apply v2: 4 consecutive float4 elements per thread per iteration (4
// independent loads in flight; the v1 grid-stride-1 loop holds one).
// Channel-quad params resolve per element via the pow2 mask (all shipped
// C are pow2).
// ---------------------------------------------------------------------------
__global__ void bn_apply_v2_kernel(const float* __restrict__ x,
                                   const float* __restrict__ residual,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd,
                                   const float* __restrict__ weight,
                                   const float* __restrict__ bias,
                                   float* __restrict__ y,
                                   int64_t m, int c, int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* r4 = reinterpret_cast<const float4*>(residual);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* b4 = reinterpret_cast<const float4*>(bias);
  float4* y4 = reinterpret_cast<float4*>(y);
  const int64_t stride4 = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n4; i0 += stride4) {
    float4 v[4], r[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      if (i0 + j < n4) v[j] = x4[i0 + j];
    if (residual != nullptr) {
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        if (i0 + j < n4) r[j] = r4[i0 + j];
    }
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int64_t i = i0 + j;
      if (i >= n4) break;
      const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
      const float4 mu = mean4[q], is = inv4[q], w = w4[q], b = b4[q];
      float4 o;
      o.x = fmaf((v[j].x - mu.x) * is.x, w.x, b.x);
      o.y = fmaf((v[j].y - mu.y) * is.y, w.y, b.y);
      o.z = fmaf((v[j].z - mu.z) * is.z, w.z, b.z);
      o.w = fmaf((v[j].w - mu.w) * is.w, w.w, b.w);
      if (residual != nullptr) {
        o.x += r[j].x; o.y += r[j].y; o.z += r[j].z; o.w += r[j].w;
      }
      if (relu) {
        o.x = fmaxf(o.x, 0.f); o.y = fmaxf(o.y, 0.f);
        o.z = fmaxf(o.z, 0.f); o.w = fmaxf(o.w, 0.f);
      }
      y4[i] = o;
    }
  }
}

__global__ void bn_bwd_apply_v2_kernel(const float* __restrict__ dy,
                                       const float* __restrict__ y,
                                       const float* __restrict__ x,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ invstd,
                                       const float* __restrict__ weight,
                                       const float* __restrict__ red,
                                       float* __restrict__ dx,
                                       float* __restrict__ dresidual,
                                       float inv_count, int64_t m, int c,
                                       int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* dy4 = reinterpret_cast<const float4*>(dy);
  const float4* y4 = reinterpret_cast<const float4*>(y);
  const float4* x4 = reinterpret_cast<const float4*>(x);
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* r1 = reinterpret_cast<const float4*>(red);
  const float4* r2 = reinterpret_cast<const float4*>(red + c);
  float4* dx4 = reinterpret_cast<float4*>(dx);
  float4* dr4 = reinterpret_cast<float4*>(dresidual);
  const int64_t stride4 = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n4; i0 += stride4) {
    float4 g[4], yy[4], v[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      if (i0 + j < n4) {
        g[j] = dy4[i0 + j];
        v[j] = x4[i0 + j];
        if (relu) yy[j] = y4[i0 + j];
      }
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int64_t i = i0 + j;
      if (i >= n4) break;
      if (relu) {
        g[j].x = yy[j].x > 0.f ? g[j].x : 0.f;
        g[j].y = yy[j].y > 0.f ? g[j].y : 0.f;
        g[j].z = yy[j].z > 0.f ? g[j].z : 0.f;
        g[j].w = yy[j].w > 0.f ? g[j].w : 0.f;
      }
      if (dresidual != nullptr) dr4[i] = g[j];
      const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
      const float4 mu = mean4[q], is = inv4[q], w = w4[q];
      const float4 m1 = r1[q], m2 = r2[q];
      float4 o;
      o.x = w.x * is.x * (g[j].x - m1.x * inv_count
                          - (v[j].x - mu.x) * is.x * m2.x * inv_count);
      o.y = w.y * is.y * (g[j].y - m1.y * inv_count
                          - (v[j].y - mu.y) * is.y * m2.y * inv_count);
      o.z = w.z * is.z * (g[j].z - m1.z * inv_count
                          - (v[j].z - mu.z) * is.z * m2.z * inv_count);
      o.w = w.w * is.w * (g[j].w - m1.w * inv_count
                          - (v[j].w - mu.w) * is.w * m2.w * inv_count);
      dx4[i] = o;
    }
  }
}

void launch_bn_apply_v2(const float* x, const float* residual,
                        const float* mean, const float* invstd,
                        const float* weight, const float* bias, float* y,
                        int64_t m, int c, int relu, int grid,
                        hipStream_t stream) {
  hipLaunchKernelGGL(bn_apply_v2_kernel, dim3(grid), dim3(256), 0, stream,
                     x, residual, mean, invstd, weight, bias, y, m, c,
                     relu);
}

void launch_bn_bwd_apply_v2(const float* dy, const float* y, const float* x,
                            const float* mean, const float* invstd,
                            const float* weight, const float* red,
                            float* dx, float* dresidual, float inv_count,
                            int64_t m, int c, int relu, int grid,
                            hipStream_t stream) {
  hipLaunchKernelGGL(bn_bwd_apply_v2_kernel, dim3(grid), dim3(256), 0,
                     stream, dy, y, x, mean, invstd, weight, red, dx,
                     dresidual, inv_count, m, c, relu);
}
