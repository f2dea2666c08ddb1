// bf16-I/O variants of the fused BatchNorm kernels (default on the
// --half path since r2: 2805 img/s vs 2323 composed at bs=1024).
//
// The --half (bf16 autocast) path currently falls back to composed fp32
// ATen BN, which re-upcasts every activation: at bs=1024 that fp32 BN
// traffic dominates the bf16 step.  These kernels read/write bf16
// activations (HALF the bytes) while keeping all statistics, parameters
// and gradients in fp32 — the same accuracy contract as autocast BN except
// the OUTPUT is bf16 (what the next bf16 conv consumes anyway).
//
// Structure mirrors the validated fp32 kernels in bn_fused.hip (slotted
// atomics, compile-time KQ unroll, 4-row MLP unroll); loads are ushort4
// (4 bf16 channels = 8 B/lane) so all quad indexing stays identical.
// fp32 helpers bn_finalize / bn_reduce_slots are shared.
#include "common.h"
#include <hip/hip_bf16.h>

#define MAXQ_BF 8

struct QuadBF {
  float4 s;
  float4 ss;
};

__device__ __forceinline__ float4 load_bf4(const ushort* p) {
  const ushort4 u = *reinterpret_cast<const ushort4*>(p);
  float4 v;
  v.x = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&u.x));
  v.y = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&u.y));
  v.z = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&u.z));
  v.w = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&u.w));
  return v;
}

__device__ __forceinline__ void store_bf4(ushort* p, float4 v) {
  ushort4 u;
  __hip_bfloat16 t;
  t = __float2bfloat16(v.x); u.x = *reinterpret_cast<ushort*>(&t);
  t = __float2bfloat16(v.y); u.y = *reinterpret_cast<ushort*>(&t);
  t = __float2bfloat16(v.z); u.z = *reinterpret_cast<ushort*>(&t);
  t = __float2bfloat16(v.w); u.w = *reinterpret_cast<ushort*>(&t);
  *reinterpret_cast<ushort4*>(p) = u;
}

// ---------------------------------------------------------------------------
template <int KQ>
__global__ void bn_stats_bf16_kernel(const ushort* __restrict__ x,
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

  // 8-row unroll + shfl tail: the structure the fp32 v2 sweep won with
  // (r2 call 4) ported to bf16 I/O
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
              const float4 v = load_bf4(x + (base + q) * 4);
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

  __shared__ QuadBF scratch[256];
  const bool cw_pow2 = (cw & (cw - 1)) == 0;
  const int riw = (cw_pow2 && cw < 64) ? (64 / cw) : 1;
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    float4 ts = s[k], tss = ss[k];
    for (int off = cw * (riw >> 1); off >= cw && off > 0; off >>= 1) {
      ts.x += __shfl_down(ts.x, off, 64);
      ts.y += __shfl_down(ts.y, off, 64);
      ts.z += __shfl_down(ts.z, off, 64);
      ts.w += __shfl_down(ts.w, off, 64);
      tss.x += __shfl_down(tss.x, off, 64);
      tss.y += __shfl_down(tss.y, off, 64);
      tss.z += __shfl_down(tss.z, off, 64);
      tss.w += __shfl_down(tss.w, off, 64);
    }
    scratch[threadIdx.x].s = ts;
    scratch[threadIdx.x].ss = tss;
    __syncthreads();
    if (tr == 0) {
      for (int r = riw; r < rpb; r += riw) {
        const QuadBF& o = scratch[tc + r * cw];
        ts.x += o.s.x; ts.y += o.s.y; ts.z += o.s.z; ts.w += o.s.w;
        tss.x += o.ss.x; tss.y += o.ss.y; tss.z += o.ss.z; tss.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int chn = q * 4;
        atomicAdd(&acc_slot[chn + 0], ts.x);
        atomicAdd(&acc_slot[chn + 1], ts.y);
        atomicAdd(&acc_slot[chn + 2], ts.z);
        atomicAdd(&acc_slot[chn + 3], ts.w);
        atomicAdd(&acc_slot[c + chn + 0], tss.x);
        atomicAdd(&acc_slot[c + chn + 1], tss.y);
        atomicAdd(&acc_slot[c + chn + 2], tss.z);
        atomicAdd(&acc_slot[c + chn + 3], tss.w);
      }
    }
    __syncthreads();
  }
}

__global__ void bn_apply_bf16_kernel(const ushort* __restrict__ x,
                                     const ushort* __restrict__ residual,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     const float* __restrict__ weight,
                                     const float* __restrict__ bias,
                                     ushort* __restrict__ y,
                                     int64_t m, int c, int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* b4 = reinterpret_cast<const float4*>(bias);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
    const float4 v = load_bf4(x + i * 4);
    const float4 mu = mean4[q], is = inv4[q], w = w4[q], b = b4[q];
    float4 o;
    o.x = fmaf((v.x - mu.x) * is.x, w.x, b.x);
    o.y = fmaf((v.y - mu.y) * is.y, w.y, b.y);
    o.z = fmaf((v.z - mu.z) * is.z, w.z, b.z);
    o.w = fmaf((v.w - mu.w) * is.w, w.w, b.w);
    if (residual != nullptr) {
      const float4 r = load_bf4(residual + i * 4);
      o.x += r.x; o.y += r.y; o.z += r.z; o.w += r.w;
    }
    if (relu) {
      o.x = fmaxf(o.x, 0.f); o.y = fmaxf(o.y, 0.f);
      o.z = fmaxf(o.z, 0.f); o.w = fmaxf(o.w, 0.f);
    }
    store_bf4(y + i * 4, o);
  }
}

template <int KQ>
__global__ void bn_bwd_reduce_bf16_kernel(const ushort* __restrict__ dy,
                                          const ushort* __restrict__ y,
                                          const ushort* __restrict__ x,
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
            float4 g = load_bf4(dy + (base + q) * 4);
            if (relu) {
              const float4 yy = load_bf4(y + (base + q) * 4);
              g.x = yy.x > 0.f ? g.x : 0.f;
              g.y = yy.y > 0.f ? g.y : 0.f;
              g.z = yy.z > 0.f ? g.z : 0.f;
              g.w = yy.w > 0.f ? g.w : 0.f;
            }
            const float4 v = load_bf4(x + (base + q) * 4);
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

  __shared__ QuadBF scratch[256];
  const bool cw_pow2 = (cw & (cw - 1)) == 0;
  const int riw = (cw_pow2 && cw < 64) ? (64 / cw) : 1;
  #pragma unroll
  for (int k = 0; k < KQ; ++k) {
    float4 t1 = s1[k], t2 = s2[k];
    for (int off = cw * (riw >> 1); off >= cw && off > 0; off >>= 1) {
      t1.x += __shfl_down(t1.x, off, 64);
      t1.y += __shfl_down(t1.y, off, 64);
      t1.z += __shfl_down(t1.z, off, 64);
      t1.w += __shfl_down(t1.w, off, 64);
      t2.x += __shfl_down(t2.x, off, 64);
      t2.y += __shfl_down(t2.y, off, 64);
      t2.z += __shfl_down(t2.z, off, 64);
      t2.w += __shfl_down(t2.w, off, 64);
    }
    scratch[threadIdx.x].s = t1;
    scratch[threadIdx.x].ss = t2;
    __syncthreads();
    if (tr == 0) {
      for (int r = riw; r < rpb; r += riw) {
        const QuadBF& o = scratch[tc + r * cw];
        t1.x += o.s.x; t1.y += o.s.y; t1.z += o.s.z; t1.w += o.s.w;
        t2.x += o.ss.x; t2.y += o.ss.y; t2.z += o.ss.z; t2.w += o.ss.w;
      }
      const int q = tc + k * cw;
      if (q < c4) {
        const int chn = q * 4;
        atomicAdd(&red_slot[chn + 0], t1.x);
        atomicAdd(&red_slot[chn + 1], t1.y);
        atomicAdd(&red_slot[chn + 2], t1.z);
        atomicAdd(&red_slot[chn + 3], t1.w);
        atomicAdd(&red_slot[c + chn + 0], t2.x);
        atomicAdd(&red_slot[c + chn + 1], t2.y);
        atomicAdd(&red_slot[c + chn + 2], t2.z);
        atomicAdd(&red_slot[c + chn + 3], t2.w);
      }
    }
    __syncthreads();
  }
}

__global__ void bn_bwd_apply_bf16_kernel(const ushort* __restrict__ dy,
                                         const ushort* __restrict__ y,
                                         const ushort* __restrict__ x,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ invstd,
                                         const float* __restrict__ weight,
                                         const float* __restrict__ red,
                                         ushort* __restrict__ dx,
                                         ushort* __restrict__ dresidual,
                                         float inv_count, int64_t m, int c,
                                         int relu) {
  const int c4 = c >> 2;
  const int q_mask = ((c4 & (c4 - 1)) == 0) ? (c4 - 1) : -1;
  const int64_t n4 = m * c4;
  const float4* mean4 = reinterpret_cast<const float4*>(mean);
  const float4* inv4 = reinterpret_cast<const float4*>(invstd);
  const float4* w4 = reinterpret_cast<const float4*>(weight);
  const float4* r1 = reinterpret_cast<const float4*>(red);
  const float4* r2 = reinterpret_cast<const float4*>(red + c);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    const int q = q_mask >= 0 ? (int)(i & q_mask) : (int)(i % c4);
    float4 g = load_bf4(dy + i * 4);
    if (relu) {
      const float4 yy = load_bf4(y + i * 4);
      g.x = yy.x > 0.f ? g.x : 0.f;
      g.y = yy.y > 0.f ? g.y : 0.f;
      g.z = yy.z > 0.f ? g.z : 0.f;
      g.w = yy.w > 0.f ? g.w : 0.f;
    }
    if (dresidual != nullptr) store_bf4(dresidual + i * 4, g);
    const float4 v = load_bf4(x + i * 4);
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
    store_bf4(dx + i * 4, o);
  }
}

// ---------------------------------------------------------------------------
__attribute__((unused)) static int stats_grid_bf(int64_t m, int c) {
  const int c4 = c >> 2;
  const int cw = c4 < 256 ? c4 : 256;
  const int rpb = 256 / cw;
  int64_t g = (m + rpb - 1) / rpb;
  if (g > 4096) g = 4096;
  if (g < 1) g = 1;
  return (int)g;
}

static int kq_for_bf(int c) {
  const int c4 = c >> 2;
  const int cw = c4 < 256 ? c4 : 256;
  const int kq = (c4 + cw - 1) / cw;
  if (kq <= 1) return 1;
  if (kq <= 2) return 2;
  if (kq <= 4) return 4;
  return 8;
}

void launch_bn_stats_bf16(const ushort* x, float* acc, int64_t m, int c,
                          int slot_mask, hipStream_t stream) {
  // grid knee from the fp32 v2 sweep (r2 call 4)
  const dim3 g(m > 4000000 ? 2048 : 1024), b(256);
  switch (kq_for_bf(c)) {
    case 1: hipLaunchKernelGGL(bn_stats_bf16_kernel<1>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_stats_bf16_kernel<2>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_stats_bf16_kernel<4>, g, b, 0, stream, x,
                               acc, m, c, slot_mask); break;
    default: hipLaunchKernelGGL(bn_stats_bf16_kernel<8>, g, b, 0, stream, x,
                                acc, m, c, slot_mask); break;
  }
}

void launch_bn_apply_bf16(const ushort* x, const ushort* residual,
                          const float* mean, const float* invstd,
                          const float* weight, const float* bias, ushort* y,
                          int64_t m, int c, int relu, hipStream_t stream) {
  const int64_t n4 = m * (c >> 2);
  hipLaunchKernelGGL(bn_apply_bf16_kernel, dim3(grid_1d(n4, 256)), dim3(256),
                     0, stream, x, residual, mean, invstd, weight, bias, y,
                     m, c, relu);
}

void launch_bn_bwd_reduce_bf16(const ushort* dy, const ushort* y,
                               const ushort* x, const float* mean,
                               const float* invstd, float* red, int64_t m,
                               int c, int relu, int slot_mask,
                               hipStream_t stream) {
  const dim3 g(m > 4000000 ? 2048 : 1024), b(256);
  switch (kq_for_bf(c)) {
    case 1: hipLaunchKernelGGL(bn_bwd_reduce_bf16_kernel<1>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 2: hipLaunchKernelGGL(bn_bwd_reduce_bf16_kernel<2>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    case 4: hipLaunchKernelGGL(bn_bwd_reduce_bf16_kernel<4>, g, b, 0, stream,
                               dy, y, x, mean, invstd, red, m, c, relu,
                               slot_mask); break;
    default: hipLaunchKernelGGL(bn_bwd_reduce_bf16_kernel<8>, g, b, 0, stream,
                                dy, y, x, mean, invstd, red, m, c, relu,
                                slot_mask); break;
  }
}

void launch_bn_bwd_apply_bf16(const ushort* dy, const ushort* y,
                              const ushort* x, const float* mean,
                              const float* invstd, const float* weight,
                              const float* red, ushort* dx,
                              ushort* dresidual, float inv_count, int64_t m,
                              int c, int relu, hipStream_t stream) {
  const int64_t n4 = m * (c >> 2);
  hipLaunchKernelGGL(bn_bwd_apply_bf16_kernel, dim3(grid_1d(n4, 256)),
                     dim3(256), 0, stream, dy, y, x, mean, invstd, weight,
                     red, dx, dresidual, inv_count, m, c, relu);
}
