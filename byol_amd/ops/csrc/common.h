// Common device helpers for the byol_amd CDNA4 (gfx950) kernels.
// Written MI355X-first: wave64, float4-vectorized global access, block
// reductions via DS-free wave shuffles.
#pragma once

#include <cstdint>
#include <hip/hip_runtime.h>

#define WAVE_SIZE 64

__device__ __forceinline__ float wave_reduce_sum(float v) {
  // full 64-lane butterfly reduction
  #pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE_SIZE);
  }
  return v;  // valid in lane 0 of the wave
}

// Block-level sum reduction (blockDim.x threads, multiple of 64).
// Returns the total in thread 0; `scratch` must hold blockDim.x/64 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nwaves = blockDim.x / WAVE_SIZE;
  float out = 0.f;
  if (wid == 0) {
    out = (lane < nwaves) ? scratch[lane] : 0.f;
    out = wave_reduce_sum(out);
  }
  __syncthreads();
  return out;  // thread 0
}

static inline int grid_1d(int64_t n, int block, int max_blocks = 16384) {
  int64_t g = (n + block - 1) / block;
  if (g > max_blocks) g = max_blocks;
  if (g < 1) g = 1;
  return static_cast<int>(g);
}

// ---------------------------------------------------------------------------
// shared conv-kernel experiment plumbing (conv1x1.hip / conv3x3.hip)
// ---------------------------------------------------------------------------

// XCD-aware bijective blockIdx remap (guide T1): consecutive logical tiles
// share an operand panel; grouping them per XCD makes panel re-reads L2
// hits.  Bijective also when nwg % 8 != 0.
__device__ __forceinline__ int xcd_remap(int bid, int nwg) {
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, orig = bid >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig;
}

// A/B knobs, read once per process (tools/conv_microbench.py sweeps them
// across processes): BYOL_CONV_XSWZ = XCD tile remap (T1), BYOL_CONV_PRIO =
// s_setprio around the MFMA cluster (T5).
static inline bool conv_env_flag(const char* name, bool dflt) {
  const char* v = getenv(name);
  return v ? v[0] == '1' : dflt;
}
static inline bool conv_xswz() {
  static bool f = conv_env_flag("BYOL_CONV_XSWZ", false);
  return f;
}
static inline bool conv_prio() {
  static bool f = conv_env_flag("BYOL_CONV_PRIO", false);
  return f;
}

#define LAUNCH_FAST(KERN, GRID, ...)                                       \
  do {                                                                     \
    const bool xs = conv_xswz(), pr = conv_prio();                         \
    if (xs && pr)                                                          \
      hipLaunchKernelGGL((KERN<true, true>), GRID, dim3(256), 0, stream,   \
                         __VA_ARGS__);                                     \
    else if (xs)                                                           \
      hipLaunchKernelGGL((KERN<true, false>), GRID, dim3(256), 0, stream,  \
                         __VA_ARGS__);                                     \
    else if (pr)                                                           \
      hipLaunchKernelGGL((KERN<false, true>), GRID, dim3(256), 0, stream,  \
                         __VA_ARGS__);                                     \
    else                                                                   \
      hipLaunchKernelGGL((KERN<false, false>), GRID, dim3(256), 0, stream, \
                         __VA_ARGS__);                                     \
  } while (0)
