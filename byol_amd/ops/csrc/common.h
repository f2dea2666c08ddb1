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
