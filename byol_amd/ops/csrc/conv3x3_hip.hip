#include "hip/hip_runtime.h"
// f32 MFMA forward kernel for NHWC 3x3 convolutions (pad=1, stride 1 or 2)
// — the other half of ResNet's conv FLOPs.
//
// Implicit GEMM over 9 taps: Y[M,N] = sum_tap X_shift(tap)[M,K] . Wp[tap],
// with Wp host-prepermuted to [9][K][N] so the B staging is coalesced.
// Border handling is per-row guards in the A staging (zero padding).
// Correctness-first rung: register staging, single-buffered LDS, the same
// 128x128x32 / 4-wave / 32x32x2-MFMA structure as conv1x1.  3x3 shapes are
// compute-bound (AI 9x the 1x1s), so the staging pipeline matters less
// here; the glds ladder is the next rung.
#include "common.h"

using f32x16_c3 = __attribute__((ext_vector_type(16))) float;

#define C3_BM 128
#define C3_BN 128
#define C3_BK 32

__global__ __launch_bounds__(256)
void conv3x3_fwd_kernel(const float* __restrict__ X,
                        const float* __restrict__ Wp,  // [9][K][N]
                        float* __restrict__ Y,
                        int B, int Hi, int Wi, int Ho, int Wo,
                        int K, int N, int stride) {
  __shared__ float lds_a[C3_BM][C3_BK + 1];
  __shared__ float lds_b[C3_BK][C3_BN + 1];

  const int64_t M = (int64_t)B * Ho * Wo;
  const int ntiles_n = (N + C3_BN - 1) / C3_BN;
  const int tile_m = blockIdx.x / ntiles_n;
  const int tile_n = blockIdx.x % ntiles_n;
  const int64_t m0 = (int64_t)tile_m * C3_BM;
  const int n0 = tile_n * C3_BN;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;

  // staging geometry: thread t stages row (t>>1), 16-float half (t&1)
  const int srow = threadIdx.x >> 1;
  const int scol = (threadIdx.x & 1) * 16;
  // decompose this thread's output row once; reused across taps/k-chunks
  const int64_t gm = m0 + srow;
  int sb = 0, iy0 = 0, ix0 = 0;
  bool row_in_m = gm < M;
  if (row_in_m) {
    const int64_t howo = (int64_t)Ho * Wo;
    sb = (int)(gm / howo);
    const int rem = (int)(gm % howo);
    iy0 = (rem / Wo) * stride - 1;   // pad = 1
    ix0 = (rem % Wo) * stride - 1;
  }

  f32x16_c3 acc[2][2];
  #pragma unroll
  for (int t = 0; t < 2; ++t)
    #pragma unroll
    for (int u = 0; u < 2; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) acc[t][u][e] = 0.f;

  for (int tap = 0; tap < 9; ++tap) {
    const int dy = tap / 3, dx = tap % 3;
    const int iy = iy0 + dy;
    const int ix = ix0 + dx;
    const bool ok = row_in_m && (unsigned)iy < (unsigned)Hi &&
                    (unsigned)ix < (unsigned)Wi;
    const float* arow =
        X + (((int64_t)sb * Hi + iy) * Wi + ix) * K;
    const float* wtap = Wp + (int64_t)tap * K * N;

    for (int k0 = 0; k0 < K; k0 += C3_BK) {
      // A stage: guarded (zero-pad borders)
      #pragma unroll
      for (int c4 = 0; c4 < 4; ++c4) {
        const int c = scol + c4 * 4;
        float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
        if (ok) v = *reinterpret_cast<const float4*>(arow + k0 + c);
        lds_a[srow][c + 0] = v.x;
        lds_a[srow][c + 1] = v.y;
        lds_a[srow][c + 2] = v.z;
        lds_a[srow][c + 3] = v.w;
      }
      // B stage: [32][BN] from Wp[tap][k0+r][n0+c], coalesced
      {
        const int r = threadIdx.x >> 3;
        const int cbase = (threadIdx.x & 7) * 16;
        #pragma unroll
        for (int c4 = 0; c4 < 4; ++c4) {
          const int c = cbase + c4 * 4;
          float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
          if (n0 + c + 3 < N) {
            v = *reinterpret_cast<const float4*>(
                wtap + (int64_t)(k0 + r) * N + n0 + c);
          } else if (n0 + c < N) {
            v.x = wtap[(int64_t)(k0 + r) * N + n0 + c];
            if (n0 + c + 1 < N)
              v.y = wtap[(int64_t)(k0 + r) * N + n0 + c + 1];
            if (n0 + c + 2 < N)
              v.z = wtap[(int64_t)(k0 + r) * N + n0 + c + 2];
          }
          lds_b[r][c + 0] = v.x;
          lds_b[r][c + 1] = v.y;
          lds_b[r][c + 2] = v.z;
          lds_b[r][c + 3] = v.w;
        }
      }
      __syncthreads();
      #pragma unroll
      for (int kk = 0; kk < C3_BK; kk += 2) {
        const int krow = kk + (lane >> 5);
        #pragma unroll
        for (int t = 0; t < 2; ++t) {
          const float a = lds_a[wr + t * 32 + (lane & 31)][krow];
          #pragma unroll
          for (int u = 0; u < 2; ++u) {
            const float b = lds_b[krow][wc + u * 32 + (lane & 31)];
            acc[t][u] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                a, b, acc[t][u], 0, 0, 0);
          }
        }
      }
      __syncthreads();
    }
  }

  #pragma unroll
  for (int t = 0; t < 2; ++t) {
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      #pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int row = (e & 3) + 8 * (e >> 2) + 4 * (lane >> 5);
        const int col = lane & 31;
        const int64_t om = m0 + wr + t * 32 + row;
        const int on = n0 + wc + u * 32 + col;
        if (om < M && on < N) Y[om * N + on] = acc[t][u][e];
      }
    }
  }
}

void launch_conv3x3_fwd(const float* x, const float* wp, float* y, int b,
                        int hi, int wi, int ho, int wo, int k, int n,
                        int stride, hipStream_t stream) {
  const int64_t m = (int64_t)b * ho * wo;
  const int64_t grid = ((m + C3_BM - 1) / C3_BM)
                       * ((n + C3_BN - 1) / C3_BN);
  hipLaunchKernelGGL(conv3x3_fwd_kernel, dim3((uint32_t)grid), dim3(256), 0,
                     stream, x, wp, y, b, hi, wi, ho, wo, k, n, stride);
}
