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

// ---------------------------------------------------------------------------
// Fast path (validated r2; superseded by the pad-free variant below
// for dispatch): zero-padded input copy makes every tap an
// unguarded uniform shift, so A and B both stage through glds
// (global_load_lds) with a 2-deep double buffer across the (tap, k0)
// sequence — the conv1x1 fast-path structure with 9x the K depth.
// Requires M%128==0, N%128==0, K%32==0.
// ---------------------------------------------------------------------------

__global__ void pad_nhwc_kernel(const float* __restrict__ x,
                                float* __restrict__ xp,
                                int b, int hi, int wi, int c) {
  const int hp = hi + 2, wp_ = wi + 2;
  const int c4 = c >> 2;
  const int64_t total = (int64_t)b * hp * wp_ * c4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  float4* o4 = reinterpret_cast<float4*>(xp);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int q = (int)(i % c4);
    const int64_t pix = i / c4;
    const int xw = (int)(pix % wp_);
    const int yh = (int)((pix / wp_) % hp);
    const int bb = (int)(pix / ((int64_t)wp_ * hp));
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    if (yh >= 1 && yh <= hi && xw >= 1 && xw <= wi) {
      v = x4[(((int64_t)bb * hi + (yh - 1)) * wi + (xw - 1)) * c4 + q];
    }
    o4[i] = v;
  }
}

#define C3F_LDSW_A 4096
#define C3F_LDSW_B 4096
#define C3F_LDSW_BUF (C3F_LDSW_A + C3F_LDSW_B)

template <bool XSWZ, bool PRIO>
__global__ __launch_bounds__(256)
void conv3x3_fwd_fast_kernel(const float* __restrict__ Xp,  // padded NHWC
                             const float* __restrict__ Wp,  // [9][K][N]
                             float* __restrict__ Y,
                             int B, int Hp, int Wpp,  // padded dims
                             int Ho, int Wo, int K, int N, int stride) {
  (void)B;  // batch rides inside the flattened m decomposition
  __shared__ __attribute__((aligned(16))) float lds[2 * C3F_LDSW_BUF];

  const int ntiles_n = N / C3_BN;
  const int bid = XSWZ ? xcd_remap(blockIdx.x, gridDim.x) : blockIdx.x;
  const int tile_m = bid / ntiles_n;
  const int tile_n = bid % ntiles_n;
  const int64_t m0 = (int64_t)tile_m * C3_BM;
  const int n0 = tile_n * C3_BN;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;

  // per-thread A staging geometry: 4 glds issues, each covering LDS words
  // (wave*4+i)*256 + lane*4 -> row r (of 128), quad q (of 8).  The output
  // row r is fixed per issue, so its padded-source base is precomputed.
  const float* abase[4];
  int aq[4];
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int off = (wave * 4 + i) * 256 + lane * 4;
    const int r = off >> 5;
    const int q = (off & 31) >> 2;
    aq[i] = ((q - r) & 7) << 2;  // quad-rotated source column (rule 21)
    const int64_t gm = m0 + r;
    const int64_t howo = (int64_t)Ho * Wo;
    const int bb = (int)(gm / howo);
    const int rem = (int)(gm % howo);
    const int iy = (rem / Wo) * stride;   // padded coords: no -1, no guard
    const int ix = (rem % Wo) * stride;
    abase[i] = Xp + (((int64_t)bb * Hp + iy) * Wpp + ix) * K;
  }

  f32x16_c3 acc[2][2];
  #pragma unroll
  for (int t = 0; t < 2; ++t)
    #pragma unroll
    for (int u = 0; u < 2; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) acc[t][u][e] = 0.f;

  const int nk = K / C3_BK;
  const int nsteps = 9 * nk;

  auto stage = [&](int buf, int step) {
    const int tap = step / nk;
    const int k0 = (step % nk) * C3_BK;
    const int dy = tap / 3, dx = tap % 3;
    const int64_t tap_off = ((int64_t)dy * Wpp + dx) * K;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const float* src = abase[i] + tap_off + k0 + aq[i];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * C3F_LDSW_BUF + (wave * 4 + i) * 256],
          16, 0, 0);
    }
    const float* wtap = Wp + ((int64_t)tap * K + k0) * N;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int off = (wave * 4 + i) * 256 + lane * 4;
      const int r = off >> 7;
      const int c = off & 127;
      const float* src = wtap + (int64_t)r * N + n0 + c;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * C3F_LDSW_BUF + C3F_LDSW_A + (wave * 4 + i) * 256],
          16, 0, 0);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int buf = 0;
  for (int step = 0; step < nsteps; ++step) {
    if (step + 1 < nsteps) stage(buf ^ 1, step + 1);
    const float* la = &lds[buf * C3F_LDSW_BUF];
    const float* lb = &lds[buf * C3F_LDSW_BUF + C3F_LDSW_A];
    if (PRIO) __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int kk = 0; kk < C3_BK; kk += 2) {
      const int krow = kk + (lane >> 5);
      #pragma unroll
      for (int t = 0; t < 2; ++t) {
        const int row = wr + t * 32 + (lane & 31);
        const float a = la[row * 32 + ((((krow >> 2) + row) & 7) << 2)
                           + (krow & 3)];
        #pragma unroll
        for (int u = 0; u < 2; ++u) {
          const float b = lb[krow * C3_BN + wc + u * 32 + (lane & 31)];
          acc[t][u] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc[t][u],
                                                           0, 0, 0);
        }
      }
    }
    if (PRIO) __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

  // LDS-restaged coalesced epilogue (same as conv1x1 fast)
  __syncthreads();
  float* cw = &lds[wave * 4096];
  #pragma unroll
  for (int t = 0; t < 2; ++t)
    #pragma unroll
    for (int u = 0; u < 2; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int lrow = t * 32 + (e & 3) + 8 * (e >> 2) + 4 * (lane >> 5);
        const int lcol = u * 32 + (lane & 31);
        cw[lrow * 64 + lcol] = acc[t][u][e];
      }
  __builtin_amdgcn_s_barrier();
  const int64_t gm_base = m0 + wr;
  const int gn_base = n0 + wc;
  #pragma unroll
  for (int p = 0; p < 16; ++p) {
    const int lrow = p * 4 + (lane >> 4);
    const int lcol = (lane & 15) * 4;
    const float4 v = *reinterpret_cast<const float4*>(&cw[lrow * 64 + lcol]);
    *reinterpret_cast<float4*>(
        &Y[(gm_base + lrow) * N + gn_base + lcol]) = v;
  }
}

// 128x64 fast variant for N=64-class widths (layer1's 3x3 at C=64 — the
// shape where the guarded slow kernel was 2x MIOpen).  Same padded-glds
// pipeline; 4 waves each own a 32x64 strip; LDS/buf = 16 KB A + 8 KB B.
#define C3F64_LDSW_B 2048
#define C3F64_LDSW_BUF (C3F_LDSW_A + C3F64_LDSW_B)

template <bool XSWZ, bool PRIO>
__global__ __launch_bounds__(256)
void conv3x3_fwd_fast64_kernel(const float* __restrict__ Xp,
                               const float* __restrict__ Wp,  // [9][K][N]
                               float* __restrict__ Y,
                               int B, int Hp, int Wpp,
                               int Ho, int Wo, int K, int N, int stride) {
  (void)B;  // batch rides inside the flattened m decomposition
  __shared__ __attribute__((aligned(16))) float lds[2 * C3F64_LDSW_BUF];

  const int ntiles_n = N / 64;
  const int bid = XSWZ ? xcd_remap(blockIdx.x, gridDim.x) : blockIdx.x;
  const int tile_m = bid / ntiles_n;
  const int tile_n = bid % ntiles_n;
  const int64_t m0 = (int64_t)tile_m * C3_BM;
  const int n0 = tile_n * 64;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave * 32;

  const float* abase[4];
  int aq[4];
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int off = (wave * 4 + i) * 256 + lane * 4;
    const int r = off >> 5;
    const int q = (off & 31) >> 2;
    aq[i] = ((q - r) & 7) << 2;
    const int64_t gm = m0 + r;
    const int64_t howo = (int64_t)Ho * Wo;
    const int bb = (int)(gm / howo);
    const int rem = (int)(gm % howo);
    const int iy = (rem / Wo) * stride;
    const int ix = (rem % Wo) * stride;
    abase[i] = Xp + (((int64_t)bb * Hp + iy) * Wpp + ix) * K;
  }

  f32x16_c3 acc[2];
  #pragma unroll
  for (int u = 0; u < 2; ++u)
    #pragma unroll
    for (int e = 0; e < 16; ++e) acc[u][e] = 0.f;

  const int nk = K / C3_BK;
  const int nsteps = 9 * nk;

  auto stage = [&](int buf, int step) {
    const int tap = step / nk;
    const int k0 = (step % nk) * C3_BK;
    const int dy = tap / 3, dx = tap % 3;
    const int64_t tap_off = ((int64_t)dy * Wpp + dx) * K;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const float* src = abase[i] + tap_off + k0 + aq[i];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * C3F64_LDSW_BUF + (wave * 4 + i) * 256],
          16, 0, 0);
    }
    const float* wtap = Wp + ((int64_t)tap * K + k0) * N;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int off = (wave * 2 + i) * 256 + lane * 4;
      const int r = off >> 6;
      const int c = off & 63;
      const float* src = wtap + (int64_t)r * N + n0 + c;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * C3F64_LDSW_BUF + C3F_LDSW_A + (wave * 2 + i) * 256],
          16, 0, 0);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int buf = 0;
  for (int step = 0; step < nsteps; ++step) {
    if (step + 1 < nsteps) stage(buf ^ 1, step + 1);
    const float* la = &lds[buf * C3F64_LDSW_BUF];
    const float* lb = &lds[buf * C3F64_LDSW_BUF + C3F_LDSW_A];
    if (PRIO) __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int kk = 0; kk < C3_BK; kk += 2) {
      const int krow = kk + (lane >> 5);
      const int row = wr + (lane & 31);
      const float a = la[row * 32 + ((((krow >> 2) + row) & 7) << 2)
                         + (krow & 3)];
      #pragma unroll
      for (int u = 0; u < 2; ++u) {
        const float b = lb[krow * 64 + u * 32 + (lane & 31)];
        acc[u] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc[u], 0, 0, 0);
      }
    }
    if (PRIO) __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

  __syncthreads();
  float* cw = &lds[wave * 2048];
  #pragma unroll
  for (int u = 0; u < 2; ++u)
    #pragma unroll
    for (int e = 0; e < 16; ++e) {
      const int lrow = (e & 3) + 8 * (e >> 2) + 4 * (lane >> 5);
      const int lcol = u * 32 + (lane & 31);
      cw[lrow * 64 + lcol] = acc[u][e];
    }
  __builtin_amdgcn_s_barrier();
  const int64_t gm_base = m0 + wr;
  #pragma unroll
  for (int p = 0; p < 8; ++p) {
    const int lrow = p * 4 + (lane >> 4);
    const int lcol = (lane & 15) * 4;
    const float4 v = *reinterpret_cast<const float4*>(&cw[lrow * 64 + lcol]);
    *reinterpret_cast<float4*>(
        &Y[(gm_base + lrow) * N + n0 + lcol]) = v;
  }
}

void launch_pad_nhwc(const float* x, float* xp, int b, int hi, int wi,
                     int c, hipStream_t stream) {
  const int64_t total = (int64_t)b * (hi + 2) * (wi + 2) * (c >> 2);
  hipLaunchKernelGGL(pad_nhwc_kernel, dim3(grid_1d(total, 256)), dim3(256),
                     0, stream, x, xp, b, hi, wi, c);
}

void launch_conv3x3_fwd_fast(const float* xp, const float* wp, float* y,
                             int b, int hi, int wi, int ho, int wo, int k,
                             int n, int stride, hipStream_t stream) {
  const int64_t m = (int64_t)b * ho * wo;
  if (n % C3_BN == 0) {
    const int64_t grid = (m / C3_BM) * (n / C3_BN);
    LAUNCH_FAST(conv3x3_fwd_fast_kernel, dim3((uint32_t)grid), xp, wp, y, b,
                hi + 2, wi + 2, ho, wo, k, n, stride);
  } else {
    const int64_t grid = (m / C3_BM) * (n / 64);
    LAUNCH_FAST(conv3x3_fwd_fast64_kernel, dim3((uint32_t)grid), xp, wp, y,
                b, hi + 2, wi + 2, ho, wo, k, n, stride);
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

// ---------------------------------------------------------------------------
// 3x3 wgrad: dW[N,K,3,3] = sum_m dY[m,n] * Xpad(tap)[m,k] — nine per-tap
// K-reduction GEMMs sharing the conv1x1 wgrad-v3 structure (glds
// double-buffered m-step pipeline + atomic epilogue).  tap lives in the
// GRID (each block owns one (n-tile, k-tile, tap, m-chunk)); the padded
// input makes every tap an unguarded uniform shift.  Output goes to a
// [9][N][K] workspace, permuted to torch's [N][K][3][3] by a tiny kernel.
// Requires M%32==0, N%64==0, K%64==0.
// ---------------------------------------------------------------------------
template <int TN, int TK>
__global__ __launch_bounds__(256)
void conv3x3_wgrad_kernel(const float* __restrict__ dY,
                          const float* __restrict__ Xp,  // padded NHWC
                          float* __restrict__ dW9,       // [9][N][K]
                          int B, int Hp, int Wpp, int Ho, int Wo,
                          int K, int N, int stride, int64_t chunk) {
  constexpr int AT = TN / 64;
  constexpr int AU = TK / 64;
  constexpr int LDSY = 32 * TN;
  constexpr int LDSX = 32 * TK;
  __shared__ __attribute__((aligned(16))) float lds[2 * (LDSY + LDSX)];

  const int64_t M = (int64_t)B * Ho * Wo;
  const int ntiles_k = K / TK;
  const int ntiles_n = N / TN;
  const int tile_k = blockIdx.x % ntiles_k;
  const int tile_n = (blockIdx.x / ntiles_k) % ntiles_n;
  const int tap = (blockIdx.x / (ntiles_k * ntiles_n)) % 9;
  const int64_t mchunk = blockIdx.x / (ntiles_k * ntiles_n * 9);
  const int n0 = tile_n * TN;
  const int k0 = tile_k * TK;
  const int dy_t = tap / 3, dx_t = tap % 3;
  const int64_t mstart = mchunk * chunk;
  const int64_t mend = (mstart + chunk < M) ? mstart + chunk : M;
  if (mstart >= M) return;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * (TN / 2);
  const int wc = (wave & 1) * (TK / 2);
  const int64_t howo = (int64_t)Ho * Wo;

  f32x16_c3 acc[AT][AU];
  #pragma unroll
  for (int t = 0; t < AT; ++t)
    #pragma unroll
    for (int u = 0; u < AU; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) acc[t][u][e] = 0.f;

  auto stage = [&](int buf, int64_t m0) {
    #pragma unroll
    for (int i = 0; i < TN / 32; ++i) {  // dY tile [32][TN]
      const int off = (wave * (TN / 32) + i) * 256 + lane * 4;
      const int r = off / TN;
      const int c = off % TN;
      const float* src = dY + (m0 + r) * N + n0 + c;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * (LDSY + LDSX) + (wave * (TN / 32) + i) * 256],
          16, 0, 0);
    }
    #pragma unroll
    for (int i = 0; i < TK / 32; ++i) {  // Xp(tap) tile [32][TK]
      const int off = (wave * (TK / 32) + i) * 256 + lane * 4;
      const int r = off / TK;
      const int c = off % TK;
      const int64_t gm = m0 + r;
      const int bb = (int)(gm / howo);
      const int rem = (int)(gm % howo);
      const int iy = (rem / Wo) * stride + dy_t;
      const int ix = (rem % Wo) * stride + dx_t;
      const float* src =
          Xp + (((int64_t)bb * Hp + iy) * Wpp + ix) * K + k0 + c;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * (LDSY + LDSX) + LDSY
                   + (wave * (TK / 32) + i) * 256],
          16, 0, 0);
    }
  };

  stage(0, mstart);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int buf = 0;
  for (int64_t m0 = mstart; m0 < mend; m0 += 32) {
    if (m0 + 32 < mend) stage(buf ^ 1, m0 + 32);
    const float* ldy = &lds[buf * (LDSY + LDSX)];
    const float* ldx = &lds[buf * (LDSY + LDSX) + LDSY];
    #pragma unroll
    for (int kk = 0; kk < 32; kk += 2) {
      const int mrow = kk + (lane >> 5);
      #pragma unroll
      for (int t = 0; t < AT; ++t) {
        const float a = ldy[mrow * TN + wr + t * 32 + (lane & 31)];
        #pragma unroll
        for (int u = 0; u < AU; ++u) {
          const float b = ldx[mrow * TK + wc + u * 32 + (lane & 31)];
          acc[t][u] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc[t][u],
                                                           0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

  float* slab = dW9 + (int64_t)tap * N * K;
  #pragma unroll
  for (int t = 0; t < AT; ++t) {
    #pragma unroll
    for (int u = 0; u < AU; ++u) {
      #pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int row = (e & 3) + 8 * (e >> 2) + 4 * (lane >> 5);
        const int col = lane & 31;
        const int gn = n0 + wr + t * 32 + row;
        const int gk = k0 + wc + u * 32 + col;
        atomicAdd(&slab[(int64_t)gn * K + gk], acc[t][u][e]);
      }
    }
  }
}

// [9][N][K] workspace -> torch [N][K][3][3]
__global__ void wgrad9_permute_kernel(const float* __restrict__ dW9,
                                      float* __restrict__ dw,
                                      int64_t nk) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nk;
       i += stride) {
    #pragma unroll
    for (int tap = 0; tap < 9; ++tap)
      dw[i * 9 + tap] = dW9[(int64_t)tap * nk + i];
  }
}

template <int TN, int TK>
static void launch_c3wg_tile(const float* dy, const float* xp, float* dw9,
                             int b, int hp, int wpp, int ho, int wo, int k,
                             int n, int stride, hipStream_t stream) {
  const int64_t m = (int64_t)b * ho * wo;
  const int64_t tiles = (int64_t)(n / TN) * (k / TK) * 9;
  int64_t chunk = 1024;
  while ((m + chunk - 1) / chunk * tiles > 8192) chunk *= 2;
  const int64_t grid = tiles * ((m + chunk - 1) / chunk);
  hipLaunchKernelGGL((conv3x3_wgrad_kernel<TN, TK>), dim3((uint32_t)grid),
                     dim3(256), 0, stream, dy, xp, dw9, b, hp, wpp, ho, wo,
                     k, n, stride, chunk);
}

void launch_conv3x3_wgrad(const float* dy, const float* xp, float* dw9,
                          float* dw, int b, int hi, int wi, int ho, int wo,
                          int k, int n, int stride, hipStream_t stream) {
  const int hp = hi + 2, wpp = wi + 2;
  const bool n128 = n % 128 == 0, k128 = k % 128 == 0;
  if (n128 && k128)
    launch_c3wg_tile<128, 128>(dy, xp, dw9, b, hp, wpp, ho, wo, k, n,
                               stride, stream);
  else if (n128)
    launch_c3wg_tile<128, 64>(dy, xp, dw9, b, hp, wpp, ho, wo, k, n,
                              stride, stream);
  else if (k128)
    launch_c3wg_tile<64, 128>(dy, xp, dw9, b, hp, wpp, ho, wo, k, n,
                              stride, stream);
  else
    launch_c3wg_tile<64, 64>(dy, xp, dw9, b, hp, wpp, ho, wo, k, n,
                             stride, stream);
  const int64_t nk = (int64_t)n * k;
  hipLaunchKernelGGL(wgrad9_permute_kernel, dim3(grid_1d(nk, 256)),
                     dim3(256), 0, stream, dw9, dw, nk);
}

// ---------------------------------------------------------------------------
// Pad-free fast forward: reads the UNPADDED input; out-of-bounds tap lanes
// redirect their glds source to a 16-B zero page instead of a padded copy
// (the pad pass cost 0.07-0.28 ms/call = most of the gap to MIOpen on the
// stride-2 shapes).  Same pipeline as conv3x3_fwd_fast_kernel otherwise.
// ---------------------------------------------------------------------------
template <int TBN>
__global__ __launch_bounds__(256)
void conv3x3_fwd_nopad_kernel(const float* __restrict__ X,  // unpadded
                              const float* __restrict__ Wp,  // [9][K][N]
                              float* __restrict__ Y,
                              const float* __restrict__ zpage,
                              int B, int Hi, int Wi,
                              int Ho, int Wo, int K, int N, int stride) {
  (void)B;  // batch rides inside the flattened m decomposition
  constexpr int LDSB = 32 * TBN;
  constexpr int LDSBUF = C3F_LDSW_A + LDSB;
  __shared__ __attribute__((aligned(16))) float lds[2 * LDSBUF];

  const int ntiles_n = N / TBN;
  const int tile_m = blockIdx.x / ntiles_n;
  const int tile_n = blockIdx.x % ntiles_n;
  const int64_t m0 = (int64_t)tile_m * C3_BM;
  const int n0 = tile_n * TBN;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // wave tile: TBN=128 -> 2x2 quadrants of 64x64; TBN=64 -> 4x1 strips
  const int wr = (TBN == 128) ? (wave >> 1) * 64 : wave * 32;
  const int wc = (TBN == 128) ? (wave & 1) * 64 : 0;
  constexpr int AT = (TBN == 128) ? 2 : 1;
  constexpr int AU = 2;

  const float* abase[4];
  int aq[4], aiy[4], aix[4];
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int off = (wave * 4 + i) * 256 + lane * 4;
    const int r = off >> 5;
    const int q = (off & 31) >> 2;
    aq[i] = ((q - r) & 7) << 2;
    const int64_t gm = m0 + r;
    const int64_t howo = (int64_t)Ho * Wo;
    const int bb = (int)(gm / howo);
    const int rem = (int)(gm % howo);
    aiy[i] = (rem / Wo) * stride - 1;  // unpadded coords incl. pad offset
    aix[i] = (rem % Wo) * stride - 1;
    abase[i] = X + (((int64_t)bb * Hi + aiy[i]) * Wi + aix[i]) * K;
  }

  f32x16_c3 acc[AT][AU];
  #pragma unroll
  for (int t = 0; t < AT; ++t)
    #pragma unroll
    for (int u = 0; u < AU; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) acc[t][u][e] = 0.f;

  const int nk = K / C3_BK;
  const int nsteps = 9 * nk;

  auto stage = [&](int buf, int step) {
    const int tap = step / nk;
    const int k0 = (step % nk) * C3_BK;
    const int dy = tap / 3, dx = tap % 3;
    const int64_t tap_off = ((int64_t)dy * Wi + dx) * K;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const bool ok = (unsigned)(aiy[i] + dy) < (unsigned)Hi &&
                      (unsigned)(aix[i] + dx) < (unsigned)Wi;
      const float* src =
          ok ? abase[i] + tap_off + k0 + aq[i] : zpage;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * LDSBUF + (wave * 4 + i) * 256],
          16, 0, 0);
    }
    const float* wtap = Wp + ((int64_t)tap * K + k0) * N;
    #pragma unroll
    for (int i = 0; i < TBN / 32; ++i) {  // 32*TBN floats / (4 waves * 1 KiB)
      const int off = (wave * (TBN / 32) + i) * 256 + lane * 4;
      const int r = off / TBN;
      const int c = off % TBN;
      const float* src = wtap + (int64_t)r * N + n0 + c;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)src,
          (__attribute__((address_space(3))) uint32_t*)
              &lds[buf * LDSBUF + C3F_LDSW_A + (wave * (TBN / 32) + i)
                   * 256],
          16, 0, 0);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int buf = 0;
  for (int step = 0; step < nsteps; ++step) {
    if (step + 1 < nsteps) stage(buf ^ 1, step + 1);
    const float* la = &lds[buf * LDSBUF];
    const float* lb = &lds[buf * LDSBUF + C3F_LDSW_A];
    #pragma unroll
    for (int kk = 0; kk < C3_BK; kk += 2) {
      const int krow = kk + (lane >> 5);
      #pragma unroll
      for (int t = 0; t < AT; ++t) {
        const int row = wr + t * 32 + (lane & 31);
        const float a = la[row * 32 + ((((krow >> 2) + row) & 7) << 2)
                           + (krow & 3)];
        #pragma unroll
        for (int u = 0; u < AU; ++u) {
          const float b = lb[krow * TBN + wc + u * 32 + (lane & 31)];
          acc[t][u] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc[t][u],
                                                           0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

  __syncthreads();
  float* cw = &lds[wave * (TBN == 128 ? 4096 : 2048)];
  #pragma unroll
  for (int t = 0; t < AT; ++t)
    #pragma unroll
    for (int u = 0; u < AU; ++u)
      #pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int lrow = t * 32 + (e & 3) + 8 * (e >> 2) + 4 * (lane >> 5);
        const int lcol = u * 32 + (lane & 31);
        cw[lrow * 64 + lcol] = acc[t][u][e];
      }
  __builtin_amdgcn_s_barrier();
  const int64_t gm_base = m0 + wr;
  const int gn_base = n0 + wc;
  #pragma unroll
  for (int p = 0; p < (TBN == 128 ? 16 : 8); ++p) {
    const int lrow = p * 4 + (lane >> 4);
    const int lcol = (lane & 15) * 4;
    const float4 v = *reinterpret_cast<const float4*>(&cw[lrow * 64 + lcol]);
    *reinterpret_cast<float4*>(
        &Y[(gm_base + lrow) * N + gn_base + lcol]) = v;
  }
}

void launch_conv3x3_fwd_nopad(const float* x, const float* wp, float* y,
                              const float* zpage, int b, int hi, int wi,
                              int ho, int wo, int k, int n, int stride,
                              hipStream_t stream) {
  const int64_t m = (int64_t)b * ho * wo;
  if (n % C3_BN == 0) {
    const int64_t grid = (m / C3_BM) * (n / C3_BN);
    hipLaunchKernelGGL((conv3x3_fwd_nopad_kernel<128>),
                       dim3((uint32_t)grid), dim3(256), 0, stream, x, wp, y,
                       zpage, b, hi, wi, ho, wo, k, n, stride);
  } else {
    const int64_t grid = (m / C3_BM) * (n / 64);
    hipLaunchKernelGGL((conv3x3_fwd_nopad_kernel<64>),
                       dim3((uint32_t)grid), dim3(256), 0, stream, x, wp, y,
                       zpage, b, hi, wi, ho, wo, k, n, stride);
  }
}
