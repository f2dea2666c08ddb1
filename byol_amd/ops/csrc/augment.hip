// GPU two-view augmentation pipeline (the reference's DALI replacement,
// SURVEY.md K18; reference pipeline /root/reference/main.py:347-400).
//
// Kernel 1 (aug_sample): per-sample RandomResizedCrop (bilinear, torch
// F.interpolate align_corners=False semantics) + horizontal flip, NHWC fp32
// out, and accumulates the per-sample grayscale sum the contrast jitter
// needs (torchvision blends against the mean gray of the *cropped* image).
//
// Kernel 2 (aug_color): per-sample ColorJitter with torchvision semantics —
// brightness/contrast/saturation blends + HSV hue shift, applied in a
// per-sample random ORDER — then optional grayscale.  All parameters are
// sampled host-side (one small [B, x] tensor), so the kernels are
// deterministic given the parameter tensors.
//
// Gaussian blur runs as a grouped torch conv upstream (per-sample sigma).
#include "common.h"

#define GRAY_R 0.299f
#define GRAY_G 0.587f
#define GRAY_B 0.114f

// crop params per sample: [y0, x0, ch, cw, flip] (floats, source pixels)
__global__ void aug_sample_kernel(const float* __restrict__ src,
                                  float* __restrict__ dst,
                                  float* __restrict__ gray_sum,
                                  const float* __restrict__ crop,
                                  int b, int hs, int ws, int s) {
  const int64_t total = (int64_t)b * s * s;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int ox = (int)(i % s);
    const int oy = (int)((i / s) % s);
    const int bi = (int)(i / ((int64_t)s * s));
    const float* cp = crop + bi * 5;
    const float y0 = cp[0], x0 = cp[1], ch = cp[2], cw = cp[3];
    const int flip = (int)cp[4];
    const int sx = flip ? (s - 1 - ox) : ox;
    // align_corners=False bilinear source coords within the crop box
    float fy = y0 + ((float)oy + 0.5f) * ch / (float)s - 0.5f;
    float fx = x0 + ((float)sx + 0.5f) * cw / (float)s - 0.5f;
    const int y1i = (int)floorf(fy);
    const int x1i = (int)floorf(fx);
    const float wy = fy - (float)y1i;
    const float wx = fx - (float)x1i;
    const int ya = min(max(y1i, 0), hs - 1);
    const int yb = min(max(y1i + 1, 0), hs - 1);
    const int xa = min(max(x1i, 0), ws - 1);
    const int xb = min(max(x1i + 1, 0), ws - 1);
    const float* sb = src + (int64_t)bi * hs * ws * 3;
    float out[3];
    #pragma unroll
    for (int col = 0; col < 3; ++col) {
      const float v00 = sb[((int64_t)ya * ws + xa) * 3 + col];
      const float v01 = sb[((int64_t)ya * ws + xb) * 3 + col];
      const float v10 = sb[((int64_t)yb * ws + xa) * 3 + col];
      const float v11 = sb[((int64_t)yb * ws + xb) * 3 + col];
      const float v0 = v00 + wx * (v01 - v00);
      const float v1 = v10 + wx * (v11 - v10);
      float v = v0 + wy * (v1 - v0);
      v = fminf(fmaxf(v, 0.f), 1.f);
      out[col] = v;
      dst[i * 3 + col] = v;
    }
    const float g = GRAY_R * out[0] + GRAY_G * out[1] + GRAY_B * out[2];
    atomicAdd(&gray_sum[bi], g);
  }
}

__device__ __forceinline__ void hue_shift(float& r, float& g, float& b,
                                          float shift) {
  const float maxc = fmaxf(r, fmaxf(g, b));
  const float minc = fminf(r, fminf(g, b));
  const float v = maxc;
  const float d = maxc - minc;
  const float sat = maxc > 0.f ? d / fmaxf(maxc, 1e-12f) : 0.f;
  const float dz = fmaxf(d, 1e-12f);
  float h;
  if (r == maxc) h = (maxc - b) / dz - (maxc - g) / dz;
  else if (g == maxc) h = 2.f + (maxc - r) / dz - (maxc - b) / dz;
  else h = 4.f + (maxc - g) / dz - (maxc - r) / dz;
  h = h / 6.f;
  h = h - floorf(h);
  if (d <= 0.f) h = 0.f;
  h = h + shift;
  h = h - floorf(h);
  const float i6 = floorf(h * 6.f);
  const float f = h * 6.f - i6;
  const float p = v * (1.f - sat);
  const float q = v * (1.f - f * sat);
  const float t = v * (1.f - (1.f - f) * sat);
  const int ii = ((int)i6) % 6;
  switch (ii) {
    case 0: r = v; g = t; b = p; break;
    case 1: r = q; g = v; b = p; break;
    case 2: r = p; g = v; b = t; break;
    case 3: r = p; g = q; b = v; break;
    case 4: r = t; g = p; b = v; break;
    default: r = v; g = p; b = q; break;
  }
  r = fminf(fmaxf(r, 0.f), 1.f);
  g = fminf(fmaxf(g, 0.f), 1.f);
  b = fminf(fmaxf(b, 0.f), 1.f);
}

// color params per sample, 10 floats:
// [do_jitter, f_bright, f_contrast, f_sat, hue, do_gray, op0, op1, op2, op3]
__global__ void aug_color_kernel(float* __restrict__ img,
                                 const float* __restrict__ gray_sum,
                                 const float* __restrict__ cparam,
                                 int b, int s) {
  const int64_t total = (int64_t)b * s * s;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv_px = 1.f / (float)(s * s);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int bi = (int)(i / ((int64_t)s * s));
    const float* p = cparam + bi * 10;
    float r = img[i * 3 + 0];
    float g = img[i * 3 + 1];
    float bl = img[i * 3 + 2];
    if (p[0] > 0.5f) {
      const float gmean = gray_sum[bi] * inv_px;
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        const int op = (int)p[6 + k];
        if (op == 0 && p[1] >= 0.f) {          // brightness
          const float f = p[1];
          r = fminf(fmaxf(f * r, 0.f), 1.f);
          g = fminf(fmaxf(f * g, 0.f), 1.f);
          bl = fminf(fmaxf(f * bl, 0.f), 1.f);
        } else if (op == 1 && p[2] >= 0.f) {   // contrast (vs mean gray)
          const float f = p[2];
          r = fminf(fmaxf(f * r + (1.f - f) * gmean, 0.f), 1.f);
          g = fminf(fmaxf(f * g + (1.f - f) * gmean, 0.f), 1.f);
          bl = fminf(fmaxf(f * bl + (1.f - f) * gmean, 0.f), 1.f);
        } else if (op == 2 && p[3] >= 0.f) {   // saturation (vs pixel gray)
          const float f = p[3];
          const float gr = GRAY_R * r + GRAY_G * g + GRAY_B * bl;
          r = fminf(fmaxf(f * r + (1.f - f) * gr, 0.f), 1.f);
          g = fminf(fmaxf(f * g + (1.f - f) * gr, 0.f), 1.f);
          bl = fminf(fmaxf(f * bl + (1.f - f) * gr, 0.f), 1.f);
        } else if (op == 3 && p[4] > -1.f) {   // hue
          hue_shift(r, g, bl, p[4]);
        }
      }
    }
    if (p[5] > 0.5f) {  // grayscale
      const float gr = GRAY_R * r + GRAY_G * g + GRAY_B * bl;
      r = gr; g = gr; bl = gr;
    }
    img[i * 3 + 0] = r;
    img[i * 3 + 1] = g;
    img[i * 3 + 2] = bl;
  }
}

// v2 (DEFAULT since r2: measured 7.9x v1): sample-major blocks — each block covers a pixel
// chunk of ONE sample, reduces its gray contribution in LDS and issues a
// single atomicAdd, instead of one atomic per pixel (s*s serialized RMWs
// per sample address in v1).
__global__ void aug_sample_v2_kernel(const float* __restrict__ src,
                                     float* __restrict__ dst,
                                     float* __restrict__ gray_sum,
                                     const float* __restrict__ crop,
                                     int hs, int ws, int s,
                                     int chunks_per_sample) {
  __shared__ float scratch[4];
  const int bi = blockIdx.x / chunks_per_sample;
  const int chunk = blockIdx.x % chunks_per_sample;
  const int npix = s * s;
  const int per_chunk = (npix + chunks_per_sample - 1) / chunks_per_sample;
  const int start = chunk * per_chunk;
  const int end = min(start + per_chunk, npix);

  const float* cp = crop + bi * 5;
  const float y0 = cp[0], x0 = cp[1], ch = cp[2], cw = cp[3];
  const int flip = (int)cp[4];
  const float* sb = src + (int64_t)bi * hs * ws * 3;
  float* db = dst + (int64_t)bi * npix * 3;

  float gacc = 0.f;
  for (int p = start + threadIdx.x; p < end; p += blockDim.x) {
    const int ox = p % s;
    const int oy = p / s;
    const int sx = flip ? (s - 1 - ox) : ox;
    float fy = y0 + ((float)oy + 0.5f) * ch / (float)s - 0.5f;
    float fx = x0 + ((float)sx + 0.5f) * cw / (float)s - 0.5f;
    const int y1i = (int)floorf(fy);
    const int x1i = (int)floorf(fx);
    const float wy = fy - (float)y1i;
    const float wx = fx - (float)x1i;
    const int ya = min(max(y1i, 0), hs - 1);
    const int yb = min(max(y1i + 1, 0), hs - 1);
    const int xa = min(max(x1i, 0), ws - 1);
    const int xb = min(max(x1i + 1, 0), ws - 1);
    float out[3];
    #pragma unroll
    for (int col = 0; col < 3; ++col) {
      const float v00 = sb[((int64_t)ya * ws + xa) * 3 + col];
      const float v01 = sb[((int64_t)ya * ws + xb) * 3 + col];
      const float v10 = sb[((int64_t)yb * ws + xa) * 3 + col];
      const float v11 = sb[((int64_t)yb * ws + xb) * 3 + col];
      const float v0 = v00 + wx * (v01 - v00);
      const float v1 = v10 + wx * (v11 - v10);
      float v = v0 + wy * (v1 - v0);
      v = fminf(fmaxf(v, 0.f), 1.f);
      out[col] = v;
      db[(int64_t)p * 3 + col] = v;
    }
    gacc += GRAY_R * out[0] + GRAY_G * out[1] + GRAY_B * out[2];
  }
  const float tot = block_reduce_sum(gacc, scratch);
  if (threadIdx.x == 0) atomicAdd(&gray_sum[bi], tot);
}

void launch_aug_sample(const float* src, float* dst, float* gray_sum,
                       const float* crop, int b, int hs, int ws, int s,
                       int use_v2, hipStream_t stream) {
  if (use_v2) {  // default: sample-major, one gray atomic per block
    int chunks = (s * s + 8191) / 8192;
    if (chunks < 1) chunks = 1;
    hipLaunchKernelGGL(aug_sample_v2_kernel, dim3(b * chunks), dim3(256), 0,
                       stream, src, dst, gray_sum, crop, hs, ws, s, chunks);
    return;
  }
  const int64_t total = (int64_t)b * s * s;
  hipLaunchKernelGGL(aug_sample_kernel, dim3(grid_1d(total, 256)), dim3(256),
                     0, stream, src, dst, gray_sum, crop, b, hs, ws, s);
}

void launch_aug_color(float* img, const float* gray_sum, const float* cparam,
                      int b, int s, hipStream_t stream) {
  const int64_t total = (int64_t)b * s * s;
  hipLaunchKernelGGL(aug_color_kernel, dim3(grid_1d(total, 256)), dim3(256),
                     0, stream, img, gray_sum, cparam, b, s);
}

// ---------------------------------------------------------------------------
// Per-sample-sigma separable Gaussian blur, NHWC [B,S,S,3] in [0,1].
// Replaces the composed grouped-conv pair (B*C groups) the Python path used:
// 3 launches total — weights, vertical pass, horizontal pass.  sigma[b]==0
// rows get the identity kernel (pass-through).  Reflect padding, matching
// torchvision GaussianBlur / data/transforms.py.
// ---------------------------------------------------------------------------

__global__ void blur_weights_kernel(const float* __restrict__ sigma,
                                    float* __restrict__ wts,  // [B][ksize]
                                    int b, int ksize) {
  const int bi = blockIdx.x * blockDim.x + threadIdx.x;
  if (bi >= b) return;
  const float sg = sigma[bi];
  float* w = wts + (int64_t)bi * ksize;
  const int half = ksize / 2;
  if (sg <= 0.f) {
    for (int k = 0; k < ksize; ++k) w[k] = (k == half) ? 1.f : 0.f;
    return;
  }
  float tot = 0.f;
  for (int k = 0; k < ksize; ++k) {
    const float d = (float)(k - half);
    const float v = __expf(-d * d / (2.f * sg * sg));
    w[k] = v;
    tot += v;
  }
  const float inv = 1.f / tot;
  for (int k = 0; k < ksize; ++k) w[k] *= inv;
}

// reflect index into [0, n): torchvision "reflect" (no edge repeat)
__device__ __forceinline__ int reflect_idx(int i, int n) {
  if (i < 0) i = -i;
  if (i >= n) i = 2 * n - 2 - i;
  return i;
}

// VERT = blur along y (else along x).  One thread per output element; the
// 64-lane wave walks x*c contiguously so every tap row is a coalesced read.
template <bool VERT>
__global__ void blur_pass_kernel(const float* __restrict__ src,
                                 float* __restrict__ dst,
                                 const float* __restrict__ wts,
                                 int b, int s, int ksize) {
  const int64_t per_img = (int64_t)s * s * 3;
  const int64_t total = (int64_t)b * per_img;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int half = ksize / 2;
  const int rowlen = s * 3;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int bi = (int)(i / per_img);
    const int64_t rem = i % per_img;
    const int y = (int)(rem / rowlen);
    const int xc = (int)(rem % rowlen);
    const float* w = wts + (int64_t)bi * ksize;
    const float* img = src + (int64_t)bi * per_img;
    float acc = 0.f;
    if (VERT) {
      #pragma unroll 4
      for (int k = 0; k < ksize; ++k) {
        const int yy = reflect_idx(y + k - half, s);
        acc = fmaf(w[k], img[(int64_t)yy * rowlen + xc], acc);
      }
    } else {
      const int x = xc / 3, c = xc - 3 * x;
      #pragma unroll 4
      for (int k = 0; k < ksize; ++k) {
        const int xx = reflect_idx(x + k - half, s);
        acc = fmaf(w[k], img[(int64_t)y * rowlen + xx * 3 + c], acc);
      }
    }
    dst[i] = acc < 0.f ? 0.f : (acc > 1.f ? 1.f : acc);
  }
}

void launch_aug_blur(const float* img, float* tmp, float* out,
                     const float* sigma, float* wts, int b, int s,
                     int ksize, hipStream_t stream) {
  hipLaunchKernelGGL(blur_weights_kernel, dim3((b + 255) / 256), dim3(256),
                     0, stream, sigma, wts, b, ksize);
  const int64_t total = (int64_t)b * s * s * 3;
  hipLaunchKernelGGL(blur_pass_kernel<true>, dim3(grid_1d(total, 256)),
                     dim3(256), 0, stream, img, tmp, wts, b, s, ksize);
  hipLaunchKernelGGL(blur_pass_kernel<false>, dim3(grid_1d(total, 256)),
                     dim3(256), 0, stream, tmp, out, wts, b, s, ksize);
}
