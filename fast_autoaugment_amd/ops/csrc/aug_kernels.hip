// GPU-resident augmentation pipeline (replaces the reference's 8 CPU
// DataLoader workers doing PIL ops per image, reference data.py:214-224,
// augmentations.py:13-182).
//
// One workgroup processes one image end-to-end:
//   gather from the HBM-resident uint8 dataset (sel index)
//   -> execute the compiled op program (aug/ops.py semantics; all RNG was
//      drawn on host, so execution is deterministic and comparable
//      bit-for-bit with the numpy CPU executor aug/cpu_exec.py)
//   -> pad-crop / hflip / normalize / cutout epilogue
//   -> write bf16 or fp32 NHWC (torch channels_last)
//
// Images <= 64x64 live entirely in LDS (two ping-pong RGBA u32 buffers +
// per-channel histograms); larger images ping-pong through a global
// workspace with the same device functions (flat addressing).
//
// Pixel math matches aug/cpu_exec.py exactly: fp64 affine coords with
// contraction disabled (numpy does separate mul/add), rintf for PIL-blend
// rounding (= np.round half-even), integer histogram/LUT ops.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

constexpr int PROG_SLOTS = 6;
constexpr int PROG_WIDTH = 7;

enum OpCode {
  OP_NOP = 0, OP_AFFINE = 1, OP_AUTOCONTRAST = 2, OP_INVERT = 3, OP_EQUALIZE = 4,
  OP_FLIP = 5, OP_SOLARIZE = 6, OP_POSTERIZE = 7, OP_CONTRAST = 8, OP_COLOR = 9,
  OP_BRIGHTNESS = 10, OP_SHARPNESS = 11, OP_CUTOUT = 12,
  OP_PAIRING = 13,
};

__device__ __forceinline__ uint32_t pack_rgb(int r, int g, int b) {
  return (uint32_t)r | ((uint32_t)g << 8) | ((uint32_t)b << 16);
}
__device__ __forceinline__ int ch_r(uint32_t p) { return p & 0xFF; }
__device__ __forceinline__ int ch_g(uint32_t p) { return (p >> 8) & 0xFF; }
__device__ __forceinline__ int ch_b(uint32_t p) { return (p >> 16) & 0xFF; }

// PIL RGB->L luminance (matches cpu_exec._luminance)
__device__ __forceinline__ int lum(uint32_t p) {
  return (int)((ch_r(p) * 19595u + ch_g(p) * 38470u + ch_b(p) * 7471u + 0x8000u) >> 16);
}

__device__ __forceinline__ int blend1(int deg, int img, float f) {
  // rintf == np.round (round-half-even); clip to [0,255]
  float v = rintf((float)deg + f * ((float)img - (float)deg));
  return (int)fminf(fmaxf(v, 0.0f), 255.0f);
}

// ---------------------------------------------------------------- device ops

__device__ void op_affine(const uint32_t* src, uint32_t* dst, int W, int H,
                          const float* p) {
  #pragma clang fp contract(off)
  double a = p[0], b = p[1], c = p[2], d = p[3], e = p[4], f = p[5];
  for (int i = threadIdx.x; i < W * H; i += blockDim.x) {
    int x = i % W, y = i / W;
    double xo = x + 0.5, yo = y + 0.5;
    double t1 = a * xo;
    double t2 = b * yo;
    double xin = floor(t1 + t2 + c);
    double t3 = d * xo;
    double t4 = e * yo;
    double yin = floor(t3 + t4 + f);
    uint32_t out = 0;
    if (xin >= 0 && xin < W && yin >= 0 && yin < H)
      out = src[(int)yin * W + (int)xin];
    dst[i] = out;
  }
}

__device__ void op_pointwise_lut(const uint32_t* src, uint32_t* dst, int n,
                                 const uint8_t* lut_r, const uint8_t* lut_g,
                                 const uint8_t* lut_b) {
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t px = src[i];
    dst[i] = pack_rgb(lut_r[ch_r(px)], lut_g[ch_g(px)], lut_b[ch_b(px)]);
  }
}

__device__ void build_histogram(const uint32_t* src, int n, uint32_t* hist /*3*256*/) {
  for (int i = threadIdx.x; i < 3 * 256; i += blockDim.x) hist[i] = 0;
  __syncthreads();
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t px = src[i];
    atomicAdd(&hist[ch_r(px)], 1u);
    atomicAdd(&hist[256 + ch_g(px)], 1u);
    atomicAdd(&hist[512 + ch_b(px)], 1u);
  }
  __syncthreads();
}

// autocontrast LUT for one channel (cpu_exec.autocontrast semantics)
__device__ void autocontrast_lut(const uint32_t* hist, uint8_t* lut) {
  int lo = 256, hi = -1;
  for (int i = 0; i < 256; ++i)
    if (hist[i]) { if (lo == 256) lo = i; hi = i; }
  if (hi <= lo) {
    for (int i = 0; i < 256; ++i) lut[i] = (uint8_t)i;
    return;
  }
  double scale = 255.0 / (hi - lo);
  double offset = -lo * scale;
  for (int i = 0; i < 256; ++i) {
    int v = (int)(i * scale + offset);       // trunc toward 0 like .astype(int32)
    lut[i] = (uint8_t)min(255, max(0, v));
  }
}

// equalize LUT for one channel (cpu_exec.equalize / PIL semantics)
__device__ void equalize_lut(const uint32_t* hist, int n, uint8_t* lut) {
  int nonzero = 0, last_nonzero = 0;
  long total = 0;
  for (int i = 0; i < 256; ++i) {
    if (hist[i]) { nonzero++; last_nonzero = hist[i]; }
    total += hist[i];
  }
  if (nonzero <= 1) {
    for (int i = 0; i < 256; ++i) lut[i] = (uint8_t)i;
    return;
  }
  long step = (total - last_nonzero) / 255;
  if (step == 0) {
    for (int i = 0; i < 256; ++i) lut[i] = (uint8_t)i;
    return;
  }
  long acc = step / 2;
  for (int i = 0; i < 256; ++i) {
    long v = acc / step;
    lut[i] = (uint8_t)(v > 255 ? 255 : v);
    acc += hist[i];
  }
}

__device__ void op_blend_const(const uint32_t* src, uint32_t* dst, int n,
                               int dr, int dg, int db, float f) {
  #pragma clang fp contract(off)
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t px = src[i];
    dst[i] = pack_rgb(blend1(dr, ch_r(px), f), blend1(dg, ch_g(px), f),
                      blend1(db, ch_b(px), f));
  }
}

__device__ void op_color(const uint32_t* src, uint32_t* dst, int n, float f) {
  #pragma clang fp contract(off)
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t px = src[i];
    int L = lum(px);
    dst[i] = pack_rgb(blend1(L, ch_r(px), f), blend1(L, ch_g(px), f),
                      blend1(L, ch_b(px), f));
  }
}

__device__ void op_sharpness(const uint32_t* src, uint32_t* dst, int W, int H, float f) {
  #pragma clang fp contract(off)
  for (int i = threadIdx.x; i < W * H; i += blockDim.x) {
    int x = i % W, y = i / W;
    uint32_t px = src[i];
    if (x == 0 || y == 0 || x == W - 1 || y == H - 1) {
      dst[i] = px;   // PIL SMOOTH keeps the 1px border
      continue;
    }
    float accr = 0, accg = 0, accb = 0;
    #pragma unroll
    for (int dy = -1; dy <= 1; ++dy) {
      #pragma unroll
      for (int dx = -1; dx <= 1; ++dx) {
        uint32_t q = src[(y + dy) * W + (x + dx)];
        float wgt = (dx == 0 && dy == 0) ? 5.0f : 1.0f;
        accr += wgt * ch_r(q); accg += wgt * ch_g(q); accb += wgt * ch_b(q);
      }
    }
    int sr = (int)fminf(fmaxf(rintf(accr / 13.0f), 0.0f), 255.0f);
    int sg = (int)fminf(fmaxf(rintf(accg / 13.0f), 0.0f), 255.0f);
    int sb = (int)fminf(fmaxf(rintf(accb / 13.0f), 0.0f), 255.0f);
    dst[i] = pack_rgb(blend1(sr, ch_r(px), f), blend1(sg, ch_g(px), f),
                      blend1(sb, ch_b(px), f));
  }
}

__device__ void op_cutout(const uint32_t* src, uint32_t* dst, int W, int H,
                          const float* p) {
  int x0 = max(0, (int)p[0]), y0 = max(0, (int)p[1]);
  int x1 = min(W - 1, (int)p[2]), y1 = min(H - 1, (int)p[3]);
  uint32_t fill = pack_rgb(125, 123, 114);
  for (int i = threadIdx.x; i < W * H; i += blockDim.x) {
    int x = i % W, y = i / W;
    bool inside = (x >= x0 && x <= x1 && y >= y0 && y <= y1);
    dst[i] = inside ? fill : src[i];
  }
}

// SamplePairing (reference augmentations.py:147-152): blend with another
// batch slot's RAW image. PIL Image.blend = float lerp + clip + truncating
// uint8 cast.
__device__ void op_pairing(const uint32_t* src, uint32_t* dst, int n,
                           const uint8_t* raw2, float alpha) {
  #pragma clang fp contract(off)
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t px = src[i];
    float r = (float)ch_r(px) + alpha * ((float)raw2[i * 3 + 0] - (float)ch_r(px));
    float g = (float)ch_g(px) + alpha * ((float)raw2[i * 3 + 1] - (float)ch_g(px));
    float b = (float)ch_b(px) + alpha * ((float)raw2[i * 3 + 2] - (float)ch_b(px));
    dst[i] = pack_rgb((int)fminf(fmaxf(r, 0.0f), 255.0f),
                      (int)fminf(fmaxf(g, 0.0f), 255.0f),
                      (int)fminf(fmaxf(b, 0.0f), 255.0f));
  }
}

// run one image's op program over ping-pong buffers; returns final src
__device__ uint32_t* run_program(uint32_t* src, uint32_t* dst, int W, int H,
                                 const float* bp, uint32_t* hist, uint8_t* lut,
                                 uint32_t* scalar_acc,
                                 const uint8_t* images, int64_t img_stride,
                                 const int64_t* sel) {
  const int n = W * H;
  for (int s = 0; s < PROG_SLOTS; ++s) {
    const float* slot = bp + s * PROG_WIDTH;
    int code = (int)slot[0];
    const float* p = slot + 1;
    if (code == OP_NOP) continue;
    switch (code) {
      case OP_AFFINE:
        op_affine(src, dst, W, H, p);
        break;
      case OP_AUTOCONTRAST: {
        build_histogram(src, n, hist);
        if (threadIdx.x < 3) autocontrast_lut(hist + threadIdx.x * 256, lut + threadIdx.x * 256);
        __syncthreads();
        op_pointwise_lut(src, dst, n, lut, lut + 256, lut + 512);
        break;
      }
      case OP_EQUALIZE: {
        build_histogram(src, n, hist);
        if (threadIdx.x < 3) equalize_lut(hist + threadIdx.x * 256, n, lut + threadIdx.x * 256);
        __syncthreads();
        op_pointwise_lut(src, dst, n, lut, lut + 256, lut + 512);
        break;
      }
      case OP_INVERT:
        for (int i = threadIdx.x; i < n; i += blockDim.x) {
          uint32_t px = src[i];
          dst[i] = pack_rgb(255 - ch_r(px), 255 - ch_g(px), 255 - ch_b(px));
        }
        break;
      case OP_FLIP:
        for (int i = threadIdx.x; i < n; i += blockDim.x) {
          int x = i % W, y = i / W;
          dst[i] = src[y * W + (W - 1 - x)];
        }
        break;
      case OP_SOLARIZE: {
        float th = p[0];
        for (int i = threadIdx.x; i < n; i += blockDim.x) {
          uint32_t px = src[i];
          int r = ch_r(px), g = ch_g(px), bb = ch_b(px);
          dst[i] = pack_rgb((float)r >= th ? 255 - r : r,
                            (float)g >= th ? 255 - g : g,
                            (float)bb >= th ? 255 - bb : bb);
        }
        break;
      }
      case OP_POSTERIZE: {
        int bits = (int)p[0];
        uint32_t m8 = bits >= 8 ? 0xFFu : (uint32_t)(0xFF & ~((1 << (8 - bits)) - 1));
        uint32_t mask = m8 | (m8 << 8) | (m8 << 16);
        for (int i = threadIdx.x; i < n; i += blockDim.x)
          dst[i] = src[i] & mask;
        break;
      }
      case OP_CONTRAST: {
        if (threadIdx.x == 0) scalar_acc[0] = 0;
        __syncthreads();
        uint32_t part = 0;
        for (int i = threadIdx.x; i < n; i += blockDim.x) part += (uint32_t)lum(src[i]);
        atomicAdd(&scalar_acc[0], part);
        __syncthreads();
        int meanv = (int)((double)scalar_acc[0] / n + 0.5);
        op_blend_const(src, dst, n, meanv, meanv, meanv, p[0]);
        break;
      }
      case OP_COLOR:
        op_color(src, dst, n, p[0]);
        break;
      case OP_BRIGHTNESS:
        op_blend_const(src, dst, n, 0, 0, 0, p[0]);
        break;
      case OP_SHARPNESS:
        op_sharpness(src, dst, W, H, p[0]);
        break;
      case OP_CUTOUT:
        op_cutout(src, dst, W, H, p);
        break;
      case OP_PAIRING:
        op_pairing(src, dst, n, images + sel[(int)p[1]] * img_stride, p[0]);
        break;
      default:
        for (int i = threadIdx.x; i < n; i += blockDim.x) dst[i] = src[i];
    }
    __syncthreads();
    uint32_t* t = src; src = dst; dst = t;
  }
  return src;
}

// ------------------------------------------------------------- main kernel

template <typename OutT, bool IN_LDS>
__global__ void aug_pipeline_kernel(
    const uint8_t* __restrict__ images, int64_t img_stride,
    const int64_t* __restrict__ sel,
    const float* __restrict__ prog,       // [B,6,7]
    const float* __restrict__ post,       // [B,6]
    const float* __restrict__ mean3, const float* __restrict__ std3,
    OutT* __restrict__ out,               // [B,H,W,3] (channels_last)
    uint32_t* __restrict__ gws,           // [B,2,H*W] when !IN_LDS
    int B, int H, int W) {
  extern __shared__ uint32_t smem[];
  const int n = H * W;
  uint32_t* bufA;
  uint32_t* bufB;
  uint32_t* hist;
  __shared__ uint8_t lut[3 * 256];
  __shared__ uint32_t scalar_acc[1];
  if (IN_LDS) {
    bufA = smem;
    bufB = smem + n;
    hist = smem + 2 * n;
  } else {
    hist = smem;
    bufA = gws + (int64_t)blockIdx.x * 2 * n;
    bufB = bufA + n;
  }

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    // ---- gather input image ------------------------------------------------
    const uint8_t* img = images + sel[b] * img_stride;
    for (int i = threadIdx.x; i < n; i += blockDim.x)
      bufA[i] = pack_rgb(img[i * 3], img[i * 3 + 1], img[i * 3 + 2]);
    __syncthreads();

    uint32_t* src = bufA;
    uint32_t* dst = bufB;
    const float* bp = prog + (int64_t)b * PROG_SLOTS * PROG_WIDTH;
    src = run_program(src, dst, W, H, bp, hist, lut, scalar_acc,
                      images, img_stride, sel);
    dst = (src == bufA) ? bufB : bufA;

    // ---- post stage: crop-shift, hflip, normalize, cutout-to-zero ---------
    const float* pp = post + (int64_t)b * 6;
    int dx = (int)pp[0], dy = (int)pp[1];
    bool flip = pp[2] > 0.5f;
    int clen = (int)pp[5];
    int cy1 = 0, cy2 = 0, cx1 = 0, cx2 = 0;
    if (clen > 0) {
      int cx = (int)pp[3], cy = (int)pp[4];
      cy1 = min(max(cy - clen / 2, 0), H); cy2 = min(max(cy + clen / 2, 0), H);
      cx1 = min(max(cx - clen / 2, 0), W); cx2 = min(max(cx + clen / 2, 0), W);
    }
    float m0 = mean3[0], m1 = mean3[1], m2 = mean3[2];
    float s0 = std3[0], s1 = std3[1], s2 = std3[2];
    OutT* orow = out + (int64_t)b * n * 3;
    for (int i = threadIdx.x; i < n; i += blockDim.x) {
      int x = i % W, y = i / W;
      int xx = flip ? (W - 1 - x) : x;
      int sx = xx + dx, sy = y + dy;
      float r = 0, g = 0, bb = 0;
      if (sx >= 0 && sx < W && sy >= 0 && sy < H) {
        uint32_t px = src[sy * W + sx];
        r = (float)ch_r(px); g = (float)ch_g(px); bb = (float)ch_b(px);
      }
      float fr = (r / 255.0f - m0) / s0;
      float fg = (g / 255.0f - m1) / s1;
      float fb = (bb / 255.0f - m2) / s2;
      if (clen > 0 && y >= cy1 && y < cy2 && x >= cx1 && x < cx2) {
        fr = 0.0f; fg = 0.0f; fb = 0.0f;
      }
      faa_from_float(fr, &orow[i * 3 + 0]);
      faa_from_float(fg, &orow[i * 3 + 1]);
      faa_from_float(fb, &orow[i * 3 + 2]);
    }
    __syncthreads();
  }
}


// ------------------------------------------------------- imagenet pipeline
// phase A program ops at source res -> box-crop bicubic resize (+flip) to
// (OH,OW) -> ColorJitter ops -> normalize with lighting-shifted mean.
// post layout: [0]=1, [1:5]=box x0,y0,w,h, [5]=flip, [6:9]=lighting rgb,
// [9:15]=3 x (jitter code, factor).

__device__ __forceinline__ float cubic_pil(float x) {
  x = fabsf(x);
  if (x < 1.0f) return ((1.5f * x - 2.5f) * x) * x + 1.0f;
  if (x < 2.0f) return (((-0.5f * x) + 2.5f) * x - 4.0f) * x + 2.0f;
  return 0.0f;
}

__device__ void op_resize_box(const uint32_t* src, uint32_t* dst, int W, int H,
                              int OW, int OH, float bx0, float by0, float bw,
                              float bh, bool flip) {
  #pragma clang fp contract(off)
  float sx_scale = bw / OW, sy_scale = bh / OH;
  float ssx = sx_scale > 1.0f ? 1.0f / sx_scale : 1.0f;
  float ssy = sy_scale > 1.0f ? 1.0f / sy_scale : 1.0f;
  float supx = 2.0f * fmaxf(sx_scale, 1.0f);
  float supy = 2.0f * fmaxf(sy_scale, 1.0f);
  int ix0 = (int)bx0, iy0 = (int)by0, ibw = (int)bw, ibh = (int)bh;
  for (int i = threadIdx.x; i < OW * OH; i += blockDim.x) {
    int ox = i % OW, oy = i / OW;
    int sxp = flip ? (OW - 1 - ox) : ox;
    float cx = bx0 + (sxp + 0.5f) * sx_scale;
    float cy = by0 + (oy + 0.5f) * sy_scale;
    int xmin = (int)(cx - supx + 0.5f); if (xmin < ix0) xmin = ix0;
    int xmax = (int)(cx + supx + 0.5f); if (xmax > ix0 + ibw) xmax = ix0 + ibw;
    int ymin = (int)(cy - supy + 0.5f); if (ymin < iy0) ymin = iy0;
    int ymax = (int)(cy + supy + 0.5f); if (ymax > iy0 + ibh) ymax = iy0 + ibh;
    float accr = 0, accg = 0, accb = 0, wsum = 0;
    for (int y = ymin; y < ymax; ++y) {
      float wy = cubic_pil((y + 0.5f - cy) * ssy);
      for (int x = xmin; x < xmax; ++x) {
        float w = wy * cubic_pil((x + 0.5f - cx) * ssx);
        uint32_t px = src[y * W + x];
        accr += w * ch_r(px); accg += w * ch_g(px); accb += w * ch_b(px);
        wsum += w;
      }
    }
    float inv = wsum != 0.0f ? 1.0f / wsum : 0.0f;
    int r = (int)fminf(fmaxf(rintf(accr * inv), 0.0f), 255.0f);
    int gch = (int)fminf(fmaxf(rintf(accg * inv), 0.0f), 255.0f);
    int bb = (int)fminf(fmaxf(rintf(accb * inv), 0.0f), 255.0f);
    dst[i] = pack_rgb(r, gch, bb);
  }
}

template <typename OutT>
__global__ void aug_pipeline_in_kernel(
    const uint8_t* __restrict__ images, int64_t img_stride,
    const int64_t* __restrict__ sel,
    const float* __restrict__ prog,        // [B,6,7]
    const float* __restrict__ post,        // [B,18]
    const float* __restrict__ mean3, const float* __restrict__ std3,
    OutT* __restrict__ out,                // [B,OH,OW,3] channels_last
    uint32_t* __restrict__ gws,            // [gridDim, 2, max(HW,OHW)]
    int B, int H, int W, int OH, int OW) {
  extern __shared__ uint32_t smem[];
  uint32_t* hist = smem;
  __shared__ uint8_t lut[3 * 256];
  __shared__ uint32_t scalar_acc[1];
  const int n = H * W;
  const int no = OH * OW;
  const int nmax = n > no ? n : no;
  uint32_t* bufA = gws + (int64_t)blockIdx.x * 2 * nmax;
  uint32_t* bufB = bufA + nmax;

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const uint8_t* img = images + sel[b] * img_stride;
    for (int i = threadIdx.x; i < n; i += blockDim.x)
      bufA[i] = pack_rgb(img[i * 3], img[i * 3 + 1], img[i * 3 + 2]);
    __syncthreads();

    const float* bp = prog + (int64_t)b * PROG_SLOTS * PROG_WIDTH;
    uint32_t* src = run_program(bufA, bufB, W, H, bp, hist, lut, scalar_acc,
                                images, img_stride, sel);
    uint32_t* dst = (src == bufA) ? bufB : bufA;

    const float* pp = post + (int64_t)b * 18;
    op_resize_box(src, dst, W, H, OW, OH, pp[1], pp[2], pp[3], pp[4], pp[5] > 0.5f);
    __syncthreads();
    { uint32_t* t = src; src = dst; dst = t; }

    // jitter ops on the resized image
    for (int sjit = 0; sjit < 3; ++sjit) {
      int code = (int)pp[9 + sjit * 2];
      float f = pp[10 + sjit * 2];
      if (code == OP_BRIGHTNESS) {
        op_blend_const(src, dst, no, 0, 0, 0, f);
      } else if (code == OP_CONTRAST) {
        if (threadIdx.x == 0) scalar_acc[0] = 0;
        __syncthreads();
        uint32_t part = 0;
        for (int i = threadIdx.x; i < no; i += blockDim.x) part += (uint32_t)lum(src[i]);
        atomicAdd(&scalar_acc[0], part);
        __syncthreads();
        int meanv = (int)((double)scalar_acc[0] / no + 0.5);
        op_blend_const(src, dst, no, meanv, meanv, meanv, f);
      } else if (code == OP_COLOR) {
        op_color(src, dst, no, f);
      } else {
        continue;
      }
      __syncthreads();
      uint32_t* t = src; src = dst; dst = t;
    }

    // normalize with lighting folded into the mean
    float m0 = mean3[0] - pp[6], m1 = mean3[1] - pp[7], m2 = mean3[2] - pp[8];
    float s0 = std3[0], s1 = std3[1], s2 = std3[2];
    OutT* orow = out + (int64_t)b * no * 3;
    for (int i = threadIdx.x; i < no; i += blockDim.x) {
      uint32_t px = src[i];
      faa_from_float((ch_r(px) / 255.0f - m0) / s0, &orow[i * 3 + 0]);
      faa_from_float((ch_g(px) / 255.0f - m1) / s1, &orow[i * 3 + 1]);
      faa_from_float((ch_b(px) / 255.0f - m2) / s2, &orow[i * 3 + 2]);
    }
    __syncthreads();
  }
}

}  // namespace

torch::Tensor aug_pipeline_imagenet(torch::Tensor images, torch::Tensor sel,
                                    torch::Tensor prog, torch::Tensor post,
                                    torch::Tensor mean, torch::Tensor std,
                                    int64_t out_h, int64_t out_w, bool bf16_out) {
  TORCH_CHECK(images.is_cuda() && images.dtype() == torch::kUInt8 && images.dim() == 4);
  TORCH_CHECK(post.size(1) == 18, "imagenet post width 18 expected");
  int H = images.size(1), W = images.size(2);
  int B = sel.size(0);
  auto sel_c = sel.to(torch::kInt64).contiguous();
  auto prog_c = prog.contiguous();
  auto post_c = post.contiguous();
  auto mean_c = mean.to(torch::kFloat32).contiguous();
  auto std_c = std.to(torch::kFloat32).contiguous();
  auto opts = images.options().dtype(bf16_out ? torch::kBFloat16 : torch::kFloat32);
  auto out = torch::empty({B, 3, out_h, out_w},
                          opts.memory_format(torch::MemoryFormat::ChannelsLast));
  int64_t nmax = std::max<int64_t>((int64_t)H * W, out_h * out_w);
  int grid = B;
  auto gws = torch::empty({(int64_t)grid, 2, nmax}, images.options().dtype(torch::kInt32));
  size_t lds = 3 * 256 * 4;
  auto stream = at::hip::getCurrentHIPStream().stream();
  if (bf16_out) {
    hipLaunchKernelGGL((aug_pipeline_in_kernel<__hip_bfloat16>), dim3(grid), dim3(256),
                       lds, stream, images.data_ptr<uint8_t>(), images.stride(0),
                       sel_c.data_ptr<int64_t>(), prog_c.data_ptr<float>(),
                       post_c.data_ptr<float>(), mean_c.data_ptr<float>(),
                       std_c.data_ptr<float>(), (__hip_bfloat16*)out.data_ptr(),
                       (uint32_t*)gws.data_ptr(), B, H, W, (int)out_h, (int)out_w);
  } else {
    hipLaunchKernelGGL((aug_pipeline_in_kernel<float>), dim3(grid), dim3(256),
                       lds, stream, images.data_ptr<uint8_t>(), images.stride(0),
                       sel_c.data_ptr<int64_t>(), prog_c.data_ptr<float>(),
                       post_c.data_ptr<float>(), mean_c.data_ptr<float>(),
                       std_c.data_ptr<float>(), (float*)out.data_ptr(),
                       (uint32_t*)gws.data_ptr(), B, H, W, (int)out_h, (int)out_w);
  }
  return out;
}

torch::Tensor aug_pipeline(torch::Tensor images, torch::Tensor sel, torch::Tensor prog,
                           torch::Tensor post, torch::Tensor mean, torch::Tensor std,
                           bool bf16_out) {
  TORCH_CHECK(images.is_cuda() && images.dtype() == torch::kUInt8 && images.dim() == 4,
              "aug_pipeline: uint8 [N,H,W,3] images expected");
  TORCH_CHECK(images.is_contiguous());
  int H = images.size(1), W = images.size(2);
  TORCH_CHECK(images.size(3) == 3);
  int B = sel.size(0);
  auto sel_c = sel.to(torch::kInt64).contiguous();
  auto prog_c = prog.contiguous();
  auto post_c = post.contiguous();
  auto mean_c = mean.to(torch::kFloat32).contiguous();
  auto std_c = std.to(torch::kFloat32).contiguous();

  auto opts = images.options().dtype(bf16_out ? torch::kBFloat16 : torch::kFloat32);
  // channels_last [B,3,H,W]: storage order is NHWC, which the kernel writes
  auto out = torch::empty({B, 3, H, W}, opts.memory_format(torch::MemoryFormat::ChannelsLast));

  int n = H * W;
  bool in_lds = n <= 64 * 64;
  int block = 256;
  int grid = B;
  auto stream = at::hip::getCurrentHIPStream().stream();
  size_t lds = in_lds ? (size_t)(2 * n + 3 * 256) * 4 : (size_t)(3 * 256) * 4;

  torch::Tensor gws;
  uint32_t* gws_ptr = nullptr;
  if (!in_lds) {
    gws = torch::empty({(int64_t)grid, 2, n}, images.options().dtype(torch::kInt32));
    gws_ptr = (uint32_t*)gws.data_ptr();
  }

  #define LAUNCH(OutT, INLDS)                                                     \
    hipLaunchKernelGGL((aug_pipeline_kernel<OutT, INLDS>), dim3(grid), dim3(block), \
                       lds, stream, images.data_ptr<uint8_t>(),                   \
                       images.stride(0), sel_c.data_ptr<int64_t>(),               \
                       prog_c.data_ptr<float>(), post_c.data_ptr<float>(),        \
                       mean_c.data_ptr<float>(), std_c.data_ptr<float>(),         \
                       (OutT*)out.data_ptr(), gws_ptr, B, H, W)

  if (bf16_out) {
    if (in_lds) LAUNCH(__hip_bfloat16, true); else LAUNCH(__hip_bfloat16, false);
  } else {
    if (in_lds) LAUNCH(float, true); else LAUNCH(float, false);
  }
  #undef LAUNCH
  return out;
}
