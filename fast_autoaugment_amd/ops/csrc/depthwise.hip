// Depthwise convolution (groups == C), NHWC bf16, k in {3,5}, stride 1/2,
// asymmetric TF-SAME padding (pt/pl passed explicitly).
//
// MIOpen/CK route EfficientNet's depthwise wgrad to a grouped-GEMM kernel
// that takes ~42 ms per call on MI355X (profiles/prof_effnet summary, 88% of
// the B0 step). Depthwise work is memory-bound elementwise-with-taps — these
// kernels stream NHWC vectors with per-thread 8-channel accumulators:
//   fwd:   Y[m,c]  = sum_taps X[tap(m),c] * W[c,tap]          (+bias)
//   bwdD:  dX[m,c] = sum_taps dY[inv_tap(m),c] * W[c,tap]     (stride-aware)
//   bwdW:  dW[c,tap] = sum_m X[tap(m),c] * dY[m,c]            (per-block LDS
//          fold -> fp32 atomics -> cast back to [C][KH][KW])
// Weight memory layout: [C,1,KH,KW] is [C][KH][KW] in both torch layouts.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16v8;

__device__ __forceinline__ float dw_b2f(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}
__device__ __forceinline__ short dw_f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

struct DwGeom {
  int B, H, W, C, Ho, Wo, KH, KW, stride, pt, pl;
};

template <bool HAS_BIAS>
__global__ void dw_fwd_kernel(const short* __restrict__ X, const short* __restrict__ Wt,
                              const short* __restrict__ bias, short* __restrict__ Y,
                              DwGeom g) {
  int64_t total = (int64_t)g.B * g.Ho * g.Wo * g.C;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    int c0 = (int)(i % g.C);
    if (c0 + 8 > g.C) continue;        // host guarantees C % 8 == 0
    int64_t m = i / g.C;
    int wo = (int)(m % g.Wo);
    int64_t t = m / g.Wo;
    int ho = (int)(t % g.Ho);
    int b = (int)(t / g.Ho);
    float acc[8] = {};
    for (int kh = 0; kh < g.KH; ++kh) {
      int hi = ho * g.stride - g.pt + kh;
      if (hi < 0 || hi >= g.H) continue;
      for (int kw = 0; kw < g.KW; ++kw) {
        int wi = wo * g.stride - g.pl + kw;
        if (wi < 0 || wi >= g.W) continue;
        bf16v8 xv = *reinterpret_cast<const bf16v8*>(
            X + (((int64_t)b * g.H + hi) * g.W + wi) * g.C + c0);
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] += dw_b2f(xv[j]) * dw_b2f(Wt[(int64_t)(c0 + j) * g.KH * g.KW + kh * g.KW + kw]);
      }
    }
    bf16v8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = acc[j];
      if (HAS_BIAS) v += dw_b2f(bias[c0 + j]);
      ov[j] = dw_f2b(v);
    }
    *reinterpret_cast<bf16v8*>(Y + i) = ov;
  }
}

// Specialized fwd: compile-time K/stride, channel-octet-invariant grid
// (host aligns gridDim*256 to a multiple of C/8) so each thread's 8-channel
// weight tap set loads ONCE into registers; interior pixels skip bounds
// checks; for stride 1 each thread produces PW=2 adjacent outputs from one
// sliding window (column loads shared between the pair).
template <bool HAS_BIAS, int KH, int KW, int S, int PW>
__global__ void dw_fwd_tpl_kernel(const short* __restrict__ X, const short* __restrict__ Wt,
                                  const short* __restrict__ bias, short* __restrict__ Y,
                                  DwGeom g) {
  const int octs = g.C >> 3;
  const int wop = (g.Wo + PW - 1) / PW;
  const int64_t items = (int64_t)g.B * g.Ho * wop * octs;
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t gstride = (int64_t)gridDim.x * blockDim.x;
  const int oct = (int)(i0 % octs);       // loop-invariant: gstride % octs == 0
  const int c0 = oct << 3;

  bf16v8 wv[KH * KW];
  #pragma unroll
  for (int t = 0; t < KH * KW; ++t) {
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      wv[t][j] = Wt[(int64_t)(c0 + j) * KH * KW + t];
  }
  float bv[8] = {};
  if (HAS_BIAS) {
    bf16v8 b8 = *reinterpret_cast<const bf16v8*>(bias + c0);
    #pragma unroll
    for (int j = 0; j < 8; ++j) bv[j] = dw_b2f(b8[j]);
  }

  for (int64_t i = i0; i < items; i += gstride) {
    int64_t m = i / octs;
    int wo = (int)(m % wop) * PW;
    int64_t t2 = m / wop;
    int ho = (int)(t2 % g.Ho);
    int b = (int)(t2 / g.Ho);
    int hi0 = ho * S - g.pt;
    int wi0 = wo * S - g.pl;
    float acc0[8] = {}, acc1[8] = {};
    bool pair = (PW == 2) && (wo + 1 < g.Wo);
    bool interior = hi0 >= 0 && hi0 + KH <= g.H && wi0 >= 0
                    && wi0 + KW + (PW - 1) * S <= g.W && (PW == 1 || pair);
    if (interior) {
      const short* base = X + (((int64_t)b * g.H + hi0) * g.W + wi0) * g.C + c0;
      #pragma unroll
      for (int kh = 0; kh < KH; ++kh) {
        #pragma unroll
        for (int col = 0; col < KW + (PW - 1) * S; ++col) {
          bf16v8 xv = *reinterpret_cast<const bf16v8*>(
              base + ((int64_t)kh * g.W + col) * g.C);
          if (col < KW) {
            #pragma unroll
            for (int j = 0; j < 8; ++j)
              acc0[j] += dw_b2f(xv[j]) * dw_b2f(wv[kh * KW + col][j]);
          }
          if (PW == 2 && col >= S && col - S < KW) {
            #pragma unroll
            for (int j = 0; j < 8; ++j)
              acc1[j] += dw_b2f(xv[j]) * dw_b2f(wv[kh * KW + col - S][j]);
          }
        }
      }
    } else {
      #pragma unroll
      for (int kh = 0; kh < KH; ++kh) {
        int hi = hi0 + kh;
        if (hi < 0 || hi >= g.H) continue;
        #pragma unroll
        for (int kw = 0; kw < KW; ++kw) {
          int wi = wi0 + kw;
          if (wi >= 0 && wi < g.W) {
            bf16v8 xv = *reinterpret_cast<const bf16v8*>(
                X + (((int64_t)b * g.H + hi) * g.W + wi) * g.C + c0);
            #pragma unroll
            for (int j = 0; j < 8; ++j)
              acc0[j] += dw_b2f(xv[j]) * dw_b2f(wv[kh * KW + kw][j]);
          }
          if (PW == 2 && pair) {
            int wi1 = wi + S;
            if (wi1 >= 0 && wi1 < g.W) {
              bf16v8 xv = *reinterpret_cast<const bf16v8*>(
                  X + (((int64_t)b * g.H + hi) * g.W + wi1) * g.C + c0);
              #pragma unroll
              for (int j = 0; j < 8; ++j)
                acc1[j] += dw_b2f(xv[j]) * dw_b2f(wv[kh * KW + kw][j]);
            }
          }
        }
      }
    }
    short* out = Y + (((int64_t)b * g.Ho + ho) * g.Wo + wo) * g.C + c0;
    bf16v8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = dw_f2b(acc0[j] + (HAS_BIAS ? bv[j] : 0.0f));
    *reinterpret_cast<bf16v8*>(out) = ov;
    if (PW == 2 && pair) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = dw_f2b(acc1[j] + (HAS_BIAS ? bv[j] : 0.0f));
      *reinterpret_cast<bf16v8*>(out + g.C) = ov;
    }
  }
}

// v3: LDS-staged input tile. A block owns ONE channel octet and a TxT
// output tile of one image: the (T*S + K-1)^2 x 8ch input window is staged
// in LDS once (every input byte read ~once from HBM vs ~K^2/pw times in
// the register-sliding kernels — round-1 measured those ~9x off the HBM
// roofline), the 8ch x K^2 weights live in LDS (broadcast reads), and
// each thread does K^2 LDS vec-reads per output.
template <bool HAS_BIAS, int K, int S, int T>
__global__ __launch_bounds__(256)
void dw_fwd_v3_kernel(const short* __restrict__ X, const short* __restrict__ Wt,
                      const short* __restrict__ bias, short* __restrict__ Y,
                      DwGeom g, int tiles_h, int tiles_w) {
  constexpr int IT = T * S + K - 1;           // staged input tile edge
  __shared__ short ldsX[IT * IT * 8];
  __shared__ short ldsW[K * K * 8];
  __shared__ float ldsB[8];

  int bid = blockIdx.x;
  const int tx = bid % tiles_w; bid /= tiles_w;
  const int ty = bid % tiles_h; bid /= tiles_h;
  const int oct = bid % (g.C >> 3);
  const int b = bid / (g.C >> 3);
  const int c0 = oct << 3;
  const int tid = threadIdx.x;

  const int hi0 = ty * T * S - g.pt;
  const int wi0 = tx * T * S - g.pl;
  // stage input window (zero-filled out of bounds)
  for (int v = tid; v < IT * IT; v += 256) {
    int r = v / IT, c = v % IT;
    int hi = hi0 + r, wi = wi0 + c;
    bf16v8 xv = {0, 0, 0, 0, 0, 0, 0, 0};
    if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
      xv = *reinterpret_cast<const bf16v8*>(
          X + (((int64_t)b * g.H + hi) * g.W + wi) * g.C + c0);
    *reinterpret_cast<bf16v8*>(&ldsX[v * 8]) = xv;
  }
  if (tid < K * K) {
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      ldsW[tid * 8 + j] = Wt[(int64_t)(c0 + j) * K * K + tid];
  }
  if (HAS_BIAS && tid < 8) ldsB[tid] = dw_b2f(bias[c0 + tid]);
  __syncthreads();

  for (int p = tid; p < T * T; p += 256) {
    int hl = p / T, wl = p % T;
    int ho = ty * T + hl, wo = tx * T + wl;
    if (ho >= g.Ho || wo >= g.Wo) continue;
    float acc[8] = {};
    #pragma unroll
    for (int kh = 0; kh < K; ++kh) {
      #pragma unroll
      for (int kw = 0; kw < K; ++kw) {
        bf16v8 xv = *reinterpret_cast<const bf16v8*>(
            &ldsX[((hl * S + kh) * IT + wl * S + kw) * 8]);
        const short* wv = &ldsW[(kh * K + kw) * 8];
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] += dw_b2f(xv[j]) * dw_b2f(wv[j]);
      }
    }
    bf16v8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      ov[j] = dw_f2b(acc[j] + (HAS_BIAS ? ldsB[j] : 0.0f));
    *reinterpret_cast<bf16v8*>(
        Y + (((int64_t)b * g.Ho + ho) * g.Wo + wo) * g.C + c0) = ov;
  }
}

__global__ void dw_bwd_data_kernel(const short* __restrict__ dY, const short* __restrict__ Wt,
                                   short* __restrict__ dX, DwGeom g) {
  int64_t total = (int64_t)g.B * g.H * g.W * g.C;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    int c0 = (int)(i % g.C);
    if (c0 + 8 > g.C) continue;
    int64_t m = i / g.C;
    int wi = (int)(m % g.W);
    int64_t t = m / g.W;
    int hi = (int)(t % g.H);
    int b = (int)(t / g.H);
    float acc[8] = {};
    for (int kh = 0; kh < g.KH; ++kh) {
      int hnum = hi + g.pt - kh;
      if (hnum < 0 || hnum % g.stride) continue;
      int ho = hnum / g.stride;
      if (ho >= g.Ho) continue;
      for (int kw = 0; kw < g.KW; ++kw) {
        int wnum = wi + g.pl - kw;
        if (wnum < 0 || wnum % g.stride) continue;
        int wo = wnum / g.stride;
        if (wo >= g.Wo) continue;
        bf16v8 dv = *reinterpret_cast<const bf16v8*>(
            dY + (((int64_t)b * g.Ho + ho) * g.Wo + wo) * g.C + c0);
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] += dw_b2f(dv[j]) * dw_b2f(Wt[(int64_t)(c0 + j) * g.KH * g.KW + kh * g.KW + kw]);
      }
    }
    bf16v8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = dw_f2b(acc[j]);
    *reinterpret_cast<bf16v8*>(dX + i) = ov;
  }
}

// one (kh,kw) tap per blockIdx.y; blockIdx.x strides the output positions.
// per-thread 8-channel register accumulators -> LDS fold -> fp32 atomics.
__global__ void dw_wrw_kernel(const short* __restrict__ X, const short* __restrict__ dY,
                              float* __restrict__ dWacc, DwGeom g) {
  int cell = blockIdx.y;
  int kh = cell / g.KW, kw = cell - (cell / g.KW) * g.KW;
  int64_t total = (int64_t)g.B * g.Ho * g.Wo * g.C;
  float s[8] = {};
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    int c0 = (int)(i % g.C);
    if (c0 + 8 > g.C) continue;
    int64_t m = i / g.C;
    int wo = (int)(m % g.Wo);
    int64_t t = m / g.Wo;
    int ho = (int)(t % g.Ho);
    int b = (int)(t / g.Ho);
    int hi = ho * g.stride - g.pt + kh;
    int wi = wo * g.stride - g.pl + kw;
    if (hi < 0 || hi >= g.H || wi < 0 || wi >= g.W) continue;
    bf16v8 xv = *reinterpret_cast<const bf16v8*>(
        X + (((int64_t)b * g.H + hi) * g.W + wi) * g.C + c0);
    bf16v8 dv = *reinterpret_cast<const bf16v8*>(dY + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += dw_b2f(xv[j]) * dw_b2f(dv[j]);
  }
  // fold by channel octet (stride-aligned grid: see host)
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = s[j];
  __syncthreads();
  const int groups = g.C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  if ((int)threadIdx.x < g.C && threadIdx.x < 256) {
    int oct = threadIdx.x / 8, lane = threadIdx.x % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int tt = t0; tt < (int)blockDim.x; tt += groups)
      acc += lds[tt * 8 + lane];
    atomicAdd(&dWacc[(int64_t)cell * g.C + threadIdx.x], acc);
  }
}

// tail channels (C > 256): scalar accumulation kernel, no LDS fold
__global__ void dw_wrw_tail_kernel(const short* __restrict__ X, const short* __restrict__ dY,
                                   float* __restrict__ dWacc, DwGeom g, int c_start) {
  int cell = blockIdx.y;
  int kh = cell / g.KW, kw = cell - (cell / g.KW) * g.KW;
  int c = c_start + blockIdx.z * blockDim.x + threadIdx.x;
  if (c >= g.C) return;
  float s = 0;
  for (int64_t m = blockIdx.x; m < (int64_t)g.B * g.Ho * g.Wo; m += gridDim.x) {
    int wo = (int)(m % g.Wo);
    int64_t t = m / g.Wo;
    int ho = (int)(t % g.Ho);
    int b = (int)(t / g.Ho);
    int hi = ho * g.stride - g.pt + kh;
    int wi = wo * g.stride - g.pl + kw;
    if (hi < 0 || hi >= g.H || wi < 0 || wi >= g.W) continue;
    s += dw_b2f(X[(((int64_t)b * g.H + hi) * g.W + wi) * g.C + c])
         * dw_b2f(dY[m * g.C + c]);
  }
  atomicAdd(&dWacc[(int64_t)cell * g.C + c], s);
}

__global__ void dw_wrw_cast_kernel(const float* __restrict__ dWacc, short* __restrict__ dW,
                                   int C, int cells) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= C * cells) return;
  int c = i / cells, cell = i % cells;
  dW[i] = dw_f2b(dWacc[(int64_t)cell * C + c]);   // out layout [C][KH][KW]
}

}  // namespace

static DwGeom dw_geom(const torch::Tensor& x, int KH, int KW, int64_t stride,
                      int64_t pt, int64_t pl, int Ho, int Wo) {
  DwGeom g;
  g.B = x.size(0); g.C = x.size(1); g.H = x.size(2); g.W = x.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pt = pt; g.pl = pl;
  g.Ho = Ho; g.Wo = Wo;
  return g;
}

torch::Tensor dwconv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pt, int64_t pb, int64_t pl, int64_t pr) {
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wc = w.contiguous();   // [C,1,KH,KW] -> [C][KH][KW] either layout
  int KH = w.size(2), KW = w.size(3);
  int Ho = (xc.size(2) + pt + pb - KH) / stride + 1;
  int Wo = (xc.size(3) + pl + pr - KW) / stride + 1;
  DwGeom g = dw_geom(xc, KH, KW, stride, pt, pl, Ho, Wo);
  TORCH_CHECK(g.C % 8 == 0, "dwconv: C % 8 == 0");
  auto y = torch::empty({g.B, g.C, Ho, Wo},
                        xc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)g.B * Ho * Wo * g.C;
  auto stream = at::hip::getCurrentHIPStream().stream();
  bool hb = bias.defined() && bias.numel() > 0;
  torch::Tensor bc;
  const short* bp = nullptr;
  if (hb) { bc = bias.contiguous(); bp = (const short*)bc.data_ptr(); }

  // v3 LDS-staged tile path (default; FAA_DW_V3=0 falls back)
  const char* v3e = getenv("FAA_DW_V3");
  const bool v3_off = v3e && v3e[0] == '0';
  if (!v3_off && (KH == KW) && (KH == 3 || KH == 5)
      && (stride == 1 || stride == 2)) {
    int T = (Ho >= 12 && Wo >= 12) ? 16 : 8;
    int tiles_h = (Ho + T - 1) / T;
    int tiles_w = (Wo + T - 1) / T;
    int64_t grid = (int64_t)g.B * (g.C / 8) * tiles_h * tiles_w;
    #define DWV3(K_, S_, T_)                                                    \
      do {                                                                      \
        if (hb)                                                                 \
          hipLaunchKernelGGL((dw_fwd_v3_kernel<true, K_, S_, T_>),              \
                             dim3((unsigned)grid), dim3(256), 0, stream,        \
                             (const short*)xc.data_ptr(),                       \
                             (const short*)wc.data_ptr(), bp,                   \
                             (short*)y.data_ptr(), g, tiles_h, tiles_w);        \
        else                                                                    \
          hipLaunchKernelGGL((dw_fwd_v3_kernel<false, K_, S_, T_>),             \
                             dim3((unsigned)grid), dim3(256), 0, stream,        \
                             (const short*)xc.data_ptr(),                       \
                             (const short*)wc.data_ptr(), nullptr,              \
                             (short*)y.data_ptr(), g, tiles_h, tiles_w);        \
      } while (0)
    if (KH == 3 && stride == 1) { if (T == 16) DWV3(3, 1, 16); else DWV3(3, 1, 8); }
    else if (KH == 3)           { if (T == 16) DWV3(3, 2, 16); else DWV3(3, 2, 8); }
    else if (stride == 1)       { if (T == 16) DWV3(5, 1, 16); else DWV3(5, 1, 8); }
    else                        { if (T == 16) DWV3(5, 2, 16); else DWV3(5, 2, 8); }
    #undef DWV3
    return y;
  }

  // specialized path: K in {3,5}, stride in {1,2}; octet-invariant grid.
  // k=5 keeps 25 bf16x8 weight vectors in registers (~100 VGPRs) which cuts
  // occupancy below what the latency-bound loop needs — measured slower on
  // EfficientNet-B0 — so k=5 stays on the generic kernel unless FAA_DW_TPL5=1.
  const char* t5e = getenv("FAA_DW_TPL5");
  const bool tpl5 = t5e && t5e[0] == '1';
  bool tpl = (KH == KW) && (KH == 3 || (KH == 5 && tpl5)) && (stride == 1 || stride == 2);
  if (tpl) {
    auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
    int octs = g.C / 8;
    int pw = (stride == 1) ? 2 : 1;
    int64_t items = (int64_t)g.B * Ho * ((Wo + pw - 1) / pw) * octs;
    int q = octs / gcd(octs, 256);
    int base = (int)std::min<int64_t>(std::max<int64_t>((items + 255) / 256, 1), 2048);
    int nb = ((base + q - 1) / q) * q;
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nb), dim3(256), 0, stream,
                         (const short*)xc.data_ptr(), (const short*)wc.data_ptr(), bp,
                         (short*)y.data_ptr(), g);
    };
    if (KH == 3 && stride == 1) launch(hb ? dw_fwd_tpl_kernel<true, 3, 3, 1, 2>
                                          : dw_fwd_tpl_kernel<false, 3, 3, 1, 2>);
    else if (KH == 3)           launch(hb ? dw_fwd_tpl_kernel<true, 3, 3, 2, 1>
                                          : dw_fwd_tpl_kernel<false, 3, 3, 2, 1>);
    else if (stride == 1)       launch(hb ? dw_fwd_tpl_kernel<true, 5, 5, 1, 2>
                                          : dw_fwd_tpl_kernel<false, 5, 5, 1, 2>);
    else                        launch(hb ? dw_fwd_tpl_kernel<true, 5, 5, 2, 1>
                                          : dw_fwd_tpl_kernel<false, 5, 5, 2, 1>);
    return y;
  }

  int grid = faa_grid(total / 8 + 1, 256);
  if (hb)
    hipLaunchKernelGGL((dw_fwd_kernel<true>), dim3(grid), dim3(256), 0, stream,
                       (const short*)xc.data_ptr(), (const short*)wc.data_ptr(), bp,
                       (short*)y.data_ptr(), g);
  else
    hipLaunchKernelGGL((dw_fwd_kernel<false>), dim3(grid), dim3(256), 0, stream,
                       (const short*)xc.data_ptr(), (const short*)wc.data_ptr(), nullptr,
                       (short*)y.data_ptr(), g);
  return y;
}

torch::Tensor dwconv_bwd_data(torch::Tensor dy, torch::Tensor w, int64_t stride,
                              int64_t pt, int64_t pl, int64_t H, int64_t W) {
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wc = w.contiguous();
  int KH = w.size(2), KW = w.size(3);
  int C = dyc.size(1), B = dyc.size(0);
  DwGeom g;
  g.B = B; g.C = C; g.H = H; g.W = W; g.KH = KH; g.KW = KW;
  g.stride = stride; g.pt = pt; g.pl = pl; g.Ho = dyc.size(2); g.Wo = dyc.size(3);
  TORCH_CHECK(C % 8 == 0);
  auto dx = torch::empty({B, C, H, W},
                         dyc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)B * H * W * C;
  int grid = faa_grid(total / 8 + 1, 256);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(dw_bwd_data_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)dyc.data_ptr(), (const short*)wc.data_ptr(),
                     (short*)dx.data_ptr(), g);
  return dx;
}

torch::Tensor dwconv_bwd_weight(torch::Tensor dy, torch::Tensor x, int64_t stride,
                                int64_t pt, int64_t pl, int64_t KH, int64_t KW) {
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = xc.size(1);
  TORCH_CHECK(C % 8 == 0);
  DwGeom g = dw_geom(xc, KH, KW, stride, pt, pl, dyc.size(2), dyc.size(3));
  int cells = KH * KW;
  auto f32 = xc.options().dtype(torch::kFloat32);
  auto acc = torch::zeros({cells, C}, f32);
  auto stream = at::hip::getCurrentHIPStream().stream();
  {
    // stride-aligned x-grid so each thread's channel octet is fixed
    auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
    int q = C / gcd(C, 2048);
    int nb = ((64 + q - 1) / q) * q;
    dim3 grid(nb, cells);
    hipLaunchKernelGGL(dw_wrw_kernel, grid, dim3(256), 0, stream,
                       (const short*)xc.data_ptr(), (const short*)dyc.data_ptr(),
                       acc.data_ptr<float>(), g);
  }
  if (C > 256) {
    dim3 grid(128, cells, (C - 256 + 255) / 256);
    hipLaunchKernelGGL(dw_wrw_tail_kernel, grid, dim3(256), 0, stream,
                       (const short*)xc.data_ptr(), (const short*)dyc.data_ptr(),
                       acc.data_ptr<float>(), g, 256);
  }
  auto dw = torch::empty({C, 1, KH, KW}, xc.options());
  hipLaunchKernelGGL(dw_wrw_cast_kernel, dim3((C * cells + 255) / 256), dim3(256), 0,
                     stream, acc.data_ptr<float>(), (short*)dw.data_ptr(), C, cells);
  return dw;
}
