// MFMA implicit-GEMM convolution for NHWC bf16 (CDNA4 / gfx950).
//
// The reference delegates all conv work to cuDNN; MIOpen's
// igemm kernels on these CIFAR shapes (3x3/1x1, C<=640, 8..32px, batch 128)
// run 20-40us each plus SubTensor/cast helper launches. These kernels map
// the conv directly onto mfma_f32_16x16x32_bf16 with LDS-staged tiles:
//
//   fwd:       Y[m=(b,ho,wo)][n=cout] = sum_k A[m][k=(kh,kw,ci)] * W[n][k]
//              (weights in torch channels_last layout [Cout][KH][KW][Cin]
//               ARE [n][k] with k contiguous - no repack needed); bias fused.
//   bwd-data:  same kernel on dY with flipped/transposed weights (repacked
//              by a tiny kernel); stride-1 only (s=2 falls back to MIOpen).
//   bwd-weight:dW[k][n] = sum_m A[m][k] * dY[m][n], split-K over m-slices
//              with fp32 atomic accumulation, then cast back to the torch
//              weight layout. dbias via a column-sum kernel.
//
// Tiles: BM=64 x BN=64 x BK=32, 4 waves/block (each wave a 32x32 C-tile of
// 2x2 16x16 fragments), LDS rows padded to 40 bf16 to spread banks,
// double-buffered. A-operand per-lane layout: row=lane%16, 8 contiguous k at
// (lane/16)*8; B likewise with n=lane%16 (verified against torch in
// tests/test_gpu_conv.py with asymmetric data per the CDNA4 guide G9).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BK = 32;   // K-slab depth (BM/BN are per-launch templates)
constexpr int LDSP = 40;             // padded LDS row stride (elems)

__device__ __forceinline__ float b2f(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}
__device__ __forceinline__ short f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

struct ConvGeom {
  int B, H, Wd, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int cin_chunks;      // ceil(Cin/8)
  int kpad;            // KH*KW*cin_chunks*8
  int groups = 1;      // grouped conv (direct tiled kernel only)
};

// load 8 input channels (zero-filled out of bounds / tail) as bf16x8
__device__ __forceinline__ bf16x8 load_x8(const short* __restrict__ X,
                                          const ConvGeom g, int b, int hi, int wi,
                                          int ci0) {
  bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
  if (b < g.B && hi >= 0 && hi < g.H && wi >= 0 && wi < g.Wd && ci0 < g.Cin) {
    const short* p = X + (((int64_t)b * g.H + hi) * g.Wd + wi) * g.Cin + ci0;
    if (ci0 + 8 <= g.Cin) {
      v = *reinterpret_cast<const bf16x8*>(p);
    } else {
      for (int j = 0; j < g.Cin - ci0; ++j) v[j] = p[j];
    }
  }
  return v;
}

// ------------------------------------------------------------------- fwd

// SPLIT=false: each workgroup owns a full K loop and writes bf16 Y directly.
// SPLIT=true (occupancy aid for deep stages whose tile grid underfills the
// 256 CUs): blockIdx.y partitions the K tiles; partial C-tiles accumulate
// into the fp32 workspace Yacc with atomics and a bias+cast kernel finishes.
template <bool HAS_BIAS, int BMT, int BNT, bool SPLIT = false>
__global__ __launch_bounds__(256)
void conv_fwd_kernel(const short* __restrict__ X, const short* __restrict__ Wt,
                     const short* __restrict__ bias, short* __restrict__ Y,
                     ConvGeom g, int M, int grid_m, float* __restrict__ Yacc = nullptr) {
  // 4 waves in a fixed 2x2 grid; per-wave sub-tile (BMT/2) x (BNT/2),
  // i.e. MFRAG=BMT/32 x NFRAG=BNT/32 fragments of 16x16. Smaller tiles
  // multiply the workgroup count for shapes that underfill 256 CUs
  // (profiles/pmc_conv_r01.txt: occupancy-bound at BM=64 on 8x8 stages).
  constexpr int MFRAG = BMT / 32;
  constexpr int NFRAG = BNT / 32;
  __shared__ short ldsA[2][BMT * LDSP];
  __shared__ short ldsB[2][BNT * LDSP];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    if (q > 0) wg = (xcd < r) ? (xcd * (q + 1) + idx) : (r * (q + 1) + (xcd - r) * q + idx);
  }
  int bm = wg % grid_m;
  int bn = wg / grid_m;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * (BMT / 2);
  const int wc = (wave & 1) * (BNT / 2);

  // A staging: BMT rows x 4 k-chunks; threads beyond BMT*4 idle on A
  const int a_row = (tid >> 2) % BMT;
  const bool a_act = (tid >> 2) < BMT;
  const int a_kc = (tid & 3);
  const int m_g = bm * BMT + a_row;
  int xb = 0, xho = 0, xwo = 0;
  if (a_act && m_g < M) {
    xb = m_g / (g.Ho * g.Wo);
    int rem = m_g - xb * (g.Ho * g.Wo);
    xho = rem / g.Wo;
    xwo = rem - xho * g.Wo;
  } else {
    xb = g.B;
  }
  const int b_row = (tid >> 2) % BNT;
  const bool b_act = (tid >> 2) < BNT;
  const int n_g = bn * BNT + b_row;

  const int nk = g.kpad / BK;
  int kt_lo = 0, kt_hi = nk;
  if (SPLIT) {
    int per = (nk + gridDim.y - 1) / gridDim.y;
    kt_lo = blockIdx.y * per;
    kt_hi = min(nk, kt_lo + per);
    if (kt_lo >= kt_hi) return;
  }
  f32x4 acc[MFRAG][NFRAG] = {};
  const int CELLS = g.KH * g.KW;

  auto load_a = [&](int kt) -> bf16x8 {
    bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (!a_act) return v;
    int kc = kt * 4 + a_kc;
    int cell = kc / g.cin_chunks;
    int ci0 = (kc - cell * g.cin_chunks) * 8;
    if (cell < CELLS) {
      int kh = cell / g.KW, kw = cell - (cell / g.KW) * g.KW;
      int hi = xho * g.stride - g.pad + kh;
      int wi = xwo * g.stride - g.pad + kw;
      v = load_x8(X, g, xb, hi, wi, ci0);
    }
    return v;
  };
  auto load_b = [&](int kt) -> bf16x8 {
    bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
    if (!b_act) return v;
    int kc = kt * 4 + a_kc;
    int cell = kc / g.cin_chunks;
    int ci0 = (kc - cell * g.cin_chunks) * 8;
    if (n_g < g.Cout && ci0 < g.Cin && cell < CELLS) {
      const short* p = Wt + ((int64_t)n_g * CELLS + cell) * g.Cin + ci0;
      if (ci0 + 8 <= g.Cin) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
        for (int j = 0; j < g.Cin - ci0; ++j) v[j] = p[j];
      }
    }
    return v;
  };
  auto write_lds = [&](int buf, bf16x8 va, bf16x8 vb) {
    if (a_act)
      *reinterpret_cast<bf16x8*>(&ldsA[buf][a_row * LDSP + a_kc * 8]) = va;
    if (b_act)
      *reinterpret_cast<bf16x8*>(&ldsB[buf][b_row * LDSP + a_kc * 8]) = vb;
  };

  write_lds(0, load_a(kt_lo), load_b(kt_lo));
  __syncthreads();

  const int fr = lane & 15;
  const int kq = (lane >> 4) * 8;

  for (int kt = kt_lo; kt < kt_hi; ++kt) {
    int buf = (kt - kt_lo) & 1;
    bf16x8 na = {}, nb = {};
    if (kt + 1 < kt_hi) {   // issue next-tile global loads BEFORE the MFMAs
      na = load_a(kt + 1);
      nb = load_b(kt + 1);
    }
    bf16x8 afrag[MFRAG], bfrag[NFRAG];
    #pragma unroll
    for (int f = 0; f < MFRAG; ++f)
      afrag[f] = *reinterpret_cast<const bf16x8*>(&ldsA[buf][(wr + f * 16 + fr) * LDSP + kq]);
    #pragma unroll
    for (int f = 0; f < NFRAG; ++f)
      bfrag[f] = *reinterpret_cast<const bf16x8*>(&ldsB[buf][(wc + f * 16 + fr) * LDSP + kq]);
    #pragma unroll
    for (int fm = 0; fm < MFRAG; ++fm)
      #pragma unroll
      for (int fn = 0; fn < NFRAG; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[fm], bfrag[fn],
                                                              acc[fm][fn], 0, 0, 0);
    if (kt + 1 < kt_hi) {
      __syncthreads();
      write_lds(buf ^ 1, na, nb);
    }
    __syncthreads();
  }

  #pragma unroll
  for (int fm = 0; fm < MFRAG; ++fm) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m_loc = wr + fm * 16 + (lane >> 4) * 4 + r;
      int m = bm * BMT + m_loc;
      if (m >= M) continue;
      #pragma unroll
      for (int fn = 0; fn < NFRAG; ++fn) {
        int n = bn * BNT + wc + fn * 16 + fr;
        if (n >= g.Cout) continue;
        float v = acc[fm][fn][r];
        if (SPLIT) {
          atomicAdd(&Yacc[(int64_t)m * g.Cout + n], v);
        } else {
          if (HAS_BIAS) v += b2f(bias[n]);
          Y[(int64_t)m * g.Cout + n] = f2b(v);
        }
      }
    }
  }
}

// ------------------------------------------------- direct tiled 3x3 fwd
// conv3x3_tile_kernel: stride-1 pad-1 3x3 conv, NHWC bf16, for the CIFAR
// geometries (W in {8,16,32}, H % 8 == 0, any Cin >= 16, any Cout).
//
// Rationale vs the im2col conv_fwd_kernel above (profiles r01: 24us at
// 2.4% MfmaUtil): implicit GEMM re-reads every input pixel 9x through
// global/L2 and spends the latency-bound inner loop on per-load integer
// division. Here each block stages its (TH+2)x(TW+2) input tile (+halo)
// in LDS ONCE per cin-slab and the 9 taps become compile-time LDS
// offsets; weights live in registers (one bf16x8 per (tap, nfrag) lane
// slot). Per cin-slab a wave issues 9*MFRAG*2 MFMAs between two
// syncthreads vs 2 in the im2col loop.
//
//  * m-tile = IB*TH*TW pixels (= WM*MR*16); wave grid WM x WN over (m, n)
//  * n-tile = WN*NR*16 couts, grid_n = ceil(Cout / n-tile)
//    small-C config: WN=1 NR=2 (BN=32, max blocks); big-C config: WN=2
//    NR=2 (BN=64, halves per-block weight traffic, measured win at C>64
//    when the grid still fills the 256 CUs)
//  * K loop = cin slabs of CS=32 (zero-padded tail), 9 taps each
//  * LDS pixel stride PS=40 shorts: 16B-aligned ds_read_b128 and 2-way-
//    max bank aliasing (fr*20 mod 32 covers 8 banks x 2 lanes = free)
template <int TH, int TW, int IB, int WN, int MR, int NR, bool HAS_BIAS,
          bool SPLIT = false>
__global__ __launch_bounds__(256)
void conv3x3_tile_kernel(const short* __restrict__ X, const short* __restrict__ Wt,
                         const short* __restrict__ bias, short* __restrict__ Y,
                         ConvGeom g, int tiles_h, int grid_n,
                         float* __restrict__ Yacc = nullptr) {
  constexpr int CS = 32;             // cin slab depth (one MFMA k)
  constexpr int PS = 40;             // per-pixel LDS stride in shorts
  constexpr int XR = TH + 2;
  constexpr int XC = TW + 2;
  constexpr int WM = 4 / WN;         // wave grid: WM (m) x WN (n)
  static_assert(IB * TH * TW == WM * MR * 16, "m tile mismatch");

  __shared__ short ldsX[IB * XR * XC * PS];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {  // XCD-aware remap: consecutive n-tiles land on one XCD's L2
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    if (q > 0) wg = (xcd < r) ? (xcd * (q + 1) + idx) : (r * (q + 1) + (xcd - r) * q + idx);
  }
  const int nt = wg % grid_n;
  int t = wg / grid_n;
  const int ty = t % tiles_h;
  const int img0 = (t / tiles_h) * IB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave / WN;
  const int wn = wave % WN;
  const int fr = lane & 15;
  const int kq = (lane >> 4) * 8;
  const int n0 = nt * (WN * NR * 16) + wn * (NR * 16);

  int aoff[MR];
  #pragma unroll
  for (int mf = 0; mf < MR; ++mf) {
    int m_loc = wm * (MR * 16) + mf * 16 + fr;
    int ib = m_loc / (TH * TW);
    int pix = m_loc % (TH * TW);
    int py = pix / TW, px = pix % TW;
    aoff[mf] = ((ib * XR + py) * XC + px) * PS + kq;
  }

  f32x4 acc[MR][NR] = {};
  const int row0 = ty * TH - 1;      // pad = 1

  // grouped conv: this n-tile's group selects a cin slice; the weight's
  // inner dim is cin_per (host guarantees the n-tile never spans groups)
  const int cin_per = (g.groups > 1) ? (g.Cin / g.groups) : g.Cin;
  const int cbase = (g.groups > 1) ? (n0 / (g.Cout / g.groups)) * cin_per : 0;

  int cs_lo = 0, cs_hi = cin_per;
  if (SPLIT) {                       // blockIdx.y partitions the cin slabs
    int slabs = (cin_per + CS - 1) / CS;
    int per = (slabs + gridDim.y - 1) / gridDim.y;
    cs_lo = blockIdx.y * per * CS;
    cs_hi = min(cin_per, cs_lo + per * CS);
    if (cs_lo >= cs_hi) return;
  }
  for (int cs = cs_lo; cs < cs_hi; cs += CS) {
    // stage the X tile slab (borders/tails zero-filled)
    constexpr int NV = IB * XR * XC * 4;     // bf16x8 stores
    for (int v = tid; v < NV; v += 256) {
      int kc = v & 3;
      int cell = v >> 2;
      int col = cell % XC;
      int rowt = cell / XC;
      int row = rowt % XR;
      int ib = rowt / XR;
      int hi = row0 + row;
      int wi = col - 1;
      int ci0 = cs + kc * 8;
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.Wd && ci0 < cin_per) {
        const short* p = X + ((((int64_t)(img0 + ib)) * g.H + hi) * g.Wd + wi) * g.Cin
                         + cbase + ci0;
        if (ci0 + 8 <= cin_per) {
          val = *reinterpret_cast<const bf16x8*>(p);
        } else {
          for (int j = 0; j < cin_per - ci0; ++j) val[j] = p[j];
        }
      }
      *reinterpret_cast<bf16x8*>(&ldsX[cell * PS + kc * 8]) = val;
    }
    // weights slab -> registers (per-lane B fragments, no LDS round-trip)
    bf16x8 wreg[9][NR];
    #pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      #pragma unroll
      for (int nf = 0; nf < NR; ++nf) {
        int n = n0 + nf * 16 + fr;
        int ci0 = cs + kq;
        bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (n < g.Cout && ci0 < cin_per) {
          const short* p = Wt + ((int64_t)n * 9 + tap) * cin_per + ci0;
          if (ci0 + 8 <= cin_per) {
            v = *reinterpret_cast<const bf16x8*>(p);
          } else {
            for (int j = 0; j < cin_per - ci0; ++j) v[j] = p[j];
          }
        }
        wreg[tap][nf] = v;
      }
    }
    __syncthreads();
    #pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      #pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int toff = (kh * XC + kw) * PS;   // compile-time per (kh,kw)
        #pragma unroll
        for (int mf = 0; mf < MR; ++mf) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(&ldsX[aoff[mf] + toff]);
          #pragma unroll
          for (int nf = 0; nf < NR; ++nf)
            acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, wreg[kh * 3 + kw][nf], acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  #pragma unroll
  for (int mf = 0; mf < MR; ++mf) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      int m_loc = wm * (MR * 16) + mf * 16 + (lane >> 4) * 4 + r;
      int ib = m_loc / (TH * TW);
      int pix = m_loc % (TH * TW);
      int py = pix / TW, px = pix % TW;
      int ho = ty * TH + py;
      if (ho >= g.Ho || px >= g.Wo) continue;   // masked tail tiles
      int64_t base = ((((int64_t)(img0 + ib)) * g.Ho + ho) * g.Wo + px) * g.Cout;
      #pragma unroll
      for (int nf = 0; nf < NR; ++nf) {
        int n = n0 + nf * 16 + fr;
        if (n < g.Cout) {
          float v = acc[mf][nf][r];
          if (SPLIT) {
            int64_t m = ((int64_t)(img0 + ib) * g.Ho + ho) * g.Wo + px;
            atomicAdd(&Yacc[m * g.Cout + n], v);
          } else {
            if (HAS_BIAS) v += b2f(bias[n]);
            Y[base + n] = f2b(v);
          }
        }
      }
    }
  }
}

// bias-add + fp32 -> bf16 cast for the split-K path
template <bool HAS_BIAS>
__global__ void conv_splitk_cast_kernel(const float* __restrict__ Yacc,
                                        const short* __restrict__ bias,
                                        short* __restrict__ Y, int64_t total, int Cout) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  float v = Yacc[i];
  if (HAS_BIAS) v += b2f(bias[(int)(i % Cout)]);
  Y[i] = f2b(v);
}

// ------------------------------------------------- weight repack (bwd-data)
// W[co][kh][kw][ci] -> W2[ci][KH-1-kh][KW-1-kw][co]
__global__ void weight_flip_kernel(const short* __restrict__ W, short* __restrict__ W2,
                                   int Cout, int KH, int KW, int Cin) {
  int64_t total = (int64_t)Cout * KH * KW * Cin;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  int ci = i % Cin;
  int64_t t = i / Cin;
  int kw = t % KW; t /= KW;
  int kh = t % KH; t /= KH;
  int co = t;
  W2[(((int64_t)ci * KH + (KH - 1 - kh)) * KW + (KW - 1 - kw)) * Cout + co] = W[i];
}

// ------------------------------------------ stride-2 bwd-data (k3 s2 p1)
// Phase decomposition (no dilated-zero waste): dx[y,x] with output parity
// (py,px) = (y&1, x&1) is a DENSE stride-1 correlation of dy with the
// sub-filter taps {kh : kh === (y+1) mod 2} x {kw : ...}: even coords use
// the single kh=1 tap, odd coords the {0,2} pair. The 4 phases map 1:1
// onto the 4 waves of a block (each wave's m-fragments stay phase-uniform
// so MFMA K is uniform per fragment). dy tile staged once per cin-slab in
// LDS; flipped weights W2[ci][fh][fw][co] (fh = 2-kh, the conv_flip_all
// layout) read to registers.
//   geom g here: Cin = dy channels (orig Cout), Cout = dx channels,
//   H/Wd = dx dims, Ho/Wo = dy dims.
template <int TH, int TW>
__global__ __launch_bounds__(256)
void conv3x3s2_dx_kernel(const short* __restrict__ dY, const short* __restrict__ W2,
                         short* __restrict__ dX, ConvGeom g,
                         int tiles_h, int grid_n) {
  constexpr int CS = 32;
  constexpr int PS = 40;
  constexpr int XR = TH / 2 + 1;       // dy rows per tile (+1 forward halo)
  constexpr int XC = TW / 2 + 1;
  constexpr int MT = TH * TW;          // dx pixels per block
  constexpr int MR = MT / 64;          // fragments per wave (wave == phase)
  static_assert(MT % 64 == 0, "tile must give whole fragments per phase");

  __shared__ short ldsY[XR * XC * PS];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    if (q > 0) wg = (xcd < r) ? (xcd * (q + 1) + idx) : (r * (q + 1) + (xcd - r) * q + idx);
  }
  const int nt = wg % grid_n;
  int t = wg / grid_n;
  const int ty = t % tiles_h;
  const int b = t / tiles_h;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // == phase index (py*2+px)
  const int py = wave >> 1;
  const int px = wave & 1;
  const int fr = lane & 15;
  const int kq = (lane >> 4) * 8;
  const int n0 = nt * 32;

  // per-fragment LDS base: lane's dx pixel inside the phase quarter
  int aoff[MR];
  #pragma unroll
  for (int mf = 0; mf < MR; ++mf) {
    int rest = mf * 16 + fr;
    int ry = rest / (TW / 2);
    int rx = rest % (TW / 2);
    aoff[mf] = (ry * XC + rx) * PS + kq;
  }

  // taps for this wave's phase: doy/dox are the +0/+1 dy offsets and
  // (fh,fw) index the FLIPPED weight layout (fh = 2-kh)
  int tap_doy[2], tap_fh[2], nty;
  if (py == 0) { nty = 1; tap_doy[0] = 0; tap_fh[0] = 1; }
  else { nty = 2; tap_doy[0] = 1; tap_fh[0] = 2; tap_doy[1] = 0; tap_fh[1] = 0; }
  int tap_dox[2], tap_fw[2], ntx;
  if (px == 0) { ntx = 1; tap_dox[0] = 0; tap_fw[0] = 1; }
  else { ntx = 2; tap_dox[0] = 1; tap_fw[0] = 2; tap_dox[1] = 0; tap_fw[1] = 0; }

  f32x4 acc[MR][2] = {};
  const int oy0 = (ty * TH) >> 1;      // first dy row of this tile

  for (int cs = 0; cs < g.Cin; cs += CS) {
    // stage the dy tile slab (zero-filled beyond Ho/Wo and channel tail)
    constexpr int NV = XR * XC * 4;
    for (int v = tid; v < NV; v += 256) {
      int kc = v & 3;
      int cell = v >> 2;
      int col = cell % XC;
      int row = cell / XC;
      int oy = oy0 + row;
      int ci0 = cs + kc * 8;
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      if (oy < g.Ho && col < g.Wo && ci0 < g.Cin) {
        const short* p = dY + (((int64_t)b * g.Ho + oy) * g.Wo + col) * g.Cin + ci0;
        if (ci0 + 8 <= g.Cin) {
          val = *reinterpret_cast<const bf16x8*>(p);
        } else {
          for (int j = 0; j < g.Cin - ci0; ++j) val[j] = p[j];
        }
      }
      *reinterpret_cast<bf16x8*>(&ldsY[cell * PS + kc * 8]) = val;
    }
    // flipped-weight fragments for this phase's taps
    bf16x8 wreg[4][2];
    for (int tyi = 0; tyi < nty; ++tyi) {
      for (int txi = 0; txi < ntx; ++txi) {
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          int n = n0 + nf * 16 + fr;   // dx channel
          int ci0 = cs + kq;
          bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
          if (n < g.Cout && ci0 < g.Cin) {
            const short* p = W2 + (((int64_t)n * 3 + tap_fh[tyi]) * 3
                                   + tap_fw[txi]) * g.Cin + ci0;
            if (ci0 + 8 <= g.Cin) {
              v = *reinterpret_cast<const bf16x8*>(p);
            } else {
              for (int j = 0; j < g.Cin - ci0; ++j) v[j] = p[j];
            }
          }
          wreg[tyi * 2 + txi][nf] = v;
        }
      }
    }
    __syncthreads();
    for (int tyi = 0; tyi < nty; ++tyi) {
      for (int txi = 0; txi < ntx; ++txi) {
        const int toff = (tap_doy[tyi] * XC + tap_dox[txi]) * PS;
        #pragma unroll
        for (int mf = 0; mf < MR; ++mf) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(&ldsY[aoff[mf] + toff]);
          #pragma unroll
          for (int nf = 0; nf < 2; ++nf)
            acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a, wreg[tyi * 2 + txi][nf], acc[mf][nf], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  #pragma unroll
  for (int mf = 0; mf < MR; ++mf) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      int rest = mf * 16 + (lane >> 4) * 4 + r;
      int ry = rest / (TW / 2);
      int rx = rest % (TW / 2);
      int y = ty * TH + 2 * ry + py;
      int x = 2 * rx + px;
      int64_t base = (((int64_t)b * g.H + y) * g.Wd + x) * g.Cout;
      #pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int n = n0 + nf * 16 + fr;
        if (n < g.Cout)
          dX[base + n] = f2b(acc[mf][nf][r]);
      }
    }
  }
}

// batched weight flip: one launch refreshes EVERY conv's bwd-data repack
// (vs 31 x 4.8us weight_flip launches per step in the round-1 profile).
// table rows: [src_ptr, dst_ptr, Cout, KH, KW, Cin, elem_offset]
__global__ void flip_weights_batched_kernel(const int64_t* __restrict__ table,
                                            int nrows, int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    // binary search the row whose [offset, offset+numel) contains i
    int lo = 0, hi = nrows - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (table[mid * 7 + 6] <= i) lo = mid; else hi = mid - 1;
    }
    const int64_t* r = table + lo * 7;
    const short* W = (const short*)r[0];
    short* W2 = (short*)r[1];
    int Cout = (int)r[2], KH = (int)r[3], KW = (int)r[4], Cin = (int)r[5];
    int64_t j = i - r[6];
    int ci = j % Cin;
    int64_t t = j / Cin;
    int kw = t % KW; t /= KW;
    int kh = t % KH; t /= KH;
    int co = (int)t;
    W2[(((int64_t)ci * KH + (KH - 1 - kh)) * KW + (KW - 1 - kw)) * Cout + co] = W[j];
  }
}

// ------------------------------------------------------------- bwd-weight
// dW_acc[k=(kh,kw,ci)][n=co] += sum_m A[m][k] * dY[m][n]  (fp32 atomics)
// block tile: 32 k-rows x 64 co, m-chunks of 32, split over blockIdx.z
__global__ __launch_bounds__(256)
void conv_wrw_kernel(const short* __restrict__ X, const short* __restrict__ dY,
                     float* __restrict__ dWacc, ConvGeom g, int M, int m_slices) {
  __shared__ short ldsA[32 * 40];      // [k'=32][m=32] padded
  __shared__ short ldsB[64 * 40];      // [co=64][m=32] padded

  const int kt = blockIdx.x;           // k-tile (32 rows of kpad)
  const int nt = blockIdx.y;           // cout tile (64)
  const int slice = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  int64_t m_per = (M + m_slices - 1) / m_slices;
  int64_t m0 = (int64_t)slice * m_per;
  int64_t m1 = min((int64_t)M, m0 + m_per);

  f32x4 acc[2] = {};                   // wave covers [2x16 k'] x [16 co]
  const int fr = lane & 15;
  const int kq = (lane >> 4) * 8;

  for (int64_t mc = m0; mc < m1; mc += 32) {
    // ---- stage A^T: [k'][m] ; threads: 128 threads x 8 = 1024 elems
    if (tid < 128) {
      int m_loc = tid >> 2;            // 0..31
      int kc8 = tid & 3;               // which 8-chunk of k'
      int64_t m = mc + m_loc;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (m < m1) {
        int b = (int)(m / (g.Ho * g.Wo));
        int rem = (int)(m - (int64_t)b * g.Ho * g.Wo);
        int ho = rem / g.Wo, wo = rem - (rem / g.Wo) * g.Wo;
        int kc = kt * 4 + kc8;
        int cell = kc / g.cin_chunks;
        int ci0 = (kc - cell * g.cin_chunks) * 8;
        if (cell < g.KH * g.KW) {
          int kh = cell / g.KW, kw = cell - (cell / g.KW) * g.KW;
          int hi = ho * g.stride - g.pad + kh;
          int wi = wo * g.stride - g.pad + kw;
          v = load_x8(X, g, b, hi, wi, ci0);
        }
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        ldsA[(kc8 * 8 + j) * 40 + m_loc] = v[j];
    } else {
      // ---- stage dY^T: [co][m]; 128 threads x 16 = 2048 elems
      int t = tid - 128;
      int m_loc = t >> 2;              // 0..31
      int cg = t & 3;                  // 16-chunk of cout
      int64_t m = mc + m_loc;
      short v[16] = {};
      if (m < m1) {
        const short* p = dY + m * g.Cout + nt * 64 + cg * 16;
        if (nt * 64 + cg * 16 + 16 <= g.Cout && (g.Cout % 8) == 0) {
          *reinterpret_cast<bf16x8*>(&v[0]) = *reinterpret_cast<const bf16x8*>(p);
          *reinterpret_cast<bf16x8*>(&v[8]) = *reinterpret_cast<const bf16x8*>(p + 8);
        } else {
          for (int j = 0; j < 16; ++j) {
            int n = nt * 64 + cg * 16 + j;
            v[j] = (n < g.Cout) ? p[j] : (short)0;
          }
        }
      }
      #pragma unroll
      for (int j = 0; j < 16; ++j)
        ldsB[(cg * 16 + j) * 40 + m_loc] = v[j];
    }
    __syncthreads();

    // wave w handles co block [w*16, w*16+16), k' rows [0..32)
    bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(&ldsB[(wave * 16 + fr) * 40 + kq]);
    #pragma unroll
    for (int fk = 0; fk < 2; ++fk) {
      bf16x8 afrag = *reinterpret_cast<const bf16x8*>(&ldsA[(fk * 16 + fr) * 40 + kq]);
      acc[fk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[fk], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: D[k'][co]: k' = fk*16 + (lane>>4)*4 + r, co = wave*16 + fr
  #pragma unroll
  for (int fk = 0; fk < 2; ++fk) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      int kp = kt * 32 + fk * 16 + (lane >> 4) * 4 + r;
      int n = nt * 64 + wave * 16 + fr;
      if (kp < g.kpad && n < g.Cout)
        atomicAdd(&dWacc[(int64_t)kp * g.Cout + n], acc[fk][r]);
    }
  }
}

// ------------------------------------------------- wrw v3 (direct tiled)
// dW_tap[ci][co] = sum_p X[p + off_tap][ci] * dY[p][co] — contraction over
// OUTPUT pixels p. The round-1 conv_wrw_kernel (im2col split-K, 32-deep m
// chunks) loses 1.2-4x to MIOpen at C >= 160: per 32-m chunk it re-stages
// both operands and issues only 3 MFMAs/wave. Here a block stages a
// TRANSPOSED X tile (with halo) and dY tile once and contracts PIX_OUT
// pixels for ALL taps from LDS: per 32-pixel chunk a wave issues
// KK*KK*MR*NR MFMAs with 2 fresh LDS b-frags.
//
//  * block tile: 64 ci x 64 co, waves 2x2 (each 32x32), acc[KK*KK][2][2]
//  * pixel tiles: full image rows (TW = W), TH rows; grid.y m-slices
//  * LDS rows padded to wrw3_pitch: 16B-aligned and 2-way-max bank step
//  * fp32 atomicAdd into dWacc[tap][Cin][Cout]; cast kernel reorders to
//    the torch channels_last weight layout
constexpr int wrw3_pitch(int n) {
  for (int p = (n + 7) / 8 * 8;; p += 8) {
    int step = (p / 2) % 32;
    if (step == 4 || step == 12 || step == 20 || step == 28) return p;
  }
}

// WSPLIT=4: small-C variant (Cin,Cout <= 32). The 2x2 wave quadrants of
// the 64x64 tile would leave 3 of 4 waves entirely out of range, so all
// waves map to quadrant (0,0) and SPLIT the pixel chunks instead; their
// duplicate-position partial sums merge through the same epilogue atomics.
template <int TH, int TW, int KK, int WSPLIT = 1>
__global__ __launch_bounds__(256)
void conv_wrw3_kernel(const short* __restrict__ X, const short* __restrict__ dY,
                      float* __restrict__ dWacc, ConvGeom g,
                      int ci_tiles, int co_tiles, int tiles_per_slice) {
  constexpr int XR = (KK == 3) ? TH + 2 : TH;
  constexpr int XC = (KK == 3) ? TW + 2 : TW;
  constexpr int PIX_IN = XR * XC;
  constexpr int PIX_OUT = TH * TW;
  // row-dependent swizzle (multiples of 8 shorts = 16B): the staging
  // scatter writes 8 rows (kc*8+j, j fixed) per instruction, whose bank
  // step without a swizzle is a multiple of 32 -> up to 16-way conflicts.
  // shift(row) spreads those rows over 8 bank groups while keeping the
  // contiguous 8-pixel fragment reads aligned.
  constexpr int XTP = wrw3_pitch(PIX_IN + 56);
  constexpr int YTP = wrw3_pitch(PIX_OUT + 56);
  static_assert(PIX_OUT % 32 == 0, "pixel tile must be a multiple of 32");

  __shared__ short ldsXT[64 * XTP];
  __shared__ short ldsYT[64 * YTP];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    if (q > 0) wg = (xcd < r) ? (xcd * (q + 1) + idx) : (r * (q + 1) + (xcd - r) * q + idx);
  }
  const int ci_t = wg % ci_tiles;
  const int co_t = wg / ci_tiles;
  const int ci0g = ci_t * 64;
  const int co0g = co_t * 64;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (WSPLIT == 1) ? (wave >> 1) : 0;   // ci half
  const int wn = (WSPLIT == 1) ? (wave & 1) : 0;    // co half
  const int fr = lane & 15;
  const int kq = (lane >> 4) * 8;

  const int tiles_h = g.H / TH;
  const int tiles_total = g.B * tiles_h;
  int t0 = blockIdx.y * tiles_per_slice;
  int t1 = min(tiles_total, t0 + tiles_per_slice);

  f32x4 acc[KK * KK][2][2] = {};

  for (int tile = t0; tile < t1; ++tile) {
    const int b = tile / tiles_h;
    const int ty = tile % tiles_h;
    const int row0 = ty * TH - (KK == 3 ? 1 : 0);
    // ---- stage X^T [ci][input pixel] (zero-filled borders/tails)
    for (int v = tid; v < PIX_IN * 8; v += 256) {
      int kc = v & 7;
      int cell = v >> 3;
      int row = cell / XC, col = cell % XC;
      int hi = row0 + row;
      int wi = col - (KK == 3 ? 1 : 0);
      int ci0 = ci0g + kc * 8;
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.Wd && ci0 < g.Cin) {
        const short* p = X + (((int64_t)b * g.H + hi) * g.Wd + wi) * g.Cin + ci0;
        if (ci0 + 8 <= g.Cin) {
          val = *reinterpret_cast<const bf16x8*>(p);
        } else {
          for (int j = 0; j < g.Cin - ci0; ++j) val[j] = p[j];
        }
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = kc * 8 + j;
        int sh = (((row >> 3) + (row & 7)) & 7) * 8;
        ldsXT[row * XTP + sh + cell] = val[j];
      }
    }
    // ---- stage dY^T [co][output pixel]
    for (int v = tid; v < PIX_OUT * 8; v += 256) {
      int kc = v & 7;
      int cell = v >> 3;
      int row = cell / TW, col = cell % TW;
      int ho = ty * TH + row;
      int co0 = co0g + kc * 8;
      bf16x8 val = {0, 0, 0, 0, 0, 0, 0, 0};
      if (ho < g.Ho && co0 < g.Cout) {
        const short* p = dY + (((int64_t)b * g.Ho + ho) * g.Wo + col) * g.Cout + co0;
        if (co0 + 8 <= g.Cout) {
          val = *reinterpret_cast<const bf16x8*>(p);
        } else {
          for (int j = 0; j < g.Cout - co0; ++j) val[j] = p[j];
        }
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = kc * 8 + j;
        int sh = (((row >> 3) + (row & 7)) & 7) * 8;
        ldsYT[row * YTP + sh + cell] = val[j];
      }
    }
    __syncthreads();

    #pragma unroll
    for (int pc = (WSPLIT == 1 ? 0 : wave); pc < PIX_OUT / 32;
         pc += (WSPLIT == 1 ? 1 : 4)) {
      const int p8 = pc * 32 + kq;       // this lane's first pixel
      const int py = p8 / TW, px = p8 % TW;
      bf16x8 bfrag[2];
      #pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        int row = wn * 32 + nf * 16 + fr;
        int sh = (((row >> 3) + (row & 7)) & 7) * 8;
        bfrag[nf] = *reinterpret_cast<const bf16x8*>(
            &ldsYT[row * YTP + sh + pc * 32 + kq]);
      }
      #pragma unroll
      for (int kh = 0; kh < KK; ++kh) {
        #pragma unroll
        for (int kw = 0; kw < KK; ++kw) {
          const int ip = (py + kh) * XC + (px + kw);
          #pragma unroll
          for (int mf = 0; mf < 2; ++mf) {
            int arow = wm * 32 + mf * 16 + fr;
            int ash = (((arow >> 3) + (arow & 7)) & 7) * 8;
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &ldsXT[arow * XTP + ash + ip]);
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf)
              acc[kh * KK + kw][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a, bfrag[nf], acc[kh * KK + kw][mf][nf], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  if (WSPLIT == 4) {
    // the 4 waves hold duplicate-position partials: fold them through LDS
    // (reusing ldsXT, 256 lanes x 8 floats = 8 KB per round, within the
    // smallest instantiation's 9.2 KB) so only wave 0 issues atomics
    float* red = reinterpret_cast<float*>(ldsXT);
    for (int tap = 0; tap < KK * KK; ++tap) {
      for (int mf = 0; mf < 2; ++mf) {
        __syncthreads();
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          #pragma unroll
          for (int r = 0; r < 4; ++r)
            red[(wave * 64 + lane) * 8 + nf * 4 + r] = acc[tap][mf][nf][r];
        __syncthreads();
        if (wave == 0) {
          #pragma unroll
          for (int nf = 0; nf < 2; ++nf)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
              int j = nf * 4 + r;
              acc[tap][mf][nf][r] = red[lane * 8 + j]
                  + red[(64 + lane) * 8 + j]
                  + red[(128 + lane) * 8 + j]
                  + red[(192 + lane) * 8 + j];
            }
        }
      }
    }
    __syncthreads();
    if (wave != 0) return;
  }

  // epilogue: one atomicAdd per accumulator element
  #pragma unroll
  for (int tap = 0; tap < KK * KK; ++tap) {
    #pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int ci = ci0g + wm * 32 + mf * 16 + (lane >> 4) * 4 + r;
        if (ci >= g.Cin) continue;
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
          int co = co0g + wn * 32 + nf * 16 + fr;
          if (co < g.Cout)
            atomicAdd(&dWacc[((int64_t)tap * g.Cin + ci) * g.Cout + co],
                      acc[tap][mf][nf][r]);
        }
      }
    }
  }
}

// dWacc[tap][Cin][Cout] -> torch channels_last weight grad [Cout][KH][KW][Cin]
__global__ void wrw3_cast_kernel(const float* __restrict__ dWacc, short* __restrict__ dW,
                                 int Cout, int KK2, int Cin) {
  int64_t total = (int64_t)Cout * KK2 * Cin;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  int ci = i % Cin;
  int64_t t = i / Cin;
  int tap = t % KK2;
  int co = t / KK2;
  dW[i] = f2b(dWacc[((int64_t)tap * Cin + ci) * Cout + co]);
}

// cast dW accumulator [kpad][Cout] back to torch layout [Cout][KH][KW][Cin]
__global__ void wrw_cast_kernel(const float* __restrict__ dWacc, short* __restrict__ dW,
                                ConvGeom g) {
  int64_t total = (int64_t)g.Cout * g.KH * g.KW * g.Cin;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  int ci = i % g.Cin;
  int64_t t = i / g.Cin;
  int cell = t % (g.KH * g.KW);
  int co = t / (g.KH * g.KW);
  int kp = cell * (g.cin_chunks * 8) + ci;
  dW[i] = f2b(dWacc[(int64_t)kp * g.Cout + co]);
}

// column sum: dbias[n] = sum_m dY[m][n] (bf16 in, fp32 partials).
// per-thread register accumulation -> per-block LDS fold by channel octet
// -> ONE atomic per channel per block (per-thread atomics serialize badly).
__global__ void colsum_kernel(const short* __restrict__ dY, float* __restrict__ partial,
                              int64_t M, int C) {
  float s[8] = {};
  int64_t total = M * C;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(dY + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += b2f(v[j]);
  }
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = s[j];
  __syncthreads();
  const int groups = C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  if ((int)threadIdx.x < C) {
    int oct = threadIdx.x / 8, lane = threadIdx.x % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + lane];
    atomicAdd(&partial[threadIdx.x], acc);
  }
}

__global__ void cast_f32_bf16_kernel(const float* __restrict__ in, short* __restrict__ out,
                                     int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = f2b(in[i]);
}

// colsum v2 — replay-safe column sum (dbias) for hipGraph capture.
// The legacy colsum_kernel (atomicAdd into a torch::zeros workspace) was
// implicated in hipGraph-replay corruption (tools/nan_flake.py bisection,
// round 1). v2 removes BOTH suspects: per-block partials are plain stores
// into a fully-overwritten torch::empty workspace (no memset node in the
// graph, no atomics), and a second tiny kernel reduces nb x C -> C and
// casts to bf16 in one pass (also removing the separate cast launch).
__global__ void colsum_partial_kernel(const short* __restrict__ dY,
                                      float* __restrict__ partial,
                                      int64_t M, int C) {
  float s[8] = {};
  int64_t total = M * C;
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(dY + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += b2f(v[j]);
  }
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = s[j];
  __syncthreads();
  const int groups = C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  // each thread folds one or more output channels (supports C > blockDim)
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    int oct = c / 8, ln = c % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + ln];
    partial[(int64_t)blockIdx.x * C + c] = acc;
  }
}

__global__ void colsum_finish_kernel(const float* __restrict__ partial,
                                     short* __restrict__ out, int nb, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float acc = 0;
  for (int b = 0; b < nb; ++b) acc += partial[(int64_t)b * C + c];
  out[c] = f2b(acc);
}

}  // namespace

static ConvGeom make_geom(const torch::Tensor& x, const torch::Tensor& w,
                          int64_t stride, int64_t pad) {
  ConvGeom g;
  g.B = x.size(0); g.Cin = x.size(1); g.H = x.size(2); g.Wd = x.size(3);
  g.Cout = w.size(0); g.KH = w.size(2); g.KW = w.size(3);
  g.stride = stride; g.pad = pad;
  g.Ho = (g.H + 2 * g.pad - g.KH) / g.stride + 1;
  g.Wo = (g.Wd + 2 * g.pad - g.KW) / g.stride + 1;
  g.cin_chunks = (g.Cin + 7) / 8;
  g.kpad = g.KH * g.KW * g.cin_chunks * 8;
  // kpad must be a multiple of BK for the k-loop
  g.kpad = (g.kpad + BK - 1) / BK * BK;
  return g;
}

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && w.scalar_type() == torch::kBFloat16,
              "conv2d_fwd: bf16 only");
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wc = w.contiguous(torch::MemoryFormat::ChannelsLast);
  ConvGeom g = make_geom(xc, wc, stride, pad);
  int M = g.B * g.Ho * g.Wo;
  auto y = torch::empty({g.B, g.Cout, g.Ho, g.Wo},
                        xc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int bnt = (g.Cout <= 32) ? 32 : 64;
  int bmt = 64;
  {
    auto blocks = [&](int bm_, int bn_) {
      return (int64_t)((M + bm_ - 1) / bm_) * ((g.Cout + bn_ - 1) / bn_);
    };
    if (blocks(64, bnt) < 1024 && blocks(32, bnt) <= 4096) bmt = 32;
  }
  // FAA_CONV_TILE=BMxBN forces a tile pair (tuning sweeps, tools/conv_tune.py)
  if (const char* e = getenv("FAA_CONV_TILE")) {
    int fm = 0, fn = 0;
    if (sscanf(e, "%dx%d", &fm, &fn) == 2 && (fm == 32 || fm == 64)
        && (fn == 32 || fn == 64)) {
      bmt = fm;
      bnt = fn;
    }
  }
  int grid_m = (M + bmt - 1) / bmt;
  int grid_n = (g.Cout + bnt - 1) / bnt;
  dim3 grid(grid_m * grid_n);
  auto stream = at::hip::getCurrentHIPStream().stream();
  bool has_bias = bias.defined() && bias.numel() > 0;
  const short* bptr = nullptr;
  torch::Tensor bc;
  if (has_bias) {
    bc = bias.contiguous();
    TORCH_CHECK(bc.scalar_type() == torch::kBFloat16);
    bptr = (const short*)bc.data_ptr();
  }

  // direct tiled 3x3 path: CIFAR geometries, any channel count.
  // FAA_CONV_DIRECT=0 falls back to the im2col kernel below; =small/=big
  // force the narrow (BN=32) / wide (BN=64) wave configs for sweeps.
  {
    const char* de = getenv("FAA_CONV_DIRECT");
    bool want = !(de && de[0] == '0');
    // tile width covering Wd (exact for the CIFAR sizes; masked tails for
    // the ResNet-50 ImageNet spatials 56/28/14/7 — epilogue guards skip
    // out-of-range rows/cols, staging zero-fills them)
    int twsel = (g.Wd <= 8) ? 8 : (g.Wd <= 16) ? 16 : (g.Wd <= 28) ? 28
                : (g.Wd <= 32) ? 32 : (g.Wd <= 56) ? 56 : 0;
    bool geom_ok = (g.KH == 3 && g.KW == 3 && g.stride == 1 && g.pad == 1
                    && g.Cin >= 16 && twsel > 0
                    && (twsel != 8 || g.B % 2 == 0));
    if (want && geom_ok) {
      int tiles_h = (g.H + 7) / 8;
      int ib = (twsel == 8) ? 2 : 1;
      // wide config (BN=64): fewer weight re-reads per FLOP; only when the
      // grid still fills the chip and for W=32 the TH=4 variant exists
      int gn64 = (g.Cout + 63) / 64;
      int tiles_h4 = (g.H + 3) / 4;
      int64_t blocks64 = (twsel == 32)
          ? (int64_t)g.B * tiles_h4 * gn64
          : (int64_t)(g.B / ib) * tiles_h * gn64;
      bool big = (g.Cin > 64) && (twsel != 56) && (twsel != 28)
                 && (blocks64 >= (twsel == 8 ? 192 : 512));
      if (de && de[0] == 'b') big = true;
      if (de && de[0] == 's') big = false;
      #define CT_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, HB_, GRID_, THTILES_, GN_)   \
        hipLaunchKernelGGL((conv3x3_tile_kernel<TH_, TW_, IB_, WN_, MR_, NR_, HB_>),\
                           dim3((unsigned)(GRID_)), dim3(256), 0, stream,          \
                           (const short*)xc.data_ptr(),                            \
                           (const short*)wc.data_ptr(), bptr,                      \
                           (short*)y.data_ptr(), g, THTILES_, GN_)
      #define CT_BOTH(TH_, TW_, IB_, WN_, MR_, NR_, GRID_, THTILES_, GN_)          \
        do {                                                                       \
          if (has_bias) CT_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, true,              \
                                  GRID_, THTILES_, GN_);                           \
          else CT_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, false,                      \
                         GRID_, THTILES_, GN_);                                    \
        } while (0)
      if (big) {
        if (twsel == 32) {
          CT_BOTH(4, 32, 1, 2, 4, 2, (int64_t)g.B * tiles_h4 * gn64, tiles_h4, gn64);
        } else if (twsel == 16) {
          CT_BOTH(8, 16, 1, 2, 4, 2, (int64_t)g.B * tiles_h * gn64, tiles_h, gn64);
        } else {
          // W=8 deep stages: M is tiny (B*64 px) and K is huge, so weight
          // re-reads dominate. IB=4 (M=256/block) halves per-FLOP weight
          // traffic; split-K over cin slabs restores grid occupancy.
          // FAA_CONV_D8 = ib2 | ib4 | ib2sk2 | ib4sk2 | ib4sk4 forces.
          const char* d8 = getenv("FAA_CONV_D8");
          int ibb = 2, sk = 1;
          if (d8) {
            ibb = (d8[2] == '4') ? 4 : 2;
            sk = (d8[3] == 's') ? (d8[5] - '0') : 1;
          } else {
            // measured (gpurun_out/call4.log D8 sweep): plain ib2 wins to
            // C=384 (35us vs 75 MIOpen at 256), ib4 wins at C>=512
            // (150 vs 159 at 640); every split-K variant loses.
            ibb = (g.Cin >= 512) ? 4 : 2;
          }
          if (ibb == 4 && g.B % 4 != 0) ibb = 2;
          if (sk > 1) {
            auto yacc = torch::zeros({(int64_t)M, (int64_t)g.Cout},
                                     xc.options().dtype(torch::kFloat32));
            dim3 sgrid((unsigned)((g.B / ibb) * gn64), sk);
            #define CT_SK(IB_, MR_)                                               \
              hipLaunchKernelGGL((conv3x3_tile_kernel<8, 8, IB_, 2, MR_, 2,       \
                                  false, true>), sgrid, dim3(256), 0, stream,     \
                                 (const short*)xc.data_ptr(),                     \
                                 (const short*)wc.data_ptr(), nullptr,            \
                                 (short*)y.data_ptr(), g, 1, gn64,                \
                                 yacc.data_ptr<float>())
            if (ibb == 4) CT_SK(4, 8); else CT_SK(2, 4);
            #undef CT_SK
            int64_t total = (int64_t)M * g.Cout;
            dim3 cgrid((unsigned)((total + 255) / 256));
            if (has_bias)
              hipLaunchKernelGGL((conv_splitk_cast_kernel<true>), cgrid, dim3(256),
                                 0, stream, yacc.data_ptr<float>(), bptr,
                                 (short*)y.data_ptr(), total, g.Cout);
            else
              hipLaunchKernelGGL((conv_splitk_cast_kernel<false>), cgrid, dim3(256),
                                 0, stream, yacc.data_ptr<float>(), nullptr,
                                 (short*)y.data_ptr(), total, g.Cout);
          } else if (ibb == 4) {
            CT_BOTH(8, 8, 4, 2, 8, 2, (int64_t)(g.B / 4) * gn64, 1, gn64);
          } else {
            CT_BOTH(8, 8, 2, 2, 4, 2, (int64_t)(g.B / 2) * gn64, 1, gn64);
          }
        }
      } else {
        int gn = (g.Cout + 31) / 32;
        int64_t grid32 = (int64_t)(g.B / ib) * tiles_h * gn;
        if (twsel == 56) CT_BOTH(8, 56, 1, 1, 7, 2, grid32, tiles_h, gn);
        else if (twsel == 28) {
          // TH=16 x TW=28 (no column waste for the ResNet 28^2 stage)
          int th16 = (g.H + 15) / 16;
          int64_t grid28 = (int64_t)g.B * th16 * gn;
          CT_BOTH(16, 28, 1, 1, 7, 2, grid28, th16, gn);
        }
        else if (twsel == 32) CT_BOTH(8, 32, 1, 1, 4, 2, grid32, tiles_h, gn);
        else if (twsel == 16) CT_BOTH(8, 16, 1, 1, 2, 2, grid32, tiles_h, gn);
        else CT_BOTH(8, 8, 2, 1, 2, 2, grid32, tiles_h, gn);
      }
      #undef CT_BOTH
      #undef CT_LAUNCH
      return y;
    }
  }
  // FAA_CONV_SPLITK=1: partition K across blockIdx.y when the tile grid
  // underfills the chip (deep 8x8 stages run ~256 WGs = 1/CU; PMC showed
  // 9% occupancy there). fp32-atomic workspace + bias/cast epilogue.
  // Env-gated default-off pending round-2 measurement.
  {
    const char* e = getenv("FAA_CONV_SPLITK");
    int nk = g.kpad / BK;
    int blocks = grid_m * grid_n;
    if (e && e[0] == '1' && blocks < 1024 && nk >= 2) {
      int sk = std::min(std::min((1024 + blocks - 1) / blocks, nk), 8);
      if (sk > 1) {
        auto yacc = torch::zeros({(int64_t)M, (int64_t)g.Cout},
                                 xc.options().dtype(torch::kFloat32));
        dim3 grid2(grid_m * grid_n, sk);
        #define CF_LAUNCH_SK(BMT_, BNT_)                                         \
          hipLaunchKernelGGL((conv_fwd_kernel<false, BMT_, BNT_, true>), grid2,  \
                             dim3(256), 0, stream, (const short*)xc.data_ptr(),  \
                             (const short*)wc.data_ptr(), nullptr,               \
                             (short*)y.data_ptr(), g, M, grid_m,                 \
                             yacc.data_ptr<float>())
        if (bmt == 32 && bnt == 32) CF_LAUNCH_SK(32, 32);
        else if (bmt == 32) CF_LAUNCH_SK(32, 64);
        else if (bnt == 32) CF_LAUNCH_SK(64, 32);
        else CF_LAUNCH_SK(64, 64);
        #undef CF_LAUNCH_SK
        int64_t total = (int64_t)M * g.Cout;
        dim3 cgrid((unsigned)((total + 255) / 256));
        if (has_bias)
          hipLaunchKernelGGL((conv_splitk_cast_kernel<true>), cgrid, dim3(256), 0,
                             stream, yacc.data_ptr<float>(), bptr,
                             (short*)y.data_ptr(), total, g.Cout);
        else
          hipLaunchKernelGGL((conv_splitk_cast_kernel<false>), cgrid, dim3(256), 0,
                             stream, yacc.data_ptr<float>(), nullptr,
                             (short*)y.data_ptr(), total, g.Cout);
        return y;
      }
    }
  }
  #define CF_LAUNCH(HB, BMT_, BNT_)                                            \
    hipLaunchKernelGGL((conv_fwd_kernel<HB, BMT_, BNT_>), grid, dim3(256), 0,  \
                       stream, (const short*)xc.data_ptr(),                    \
                       (const short*)wc.data_ptr(), bptr,                      \
                       (short*)y.data_ptr(), g, M, grid_m)
  if (has_bias) {
    if (bmt == 32 && bnt == 32) CF_LAUNCH(true, 32, 32);
    else if (bmt == 32) CF_LAUNCH(true, 32, 64);
    else if (bnt == 32) CF_LAUNCH(true, 64, 32);
    else CF_LAUNCH(true, 64, 64);
  } else {
    if (bmt == 32 && bnt == 32) CF_LAUNCH(false, 32, 32);
    else if (bmt == 32) CF_LAUNCH(false, 32, 64);
    else if (bnt == 32) CF_LAUNCH(false, 64, 32);
    else CF_LAUNCH(false, 64, 64);
  }
  #undef CF_LAUNCH
  return y;
}

torch::Tensor conv2d_bwd_data(torch::Tensor dy, torch::Tensor w,
                              int64_t stride, int64_t pad,
                              int64_t H, int64_t W_in) {
  TORCH_CHECK(stride == 1, "conv2d_bwd_data: stride-1 only (caller falls back)");
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wc = w.contiguous(torch::MemoryFormat::ChannelsLast);
  int Cout = wc.size(0), Cin = wc.size(1), KH = wc.size(2), KW = wc.size(3);
  auto stream = at::hip::getCurrentHIPStream().stream();
  // repack: W2 [Cin][KH][KW][Cout] flipped
  auto w2 = torch::empty({Cin, Cout, KH, KW},
                         wc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)Cout * KH * KW * Cin;
  hipLaunchKernelGGL(weight_flip_kernel, dim3((total + 255) / 256), dim3(256), 0, stream,
                     (const short*)wc.data_ptr(), (short*)w2.data_ptr(),
                     Cout, KH, KW, Cin);
  // full-correlation: dX = conv(dY, W2, pad = KH-1-pad)
  return conv2d_fwd(dyc, w2, torch::Tensor(), 1, KH - 1 - pad);
}

torch::Tensor conv2d_fwd_grouped(torch::Tensor x, torch::Tensor w,
                                  torch::Tensor bias, int64_t stride,
                                  int64_t pad, int64_t groups) {
  // grouped 3x3 s1 conv through the direct tiled kernel (ShakeResNeXt
  // cardinality-4 branches, reference shake_resnext.py:34). Python gates
  // eligibility; hard requirements checked here.
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && w.scalar_type() == torch::kBFloat16);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto wc = w.contiguous(torch::MemoryFormat::ChannelsLast);
  ConvGeom g;
  g.B = xc.size(0); g.Cin = xc.size(1); g.H = xc.size(2); g.Wd = xc.size(3);
  g.Cout = wc.size(0); g.KH = wc.size(2); g.KW = wc.size(3);
  g.stride = stride; g.pad = pad; g.groups = (int)groups;
  g.Ho = g.H; g.Wo = g.Wd;
  g.cin_chunks = 0; g.kpad = 0;
  int cin_per = wc.size(1), cout_per = g.Cout / (int)groups;
  TORCH_CHECK(g.Cin == cin_per * groups && g.Cout % groups == 0,
              "grouped geometry mismatch");
  TORCH_CHECK(g.KH == 3 && g.KW == 3 && stride == 1 && pad == 1
              && cin_per >= 16 && cout_per % 32 == 0
              && ((g.Wd == 32 && g.H % 8 == 0) || (g.Wd == 16 && g.H % 8 == 0)
                  || (g.Wd == 8 && g.H == 8 && g.B % 2 == 0)),
              "conv2d_fwd_grouped: unsupported geometry");
  auto y = torch::empty({g.B, g.Cout, g.Ho, g.Wo},
                        xc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  auto stream = at::hip::getCurrentHIPStream().stream();
  bool has_bias = bias.defined() && bias.numel() > 0;
  torch::Tensor bc;
  const short* bptr = nullptr;
  if (has_bias) {
    bc = bias.contiguous();
    bptr = (const short*)bc.data_ptr();
  }
  int tiles_h = g.H / 8;
  int ib = (g.Wd == 8) ? 2 : 1;
  bool big = (cin_per > 64) && (cout_per % 64 == 0);
  #define CTG_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, HB_, GRID_, THTILES_, GN_)   \
    hipLaunchKernelGGL((conv3x3_tile_kernel<TH_, TW_, IB_, WN_, MR_, NR_, HB_>),\
                       dim3((unsigned)(GRID_)), dim3(256), 0, stream,           \
                       (const short*)xc.data_ptr(),                             \
                       (const short*)wc.data_ptr(), bptr,                       \
                       (short*)y.data_ptr(), g, THTILES_, GN_)
  #define CTG_BOTH(TH_, TW_, IB_, WN_, MR_, NR_, GRID_, THTILES_, GN_)          \
    do {                                                                        \
      if (has_bias) CTG_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, true,              \
                               GRID_, THTILES_, GN_);                           \
      else CTG_LAUNCH(TH_, TW_, IB_, WN_, MR_, NR_, false,                      \
                      GRID_, THTILES_, GN_);                                    \
    } while (0)
  if (big) {
    int gn64 = g.Cout / 64;
    int tiles_h4 = g.H / 4;
    if (g.Wd == 32) CTG_BOTH(4, 32, 1, 2, 4, 2, (int64_t)g.B * tiles_h4 * gn64, tiles_h4, gn64);
    else if (g.Wd == 16) CTG_BOTH(8, 16, 1, 2, 4, 2, (int64_t)g.B * tiles_h * gn64, tiles_h, gn64);
    else CTG_BOTH(8, 8, 2, 2, 4, 2, (int64_t)(g.B / 2) * gn64, 1, gn64);
  } else {
    int gn = g.Cout / 32;
    int64_t grid32 = (int64_t)(g.B / ib) * tiles_h * gn;
    if (g.Wd == 32) CTG_BOTH(8, 32, 1, 1, 4, 2, grid32, tiles_h, gn);
    else if (g.Wd == 16) CTG_BOTH(8, 16, 1, 1, 2, 2, grid32, tiles_h, gn);
    else CTG_BOTH(8, 8, 2, 1, 2, 2, grid32, tiles_h, gn);
  }
  #undef CTG_BOTH
  #undef CTG_LAUNCH
  return y;
}

torch::Tensor conv2d_bwd_data_s2(torch::Tensor dy, torch::Tensor w,
                                  int64_t H, int64_t W, bool pre_flipped) {
  // stride-2 3x3 pad-1 bwd-data by phase decomposition (conv3x3s2_dx_kernel).
  // w: either the conv weight [Cout][Cin][3][3] (channels_last) or, when
  // pre_flipped, the conv_flip_all repack [Cin][Cout][3][3] (channels_last,
  // i.e. [ci][fh][fw][co] in memory).
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int B = dyc.size(0), Cdy = dyc.size(1), Ho = dyc.size(2), Wo = dyc.size(3);
  TORCH_CHECK((W == 16 || W == 32) && H % 8 == 0 && H == 2 * Ho && W == 2 * Wo,
              "conv2d_bwd_data_s2: unsupported geometry");
  auto stream = at::hip::getCurrentHIPStream().stream();
  torch::Tensor w2;
  int Cdx;
  if (pre_flipped) {
    w2 = w;
    Cdx = w.size(0);
    TORCH_CHECK(w.size(1) == Cdy && w.is_contiguous(torch::MemoryFormat::ChannelsLast));
  } else {
    auto wc = w.contiguous(torch::MemoryFormat::ChannelsLast);
    Cdx = wc.size(1);
    TORCH_CHECK(wc.size(0) == Cdy && wc.size(2) == 3 && wc.size(3) == 3);
    w2 = torch::empty({Cdx, Cdy, 3, 3},
                      wc.options().memory_format(torch::MemoryFormat::ChannelsLast));
    int64_t total = (int64_t)Cdy * 9 * Cdx;
    hipLaunchKernelGGL(weight_flip_kernel, dim3((total + 255) / 256), dim3(256),
                       0, stream, (const short*)wc.data_ptr(), (short*)w2.data_ptr(),
                       Cdy, 3, 3, Cdx);
  }
  ConvGeom g;
  g.B = B; g.Cin = Cdy; g.Cout = Cdx; g.H = H; g.Wd = W;
  g.Ho = Ho; g.Wo = Wo; g.KH = 3; g.KW = 3; g.stride = 2; g.pad = 1;
  g.cin_chunks = (Cdy + 7) / 8; g.kpad = 0;
  auto dx = torch::empty({B, Cdx, H, W},
                         dyc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int grid_n = (Cdx + 31) / 32;
  int tiles_h = H / 8;
  dim3 grid((unsigned)((int64_t)B * tiles_h * grid_n));
  if (W == 32)
    hipLaunchKernelGGL((conv3x3s2_dx_kernel<8, 32>), grid, dim3(256), 0, stream,
                       (const short*)dyc.data_ptr(), (const short*)w2.data_ptr(),
                       (short*)dx.data_ptr(), g, tiles_h, grid_n);
  else
    hipLaunchKernelGGL((conv3x3s2_dx_kernel<8, 16>), grid, dim3(256), 0, stream,
                       (const short*)dyc.data_ptr(), (const short*)w2.data_ptr(),
                       (short*)dx.data_ptr(), g, tiles_h, grid_n);
  return dx;
}

torch::Tensor colsum_bf16(torch::Tensor dy) {
  // dbias[n] = sum over rows of a [*, C] bf16 channels-last tensor.
  // v2 two-kernel scheme: no atomics, no memset — every byte read was
  // written earlier in the same stream/replay (hipGraph-safe).
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = dyc.size(1);
  int64_t M = dyc.numel() / C;
  auto stream = at::hip::getCurrentHIPStream().stream();
  TORCH_CHECK(C % 8 == 0, "colsum_bf16: C % 8 == 0 expected");
  auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
  int q = C / gcd(C, 2048);
  int nb = ((64 + q - 1) / q) * q;   // multiple of q => per-thread channel set
                                     // is invariant across grid-stride iters
  auto part = torch::empty({nb, C}, dyc.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(colsum_partial_kernel, dim3(nb), dim3(256), 0, stream,
                     (const short*)dyc.data_ptr(), part.data_ptr<float>(),
                     M, C);
  auto out = torch::empty({C}, dyc.options());
  hipLaunchKernelGGL(colsum_finish_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     stream, part.data_ptr<float>(), (short*)out.data_ptr(), nb, C);
  return out;
}

void colsum_bf16_ws(torch::Tensor dy, torch::Tensor part, torch::Tensor out) {
  // colsum v2 into caller-provided buffers (hipGraph bisect variant E:
  // workspace + out live OUTSIDE the capture pool).
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = dyc.size(1);
  int64_t M = dyc.numel() / C;
  auto stream = at::hip::getCurrentHIPStream().stream();
  TORCH_CHECK(C % 8 == 0, "colsum_bf16_ws: C % 8 == 0 expected");
  auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
  int q = C / gcd(C, 2048);
  int nb = ((64 + q - 1) / q) * q;
  TORCH_CHECK(part.numel() >= (int64_t)nb * C && part.scalar_type() == torch::kFloat32,
              "colsum_bf16_ws: workspace too small");
  TORCH_CHECK(out.numel() == C && out.scalar_type() == torch::kBFloat16);
  hipLaunchKernelGGL(colsum_partial_kernel, dim3(nb), dim3(256), 0, stream,
                     (const short*)dyc.data_ptr(), part.data_ptr<float>(), M, C);
  hipLaunchKernelGGL(colsum_finish_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     stream, part.data_ptr<float>(), (short*)out.data_ptr(), nb, C);
}

void flip_weights_batched(torch::Tensor table, int64_t total) {
  TORCH_CHECK(table.scalar_type() == torch::kInt64 && table.size(1) == 7,
              "flip table must be int64 [n,7]");
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL(flip_weights_batched_kernel, dim3(grid), dim3(256), 0,
                     stream, table.data_ptr<int64_t>(), (int)table.size(0), total);
}

torch::Tensor colsum_bf16_legacy(torch::Tensor dy) {
  // round-1 atomic+zeros implementation, kept ONLY for the hipGraph
  // corruption bisect (tools/nan_flake.py); not used on any dispatch path.
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = dyc.size(1);
  int64_t M = dyc.numel() / C;
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto part = torch::zeros({C}, dyc.options().dtype(torch::kFloat32));
  TORCH_CHECK(C % 8 == 0, "colsum_bf16: C % 8 == 0 expected");
  auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
  int q = C / gcd(C, 2048);
  int nb = ((64 + q - 1) / q) * q;
  hipLaunchKernelGGL(colsum_kernel, dim3(nb), dim3(256), 0, stream,
                     (const short*)dyc.data_ptr(), part.data_ptr<float>(),
                     M, C);
  auto out = torch::empty({C}, dyc.options());
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     stream, part.data_ptr<float>(), (short*)out.data_ptr(), C);
  return out;
}

std::vector<torch::Tensor> conv2d_bwd_weight(torch::Tensor dy, torch::Tensor x,
                                             int64_t stride, int64_t pad,
                                             int64_t KH, int64_t KW, bool want_bias) {
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  int Cout = dyc.size(1);
  ConvGeom g;
  g.B = xc.size(0); g.Cin = xc.size(1); g.H = xc.size(2); g.Wd = xc.size(3);
  g.Cout = Cout; g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.Ho = dyc.size(2); g.Wo = dyc.size(3);
  g.cin_chunks = (g.Cin + 7) / 8;
  g.kpad = (g.KH * g.KW * g.cin_chunks * 8 + BK - 1) / BK * BK;
  int M = g.B * g.Ho * g.Wo;
  auto stream = at::hip::getCurrentHIPStream().stream();

  auto f32 = xc.options().dtype(torch::kFloat32);

  // wrw v3 direct tiled path (stride-1 3x3/1x1 CIFAR geometries).
  // Default OFF pending tuning: call4 measured the v2 split-K kernel (and
  // MIOpen) ahead at small C — v3's epilogue atomics dominate when the
  // (ci,co) grid is tiny and every block covers one pixel tile.
  {
    // auto: v3 wins at Cin>=160 on 32px tiles (232 vs 278us MIOpen /
    // 1032us v2 at 160^3, call5 WRWSWEEP); v2 keeps the small-C shapes.
    const char* e = getenv("FAA_WRW_V3");
    bool want = e ? (e[0] == '1') : (g.Cin >= 160 && g.Wd == 32);
    bool k3 = (KH == 3 && KW == 3 && pad == 1);
    bool k1 = (KH == 1 && KW == 1 && pad == 0);
    bool geom_ok = stride == 1 && (k3 || k1)
        && ((g.Wd == 32 && g.H % 4 == 0)
            || ((g.Wd == 16 || g.Wd == 8) && g.H % 8 == 0));
    if (want && geom_ok) {
      int KK = k3 ? 3 : 1;
      int ci_tiles = (g.Cin + 63) / 64;
      int co_tiles = (Cout + 63) / 64;
      int TH = (g.Wd == 32) ? 4 : 8;
      int tiles_total = g.B * (g.H / TH);
      int blocks_xy = ci_tiles * co_tiles;
      // amortize the per-block epilogue (64x64xKK^2 atomics) over >=2 tiles
      int slices = std::max(1, std::min(512 / blocks_xy, tiles_total / 2));
      if (const char* se = getenv("FAA_WRW3_SLICES")) {
        int f = atoi(se);
        if (f > 0) slices = std::min(f, tiles_total);
      }
      int tps = (tiles_total + slices - 1) / slices;
      slices = (tiles_total + tps - 1) / tps;
      auto dwacc = torch::zeros({(int64_t)KK * KK, g.Cin, (int64_t)Cout}, f32);
      dim3 grid(blocks_xy, slices);
      // WSPLIT=4 measured SLOWER at small C (call7: 4x epilogue atomics
      // dominate); kept behind an env for future cross-wave-reduce work
      const char* wse = getenv("FAA_WRW3_WSPLIT");
      bool small = (wse && wse[0] == '1') && (g.Cin <= 32 && Cout <= 32);
      #define WRW3_LAUNCH(TH_, TW_, KK_, WS_)                                   \
        hipLaunchKernelGGL((conv_wrw3_kernel<TH_, TW_, KK_, WS_>), grid,        \
                           dim3(256), 0, stream, (const short*)xc.data_ptr(),   \
                           (const short*)dyc.data_ptr(),                        \
                           dwacc.data_ptr<float>(), g, ci_tiles, co_tiles, tps)
      #define WRW3_PICK(TH_, TW_, KK_)                                          \
        do {                                                                    \
          if (small) WRW3_LAUNCH(TH_, TW_, KK_, 4);                             \
          else WRW3_LAUNCH(TH_, TW_, KK_, 1);                                   \
        } while (0)
      if (k3) {
        if (g.Wd == 32) WRW3_PICK(4, 32, 3);
        else if (g.Wd == 16) WRW3_PICK(8, 16, 3);
        else WRW3_PICK(8, 8, 3);
      } else {
        if (g.Wd == 32) WRW3_PICK(4, 32, 1);
        else if (g.Wd == 16) WRW3_PICK(8, 16, 1);
        else WRW3_PICK(8, 8, 1);
      }
      #undef WRW3_PICK
      #undef WRW3_LAUNCH
      auto dw = torch::empty({Cout, g.Cin, (int64_t)KH, (int64_t)KW},
                             xc.options().memory_format(torch::MemoryFormat::ChannelsLast));
      int64_t total3 = (int64_t)Cout * KK * KK * g.Cin;
      hipLaunchKernelGGL(wrw3_cast_kernel, dim3((total3 + 255) / 256), dim3(256),
                         0, stream, dwacc.data_ptr<float>(), (short*)dw.data_ptr(),
                         Cout, KK * KK, g.Cin);
      torch::Tensor dbias;
      if (want_bias) {
        if (Cout % 8 == 0) dbias = colsum_bf16(dyc);
        else dbias = dyc.sum(/*dim=*/{0, 2, 3}).to(xc.scalar_type());
      }
      return {dw, dbias};
    }
  }

  auto dwacc = torch::zeros({g.kpad, Cout}, f32);
  int grid_k = g.kpad / 32;
  int grid_n = (Cout + 63) / 64;
  // pick m-slices for ~>=512 blocks without over-splitting tiny workloads
  int m_slices = 1;
  while (grid_k * grid_n * m_slices < 512 && (int64_t)m_slices * 2 * 32 <= M)
    m_slices *= 2;
  dim3 grid(grid_k, grid_n, m_slices);
  hipLaunchKernelGGL(conv_wrw_kernel, grid, dim3(256), 0, stream,
                     (const short*)xc.data_ptr(), (const short*)dyc.data_ptr(),
                     dwacc.data_ptr<float>(), g, M, m_slices);
  auto dw = torch::empty({Cout, g.Cin, (int64_t)KH, (int64_t)KW},
                         xc.options().memory_format(torch::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)Cout * KH * KW * g.Cin;
  hipLaunchKernelGGL(wrw_cast_kernel, dim3((total + 255) / 256), dim3(256), 0, stream,
                     dwacc.data_ptr<float>(), (short*)dw.data_ptr(), g);

  torch::Tensor dbias;
  if (want_bias) {
    if (Cout % 8 == 0) {
      dbias = colsum_bf16(dyc);
    } else {
      dbias = dyc.sum(/*dim=*/{0, 2, 3}).to(xc.scalar_type());
    }
  }
  return {dw, dbias};
}
