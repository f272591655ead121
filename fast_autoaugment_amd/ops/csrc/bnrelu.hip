// Fused NHWC BatchNorm2d + ReLU, training fwd/bwd + eval fwd.
//
// The pre-activation BN->ReLU pair is the second-hottest op family of the
// CIFAR nets after conv (reference wideresnet.py:37-38, pyramidnet.py:83-96).
// MIOpen runs BN and ReLU as separate NCHW kernels; here both fuse into one
// NHWC pass with fp32 stats.
//
// Reduction scheme (HBM-bound, per G13): the NHWC storage [rows][C] is read
// LINEARLY in 8-element bf16 vectors. With C a multiple of 8 and the thread
// stride (blockDim*8) a multiple of C, every thread touches a FIXED set of 8
// channels -> 2x8 fp32 accumulators live in registers, no LDS/global atomics
// in the main loop. Per-block partials land in scratch[block][2C] (no
// zero-fill needed); a small finalize kernel folds partials + running-stat
// update. C not divisible by 8 (e.g. stem C=3 is never BN'd here) falls back
// to a strided scalar path.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

typedef __attribute__((ext_vector_type(4))) short short4v;

__device__ __forceinline__ float bf16_bits_to_f(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f_to_bf16_bits(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// ------------------------------------------------- last-block finalize
// FAA_BN_LASTBLOCK=1 (staged for round 2, default off): the LAST reduce
// block to finish performs the finalize inline instead of a separate
// 4.7us kernel launch per BN site (~66 launches/step on WRN-40-2).
// Pattern: write partials -> __threadfence -> atomicAdd(counter); the last
// arrival re-reduces scratch and resets the counter, so the persistent
// counter never needs per-call zeroing.
__device__ __forceinline__ void bn_finalize_inblock(
    const float* __restrict__ scratch, int nblocks, float* __restrict__ mean,
    float* __restrict__ invstd, float* __restrict__ running_mean,
    float* __restrict__ running_var, int C, int64_t count, float eps,
    float momentum) {
  for (int c = threadIdx.x; c < C; c += (int)blockDim.x) {
    float s = 0, ss = 0;
    for (int b = 0; b < nblocks; ++b) {
      s += scratch[(int64_t)b * 2 * C + c];
      ss += scratch[(int64_t)b * 2 * C + C + c];
    }
    float m = s / count;
    float var = fmaxf(ss / count - m * m, 0.0f);
    mean[c] = m;
    invstd[c] = rsqrtf(var + eps);
    if (running_mean != nullptr) {
      float unbiased = count > 1 ? var * (float)count / (float)(count - 1) : var;
      running_mean[c] = (1.0f - momentum) * running_mean[c] + momentum * m;
      running_var[c] = (1.0f - momentum) * running_var[c] + momentum * unbiased;
    }
  }
}

struct BnFinTail {          // nullptr counter = disabled
  unsigned* counter;
  float* mean; float* invstd; float* running_mean; float* running_var;
  int64_t count; float eps; float momentum;
};

__device__ __forceinline__ void bn_fin_tail_run(const BnFinTail& t,
                                                const float* scratch, int C) {
  if (t.counter == nullptr) return;
  __threadfence();                      // release: partials visible before tick
  __shared__ unsigned order;
  if (threadIdx.x == 0) order = atomicAdd(t.counter, 1u);
  __syncthreads();
  if (order == gridDim.x - 1) {
    __threadfence();                    // acquire: see every block's partials
    bn_finalize_inblock(scratch, gridDim.x, t.mean, t.invstd, t.running_mean,
                        t.running_var, C, t.count, t.eps, t.momentum);
    __syncthreads();
    if (threadIdx.x == 0) *t.counter = 0;
  }
}

template <typename GT>
struct BnBwdFinTail {      // nullptr counter = disabled
  unsigned* counter;
  float* sums; GT* dbeta; GT* dgamma;
};

template <typename GT>
__device__ __forceinline__ void bn_bwd_fin_tail_run(const BnBwdFinTail<GT>& t,
                                                    const float* scratch, int C) {
  if (t.counter == nullptr) return;
  __threadfence();
  __shared__ unsigned order;
  if (threadIdx.x == 0) order = atomicAdd(t.counter, 1u);
  __syncthreads();
  if (order == gridDim.x - 1) {
    __threadfence();                    // acquire: see every block's partials
    for (int c = threadIdx.x; c < 2 * C; c += (int)blockDim.x) {
      float s = 0;
      for (int b = 0; b < (int)gridDim.x; ++b)
        s += scratch[(int64_t)b * 2 * C + c];
      t.sums[c] = s;
      if (c < C) faa_from_float(s, &t.dbeta[c]);
      else faa_from_float(s, &t.dgamma[c - C]);
    }
    __syncthreads();
    if (threadIdx.x == 0) *t.counter = 0;
  }
}

// ---------------------------------------------------------------- fwd reduce
// block: 256 threads; each thread owns channels [c0, c0+8) with
// c0 = (tid*8) % C. partials: scratch[blockIdx.x*2C + {c, C+c}]
template <typename T>
__global__ void bn_reduce_vec_kernel(const T* __restrict__ x, float* __restrict__ scratch,
                                     int64_t total, int C, BnFinTail tail = {}) {
  // host guarantees (gridDim*blockDim*8) % C == 0, so each thread's channel
  // octet c0 is FIXED across its grid-stride loop.
  float s[8] = {0}, ss[8] = {0};
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  if constexpr (sizeof(T) == 2) {
    for (int64_t i = i0; i < total; i += stride) {
      short4v v0 = *reinterpret_cast<const short4v*>(reinterpret_cast<const short*>(x) + i);
      short4v v1 = *reinterpret_cast<const short4v*>(reinterpret_cast<const short*>(x) + i + 4);
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        float f0 = bf16_bits_to_f(v0[k]);
        float f1 = bf16_bits_to_f(v1[k]);
        s[k] += f0; ss[k] += f0 * f0;
        s[k + 4] += f1; ss[k + 4] += f1 * f1;
      }
    }
  } else {
    for (int64_t i = i0; i < total; i += stride) {
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = faa_to_float(x[i + k]);
        s[k] += f; ss[k] += f * f;
      }
    }
  }
  // fold threads with the same channel octet through LDS (no atomics).
  // thread t owns octet ((bid*blockDim + t) % groups); for channel c the
  // owner threads are t = ((c/8 - bid*blockDim) mod groups) + k*groups.
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = s[k];
  __syncthreads();
  float* out = scratch + (int64_t)blockIdx.x * 2 * C;
  const int groups = C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  // strided channel loop: C can exceed blockDim (up to 2048 channels/block)
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + lane];
    out[ch] = acc;
  }
  __syncthreads();
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = ss[k];
  __syncthreads();
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + lane];
    out[C + ch] = acc;
  }
  bn_fin_tail_run(tail, scratch, C);
}

// any-C scalar reduce (C % 8 != 0, e.g. PyramidNet's rounded widths): the
// grid stride (nblocks*256) is chosen a multiple of C on the host, so each
// thread's channel c = i0 % C is loop-invariant; per-block partials are
// folded through LDS (no atomics) and finalized like the vec path.
template <typename T>
__global__ void bn_reduce_anyc_kernel(const T* __restrict__ x, float* __restrict__ scratch,
                                      int64_t total, int C, BnFinTail tail = {}) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float s = 0, ss = 0;
  for (int64_t i = i0; i < total; i += stride) {
    float f = faa_to_float(x[i]);
    s += f; ss += f * f;
  }
  // thread t owns channel (bid*256+t) % C; fold same-channel threads in LDS.
  // Channels with no owner thread in this block get an explicit 0 partial.
  __shared__ float lds[256];
  lds[threadIdx.x] = s;
  __syncthreads();
  float* out = scratch + (int64_t)blockIdx.x * 2 * C;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % C);
  for (int c = threadIdx.x; c < C; c += (int)blockDim.x) {
    int t0 = (c - shift + C) % C;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += C) acc += lds[t];
    out[c] = acc;
  }
  __syncthreads();
  lds[threadIdx.x] = ss;
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += (int)blockDim.x) {
    int t0 = (c - shift + C) % C;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += C) acc += lds[t];
    out[C + c] = acc;
  }
  bn_fin_tail_run(tail, scratch, C);
}

// residual add fused with the FOLLOWING BatchNorm's fwd reduce: the join
// points of the pre-act nets (WideBasic's `out + shortcut`, reference
// wideresnet.py:41) feed straight into the next block's bn1, so summing
// sum/sumsq while writing the add result removes bn_reduce's separate
// full-tensor read pass (profiles/step_profile_wrn40_2_r02.txt: at::add
// 3.1% + bn_reduce_vec 5.1%). Same channel-invariance contract as
// bn_reduce_vec_kernel (host picks nb via bn_nblocks).
template <typename T>
__global__ void add_bn_reduce_vec_kernel(const T* __restrict__ a,
                                         const T* __restrict__ b,
                                         T* __restrict__ out,
                                         float* __restrict__ scratch,
                                         int64_t total, int C) {
  float s[8] = {0}, ss[8] = {0};
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  if constexpr (sizeof(T) == 2) {
    const short* as = reinterpret_cast<const short*>(a);
    const short* bs = reinterpret_cast<const short*>(b);
    short* os = reinterpret_cast<short*>(out);
    for (int64_t i = i0; i < total; i += stride) {
      short4v a0 = *reinterpret_cast<const short4v*>(as + i);
      short4v a1 = *reinterpret_cast<const short4v*>(as + i + 4);
      short4v b0 = *reinterpret_cast<const short4v*>(bs + i);
      short4v b1 = *reinterpret_cast<const short4v*>(bs + i + 4);
      short4v o0, o1;
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        float f0 = bf16_bits_to_f(a0[k]) + bf16_bits_to_f(b0[k]);
        float f1 = bf16_bits_to_f(a1[k]) + bf16_bits_to_f(b1[k]);
        o0[k] = f_to_bf16_bits(f0);
        o1[k] = f_to_bf16_bits(f1);
        // accumulate the ROUNDED value: the next BN reads the bf16 result
        f0 = bf16_bits_to_f(o0[k]);
        f1 = bf16_bits_to_f(o1[k]);
        s[k] += f0; ss[k] += f0 * f0;
        s[k + 4] += f1; ss[k + 4] += f1 * f1;
      }
      *reinterpret_cast<short4v*>(os + i) = o0;
      *reinterpret_cast<short4v*>(os + i + 4) = o1;
    }
  } else {
    for (int64_t i = i0; i < total; i += stride) {
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = faa_to_float(a[i + k]) + faa_to_float(b[i + k]);
        faa_from_float(f, &out[i + k]);
        s[k] += f; ss[k] += f * f;
      }
    }
  }
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = s[k];
  __syncthreads();
  float* outp = scratch + (int64_t)blockIdx.x * 2 * C;
  const int groups = C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + lane];
    outp[ch] = acc;
  }
  __syncthreads();
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = ss[k];
  __syncthreads();
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups)
      acc += lds[t * 8 + lane];
    outp[C + ch] = acc;
  }
}

// one 64-lane wave per channel: lanes stride the partial blocks in parallel
// (a serial per-thread loop over ~64 partials costs ~16us in pure latency)
__global__ void bn_finalize_kernel(const float* __restrict__ scratch, int nblocks,
                                   float* __restrict__ mean, float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, int64_t count, float eps, float momentum) {
  int c = blockIdx.x;
  if (c >= C) return;
  float s = 0, ss = 0;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x) {
    s += scratch[(int64_t)b * 2 * C + c];
    ss += scratch[(int64_t)b * 2 * C + C + c];
  }
  s = faa_warp_reduce_sum(s);
  ss = faa_warp_reduce_sum(ss);
  if (threadIdx.x != 0) return;
  float m = s / count;
  float var = fmaxf(ss / count - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    float unbiased = count > 1 ? var * (float)count / (float)(count - 1) : var;
    running_mean[c] = (1.0f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.0f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// ---------------------------------------------------------------- fwd apply

// fused activation after the affine transform: 0 = none, 1 = relu, 2 = swish
template <int ACT>
__device__ __forceinline__ float faa_bn_act(float z) {
  if constexpr (ACT == 1) {
    return fmaxf(z, 0.0f);
  } else if constexpr (ACT == 2) {
    return z / (1.0f + __expf(-z));
  }
  return z;
}

// dz/dy for the fused activation. relu gates on the saved output; swish
// recomputes the pre-activation z = (x-mean)*invstd*gamma+beta.
template <int ACT>
__device__ __forceinline__ float faa_bn_act_grad(float g, float out_v, float z) {
  if constexpr (ACT == 1) {
    return out_v > 0.0f ? g : 0.0f;
  } else if constexpr (ACT == 2) {
    float s = 1.0f / (1.0f + __expf(-z));
    return g * s * (1.0f + z * (1.0f - s));
  }
  (void)out_v; (void)z;
  return g;
}

template <typename T, typename GT, int ACT>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ out,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const GT* __restrict__ gamma,
                                const GT* __restrict__ beta,
                                int64_t total, int C) {
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = i0; i < total; i += stride) {
    int c0 = (int)(i % C);
    if constexpr (sizeof(T) == 2) {
      if (i + 8 <= total) {
        const short* xs = reinterpret_cast<const short*>(x);
        short4v v0 = *reinterpret_cast<const short4v*>(xs + i);
        short4v v1 = *reinterpret_cast<const short4v*>(xs + i + 4);
        short4v o0, o1;
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
          int ca = c0 + k; while (ca >= C) ca -= C;
          int cb = c0 + k + 4; while (cb >= C) cb -= C;
          float a = (bf16_bits_to_f(v0[k]) - mean[ca]) * invstd[ca] * faa_to_float(gamma[ca]) + faa_to_float(beta[ca]);
          float b = (bf16_bits_to_f(v1[k]) - mean[cb]) * invstd[cb] * faa_to_float(gamma[cb]) + faa_to_float(beta[cb]);
          o0[k] = f_to_bf16_bits(faa_bn_act<ACT>(a));
          o1[k] = f_to_bf16_bits(faa_bn_act<ACT>(b));
        }
        short* os = reinterpret_cast<short*>(out);
        *reinterpret_cast<short4v*>(os + i) = o0;
        *reinterpret_cast<short4v*>(os + i + 4) = o1;
        continue;
      }
    }
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      int64_t j = i + k;
      if (j < total) {
        int c = c0 + k; while (c >= C) c -= C;
        float v = (faa_to_float(x[j]) - mean[c]) * invstd[c] * faa_to_float(gamma[c]) + faa_to_float(beta[c]);
        faa_from_float(faa_bn_act<ACT>(v), &out[j]);
      }
    }
  }
}

// ---------------------------------------------------------------- bwd reduce
template <typename T, typename GT, int ACT>
__global__ void bn_bwd_reduce_vec_kernel(const T* __restrict__ x, const T* __restrict__ out,
                                         const T* __restrict__ dy,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ invstd,
                                         const GT* __restrict__ gamma,
                                         const GT* __restrict__ beta,
                                         float* __restrict__ scratch,
                                         int64_t total, int C,
                                         BnBwdFinTail<GT> tail = {}) {
  float sdy[8] = {0}, sdyx[8] = {0};
  const int c0 = (int)((((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8) % C);
  float m[8], is[8], ga[8], be[8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) {
    int c = (c0 + k) % C;
    m[k] = mean[c]; is[k] = invstd[c];
    ga[k] = faa_to_float(gamma[c]); be[k] = faa_to_float(beta[c]);
  }
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  if constexpr (sizeof(T) == 2) {
    const short* xs = reinterpret_cast<const short*>(x);
    const short* os = reinterpret_cast<const short*>(out);
    const short* ds = reinterpret_cast<const short*>(dy);
    for (int64_t i = i0; i < total; i += stride) {
      short4v xv0 = *reinterpret_cast<const short4v*>(xs + i);
      short4v xv1 = *reinterpret_cast<const short4v*>(xs + i + 4);
      short4v ov0 = *reinterpret_cast<const short4v*>(os + i);
      short4v ov1 = *reinterpret_cast<const short4v*>(os + i + 4);
      short4v dv0 = *reinterpret_cast<const short4v*>(ds + i);
      short4v dv1 = *reinterpret_cast<const short4v*>(ds + i + 4);
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        float xh0 = (bf16_bits_to_f(xv0[k]) - m[k]) * is[k];
        float xh1 = (bf16_bits_to_f(xv1[k]) - m[k + 4]) * is[k + 4];
        float g0 = faa_bn_act_grad<ACT>(bf16_bits_to_f(dv0[k]), bf16_bits_to_f(ov0[k]),
                                        xh0 * ga[k] + be[k]);
        float g1 = faa_bn_act_grad<ACT>(bf16_bits_to_f(dv1[k]), bf16_bits_to_f(ov1[k]),
                                        xh1 * ga[k + 4] + be[k + 4]);
        sdy[k] += g0;
        sdyx[k] += g0 * xh0;
        sdy[k + 4] += g1;
        sdyx[k + 4] += g1 * xh1;
      }
    }
  } else {
    for (int64_t i = i0; i < total; i += stride) {
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float xh = (faa_to_float(x[i + k]) - m[k]) * is[k];
        float g = faa_bn_act_grad<ACT>(faa_to_float(dy[i + k]), faa_to_float(out[i + k]),
                                       xh * ga[k] + be[k]);
        sdy[k] += g;
        sdyx[k] += g * xh;
      }
    }
  }
  __shared__ float lds[256 * 8];
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = sdy[k];
  __syncthreads();
  float* outp = scratch + (int64_t)blockIdx.x * 2 * C;
  const int groups = C / 8;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % groups);
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups) acc += lds[t * 8 + lane];
    outp[ch] = acc;
  }
  __syncthreads();
  #pragma unroll
  for (int k = 0; k < 8; ++k) lds[threadIdx.x * 8 + k] = sdyx[k];
  __syncthreads();
  for (int ch = threadIdx.x; ch < C; ch += (int)blockDim.x) {
    int oct = ch / 8, lane = ch % 8;
    int t0 = (oct - shift + groups) % groups;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += groups) acc += lds[t * 8 + lane];
    outp[C + ch] = acc;
  }
  bn_bwd_fin_tail_run(tail, scratch, C);
}

// any-C bwd reduce: same channel-invariant decomposition as bn_reduce_anyc.
template <typename T, typename GT, int ACT>
__global__ void bn_bwd_reduce_anyc_kernel(const T* __restrict__ x, const T* __restrict__ out,
                                          const T* __restrict__ dy,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          const GT* __restrict__ gamma,
                                          const GT* __restrict__ beta,
                                          float* __restrict__ scratch,
                                          int64_t total, int C,
                                          BnBwdFinTail<GT> tail = {}) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int cown = (int)(i0 % C);
  const float m = mean[cown], is = invstd[cown];
  const float ga = faa_to_float(gamma[cown]), be = faa_to_float(beta[cown]);
  float sdy = 0, sdyx = 0;
  for (int64_t i = i0; i < total; i += stride) {
    float xh = (faa_to_float(x[i]) - m) * is;
    float g = faa_bn_act_grad<ACT>(faa_to_float(dy[i]), faa_to_float(out[i]),
                                   xh * ga + be);
    sdy += g;
    sdyx += g * xh;
  }
  __shared__ float lds[256];
  lds[threadIdx.x] = sdy;
  __syncthreads();
  float* outp = scratch + (int64_t)blockIdx.x * 2 * C;
  const int shift = (int)(((int64_t)blockIdx.x * blockDim.x) % C);
  for (int c = threadIdx.x; c < C; c += (int)blockDim.x) {
    int t0 = (c - shift + C) % C;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += C) acc += lds[t];
    outp[c] = acc;
  }
  __syncthreads();
  lds[threadIdx.x] = sdyx;
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += (int)blockDim.x) {
    int t0 = (c - shift + C) % C;
    float acc = 0;
    for (int t = t0; t < (int)blockDim.x; t += C) acc += lds[t];
    outp[C + c] = acc;
  }
  bn_bwd_fin_tail_run(tail, scratch, C);
}

template <typename GT>
__global__ void bn_bwd_finalize_kernel(const float* __restrict__ scratch, int nblocks,
                                       float* __restrict__ sums,
                                       GT* __restrict__ dbeta, GT* __restrict__ dgamma,
                                       int C) {
  // sums: [2C] = {sum_dy (=dbeta), sum_dy_xhat (=dgamma)}; one wave per element
  int c = blockIdx.x;
  if (c >= 2 * C) return;
  float s = 0;
  for (int b = threadIdx.x; b < nblocks; b += blockDim.x)
    s += scratch[(int64_t)b * 2 * C + c];
  s = faa_warp_reduce_sum(s);
  if (threadIdx.x == 0) {
    sums[c] = s;
    if (c < C) faa_from_float(s, &dbeta[c]);
    else faa_from_float(s, &dgamma[c - C]);
  }
}

template <typename T, typename GT, int ACT>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x, const T* __restrict__ out,
                                    const T* __restrict__ dy, T* __restrict__ dx,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const GT* __restrict__ gamma,
                                    const GT* __restrict__ beta,
                                    const float* __restrict__ sums,
                                    int64_t total, int C, int64_t count, int training) {
  int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  float inv_count = 1.0f / (float)count;
  for (int64_t i = i0; i < total; i += stride) {
    int c0 = (int)(i % C);
    if constexpr (sizeof(T) == 2) {
      if (i + 8 <= total) {
        const short* xs = reinterpret_cast<const short*>(x);
        const short* os = reinterpret_cast<const short*>(out);
        const short* ds = reinterpret_cast<const short*>(dy);
        short4v xv0 = *reinterpret_cast<const short4v*>(xs + i);
        short4v xv1 = *reinterpret_cast<const short4v*>(xs + i + 4);
        short4v ov0 = *reinterpret_cast<const short4v*>(os + i);
        short4v ov1 = *reinterpret_cast<const short4v*>(os + i + 4);
        short4v dv0 = *reinterpret_cast<const short4v*>(ds + i);
        short4v dv1 = *reinterpret_cast<const short4v*>(ds + i + 4);
        short4v r0, r1;
        #pragma unroll
        for (int k = 0; k < 4; ++k) {
          int ca = c0 + k; while (ca >= C) ca -= C;
          int cb = c0 + k + 4; while (cb >= C) cb -= C;
          float isa = invstd[ca], isb = invstd[cb];
          float gca = faa_to_float(gamma[ca]), gcb = faa_to_float(gamma[cb]);
          float xa = (bf16_bits_to_f(xv0[k]) - mean[ca]) * isa;
          float xb = (bf16_bits_to_f(xv1[k]) - mean[cb]) * isb;
          float ga = faa_bn_act_grad<ACT>(bf16_bits_to_f(dv0[k]), bf16_bits_to_f(ov0[k]),
                                          xa * gca + faa_to_float(beta[ca]));
          float gb = faa_bn_act_grad<ACT>(bf16_bits_to_f(dv1[k]), bf16_bits_to_f(ov1[k]),
                                          xb * gcb + faa_to_float(beta[cb]));
          float ra, rb;
          if (training) {
            ra = gca * isa * (ga - sums[ca] * inv_count - xa * sums[C + ca] * inv_count);
            rb = gcb * isb * (gb - sums[cb] * inv_count - xb * sums[C + cb] * inv_count);
          } else {
            ra = gca * isa * ga;
            rb = gcb * isb * gb;
          }
          r0[k] = f_to_bf16_bits(ra);
          r1[k] = f_to_bf16_bits(rb);
        }
        short* dxs = reinterpret_cast<short*>(dx);
        *reinterpret_cast<short4v*>(dxs + i) = r0;
        *reinterpret_cast<short4v*>(dxs + i + 4) = r1;
        continue;
      }
    }
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      int64_t j = i + k;
      if (j < total) {
        int c = c0 + k; while (c >= C) c -= C;
        float is = invstd[c];
        float gc = faa_to_float(gamma[c]);
        float xhat = (faa_to_float(x[j]) - mean[c]) * is;
        float g = faa_bn_act_grad<ACT>(faa_to_float(dy[j]), faa_to_float(out[j]),
                                       xhat * gc + faa_to_float(beta[c]));
        float res;
        if (training) {
          res = gc * is * (g - sums[c] * inv_count - xhat * sums[C + c] * inv_count);
        } else {
          res = gc * is * g;
        }
        faa_from_float(res, &dx[j]);
      }
    }
  }
}

}  // namespace

#define DISPATCH_FB(TYPE, NAME, ...)                                           \
  [&] {                                                                        \
    if (TYPE == torch::kFloat32) { using scalar_t = float; return __VA_ARGS__(); } \
    else if (TYPE == torch::kBFloat16) { using scalar_t = __hip_bfloat16; return __VA_ARGS__(); } \
    else { TORCH_CHECK(false, NAME ": unsupported dtype"); }                  \
  }()

// partial-sum block count: the grid stride (nblocks*256*8) must be a
// multiple of C so each thread's channel octet is loop-invariant.
// persistent device counter for the last-block finalize (allocated on
// first eager use, i.e. before any graph capture; reset by the last block)
static unsigned* bn_lastblock_counter() {
  static unsigned* p = nullptr;
  if (p == nullptr) {
    (void)hipMalloc(&p, sizeof(unsigned));
    (void)hipMemset(p, 0, sizeof(unsigned));
  }
  return p;
}

static bool bn_lastblock_enabled() {
  const char* e = getenv("FAA_BN_LASTBLOCK");
  return e && e[0] == '1';
}

static int bn_nblocks(int C, int64_t total) {
  auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
  int q = C / gcd(C, 2048);
  // target ~8 vector iterations per thread, clamp [64, 512] then align to q
  int64_t want = total / 8 / 256 / 8;
  int base = (int)std::min<int64_t>(std::max<int64_t>(want, 64), 512);
  return ((base + q - 1) / q) * q;
}

// 3-way activation dispatch: f receives integral_constant<int, ACT>
template <typename F>
static void act_dispatch(int act, F&& f) {
  if (act == 1) f(std::integral_constant<int, 1>{});
  else if (act == 2) f(std::integral_constant<int, 2>{});
  else f(std::integral_constant<int, 0>{});
}

// any-C variant: grid stride is nblocks*256 (scalar elements), must be a
// multiple of C for channel invariance
static int bn_nblocks_anyc(int C, int64_t total) {
  auto gcd = [](int a, int b) { while (b) { int t = a % b; a = b; b = t; } return a; };
  int q = C / gcd(C, 256);
  int64_t want = total / 256 / 32;
  int base = (int)std::min<int64_t>(std::max<int64_t>(want, 64), 1024);
  return ((base + q - 1) / q) * q;
}

std::vector<torch::Tensor> residual_add_bn_stats(torch::Tensor a, torch::Tensor b) {
  // out = a + b (NHWC) + per-block BN partials for the FOLLOWING bn_relu_fwd
  auto ac = a.contiguous(torch::MemoryFormat::ChannelsLast);
  auto bc = b.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = ac.size(1);
  int64_t total = ac.numel();
  TORCH_CHECK(C % 8 == 0 && total % 8 == 0 && C <= 2048,
              "residual_add_bn_stats: vec-path geometry required");
  auto out = torch::empty_like(ac);
  int nb = bn_nblocks(C, total);
  auto scratch = torch::empty({nb, 2 * C}, ac.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream().stream();
  DISPATCH_FB(ac.scalar_type(), "add_bn_reduce", [&] {
    hipLaunchKernelGGL((add_bn_reduce_vec_kernel<scalar_t>), dim3(nb), dim3(256),
                       0, stream, (const scalar_t*)ac.data_ptr(),
                       (const scalar_t*)bc.data_ptr(), (scalar_t*)out.data_ptr(),
                       scratch.data_ptr<float>(), total, C);
  });
  return {out, scratch};
}

std::vector<torch::Tensor> bn_relu_fwd(torch::Tensor x, torch::Tensor gamma,
                                       torch::Tensor beta, torch::Tensor running_mean,
                                       torch::Tensor running_var, bool training,
                                       double momentum, double eps, int64_t act,
                                       torch::Tensor pre_scratch) {
  TORCH_CHECK(x.dim() == 4, "bn_relu: 4-D input expected");
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = xc.size(1);
  int64_t total = xc.numel();
  int64_t rows = total / C;
  auto out = torch::empty_like(xc);
  auto f32 = xc.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto invstd = torch::empty({C}, f32);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto g = gamma.contiguous();
  auto bta = beta.contiguous();
  TORCH_CHECK(g.scalar_type() == bta.scalar_type(), "gamma/beta dtype mismatch");

  if (training) {
    bool vec = (C % 8 == 0) && (total % 8 == 0) && (C <= 2048);
    BnFinTail tail{};
    bool lastblock = bn_lastblock_enabled();
    if (lastblock) {
      tail.counter = bn_lastblock_counter();
      tail.mean = mean.data_ptr<float>();
      tail.invstd = invstd.data_ptr<float>();
      tail.running_mean = running_mean.defined() ? running_mean.data_ptr<float>() : nullptr;
      tail.running_var = running_var.defined() ? running_var.data_ptr<float>() : nullptr;
      tail.count = rows;
      tail.eps = (float)eps;
      tail.momentum = (float)momentum;
    }
    bool have_pre = pre_scratch.defined() && pre_scratch.numel() > 0
        && pre_scratch.dim() == 2 && pre_scratch.size(1) == 2 * C;
    if (have_pre && !lastblock) {
      // partials precomputed by residual_add_bn_stats: skip the reduce pass
      int nb = pre_scratch.size(0);
      hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(64), 0, stream,
                         pre_scratch.data_ptr<float>(), nb, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(),
                         running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                         running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                         C, rows, (float)eps, (float)momentum);
    } else if (vec) {
      int nb = bn_nblocks(C, total);
      auto scratch = torch::empty({nb, 2 * C}, f32);
      DISPATCH_FB(xc.scalar_type(), "bn_reduce", [&] {
        hipLaunchKernelGGL((bn_reduce_vec_kernel<scalar_t>), dim3(nb), dim3(256), 0,
                           stream, (const scalar_t*)xc.data_ptr(),
                           scratch.data_ptr<float>(), total, C, tail);
      });
      if (!lastblock)
        hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(64), 0, stream,
                           scratch.data_ptr<float>(), nb, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(),
                           running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                           running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                           C, rows, (float)eps, (float)momentum);
    } else {
      int nb = bn_nblocks_anyc(C, total);
      auto scratch = torch::empty({nb, 2 * C}, f32);
      DISPATCH_FB(xc.scalar_type(), "bn_reduce_s", [&] {
        hipLaunchKernelGGL((bn_reduce_anyc_kernel<scalar_t>), dim3(nb), dim3(256), 0,
                           stream, (const scalar_t*)xc.data_ptr(),
                           scratch.data_ptr<float>(), total, C, tail);
      });
      if (!lastblock)
        hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(64), 0, stream,
                           scratch.data_ptr<float>(), nb, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(),
                           running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                           running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                           C, rows, (float)eps, (float)momentum);
    }
  } else {
    mean.copy_(running_mean);
    invstd.copy_(torch::rsqrt(running_var + eps));
  }
  int grid1 = faa_grid(total / 8 + 1, 256);
  DISPATCH_FB(xc.scalar_type(), "bn_apply", [&] {
    using data_t = scalar_t;
    DISPATCH_FB(g.scalar_type(), "bn_apply_g", [&] {
      act_dispatch((int)act, [&](auto A) {
        hipLaunchKernelGGL((bn_apply_kernel<data_t, scalar_t, decltype(A)::value>),
                           dim3(grid1), dim3(256), 0,
                           stream, (const data_t*)xc.data_ptr(), (data_t*)out.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           (const scalar_t*)g.data_ptr(), (const scalar_t*)bta.data_ptr(),
                           total, C);
      });
    });
  });
  return {out, mean, invstd};
}

std::vector<torch::Tensor> bn_relu_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor out, torch::Tensor mean,
                                       torch::Tensor invstd, torch::Tensor gamma,
                                       torch::Tensor beta, bool training, int64_t act) {
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto oc = out.contiguous(torch::MemoryFormat::ChannelsLast);
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = xc.size(1);
  int64_t total = xc.numel();
  int64_t rows = total / C;
  auto f32 = xc.options().dtype(torch::kFloat32);
  auto dx = torch::empty_like(xc);
  auto g = gamma.contiguous();
  auto bta = beta.contiguous();
  TORCH_CHECK(g.scalar_type() == bta.scalar_type(), "gamma/beta dtype mismatch");
  auto sums = torch::empty({2 * C}, f32);
  auto dgamma = torch::empty({C}, xc.options().dtype(g.scalar_type()));
  auto dbeta = torch::empty({C}, xc.options().dtype(g.scalar_type()));
  auto stream = at::hip::getCurrentHIPStream().stream();

  bool vec = (C % 8 == 0) && (total % 8 == 0) && (C <= 2048);
  int nb = vec ? bn_nblocks(C, total) : bn_nblocks_anyc(C, total);
  auto scratch = torch::empty({nb, 2 * C}, f32);
  bool lastblock = bn_lastblock_enabled();
  DISPATCH_FB(xc.scalar_type(), "bn_bwd_reduce", [&] {
    using data_t = scalar_t;
    DISPATCH_FB(g.scalar_type(), "bn_bwd_reduce_g", [&] {
      BnBwdFinTail<scalar_t> tail{};
      if (lastblock) {
        tail.counter = bn_lastblock_counter();
        tail.sums = sums.data_ptr<float>();
        tail.dbeta = (scalar_t*)dbeta.data_ptr();
        tail.dgamma = (scalar_t*)dgamma.data_ptr();
      }
      act_dispatch((int)act, [&](auto A) {
        constexpr int kAct = decltype(A)::value;
        if (vec)
          hipLaunchKernelGGL((bn_bwd_reduce_vec_kernel<data_t, scalar_t, kAct>),
                             dim3(nb), dim3(256), 0,
                             stream, (const data_t*)xc.data_ptr(), (const data_t*)oc.data_ptr(),
                             (const data_t*)dyc.data_ptr(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), (const scalar_t*)g.data_ptr(),
                             (const scalar_t*)bta.data_ptr(), scratch.data_ptr<float>(),
                             total, C, tail);
        else
          hipLaunchKernelGGL((bn_bwd_reduce_anyc_kernel<data_t, scalar_t, kAct>),
                             dim3(nb), dim3(256), 0,
                             stream, (const data_t*)xc.data_ptr(), (const data_t*)oc.data_ptr(),
                             (const data_t*)dyc.data_ptr(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), (const scalar_t*)g.data_ptr(),
                             (const scalar_t*)bta.data_ptr(), scratch.data_ptr<float>(),
                             total, C, tail);
      });
    });
  });
  if (!lastblock)
    DISPATCH_FB(g.scalar_type(), "bn_bwd_fin", [&] {
      hipLaunchKernelGGL((bn_bwd_finalize_kernel<scalar_t>), dim3(2 * C), dim3(64), 0,
                         stream, scratch.data_ptr<float>(), nb, sums.data_ptr<float>(),
                         (scalar_t*)dbeta.data_ptr(), (scalar_t*)dgamma.data_ptr(), C);
    });
  int grid1 = faa_grid(total / 8 + 1, 256);
  DISPATCH_FB(xc.scalar_type(), "bn_bwd_apply", [&] {
    using data_t = scalar_t;
    DISPATCH_FB(g.scalar_type(), "bn_bwd_apply_g", [&] {
      act_dispatch((int)act, [&](auto A) {
        hipLaunchKernelGGL((bn_bwd_apply_kernel<data_t, scalar_t, decltype(A)::value>),
                           dim3(grid1), dim3(256), 0,
                           stream, (const data_t*)xc.data_ptr(), (const data_t*)oc.data_ptr(),
                           (const data_t*)dyc.data_ptr(), (data_t*)dx.data_ptr(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           (const scalar_t*)g.data_ptr(), (const scalar_t*)bta.data_ptr(),
                           sums.data_ptr<float>(), total, C, rows, training ? 1 : 0);
      });
    });
  });
  return {dx, dgamma, dbeta};
}
