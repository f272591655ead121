// Shared helpers for the fast_autoaugment_amd CDNA4 (gfx950) kernels.
//
// Conventions (see /opt/skills/guides/cdna_hip_programming.md):
//  * wavefront = 64 lanes; block sizes are multiples of 64
//  * memory-bound kernels vectorize bf16 as ushort-packed loads (G13)
//  * grid-stride loops capped near 2048 blocks keep launch overhead low (G11)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define FAA_WAVE 64
#define FAA_CHECK(cond, msg) TORCH_CHECK(cond, msg)

static inline int faa_grid(int64_t total, int block, int cap = 2048) {
  int64_t g = (total + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

__device__ __forceinline__ float faa_warp_reduce_sum(float v) {
  #pragma unroll
  for (int off = FAA_WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, FAA_WAVE);
  return v;
}

// block-level sum into lds scratch (needs blockDim.x/64 floats of lds)
__device__ __forceinline__ float faa_block_reduce_sum(float v, float* lds) {
  int lane = threadIdx.x & (FAA_WAVE - 1);
  int wid = threadIdx.x / FAA_WAVE;
  v = faa_warp_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  int nw = blockDim.x / FAA_WAVE;
  v = ((int)threadIdx.x < nw) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) v = faa_warp_reduce_sum(v);
  return v;  // valid in thread 0
}

// ---- dtype helpers ---------------------------------------------------------
template <typename T> struct FaaVec8;          // 8-element packed load type

template <> struct FaaVec8<float> {
  using type = float4;                          // 2x float4 loads instead
};

__device__ __forceinline__ float faa_to_float(float x) { return x; }
__device__ __forceinline__ float faa_to_float(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ void faa_from_float(float v, float* o) { *o = v; }
__device__ __forceinline__ void faa_from_float(float v, __hip_bfloat16* o) { *o = __float2bfloat16(v); }
