// Fused flat-buffer optimizer step (reference update chain train.py:61-70):
//   1. wd_norm_kernel: g[:n_decay] += wd * p[:n_decay]   (manual non-BN L2)
//                      + block-partial ||g||^2 accumulated to normsq[0]
//   2. sgd_step_kernel: coef = min(1, clip/(sqrt(normsq)+1e-6)) read ON
//      DEVICE (no host sync -> hipGraph-capturable), nesterov momentum
//      update, p -= lr*(g*coef + mu*buf')  [torch SGD semantics]
//   3. ema_lerp_kernel: shadow = (1-mu)*x + mu*shadow on a flat buffer
//
// All buffers are fp32 and contiguous (parallel/flat.py), so both kernels
// stream at HBM bandwidth with float4 vector access.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

__global__ void zero1_kernel(float* p) { if (threadIdx.x == 0) p[0] = 0.0f; }

__global__ void wd_norm_kernel(float* __restrict__ g, const float* __restrict__ p,
                               float* __restrict__ normsq, int64_t n, int64_t n_decay,
                               float wd) {
  __shared__ float lds[8];
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  float acc = 0.0f;
  for (int64_t i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      float4 gv = *reinterpret_cast<float4*>(g + i);
      if (wd != 0.0f && i + 3 < n_decay) {
        const float4 pv = *reinterpret_cast<const float4*>(p + i);
        gv.x += wd * pv.x; gv.y += wd * pv.y; gv.z += wd * pv.z; gv.w += wd * pv.w;
        *reinterpret_cast<float4*>(g + i) = gv;
      } else if (wd != 0.0f && i < n_decay) {
        // straddles the decay boundary: scalar tail
        for (int v = 0; v < 4; ++v)
          if (i + v < n_decay) g[i + v] += wd * p[i + v];
        gv = *reinterpret_cast<float4*>(g + i);
      }
      acc += gv.x * gv.x + gv.y * gv.y + gv.z * gv.z + gv.w * gv.w;
    } else {
      for (int64_t j = i; j < n; ++j) {
        if (wd != 0.0f && j < n_decay) g[j] += wd * p[j];
        float gv = g[j];
        acc += gv * gv;
      }
    }
  }
  acc = faa_block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(normsq, acc);
}

__global__ void sgd_step_kernel(float* __restrict__ p, float* __restrict__ g,
                                float* __restrict__ buf, const float* __restrict__ normsq,
                                const float* __restrict__ lr_p,
                                int64_t n, float clip, float mu, int nesterov) {
  float lr = lr_p[0];     // device-resident lr -> hipGraph-replayable
  float coef = 1.0f;
  if (clip > 0.0f) {
    float norm = sqrtf(normsq[0]);
    coef = fminf(1.0f, clip / (norm + 1e-6f));
  }
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      float4 gv = *reinterpret_cast<float4*>(g + i);
      float4 bv = *reinterpret_cast<float4*>(buf + i);
      float4 pv = *reinterpret_cast<float4*>(p + i);
      #pragma unroll
      for (int v = 0; v < 4; ++v) {
        float* gp = &gv.x + v; float* bp = &bv.x + v; float* pp = &pv.x + v;
        float gg = *gp * coef;
        float bb = mu * *bp + gg;
        float upd = nesterov ? (gg + mu * bb) : bb;
        *bp = bb;
        *pp = *pp - lr * upd;
      }
      *reinterpret_cast<float4*>(buf + i) = bv;
      *reinterpret_cast<float4*>(p + i) = pv;
      // also store clipped grad so later consumers (clip-aware logs) see it
      gv.x *= coef; gv.y *= coef; gv.z *= coef; gv.w *= coef;
      *reinterpret_cast<float4*>(g + i) = gv;
    } else {
      for (int64_t j = i; j < n; ++j) {
        float gg = g[j] * coef;
        float bb = mu * buf[j] + gg;
        float upd = nesterov ? (gg + mu * bb) : bb;
        buf[j] = bb;
        p[j] -= lr * upd;
        g[j] = gg;
      }
    }
  }
}

__global__ void ema_lerp_kernel(float* __restrict__ shadow, const float* __restrict__ x,
                                int64_t n, float mu) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      float4 sv = *reinterpret_cast<float4*>(shadow + i);
      const float4 xv = *reinterpret_cast<const float4*>(x + i);
      sv.x = (1.0f - mu) * xv.x + mu * sv.x;
      sv.y = (1.0f - mu) * xv.y + mu * sv.y;
      sv.z = (1.0f - mu) * xv.z + mu * sv.z;
      sv.w = (1.0f - mu) * xv.w + mu * sv.w;
      *reinterpret_cast<float4*>(shadow + i) = sv;
    } else {
      for (int64_t j = i; j < n; ++j)
        shadow[j] = (1.0f - mu) * x[j] + mu * shadow[j];
    }
  }
}

// ---- mixed-precision variant: bf16 working weights/grads + fp32 master ----
// The model computes natively in bf16 (no autocast cast kernels); grads land
// in a flat bf16 buffer; the step reads bf16 grads, applies manual WD from
// the fp32 master, clips by the global norm, updates the fp32 master and
// re-quantizes the bf16 working copy. Norm pass is read-only (no wd write).

typedef __attribute__((ext_vector_type(4))) short short4x;

__device__ __forceinline__ float bits_to_f32(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__global__ void wd_norm_mixed_kernel(const short* __restrict__ g,
                                     const float* __restrict__ p,
                                     float* __restrict__ normsq,
                                     int64_t n, int64_t n_decay, float wd) {
  __shared__ float lds[8];
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  float acc = 0.0f;
  for (int64_t i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      short4x gv = *reinterpret_cast<const short4x*>(g + i);
      #pragma unroll
      for (int v = 0; v < 4; ++v) {
        float gg = bits_to_f32(gv[v]);
        if (wd != 0.0f && i + v < n_decay) gg += wd * p[i + v];
        acc += gg * gg;
      }
    } else {
      for (int64_t j = i; j < n; ++j) {
        float gg = bits_to_f32(g[j]);
        if (wd != 0.0f && j < n_decay) gg += wd * p[j];
        acc += gg * gg;
      }
    }
  }
  acc = faa_block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(normsq, acc);
}

__global__ void sgd_step_mixed_kernel(float* __restrict__ p, short* __restrict__ w,
                                      const short* __restrict__ g,
                                      float* __restrict__ buf,
                                      const float* __restrict__ normsq,
                                      const float* __restrict__ lr_p,
                                      int64_t n, int64_t n_decay, float wd,
                                      float clip, float mu, int nesterov) {
  float lr = lr_p[0];
  float coef = 1.0f;
  if (clip > 0.0f) {
    float norm = sqrtf(normsq[0]);
    coef = fminf(1.0f, clip / (norm + 1e-6f));
  }
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      short4x gv = *reinterpret_cast<const short4x*>(g + i);
      float4 bv = *reinterpret_cast<float4*>(buf + i);
      float4 pv = *reinterpret_cast<float4*>(p + i);
      short4x wv;
      #pragma unroll
      for (int v = 0; v < 4; ++v) {
        float* bp = &bv.x + v; float* pp = &pv.x + v;
        float gg = bits_to_f32(gv[v]);
        if (wd != 0.0f && i + v < n_decay) gg += wd * *pp;
        gg *= coef;
        float bb = mu * *bp + gg;
        float upd = nesterov ? (gg + mu * bb) : bb;
        *bp = bb;
        *pp = *pp - lr * upd;
        __hip_bfloat16 h = __float2bfloat16(*pp);
        wv[v] = *reinterpret_cast<short*>(&h);
      }
      *reinterpret_cast<float4*>(buf + i) = bv;
      *reinterpret_cast<float4*>(p + i) = pv;
      *reinterpret_cast<short4x*>(w + i) = wv;
    } else {
      for (int64_t j = i; j < n; ++j) {
        float gg = bits_to_f32(g[j]);
        if (wd != 0.0f && j < n_decay) gg += wd * p[j];
        gg *= coef;
        float bb = mu * buf[j] + gg;
        float upd = nesterov ? (gg + mu * bb) : bb;
        buf[j] = bb;
        p[j] -= lr * upd;
        __hip_bfloat16 h = __float2bfloat16(p[j]);
        w[j] = *reinterpret_cast<short*>(&h);
      }
    }
  }
}

// ---- multi-tensor grad gather: scattered autograd grads -> flat buffer ----
// table rows: [src_ptr, dst_offset, numel] (int64). One block per tensor,
// vectorized bf16x8 copy. Lets backward run with .grad=None (no per-param
// accumulation add kernels); addresses are stable under hipGraph replay.
typedef __attribute__((ext_vector_type(4))) short short4g;

__global__ void gather_grads_kernel(const int64_t* __restrict__ table, int n,
                                    short* __restrict__ flat) {
  for (int e = blockIdx.x; e < n; e += gridDim.x) {
    const short* src = (const short*)table[e * 3];
    int64_t dst = table[e * 3 + 1];
    int64_t len = table[e * 3 + 2];
    int64_t i = (int64_t)threadIdx.x * 8;
    for (; i + 8 <= len; i += (int64_t)blockDim.x * 8) {
      *reinterpret_cast<short4g*>(flat + dst + i) =
          *reinterpret_cast<const short4g*>(src + i);
      *reinterpret_cast<short4g*>(flat + dst + i + 4) =
          *reinterpret_cast<const short4g*>(src + i + 4);
    }
    if (threadIdx.x == 0)
      for (int64_t j = (len / 8) * 8; j < len; ++j) flat[dst + j] = src[j];
  }
}

// ---- fused RMSpropTF step (reference tf_port/rmsprop.py:80-99 semantics:
// ms initialized to ONES by the host, eps INSIDE the sqrt), mixed bf16
// working weights + fp32 master/ms/mom, manual WD on the decay segment,
// optional global clip via the shared norm pass.
__global__ void rmsprop_step_mixed_kernel(float* __restrict__ p, short* __restrict__ w,
                                          const short* __restrict__ g,
                                          float* __restrict__ ms, float* __restrict__ mom,
                                          const float* __restrict__ normsq,
                                          const float* __restrict__ lr_p,
                                          int64_t n, int64_t n_decay, float wd,
                                          float clip, float rho, float mu, float eps) {
  float lr = lr_p[0];
  float coef = 1.0f;
  if (clip > 0.0f) {
    float norm = sqrtf(normsq[0]);
    coef = fminf(1.0f, clip / (norm + 1e-6f));
  }
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < n; i += stride) {
    int64_t lim = min(i + 4, n);
    for (int64_t j = i; j < lim; ++j) {
      float gg = bits_to_f32(g[j]);
      if (wd != 0.0f && j < n_decay) gg += wd * p[j];
      gg *= coef;
      float m = ms[j] + (gg * gg - ms[j]) * (1.0f - rho);
      ms[j] = m;
      float mo = mu * mom[j] + lr * gg * rsqrtf(m + eps);
      mom[j] = mo;
      p[j] -= mo;
      __hip_bfloat16 h = __float2bfloat16(p[j]);
      w[j] = *reinterpret_cast<short*>(&h);
    }
  }
}

}  // namespace

void rmsprop_fused_step_mixed(torch::Tensor master, torch::Tensor work, torch::Tensor g,
                              torch::Tensor ms, torch::Tensor mom, torch::Tensor normsq,
                              torch::Tensor lr_t, int64_t n_decay, double wd,
                              double clip, double rho, double momentum, double eps) {
  int64_t n = master.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int block = 256;
  int grid = faa_grid(n / 4 + 1, block);
  if (clip > 0.0) {
    hipLaunchKernelGGL(zero1_kernel, dim3(1), dim3(64), 0, stream, normsq.data_ptr<float>());
    hipLaunchKernelGGL(wd_norm_mixed_kernel, dim3(grid), dim3(block), 0, stream,
                       (const short*)g.data_ptr(), master.data_ptr<float>(),
                       normsq.data_ptr<float>(), n, n_decay, (float)wd);
  }
  hipLaunchKernelGGL(rmsprop_step_mixed_kernel, dim3(grid), dim3(block), 0, stream,
                     master.data_ptr<float>(), (short*)work.data_ptr(),
                     (const short*)g.data_ptr(), ms.data_ptr<float>(),
                     mom.data_ptr<float>(), normsq.data_ptr<float>(),
                     lr_t.data_ptr<float>(), n, n_decay, (float)wd, (float)clip,
                     (float)rho, (float)momentum, (float)eps);
}

void gather_grads(torch::Tensor table, torch::Tensor flat) {
  TORCH_CHECK(table.dtype() == torch::kInt64 && table.is_cuda());
  TORCH_CHECK(flat.dtype() == torch::kBFloat16);
  int n = table.size(0);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(gather_grads_kernel, dim3(std::min(n, 512)), dim3(256), 0, stream,
                     table.data_ptr<int64_t>(), n, (short*)flat.data_ptr());
}

void sgd_fused_step_mixed(torch::Tensor master, torch::Tensor work, torch::Tensor g,
                          torch::Tensor buf, torch::Tensor normsq, torch::Tensor lr_t,
                          int64_t n_decay, double wd, double clip, double momentum,
                          int64_t nesterov) {
  TORCH_CHECK(master.dtype() == torch::kFloat32 && work.dtype() == torch::kBFloat16
              && g.dtype() == torch::kBFloat16);
  int64_t n = master.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int block = 256;
  int grid = faa_grid(n / 4 + 1, block);
  hipLaunchKernelGGL(zero1_kernel, dim3(1), dim3(64), 0, stream, normsq.data_ptr<float>());
  hipLaunchKernelGGL(wd_norm_mixed_kernel, dim3(grid), dim3(block), 0, stream,
                     (const short*)g.data_ptr(), master.data_ptr<float>(),
                     normsq.data_ptr<float>(), n, n_decay, (float)wd);
  hipLaunchKernelGGL(sgd_step_mixed_kernel, dim3(grid), dim3(block), 0, stream,
                     master.data_ptr<float>(), (short*)work.data_ptr(),
                     (const short*)g.data_ptr(), buf.data_ptr<float>(),
                     normsq.data_ptr<float>(), lr_t.data_ptr<float>(), n, n_decay,
                     (float)wd, (float)clip, (float)momentum, (int)nesterov);
}

void sgd_fused_step(torch::Tensor p, torch::Tensor g, torch::Tensor buf,
                    torch::Tensor normsq, torch::Tensor lr_t, int64_t n_decay,
                    double wd, double clip, double momentum, int64_t nesterov) {
  TORCH_CHECK(p.is_cuda() && p.dtype() == torch::kFloat32 && p.is_contiguous());
  int64_t n = p.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int block = 256;
  int grid = faa_grid(n / 4 + 1, block);
  hipLaunchKernelGGL(zero1_kernel, dim3(1), dim3(64), 0, stream, normsq.data_ptr<float>());
  hipLaunchKernelGGL(wd_norm_kernel, dim3(grid), dim3(block), 0, stream,
                     g.data_ptr<float>(), p.data_ptr<float>(), normsq.data_ptr<float>(),
                     n, n_decay, (float)wd);
  hipLaunchKernelGGL(sgd_step_kernel, dim3(grid), dim3(block), 0, stream,
                     p.data_ptr<float>(), g.data_ptr<float>(), buf.data_ptr<float>(),
                     normsq.data_ptr<float>(), lr_t.data_ptr<float>(), n, (float)clip,
                     (float)momentum, (int)nesterov);
}

void ema_lerp_(torch::Tensor shadow, torch::Tensor x, double mu) {
  TORCH_CHECK(shadow.is_cuda() && shadow.is_contiguous() && x.is_contiguous());
  int64_t n = shadow.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  const int block = 256;
  int grid = faa_grid(n / 4 + 1, block);
  hipLaunchKernelGGL(ema_lerp_kernel, dim3(grid), dim3(block), 0, stream,
                     shadow.data_ptr<float>(), x.data_ptr<float>(), n, (float)mu);
}
