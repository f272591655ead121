// Elementwise / broadcast kernels: the stochastic-regularizer hot paths.
//
//  scale_bcast : out[b,...] = x[b,...] * s[b]        (ShakeDrop fwd/bwd,
//                drop_connect, ShakeShake bwd — reference shakedrop.py:9-34)
//  scale_lerp  : out = a[b]*x1 + (1-a[b])*x2          (ShakeShake fwd,
//                reference shakeshake/shakeshake.py:9-18)
//  swish fwd/bwd: x*sigmoid(x) and its recompute-backward
//                (reference efficientnet_pytorch/utils.py:38-54)
//  mixup_fwd   : out[i] = lam*x[i] + (1-lam)*x[perm[i]] (aug_mixup.py:13-23)
//  pad_add     : out = x + channel-padded shortcut (pyramidnet.py:109-113)
//
// All are HBM-bound: bf16 is moved as ushort4 (8 B/lane) vectors per G13.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

// ---------------------------------------------------------------- scale_bcast
template <typename T, int VEC>
__global__ void scale_bcast_kernel(const T* __restrict__ x, const float* __restrict__ s,
                                   T* __restrict__ out, int64_t total, int64_t per_sample) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  for (int64_t i = i0; i < total; i += stride) {
    int64_t b = i / per_sample;     // VEC divides per_sample (asserted host-side)
    float sc = s[b];
    #pragma unroll
    for (int v = 0; v < VEC; ++v) {
      if (i + v < total) {
        float val = faa_to_float(x[i + v]);
        faa_from_float(val * sc, &out[i + v]);
      }
    }
  }
}

// ---------------------------------------------------------------- scale_lerp
template <typename T, int VEC>
__global__ void scale_lerp_kernel(const T* __restrict__ x1, const T* __restrict__ x2,
                                  const float* __restrict__ a, T* __restrict__ out,
                                  int64_t total, int64_t per_sample) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  for (int64_t i = i0; i < total; i += stride) {
    int64_t b = i / per_sample;
    float al = a[b];
    #pragma unroll
    for (int v = 0; v < VEC; ++v) {
      if (i + v < total) {
        float v1 = faa_to_float(x1[i + v]);
        float v2 = faa_to_float(x2[i + v]);
        faa_from_float(al * v1 + (1.0f - al) * v2, &out[i + v]);
      }
    }
  }
}

// -------------------------------------------------------------------- swish
template <typename T, int VEC>
__global__ void swish_fwd_kernel(const T* __restrict__ x, T* __restrict__ out, int64_t total) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  for (int64_t i = i0; i < total; i += stride) {
    #pragma unroll
    for (int v = 0; v < VEC; ++v) {
      if (i + v < total) {
        float xv = faa_to_float(x[i + v]);
        float s = 1.0f / (1.0f + __expf(-xv));
        faa_from_float(xv * s, &out[i + v]);
      }
    }
  }
}

template <typename T, int VEC>
__global__ void swish_bwd_kernel(const T* __restrict__ g, const T* __restrict__ x,
                                 T* __restrict__ out, int64_t total) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  for (int64_t i = i0; i < total; i += stride) {
    #pragma unroll
    for (int v = 0; v < VEC; ++v) {
      if (i + v < total) {
        float xv = faa_to_float(x[i + v]);
        float gv = faa_to_float(g[i + v]);
        float s = 1.0f / (1.0f + __expf(-xv));
        faa_from_float(gv * (s * (1.0f + xv * (1.0f - s))), &out[i + v]);
      }
    }
  }
}

// -------------------------------------------------------------------- mixup
template <typename T, int VEC>
__global__ void mixup_kernel(const T* __restrict__ x, const int64_t* __restrict__ perm,
                             T* __restrict__ out, float lam, int64_t total,
                             int64_t per_sample) {
  int64_t i0 = (int64_t)(blockIdx.x * blockDim.x + threadIdx.x) * VEC;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  for (int64_t i = i0; i < total; i += stride) {
    int64_t b = i / per_sample;
    int64_t off = i - b * per_sample;
    const T* xp = x + perm[b] * per_sample + off;
    #pragma unroll
    for (int v = 0; v < VEC; ++v) {
      if (i + v < total) {
        float a0 = faa_to_float(x[i + v]);
        float a1 = faa_to_float(xp[v]);
        faa_from_float(lam * a0 + (1.0f - lam) * a1, &out[i + v]);
      }
    }
  }
}

// ------------------------------------------------------------------ pad_add
// NHWC (channels_last) layout: out[b,h,w,c] += (c < c_short) ? short[b,h,w,c] : 0
template <typename T>
__global__ void pad_add_kernel(const T* __restrict__ x, const T* __restrict__ sc,
                               T* __restrict__ out, int64_t rows, int c_out, int c_short) {
  int64_t r0 = blockIdx.x;
  for (int64_t r = r0; r < rows; r += gridDim.x) {
    const T* xr = x + r * c_out;
    const T* sr = sc + r * c_short;
    T* orow = out + r * c_out;
    for (int c = threadIdx.x; c < c_out; c += blockDim.x) {
      float v = faa_to_float(xr[c]);
      if (c < c_short) v += faa_to_float(sr[c]);
      faa_from_float(v, &orow[c]);
    }
  }
}

template <typename T>
void launch_all(const torch::Tensor&) {}

}  // namespace

#define DISPATCH_FLOAT_BF16(TYPE, NAME, ...)                                   \
  [&] {                                                                        \
    if (TYPE == torch::kFloat32) { using scalar_t = float; return __VA_ARGS__(); } \
    else if (TYPE == torch::kBFloat16) { using scalar_t = __hip_bfloat16; return __VA_ARGS__(); } \
    else { TORCH_CHECK(false, NAME ": unsupported dtype"); }                  \
  }()

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor scale_bcast(torch::Tensor x, torch::Tensor s) {
  TORCH_CHECK(x.is_cuda() && s.is_cuda(), "scale_bcast: cuda tensors required");
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t total = xc.numel();
  int64_t per_sample = total / xc.size(0);
  auto sf = s.to(torch::kFloat32).contiguous();
  const int block = 256, vec = 8;
  int grid = faa_grid(total / vec + 1, block);
  DISPATCH_FLOAT_BF16(xc.scalar_type(), "scale_bcast", [&] {
    hipLaunchKernelGGL((scale_bcast_kernel<scalar_t, vec>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)xc.data_ptr(), sf.data_ptr<float>(),
                       (scalar_t*)out.data_ptr(), total, per_sample);
  });
  return out;
}

torch::Tensor scale_lerp(torch::Tensor x1, torch::Tensor x2, torch::Tensor a) {
  TORCH_CHECK(x1.is_cuda(), "scale_lerp: cuda tensors required");
  auto f = x1.suggest_memory_format();
  auto c1 = x1.contiguous(f);
  auto c2 = x2.contiguous(f);
  auto out = torch::empty_like(c1);
  int64_t total = c1.numel();
  int64_t per_sample = total / c1.size(0);
  auto af = a.to(torch::kFloat32).contiguous();
  const int block = 256, vec = 8;
  int grid = faa_grid(total / vec + 1, block);
  DISPATCH_FLOAT_BF16(c1.scalar_type(), "scale_lerp", [&] {
    hipLaunchKernelGGL((scale_lerp_kernel<scalar_t, vec>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)c1.data_ptr(), (const scalar_t*)c2.data_ptr(),
                       af.data_ptr<float>(), (scalar_t*)out.data_ptr(), total, per_sample);
  });
  return out;
}

torch::Tensor swish_fwd(torch::Tensor x) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t total = xc.numel();
  const int block = 256, vec = 8;
  int grid = faa_grid(total / vec + 1, block);
  DISPATCH_FLOAT_BF16(xc.scalar_type(), "swish_fwd", [&] {
    hipLaunchKernelGGL((swish_fwd_kernel<scalar_t, vec>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)xc.data_ptr(), (scalar_t*)out.data_ptr(), total);
  });
  return out;
}

torch::Tensor swish_bwd(torch::Tensor g, torch::Tensor x) {
  auto f = x.suggest_memory_format();
  auto gc = g.contiguous(f);
  auto xc = x.contiguous(f);
  auto out = torch::empty_like(xc);
  int64_t total = xc.numel();
  const int block = 256, vec = 8;
  int grid = faa_grid(total / vec + 1, block);
  DISPATCH_FLOAT_BF16(xc.scalar_type(), "swish_bwd", [&] {
    hipLaunchKernelGGL((swish_bwd_kernel<scalar_t, vec>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)gc.data_ptr(), (const scalar_t*)xc.data_ptr(),
                       (scalar_t*)out.data_ptr(), total);
  });
  return out;
}

torch::Tensor mixup_fwd(torch::Tensor x, torch::Tensor perm, float lam) {
  auto xc = x.contiguous(x.suggest_memory_format());
  auto out = torch::empty_like(xc);
  int64_t total = xc.numel();
  int64_t per_sample = total / xc.size(0);
  auto pc = perm.to(torch::kInt64).contiguous();
  const int block = 256, vec = 8;
  int grid = faa_grid(total / vec + 1, block);
  DISPATCH_FLOAT_BF16(xc.scalar_type(), "mixup", [&] {
    hipLaunchKernelGGL((mixup_kernel<scalar_t, vec>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)xc.data_ptr(), pc.data_ptr<int64_t>(),
                       (scalar_t*)out.data_ptr(), lam, total, per_sample);
  });
  return out;
}

torch::Tensor pad_add(torch::Tensor x, torch::Tensor shortcut) {
  // channels_last NHWC expected; rows = B*H*W
  TORCH_CHECK(x.dim() == 4, "pad_add: 4-D expected");
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto sc = shortcut.contiguous(torch::MemoryFormat::ChannelsLast);
  auto out = torch::empty_like(xc);
  int64_t rows = xc.size(0) * xc.size(2) * xc.size(3);
  int c_out = xc.size(1), c_short = sc.size(1);
  const int block = 128;
  int grid = faa_grid(rows, 1, 4096);
  DISPATCH_FLOAT_BF16(xc.scalar_type(), "pad_add", [&] {
    hipLaunchKernelGGL((pad_add_kernel<scalar_t>), dim3(grid), dim3(block), 0,
                       cur_stream(),
                       (const scalar_t*)xc.data_ptr(), (const scalar_t*)sc.data_ptr(),
                       (scalar_t*)out.data_ptr(), rows, c_out, c_short);
  });
  return out;
}
