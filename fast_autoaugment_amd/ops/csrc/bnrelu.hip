// Fused NHWC BatchNorm2d + ReLU, training fwd/bwd + eval fwd.
//
// The pre-activation BN->ReLU pair is the second-hottest op family of the
// CIFAR nets after conv (reference wideresnet.py:37-38, pyramidnet.py:83-96).
// MIOpen runs BN and ReLU as separate kernels over NCHW; here both fuse into
// one NHWC pass with fp32 stats: reduce (sum/sumsq per channel, wave-aligned
// 64-channel strips) -> finalize (mean/invstd + running-stat update)
// -> apply (normalize+scale+relu). Backward reduces dgamma/dbeta with the
// relu mask folded in, then applies dx in one pass.
//
// Layout: x is [N,C,H,W] channels_last => storage [rows=N*H*W][C]; lanes map
// to consecutive channels (coalesced), rows stride across the grid.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

template <typename T>
__global__ void bn_reduce_kernel(const T* __restrict__ x, float* __restrict__ scratch,
                                 int64_t rows, int C) {
  // block: 64 x 4 (channel x row-group); grid: (row_chunks, ceil(C/64))
  int c = blockIdx.y * 64 + (threadIdx.x & 63);
  int rg = threadIdx.x >> 6;      // 0..3
  float s = 0.0f, ss = 0.0f;
  if (c < C) {
    for (int64_t r = (int64_t)blockIdx.x * 4 + rg; r < rows; r += (int64_t)gridDim.x * 4) {
      float v = faa_to_float(x[r * C + c]);
      s += v; ss += v * v;
    }
  }
  __shared__ float lds_s[4][64];
  __shared__ float lds_ss[4][64];
  lds_s[rg][threadIdx.x & 63] = s;
  lds_ss[rg][threadIdx.x & 63] = ss;
  __syncthreads();
  if (rg == 0 && c < C) {
    float ts = 0, tss = 0;
    #pragma unroll
    for (int k = 0; k < 4; ++k) { ts += lds_s[k][threadIdx.x & 63]; tss += lds_ss[k][threadIdx.x & 63]; }
    atomicAdd(&scratch[c], ts);
    atomicAdd(&scratch[C + c], tss);
  }
}

__global__ void bn_finalize_kernel(const float* __restrict__ scratch,
                                   float* __restrict__ mean, float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, int64_t count, float eps, float momentum) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float m = scratch[c] / count;
  float var = scratch[C + c] / count - m * m;
  var = fmaxf(var, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    float unbiased = count > 1 ? var * (float)count / (float)(count - 1) : var;
    running_mean[c] = (1.0f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.0f - momentum) * running_var[c] + momentum * unbiased;
  }
}

template <typename T>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ out,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                int64_t total, int C) {
  int64_t i0 = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < total; i += stride) {
    int c = (int)(i % C);
    float v = (faa_to_float(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    faa_from_float(fmaxf(v, 0.0f), &out[i]);
  }
}

template <typename T>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x, const T* __restrict__ out,
                                     const T* __restrict__ dy,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ scratch,   // [2C]: sum_dy, sum_dy_xhat
                                     int64_t rows, int C) {
  int c = blockIdx.y * 64 + (threadIdx.x & 63);
  int rg = threadIdx.x >> 6;
  float sdy = 0.0f, sdyx = 0.0f;
  if (c < C) {
    float m = mean[c], is = invstd[c];
    for (int64_t r = (int64_t)blockIdx.x * 4 + rg; r < rows; r += (int64_t)gridDim.x * 4) {
      int64_t i = r * C + c;
      float mask = faa_to_float(out[i]) > 0.0f ? 1.0f : 0.0f;
      float g = faa_to_float(dy[i]) * mask;
      sdy += g;
      sdyx += g * (faa_to_float(x[i]) - m) * is;
    }
  }
  __shared__ float lds_a[4][64];
  __shared__ float lds_b[4][64];
  lds_a[rg][threadIdx.x & 63] = sdy;
  lds_b[rg][threadIdx.x & 63] = sdyx;
  __syncthreads();
  if (rg == 0 && c < C) {
    float ta = 0, tb = 0;
    #pragma unroll
    for (int k = 0; k < 4; ++k) { ta += lds_a[k][threadIdx.x & 63]; tb += lds_b[k][threadIdx.x & 63]; }
    atomicAdd(&scratch[c], ta);
    atomicAdd(&scratch[C + c], tb);
  }
}

template <typename T>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x, const T* __restrict__ out,
                                    const T* __restrict__ dy, T* __restrict__ dx,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ scratch,
                                    int64_t total, int C, int64_t count, int training) {
  int64_t i0 = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float inv_count = 1.0f / (float)count;
  for (int64_t i = i0; i < total; i += stride) {
    int c = (int)(i % C);
    float mask = faa_to_float(out[i]) > 0.0f ? 1.0f : 0.0f;
    float g = faa_to_float(dy[i]) * mask;
    float is = invstd[c];
    float res;
    if (training) {
      float xhat = (faa_to_float(x[i]) - mean[c]) * is;
      float t = g - scratch[c] * inv_count - xhat * scratch[C + c] * inv_count;
      res = gamma[c] * is * t;
    } else {
      res = gamma[c] * is * g;
    }
    faa_from_float(res, &dx[i]);
  }
}

}  // namespace

#define DISPATCH_FB(TYPE, NAME, ...)                                           \
  [&] {                                                                        \
    if (TYPE == torch::kFloat32) { using scalar_t = float; return __VA_ARGS__(); } \
    else if (TYPE == torch::kBFloat16) { using scalar_t = __hip_bfloat16; return __VA_ARGS__(); } \
    else { TORCH_CHECK(false, NAME ": unsupported dtype"); }                  \
  }()

std::vector<torch::Tensor> bn_relu_fwd(torch::Tensor x, torch::Tensor gamma,
                                       torch::Tensor beta, torch::Tensor running_mean,
                                       torch::Tensor running_var, bool training,
                                       double momentum, double eps) {
  TORCH_CHECK(x.dim() == 4, "bn_relu: 4-D input expected");
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = xc.size(1);
  int64_t rows = xc.numel() / C;
  auto out = torch::empty_like(xc);
  auto f32 = xc.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, f32);
  auto invstd = torch::empty({C}, f32);
  auto stream = at::hip::getCurrentHIPStream().stream();

  auto g = gamma.to(torch::kFloat32).contiguous();
  auto bta = beta.to(torch::kFloat32).contiguous();

  if (training) {
    auto scratch = torch::zeros({2 * C}, f32);
    dim3 block(256);
    int row_chunks = (int)std::min<int64_t>((rows + 3) / 4, 1024);
    dim3 grid(row_chunks, (C + 63) / 64);
    DISPATCH_FB(xc.scalar_type(), "bn_reduce", [&] {
      hipLaunchKernelGGL((bn_reduce_kernel<scalar_t>), grid, block, 0, stream,
                         (const scalar_t*)xc.data_ptr(), scratch.data_ptr<float>(), rows, C);
    });
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0, stream,
                       scratch.data_ptr<float>(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(),
                       running_mean.defined() ? running_mean.data_ptr<float>() : nullptr,
                       running_var.defined() ? running_var.data_ptr<float>() : nullptr,
                       C, rows, (float)eps, (float)momentum);
  } else {
    // eval: mean/invstd from running stats (computed on device, no sync)
    mean.copy_(running_mean);
    invstd.copy_(torch::rsqrt(running_var + eps));
  }
  int64_t total = xc.numel();
  int block1 = 256;
  int grid1 = faa_grid(total, block1);
  DISPATCH_FB(xc.scalar_type(), "bn_apply", [&] {
    hipLaunchKernelGGL((bn_apply_kernel<scalar_t>), dim3(grid1), dim3(block1), 0, stream,
                       (const scalar_t*)xc.data_ptr(), (scalar_t*)out.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       g.data_ptr<float>(), bta.data_ptr<float>(), total, C);
  });
  return {out, mean, invstd};
}

std::vector<torch::Tensor> bn_relu_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor out, torch::Tensor mean,
                                       torch::Tensor invstd, torch::Tensor gamma,
                                       bool training) {
  auto xc = x.contiguous(torch::MemoryFormat::ChannelsLast);
  auto oc = out.contiguous(torch::MemoryFormat::ChannelsLast);
  auto dyc = dy.contiguous(torch::MemoryFormat::ChannelsLast);
  int C = xc.size(1);
  int64_t rows = xc.numel() / C;
  int64_t total = xc.numel();
  auto f32 = xc.options().dtype(torch::kFloat32);
  auto scratch = torch::zeros({2 * C}, f32);
  auto dx = torch::empty_like(xc);
  auto g = gamma.to(torch::kFloat32).contiguous();
  auto stream = at::hip::getCurrentHIPStream().stream();

  dim3 block(256);
  int row_chunks = (int)std::min<int64_t>((rows + 3) / 4, 1024);
  dim3 grid(row_chunks, (C + 63) / 64);
  DISPATCH_FB(xc.scalar_type(), "bn_bwd_reduce", [&] {
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<scalar_t>), grid, block, 0, stream,
                       (const scalar_t*)xc.data_ptr(), (const scalar_t*)oc.data_ptr(),
                       (const scalar_t*)dyc.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), scratch.data_ptr<float>(), rows, C);
  });
  int block1 = 256;
  int grid1 = faa_grid(total, block1);
  DISPATCH_FB(xc.scalar_type(), "bn_bwd_apply", [&] {
    hipLaunchKernelGGL((bn_bwd_apply_kernel<scalar_t>), dim3(grid1), dim3(block1), 0, stream,
                       (const scalar_t*)xc.data_ptr(), (const scalar_t*)oc.data_ptr(),
                       (const scalar_t*)dyc.data_ptr(), (scalar_t*)dx.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       g.data_ptr<float>(), scratch.data_ptr<float>(),
                       total, C, rows, training ? 1 : 0);
  });
  // dgamma = scratch[C:2C], dbeta = scratch[0:C]
  auto dbeta = scratch.narrow(0, 0, C).clone();
  auto dgamma = scratch.narrow(0, C, C).clone();
  return {dx, dgamma, dbeta};
}
