// Fused log-softmax + label-smoothing cross-entropy (reference metrics.py:26-46).
//
// fwd: one block per row; saves the fp32 softmax for backward and
//      atomically accumulates the mean loss into a single scalar.
// bwd: grad = (softmax - smoothed_target) * (grad_loss / B)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "faa_common.h"

namespace {

template <typename T>
__global__ void ls_ce_fwd_kernel(const T* __restrict__ logits,
                                 const int64_t* __restrict__ target,
                                 float* __restrict__ softmax_out,
                                 float* __restrict__ loss_out,
                                 int B, int C, float eps) {
  __shared__ float lds[8];      // blockDim.x/64 partials
  int row = blockIdx.x;
  if (row >= B) return;
  const T* lrow = logits + (int64_t)row * C;
  float* srow = softmax_out + (int64_t)row * C;

  // 1) row max
  float m = -INFINITY;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    m = fmaxf(m, faa_to_float(lrow[c]));
  #pragma unroll
  for (int off = FAA_WAVE / 2; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_down(m, off, FAA_WAVE));
  if ((threadIdx.x & (FAA_WAVE - 1)) == 0) lds[threadIdx.x / FAA_WAVE] = m;
  __syncthreads();
  if (threadIdx.x < FAA_WAVE) {
    float v = (threadIdx.x < blockDim.x / FAA_WAVE) ? lds[threadIdx.x] : -INFINITY;
    #pragma unroll
    for (int off = FAA_WAVE / 2; off > 0; off >>= 1)
      v = fmaxf(v, __shfl_down(v, off, FAA_WAVE));
    if (threadIdx.x == 0) lds[0] = v;
  }
  __syncthreads();
  m = lds[0];
  __syncthreads();

  // 2) sum exp + sum of logp pieces
  float se = 0.0f;
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    se += __expf(faa_to_float(lrow[c]) - m);
  se = faa_block_reduce_sum(se, lds);
  if (threadIdx.x == 0) lds[0] = se;
  __syncthreads();
  se = lds[0];
  float lse = __logf(se) + m;
  __syncthreads();

  // 3) softmax + loss pieces:  loss = -(1-eps)*logp[t] - (eps/C)*sum logp
  int64_t t = target[row];
  float sum_logp = 0.0f;
  float logp_t = 0.0f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float lp = faa_to_float(lrow[c]) - lse;
    srow[c] = __expf(lp);
    if (eps > 0.0f) sum_logp += lp;
    if (c == (int)t) logp_t = lp;
  }
  if (eps > 0.0f) {
    sum_logp = faa_block_reduce_sum(sum_logp, lds);
  }
  // logp_t lives in exactly one thread; reduce it
  __syncthreads();
  float lt = faa_block_reduce_sum(logp_t, lds);
  if (threadIdx.x == 0) {
    float smooth = eps > 0.0f ? eps / C : 0.0f;
    float onval = 1.0f - eps + smooth;
    // -(onval*logp_t) - smooth*(sum_logp - logp_t)
    float loss = -onval * lt - (eps > 0.0f ? smooth * (sum_logp - lt) : 0.0f);
    atomicAdd(loss_out, loss / B);
  }
}

template <typename T>
__global__ void ls_ce_bwd_kernel(const float* __restrict__ softmax,
                                 const int64_t* __restrict__ target,
                                 const float* __restrict__ grad_loss,
                                 T* __restrict__ grad_out,
                                 int B, int C, float eps) {
  float gscale = grad_loss[0] / B;
  float smooth = eps > 0.0f ? eps / C : 0.0f;
  float onval = 1.0f - eps + smooth;
  int64_t total = (int64_t)B * C;
  int64_t i0 = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < total; i += stride) {
    int64_t row = i / C;
    int c = i - row * C;
    float tval = (c == (int)target[row]) ? onval : smooth;
    faa_from_float((softmax[i] - tval) * gscale, &grad_out[i]);
  }
}

}  // namespace

extern torch::Tensor scale_bcast(torch::Tensor, torch::Tensor);  // decl reuse

#define DISPATCH_FLOAT_BF16_L(TYPE, NAME, ...)                                 \
  [&] {                                                                        \
    if (TYPE == torch::kFloat32) { using scalar_t = float; return __VA_ARGS__(); } \
    else if (TYPE == torch::kBFloat16) { using scalar_t = __hip_bfloat16; return __VA_ARGS__(); } \
    else { TORCH_CHECK(false, NAME ": unsupported dtype"); }                  \
  }()

std::vector<torch::Tensor> label_smooth_ce_fwd(torch::Tensor logits, torch::Tensor target,
                                               double eps) {
  TORCH_CHECK(logits.dim() == 2, "label_smooth_ce: [B,C] expected");
  auto lc = logits.contiguous();
  auto tc = target.to(torch::kInt64).contiguous();
  int B = lc.size(0), C = lc.size(1);
  auto softmax = torch::empty({B, C}, lc.options().dtype(torch::kFloat32));
  auto loss = torch::zeros({}, lc.options().dtype(torch::kFloat32));
  int block = 256;
  auto stream = at::hip::getCurrentHIPStream().stream();
  DISPATCH_FLOAT_BF16_L(lc.scalar_type(), "ls_ce_fwd", [&] {
    hipLaunchKernelGGL((ls_ce_fwd_kernel<scalar_t>), dim3(B), dim3(block), 0, stream,
                       (const scalar_t*)lc.data_ptr(), tc.data_ptr<int64_t>(),
                       softmax.data_ptr<float>(), loss.data_ptr<float>(),
                       B, C, (float)eps);
  });
  return {loss, softmax};
}

torch::Tensor label_smooth_ce_bwd(torch::Tensor softmax, torch::Tensor target,
                                  torch::Tensor grad_loss, double eps) {
  auto sc = softmax.contiguous();
  auto tc = target.to(torch::kInt64).contiguous();
  int B = sc.size(0), C = sc.size(1);
  auto grad = torch::empty({B, C}, sc.options().dtype(torch::kFloat32));
  auto gl = grad_loss.to(torch::kFloat32).contiguous();
  int block = 256;
  int grid = faa_grid((int64_t)B * C, block);
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL((ls_ce_bwd_kernel<float>), dim3(grid), dim3(block), 0, stream,
                     sc.data_ptr<float>(), tc.data_ptr<int64_t>(),
                     gl.data_ptr<float>(), grad.data_ptr<float>(), B, C, (float)eps);
  return grad;
}
