// Fused AdamW over flat parameter buffers (K11 in SURVEY.md §2.E).
//
// The trainer keeps ONE flat fp32 master copy + exp_avg + exp_avg_sq and
// ONE flat bf16 gradient buffer (each model param / param.grad is a view
// into the flat bf16 buffers), so the whole optimizer step is:
//   1) grad_sq_sum_kernel  -> global grad norm (one number, stays on device)
//   2) adamw_kernel        -> reads gnorm from device mem (no host sync),
//                             applies clip scale, Adam moments with bias
//                             correction, decoupled weight decay, writes
//                             master fp32 AND the live bf16 params.
// Gradient pre-scale (1/world_size after RCCL SUM all-reduce, or loss
// scaling) folds into `grad_scale`.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

__global__ void grad_sq_sum_kernel(
    const uint16_t* __restrict__ grad, // [N] bf16
    float* __restrict__ out,           // [1], pre-zeroed
    int64_t N, float grad_scale) {
  __shared__ float scratch[16];
  float acc = 0.f;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; i + 7 < N;
       i += (int64_t)gridDim.x * blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(grad + i);
    short4v b = *reinterpret_cast<const short4v*>(grad + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float ga = bf16_to_f32((uint16_t)a[j]) * grad_scale;
      float gb = bf16_to_f32((uint16_t)b[j]) * grad_scale;
      acc += ga * ga + gb * gb;
    }
  }
  // tail (N not multiple of 8)
  const int64_t tail_start = (N / 8) * 8;
  for (int64_t i = tail_start + blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    float g = bf16_to_f32(grad[i]) * grad_scale;
    acc += g * g;
  }
  acc = block_reduce_sum(acc, scratch);
  // DETERMINISTIC reduction: per-block partial to its own slot; a fixed-
  // order second pass sums them. (atomicAdd here made the grad norm — and
  // through grad clipping the whole optimizer step — vary by ~1 ulp run to
  // run, breaking bit-reproducibility.)
  if (threadIdx.x == 0) out[blockIdx.x] = acc;
}

__global__ void sum_partials_kernel(const float* __restrict__ partials,
                                    float* __restrict__ out, int n) {
  // one block, fixed order: lane-strided partial sums then an ORDERED
  // tree over the 4 wave results via LDS
  __shared__ float scratch[16];
  float acc = 0.f;
  for (int i = threadIdx.x; i < n; i += blockDim.x) acc += partials[i];
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) out[0] = acc;
}

__global__ void adamw_kernel(
    const uint16_t* __restrict__ grad, // [N] bf16
    float* __restrict__ master,        // [N] fp32
    float* __restrict__ m,             // [N] fp32
    float* __restrict__ v,             // [N] fp32
    uint16_t* __restrict__ param,      // [N] bf16 (live model weights)
    const float* __restrict__ gnorm_sq,// [1] or null (no clipping)
    int64_t N, float lr, float beta1, float beta2, float eps,
    float weight_decay, float bias_c1, float bias_c2,
    float grad_clip, float grad_scale) {
  float clip_scale = 1.0f;
  if (gnorm_sq != nullptr && grad_clip > 0.f) {
    const float norm = sqrtf(*gnorm_sq);
    if (norm > grad_clip) clip_scale = grad_clip / (norm + 1e-6f);
  }
  const float gs = grad_scale * clip_scale;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    float g = bf16_to_f32(grad[i]) * gs;
    float mi = beta1 * m[i] + (1.f - beta1) * g;
    float vi = beta2 * v[i] + (1.f - beta2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float mhat = mi / bias_c1;
    const float vhat = vi / bias_c2;
    float p = master[i];
    p -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * p);
    master[i] = p;
    param[i] = f32_to_bf16(p);
  }
}

static inline hipStream_t aw_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor grad_sq_sum(torch::Tensor grad, double grad_scale) {
  TORCH_CHECK(grad.is_cuda() && grad.dtype() == torch::kBFloat16 && grad.is_contiguous());
  const int64_t N = grad.numel();
  const unsigned grid = grid_for(N / 8, 256);
  auto partials = torch::empty({(int64_t)grid}, grad.options().dtype(torch::kFloat32));
  auto out = torch::zeros({1}, grad.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(grad_sq_sum_kernel, dim3(grid), dim3(256), 0, aw_stream(),
                     (const uint16_t*)grad.data_ptr(), partials.data_ptr<float>(), N,
                     (float)grad_scale);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(sum_partials_kernel, dim3(1), dim3(256), 0, aw_stream(),
                     partials.data_ptr<float>(), out.data_ptr<float>(), (int)grid);
  HIP_CHECK_KERNEL();
  return out;
}

void adamw_step(torch::Tensor grad, torch::Tensor master, torch::Tensor m, torch::Tensor v,
                torch::Tensor param, c10::optional<torch::Tensor> gnorm_sq,
                double lr, double beta1, double beta2, double eps, double weight_decay,
                int64_t step, double grad_clip, double grad_scale) {
  const int64_t N = master.numel();
  TORCH_CHECK(grad.numel() == N && m.numel() == N && v.numel() == N && param.numel() == N);
  const float bias_c1 = 1.f - powf((float)beta1, (float)step);
  const float bias_c2 = 1.f - powf((float)beta2, (float)step);
  const float* gptr = gnorm_sq.has_value() ? gnorm_sq->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(adamw_kernel, dim3(grid_for(N, 256)), dim3(256), 0, aw_stream(),
                     (const uint16_t*)grad.data_ptr(), master.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), (uint16_t*)param.data_ptr(),
                     gptr, N, (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bias_c1, bias_c2, (float)grad_clip, (float)grad_scale);
  HIP_CHECK_KERNEL();
}
