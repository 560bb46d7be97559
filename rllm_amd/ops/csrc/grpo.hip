// Fused GRPO/PPO ratio-clip policy loss with low-variance (k3) KL penalty
// (K8-K10 in SURVEY.md §2.E). One elementwise pass computes both the
// per-token loss AND its gradient w.r.t. the new logprob, so backward is a
// single multiply — no recompute, no graph through exp/clamp.
//
// Per token (mirrors reference semantics, verl_backend.py:64-92 +
// config clip_ratio/clip_ratio_high, kl_loss_type=low_var_kl):
//   ratio = exp(lp - old_lp)
//   pg    = -min(ratio * A, clamp(ratio, 1-eps_lo, 1+eps_hi) * A)
//   k3    = exp(ref - lp) - (ref - lp) - 1          (if kl_beta > 0)
//   L     = tis_w * (pg + kl_beta * k3)
//   dL/dlp = tis_w * (pg_grad + kl_beta * (1 - exp(ref - lp)))
// where pg_grad = -ratio*A when the unclipped branch is the min, else 0.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

__global__ void grpo_loss_kernel(
    const float* __restrict__ logprob,      // [N] new policy logprob (fp32)
    const float* __restrict__ old_logprob,  // [N]
    const float* __restrict__ ref_logprob,  // [N] or null
    const float* __restrict__ advantages,   // [N]
    const float* __restrict__ tis_w,        // [N] or null (importance weights)
    float* __restrict__ loss_tok,           // [N] out
    float* __restrict__ dlp_tok,            // [N] out: dL/dlp
    float* __restrict__ clipped_tok,        // [N] out: 1.0 where clipped branch won
    int64_t N, float eps_lo, float eps_hi, float kl_beta) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float lp = logprob[i];
    const float old_lp = old_logprob[i];
    const float adv = advantages[i];
    const float w = tis_w ? tis_w[i] : 1.0f;

    const float ratio = __expf(lp - old_lp);
    const float s1 = ratio * adv;
    const float clipped_ratio = fminf(fmaxf(ratio, 1.0f - eps_lo), 1.0f + eps_hi);
    const float s2 = clipped_ratio * adv;

    float pg, dpg, was_clipped;
    if (s1 <= s2) {  // unclipped branch active: grad flows through ratio
      pg = -s1;
      dpg = -s1;  // d(-ratio*A)/dlp = -ratio*A
      was_clipped = 0.f;
    } else {        // clipped branch: constant w.r.t. lp
      pg = -s2;
      dpg = 0.f;
      was_clipped = 1.f;
    }

    float kl = 0.f, dkl = 0.f;
    if (kl_beta > 0.f && ref_logprob) {
      const float d = ref_logprob[i] - lp;
      const float ed = __expf(d);
      kl = ed - d - 1.0f;
      dkl = 1.0f - ed;
    }

    loss_tok[i] = w * (pg + kl_beta * kl);
    dlp_tok[i] = w * (dpg + kl_beta * dkl);
    clipped_tok[i] = was_clipped;
  }
}

static inline hipStream_t gl_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

std::vector<torch::Tensor> grpo_loss_fwd(
    torch::Tensor logprob, torch::Tensor old_logprob,
    c10::optional<torch::Tensor> ref_logprob, torch::Tensor advantages,
    c10::optional<torch::Tensor> tis_w,
    double eps_lo, double eps_hi, double kl_beta) {
  TORCH_CHECK(logprob.is_cuda() && logprob.dtype() == torch::kFloat32 && logprob.is_contiguous());
  const int64_t N = logprob.numel();
  auto loss_tok = torch::empty_like(logprob);
  auto dlp_tok = torch::empty_like(logprob);
  auto clipped = torch::empty_like(logprob);
  const float* ref_ptr = ref_logprob.has_value() ? ref_logprob->data_ptr<float>() : nullptr;
  const float* w_ptr = tis_w.has_value() ? tis_w->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(grpo_loss_kernel, dim3(grid_for(N, 256)), dim3(256), 0, gl_stream(),
                     logprob.data_ptr<float>(), old_logprob.data_ptr<float>(), ref_ptr,
                     advantages.data_ptr<float>(), w_ptr,
                     loss_tok.data_ptr<float>(), dlp_tok.data_ptr<float>(), clipped.data_ptr<float>(),
                     N, (float)eps_lo, (float)eps_hi, (float)kl_beta);
  HIP_CHECK_KERNEL();
  return {loss_tok, dlp_tok, clipped};
}
