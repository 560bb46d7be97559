// Fused token-logprob + entropy over vocab chunks (K7 in SURVEY.md §2.E).
//
// The [T, vocab] logits matrix (vocab = 152k) is never materialized:
// the caller GEMMs one vocab chunk at a time (hipBLASLt bf16) and these
// kernels maintain per-row online logsumexp state (m, s) plus the online
// entropy accumulator e = sum(exp(l - m) * l), and gather the target
// token's logit. Final: logprob = z_t - (m + log s);
// entropy = (m + log s) - e / s.
//
// Backward (policy gradient only; entropy is a metric):
// dlogits[t,c] = dlp[t] * (1{c==target} - softmax[t,c]), formed chunk-wise.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// One block per row; C = chunk width (multiple of 8).
__global__ void lse_chunk_update_kernel(
    const uint16_t* __restrict__ logits, // [T, C] bf16 chunk
    float* __restrict__ m_state,         // [T] running max
    float* __restrict__ s_state,         // [T] running sumexp (rel. to m)
    float* __restrict__ e_state,         // [T] running sum exp(l-m)*l (entropy; may be null)
    float* __restrict__ target_logit,    // [T]
    const int32_t* __restrict__ targets, // [T] global vocab ids (-1 = ignore)
    int64_t chunk_start, int C, float inv_temp) {
  __shared__ float scratch[16];
  const int64_t row = blockIdx.x;
  const uint16_t* lr = logits + row * (int64_t)C;

  // pass 1: chunk max
  float local_max = -INFINITY;
  for (int i = threadIdx.x * 8; i < C; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      local_max = fmaxf(local_max, bf16_to_f32((uint16_t)a[j]));
      local_max = fmaxf(local_max, bf16_to_f32((uint16_t)b[j]));
    }
  }
  float chunk_max = block_reduce_max(local_max, scratch) * inv_temp;

  const float m_old = m_state[row];
  const float m_new = fmaxf(m_old, chunk_max);

  // pass 2: sumexp (and entropy numerator) relative to m_new
  float local_s = 0.f;
  float local_e = 0.f;
  const bool want_e = (e_state != nullptr);
  for (int i = threadIdx.x * 8; i < C; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float za = bf16_to_f32((uint16_t)a[j]) * inv_temp;
      float zb = bf16_to_f32((uint16_t)b[j]) * inv_temp;
      float ea = __expf(za - m_new);
      float eb = __expf(zb - m_new);
      local_s += ea + eb;
      if (want_e) local_e += ea * za + eb * zb;
    }
  }
  __syncthreads();
  float chunk_s = block_reduce_sum(local_s, scratch);
  float chunk_e = 0.f;
  if (want_e) {
    __syncthreads();
    chunk_e = block_reduce_sum(local_e, scratch);
  }

  // target gather (single thread; at most one hit per chunk)
  const int32_t tgt = targets[row];
  const int64_t rel = (int64_t)tgt - chunk_start;
  if (threadIdx.x == 0) {
    const float rescale = __expf(m_old - m_new);
    s_state[row] = s_state[row] * rescale + chunk_s;
    if (want_e) e_state[row] = e_state[row] * rescale + chunk_e;
    m_state[row] = m_new;
    if (tgt >= 0 && rel >= 0 && rel < C) {
      target_logit[row] = bf16_to_f32(lr[rel]) * inv_temp;
    }
  }
}

// dlogits[t, c] = dlp[t] * (1{c==target} - exp(z - lse))
__global__ void ce_bwd_chunk_kernel(
    const uint16_t* __restrict__ logits, // [T, C] bf16 chunk
    const float* __restrict__ lse,       // [T] = m + log s
    const float* __restrict__ dlp,       // [T] upstream grad of logprob
    const int32_t* __restrict__ targets, // [T]
    uint16_t* __restrict__ dlogits,      // [T, C] bf16 out
    int64_t chunk_start, int C, float inv_temp) {
  const int64_t row = blockIdx.x;
  const uint16_t* lr = logits + row * (int64_t)C;
  uint16_t* dr = dlogits + row * (int64_t)C;
  const float l = lse[row];
  const float g = dlp[row];
  const int64_t rel = (int64_t)targets[row] - chunk_start;

  for (int i = threadIdx.x * 8; i < C; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
    short4v oa, ob;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pa = __expf(bf16_to_f32((uint16_t)a[j]) * inv_temp - l);
      float pb = __expf(bf16_to_f32((uint16_t)b[j]) * inv_temp - l);
      float da = -g * pa;
      float db = -g * pb;
      if ((int64_t)(i + j) == rel) da += g;
      if ((int64_t)(i + 4 + j) == rel) db += g;
      oa[j] = (short)f32_to_bf16(da * inv_temp);
      ob[j] = (short)f32_to_bf16(db * inv_temp);
    }
    *reinterpret_cast<short4v*>(dr + i) = oa;
    *reinterpret_cast<short4v*>(dr + i + 4) = ob;
  }
}

static inline hipStream_t lp_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void lse_chunk_update(torch::Tensor logits, torch::Tensor m_state, torch::Tensor s_state,
                      c10::optional<torch::Tensor> e_state, torch::Tensor target_logit,
                      torch::Tensor targets, int64_t chunk_start, double inv_temp) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 && logits.is_contiguous());
  TORCH_CHECK(targets.dtype() == torch::kInt32);
  const int64_t T = logits.size(0);
  const int C = (int)logits.size(1);
  TORCH_CHECK(C % 8 == 0, "chunk width must be a multiple of 8");
  float* e_ptr = e_state.has_value() ? e_state->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(lse_chunk_update_kernel, dim3((unsigned)T), dim3(256), 0, lp_stream(),
                     (const uint16_t*)logits.data_ptr(), m_state.data_ptr<float>(),
                     s_state.data_ptr<float>(), e_ptr, target_logit.data_ptr<float>(),
                     targets.data_ptr<int32_t>(), chunk_start, C, (float)inv_temp);
  HIP_CHECK_KERNEL();
}

torch::Tensor ce_bwd_chunk(torch::Tensor logits, torch::Tensor lse, torch::Tensor dlp,
                           torch::Tensor targets, int64_t chunk_start, double inv_temp) {
  const int64_t T = logits.size(0);
  const int C = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  hipLaunchKernelGGL(ce_bwd_chunk_kernel, dim3((unsigned)T), dim3(256), 0, lp_stream(),
                     (const uint16_t*)logits.data_ptr(), lse.data_ptr<float>(),
                     dlp.data_ptr<float>(), targets.data_ptr<int32_t>(),
                     (uint16_t*)dlogits.data_ptr(), chunk_start, C, (float)inv_temp);
  HIP_CHECK_KERNEL();
  return dlogits;
}
