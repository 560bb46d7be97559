// Fused sampling + per-token logprob capture (K4 in SURVEY.md §2.E).
//
// Single pass over each row of [B, vocab] bf16 logits:
//  * Gumbel-max trick: argmax(z + g) with g ~ Gumbel(0,1) is an exact
//    categorical sample from softmax(z) — one reduction, no CDF/sort.
//  * Online logsumexp in the same pass -> logprob of the sampled token,
//    1:1 aligned with the emitted token id (reference contract
//    types.py:167-168; gateway injects logprobs=true, middleware.py:26).
//  * temperature folds into z = logit * inv_temp; greedy (temperature 0)
//    uses inv_temp=1 for z and suppresses the gumbel noise.
// top_k / top_p: the engine pre-masks logits (set to -inf) in a separate
// kernel-free pass when requested (non-default); this kernel then samples
// the masked distribution exactly.
//
// RNG: counter-based hash of (seed, row, col) — deterministic per
// (seed, step), no state to carry.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

__global__ void sample_logprob_kernel(
    const uint16_t* __restrict__ logits, // [B, V] bf16
    int32_t* __restrict__ token_out,     // [B]
    float* __restrict__ logprob_out,     // [B]
    const uint32_t* __restrict__ step_ptr, // device step counter (hipGraph-safe) or null
    int V, float inv_temp, uint32_t seed, uint32_t step, int greedy) {
  __shared__ float red_m[4], red_s[4], red_g[4];
  __shared__ int red_i[4];

  const int64_t row = blockIdx.x;
  const uint16_t* lr = logits + row * (int64_t)V;
  const uint32_t step_eff = step_ptr ? *step_ptr : step;
  const uint32_t row_seed = hash_u32(seed, step_eff, (uint32_t)row);

  // thread-local online state
  float m = -INFINITY, s = 0.f;       // logsumexp state over z
  float best_g = -INFINITY;           // best perturbed value
  int best_i = -1;

  for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = i + j;
      const float raw = (j < 4) ? bf16_to_f32((uint16_t)a[j]) : bf16_to_f32((uint16_t)b[j - 4]);
      const float z = raw * inv_temp;
      if (z != -INFINITY) {
        // online LSE
        if (z > m) { s = s * __expf(m - z) + 1.f; m = z; }
        else { s += __expf(z - m); }
        // perturbed argmax
        float zg = z;
        if (!greedy) {
          const float u = uniform_from_u32(hash_u32(row_seed, (uint32_t)col, 0x5bd1e995u));
          zg = z - __logf(-__logf(u));
        }
        if (zg > best_g || (zg == best_g && col < best_i)) { best_g = zg; best_i = col; }
      }
    }
  }

  // ---- wave reduce: merge (m,s) and (best_g,best_i) ----
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float m_o = __shfl_xor(m, off, 64);
    const float s_o = __shfl_xor(s, off, 64);
    const float g_o = __shfl_xor(best_g, off, 64);
    const int i_o = __shfl_xor(best_i, off, 64);
    if (m_o > m) { s = s * __expf(m - m_o) + s_o; m = m_o; }
    else if (m_o != -INFINITY) { s += s_o * __expf(m_o - m); }
    if (g_o > best_g || (g_o == best_g && i_o >= 0 && (best_i < 0 || i_o < best_i))) { best_g = g_o; best_i = i_o; }
  }
  if (lane == 0) { red_m[wid] = m; red_s[wid] = s; red_g[wid] = best_g; red_i[wid] = best_i; }
  __syncthreads();

  if (threadIdx.x == 0) {
    const int nwaves = (blockDim.x + 63) / 64;
    float M = red_m[0], S = red_s[0], G = red_g[0];
    int I = red_i[0];
    for (int w = 1; w < nwaves; ++w) {
      if (red_m[w] > M) { S = S * __expf(M - red_m[w]) + red_s[w]; M = red_m[w]; }
      else if (red_m[w] != -INFINITY) { S += red_s[w] * __expf(red_m[w] - M); }
      if (red_g[w] > G || (red_g[w] == G && red_i[w] >= 0 && (I < 0 || red_i[w] < I))) { G = red_g[w]; I = red_i[w]; }
    }
    token_out[row] = I;
    const float lse = M + __logf(S);
    const float z_tok = bf16_to_f32(lr[I]) * inv_temp;
    logprob_out[row] = z_tok - lse;
  }
}

// Gather logprobs of GIVEN tokens from a [T, V] logits matrix (used for
// prompt logprobs / teacher-forced scoring in the rollout engine).
__global__ void gather_logprob_kernel(
    const uint16_t* __restrict__ logits, // [T, V]
    const int32_t* __restrict__ tokens,  // [T]
    float* __restrict__ logprob_out,     // [T]
    int V, float inv_temp) {
  __shared__ float scratch[16];
  const int64_t row = blockIdx.x;
  const uint16_t* lr = logits + row * (int64_t)V;

  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float raw = (j < 4) ? bf16_to_f32((uint16_t)a[j]) : bf16_to_f32((uint16_t)b[j - 4]);
      const float z = raw * inv_temp;
      if (z > m) { s = s * __expf(m - z) + 1.f; m = z; }
      else if (z != -INFINITY) { s += __expf(z - m); }
    }
  }
  // block merge of (m, s) via two-phase: max then sumexp relative to it
  float m_blk = block_reduce_max(m, scratch);
  __syncthreads();
  float s_adj = (m == -INFINITY) ? 0.f : s * __expf(m - m_blk);
  float s_blk = block_reduce_sum(s_adj, scratch);
  if (threadIdx.x == 0) {
    const float lse = m_blk + __logf(s_blk);
    logprob_out[row] = bf16_to_f32(lr[tokens[row]]) * inv_temp - lse;
  }
}

// Split-vocab variant: grid (B, S) partials + a merge kernel — at decode
// batch sizes a single block per row leaves 7/8 of the chip idle on the
// 152k-vocab scan.
__global__ void sample_partial_kernel(
    const uint16_t* __restrict__ logits, // [B, V]
    float* __restrict__ ws,              // [B, S, 4]: m, s, best_g, best_i(bits)
    const uint32_t* __restrict__ step_ptr,
    int V, float inv_temp, uint32_t seed, uint32_t step, int greedy) {
  __shared__ float red_m[4], red_s[4], red_g[4];
  __shared__ int red_i[4];
  const int64_t row = blockIdx.x;
  const int split = blockIdx.y;
  const int S = gridDim.y;
  const int chunk = (((V + S - 1) / S) + 7) & ~7;  // 8-aligned so no tail is skipped
  const int lo = split * chunk;
  const int hi = min(V, lo + chunk);
  const uint16_t* lr = logits + row * (int64_t)V;
  const uint32_t step_eff = step_ptr ? *step_ptr : step;
  const uint32_t row_seed = hash_u32(seed, step_eff, (uint32_t)row);

  float m = -INFINITY, s = 0.f, best_g = -INFINITY;
  int best_i = -1;
  for (int i = lo + threadIdx.x * 8; i + 7 < hi; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(lr + i);
    short4v b = *reinterpret_cast<const short4v*>(lr + i + 4);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = i + j;
      const float raw = (j < 4) ? bf16_to_f32((uint16_t)a[j]) : bf16_to_f32((uint16_t)b[j - 4]);
      const float z = raw * inv_temp;
      if (z != -INFINITY) {
        if (z > m) { s = s * __expf(m - z) + 1.f; m = z; }
        else { s += __expf(z - m); }
        float zg = z;
        if (!greedy) {
          const float u = uniform_from_u32(hash_u32(row_seed, (uint32_t)col, 0x5bd1e995u));
          zg = z - __logf(-__logf(u));
        }
        if (zg > best_g || (zg == best_g && col < best_i)) { best_g = zg; best_i = col; }
      }
    }
  }
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float m_o = __shfl_xor(m, off, 64);
    const float s_o = __shfl_xor(s, off, 64);
    const float g_o = __shfl_xor(best_g, off, 64);
    const int i_o = __shfl_xor(best_i, off, 64);
    if (m_o > m) { s = s * __expf(m - m_o) + s_o; m = m_o; }
    else if (m_o != -INFINITY) { s += s_o * __expf(m_o - m); }
    if (g_o > best_g || (g_o == best_g && i_o >= 0 && (best_i < 0 || i_o < best_i))) { best_g = g_o; best_i = i_o; }
  }
  if (lane == 0) { red_m[wid] = m; red_s[wid] = s; red_g[wid] = best_g; red_i[wid] = best_i; }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = (blockDim.x + 63) / 64;
    float M = red_m[0], Ss = red_s[0], G = red_g[0];
    int I = red_i[0];
    for (int w = 1; w < nwaves; ++w) {
      if (red_m[w] > M) { Ss = Ss * __expf(M - red_m[w]) + red_s[w]; M = red_m[w]; }
      else if (red_m[w] != -INFINITY) { Ss += red_s[w] * __expf(red_m[w] - M); }
      if (red_g[w] > G || (red_g[w] == G && red_i[w] >= 0 && (I < 0 || red_i[w] < I))) { G = red_g[w]; I = red_i[w]; }
    }
    float* out = ws + (row * S + split) * 4;
    out[0] = M; out[1] = Ss; out[2] = G;
    *reinterpret_cast<int*>(out + 3) = I;
  }
}

__global__ void sample_merge_kernel(
    const uint16_t* __restrict__ logits, // [B, V]
    const float* __restrict__ ws,        // [B, S, 4]
    int32_t* __restrict__ token_out,
    float* __restrict__ logprob_out,
    int64_t B, int V, int S, float inv_temp) {
  // one thread per row (rows are few at decode batch sizes)
  const int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  float M = -INFINITY, Ss = 0.f, G = -INFINITY;
  int I = -1;
  const float* base = ws + row * (int64_t)S * 4;
  for (int p = 0; p < S; ++p) {
    const float m = base[p * 4], s = base[p * 4 + 1], g = base[p * 4 + 2];
    const int i = *reinterpret_cast<const int*>(base + p * 4 + 3);
    if (m > M) { Ss = Ss * __expf(M - m) + s; M = m; }
    else if (m != -INFINITY) { Ss += s * __expf(m - M); }
    if (g > G || (g == G && i >= 0 && (I < 0 || i < I))) { G = g; I = i; }
  }
  token_out[row] = I;
  const float lse = M + __logf(Ss);
  logprob_out[row] = bf16_to_f32(logits[row * (int64_t)V + I]) * inv_temp - lse;
}

static inline hipStream_t sp_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

std::vector<torch::Tensor> sample_logprob(torch::Tensor logits, double temperature,
                                          int64_t seed, int64_t step,
                                          c10::optional<torch::Tensor> step_tensor) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16 && logits.is_contiguous());
  const int64_t B = logits.size(0);
  const int V = (int)logits.size(1);
  TORCH_CHECK(V % 8 == 0);
  auto tokens = torch::empty({B}, logits.options().dtype(torch::kInt32));
  auto logprobs = torch::empty({B}, logits.options().dtype(torch::kFloat32));
  const bool greedy = temperature <= 0.0;
  const float inv_temp = greedy ? 1.0f : (float)(1.0 / temperature);
  const uint32_t* step_ptr = nullptr;
  if (step_tensor.has_value()) {
    TORCH_CHECK(step_tensor->dtype() == torch::kInt32 && step_tensor->is_cuda());
    step_ptr = (const uint32_t*)step_tensor->data_ptr<int32_t>();
  }
  // split the vocab scan so the chip stays busy at small B
  int splits = (int)std::min<int64_t>(16, std::max<int64_t>(1, (2 * 256) / std::max<int64_t>(1, B)));
  if (splits > 1) {
    auto ws = torch::empty({B, splits, 4}, logits.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(sample_partial_kernel, dim3((unsigned)B, splits), dim3(256), 0, sp_stream(),
                       (const uint16_t*)logits.data_ptr(), ws.data_ptr<float>(), step_ptr,
                       V, inv_temp, (uint32_t)seed, (uint32_t)step, greedy ? 1 : 0);
    HIP_CHECK_KERNEL();
    hipLaunchKernelGGL(sample_merge_kernel, dim3((unsigned)((B + 255) / 256)), dim3(256), 0, sp_stream(),
                       (const uint16_t*)logits.data_ptr(), ws.data_ptr<float>(),
                       tokens.data_ptr<int32_t>(), logprobs.data_ptr<float>(),
                       B, V, splits, inv_temp);
    HIP_CHECK_KERNEL();
    return {tokens, logprobs};
  }
  hipLaunchKernelGGL(sample_logprob_kernel, dim3((unsigned)B), dim3(256), 0, sp_stream(),
                     (const uint16_t*)logits.data_ptr(), tokens.data_ptr<int32_t>(),
                     logprobs.data_ptr<float>(), step_ptr, V, inv_temp,
                     (uint32_t)seed, (uint32_t)step, greedy ? 1 : 0);
  HIP_CHECK_KERNEL();
  return {tokens, logprobs};
}

torch::Tensor gather_logprob(torch::Tensor logits, torch::Tensor tokens, double temperature) {
  const int64_t T = logits.size(0);
  const int V = (int)logits.size(1);
  auto out = torch::empty({T}, logits.options().dtype(torch::kFloat32));
  const float inv_temp = temperature <= 0.0 ? 1.0f : (float)(1.0 / temperature);
  hipLaunchKernelGGL(gather_logprob_kernel, dim3((unsigned)T), dim3(256), 0, sp_stream(),
                     (const uint16_t*)logits.data_ptr(), tokens.data_ptr<int32_t>(),
                     out.data_ptr<float>(), V, inv_temp);
  HIP_CHECK_KERNEL();
  return out;
}
