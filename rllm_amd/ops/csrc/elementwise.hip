// Fused elementwise / normalization kernels for the Qwen-family decoder:
// RMSNorm (fwd+bwd), rotary embedding (fwd; bwd = inverse rotation),
// SwiGLU (fwd+bwd). All bf16 I/O with fp32 math, vectorized 8-wide.
//
// Replaces what the reference delegates to vLLM/flash-attn fused layer ops
// (SURVEY.md §2.E K5).

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

// ---------------------------------------------------------------------------
// RMSNorm
// ---------------------------------------------------------------------------

// One block per row. H must be a multiple of 8 (true for all Qwen configs).
__global__ void rmsnorm_fwd_kernel(
    const uint16_t* __restrict__ x,  // [T, H] bf16
    const uint16_t* __restrict__ w,  // [H] bf16
    uint16_t* __restrict__ y,        // [T, H] bf16
    float* __restrict__ inv_rms_out, // [T] f32 (saved for backward; may be null)
    int H, float eps) {
  __shared__ float scratch[16];
  const int64_t row = blockIdx.x;
  const uint16_t* xr = x + row * (int64_t)H;
  uint16_t* yr = y + row * (int64_t)H;

  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(xr + i);
    short4v b = *reinterpret_cast<const short4v*>(xr + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float va = bf16_to_f32((uint16_t)a[j]);
      float vb = bf16_to_f32((uint16_t)b[j]);
      ss += va * va + vb * vb;
    }
  }
  ss = block_reduce_sum(ss, scratch);
  const float inv_rms = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0 && inv_rms_out != nullptr) inv_rms_out[row] = inv_rms;

  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(xr + i);
    short4v b = *reinterpret_cast<const short4v*>(xr + i + 4);
    short4v wa = *reinterpret_cast<const short4v*>(w + i);
    short4v wb = *reinterpret_cast<const short4v*>(w + i + 4);
    short4v oa, ob;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      oa[j] = (short)f32_to_bf16(bf16_to_f32((uint16_t)a[j]) * inv_rms * bf16_to_f32((uint16_t)wa[j]));
      ob[j] = (short)f32_to_bf16(bf16_to_f32((uint16_t)b[j]) * inv_rms * bf16_to_f32((uint16_t)wb[j]));
    }
    *reinterpret_cast<short4v*>(yr + i) = oa;
    *reinterpret_cast<short4v*>(yr + i + 4) = ob;
  }
}

// dx = inv_rms * (dy*w - x * inv_rms^2 * mean_h(dy*w*x))
__global__ void rmsnorm_bwd_dx_kernel(
    const uint16_t* __restrict__ dy, // [T, H]
    const uint16_t* __restrict__ x,  // [T, H]
    const uint16_t* __restrict__ w,  // [H]
    const float* __restrict__ inv_rms, // [T]
    uint16_t* __restrict__ dx,       // [T, H]
    int H) {
  __shared__ float scratch[16];
  const int64_t row = blockIdx.x;
  const uint16_t* dyr = dy + row * (int64_t)H;
  const uint16_t* xr = x + row * (int64_t)H;
  uint16_t* dxr = dx + row * (int64_t)H;
  const float ir = inv_rms[row];

  float dot = 0.f;
  for (int i = threadIdx.x * 4; i < H; i += blockDim.x * 4) {
    short4v d = *reinterpret_cast<const short4v*>(dyr + i);
    short4v a = *reinterpret_cast<const short4v*>(xr + i);
    short4v ww = *reinterpret_cast<const short4v*>(w + i);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      dot += bf16_to_f32((uint16_t)d[j]) * bf16_to_f32((uint16_t)ww[j]) * bf16_to_f32((uint16_t)a[j]);
  }
  dot = block_reduce_sum(dot, scratch);
  const float c = dot * ir * ir / (float)H;

  for (int i = threadIdx.x * 4; i < H; i += blockDim.x * 4) {
    short4v d = *reinterpret_cast<const short4v*>(dyr + i);
    short4v a = *reinterpret_cast<const short4v*>(xr + i);
    short4v ww = *reinterpret_cast<const short4v*>(w + i);
    short4v o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = bf16_to_f32((uint16_t)d[j]) * bf16_to_f32((uint16_t)ww[j]);
      float xv = bf16_to_f32((uint16_t)a[j]);
      o[j] = (short)f32_to_bf16(ir * (g - xv * c));
    }
    *reinterpret_cast<short4v*>(dxr + i) = o;
  }
}

// dw[h] = sum_t dy[t,h] * x[t,h] * inv_rms[t]  (fp32 accumulation)
__global__ void rmsnorm_bwd_dw_kernel(
    const uint16_t* __restrict__ dy,
    const uint16_t* __restrict__ x,
    const float* __restrict__ inv_rms,
    float* __restrict__ dw,          // [H] fp32, pre-zeroed
    int64_t T, int H) {
  const int h = blockIdx.x * blockDim.x + threadIdx.x;
  if (h >= H) return;
  float acc = 0.f;
  for (int64_t t = blockIdx.y; t < T; t += gridDim.y) {
    acc += bf16_to_f32(dy[t * H + h]) * bf16_to_f32(x[t * H + h]) * inv_rms[t];
  }
  atomicAdd(&dw[h], acc);
}

// ---------------------------------------------------------------------------
// Rotary embedding (NeoX rotate-half, as used by the Qwen2 family)
// ---------------------------------------------------------------------------

// In-place on q [T, Hq, D] and k [T, Hk, D]; cos/sin tables [max_pos, D/2] f32
// precomputed on host (on-device trig turns this memory-bound op VALU-bound).
// backward: apply inverse rotation (negated sin).
__global__ void rope_kernel(
    uint16_t* __restrict__ q,
    uint16_t* __restrict__ k,
    const float* __restrict__ cos_tab,
    const float* __restrict__ sin_tab,
    const int32_t* __restrict__ positions, // [T]
    int64_t T, int Hq, int Hk, int D, int neg_sin) {
  const int half = D / 2;
  const int64_t total = T * (int64_t)(Hq + Hk) * half;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int i = (int)(idx % half);
    int64_t th = idx / half;
    int h = (int)(th % (Hq + Hk));
    int64_t t = th / (Hq + Hk);

    uint16_t* base;
    if (h < Hq) base = q + (t * Hq + h) * D;
    else        base = k + (t * Hk + (h - Hq)) * D;

    const int pos = positions[t];
    float c = cos_tab[(int64_t)pos * half + i];
    float s = sin_tab[(int64_t)pos * half + i];
    if (neg_sin) s = -s;

    float a = bf16_to_f32(base[i]);
    float b = bf16_to_f32(base[i + half]);
    base[i]        = f32_to_bf16(a * c - b * s);
    base[i + half] = f32_to_bf16(b * c + a * s);
  }
}

// ---------------------------------------------------------------------------
// SwiGLU
// ---------------------------------------------------------------------------

// gateup [T, 2I] (gate | up concatenated along the feature dim) -> out [T, I]
// 16-byte (8 x bf16) loads/stores when I % 8 == 0 — the 8-byte version left
// ~4x of the stream floor on the table at decode sizes.
__global__ void swiglu_fwd_kernel8(
    const uint16_t* __restrict__ gateup,
    uint16_t* __restrict__ out,
    int64_t T, int I) {
  const int64_t total = T * (int64_t)I / 8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = idx / (I / 8);
    int i = (int)(idx % (I / 8)) * 8;
    short8v gv = *reinterpret_cast<const short8v*>(gateup + t * 2 * I + i);
    short8v uv = *reinterpret_cast<const short8v*>(gateup + t * 2 * I + I + i);
    short8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32((uint16_t)gv[j]);
      float uf = bf16_to_f32((uint16_t)uv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = (short)f32_to_bf16(gf * sig * uf);
    }
    *reinterpret_cast<short8v*>(out + t * I + i) = o;
  }
}

__global__ void swiglu_fwd_kernel(
    const uint16_t* __restrict__ gateup,
    uint16_t* __restrict__ out,
    int64_t T, int I) {
  const int64_t total = T * (int64_t)I / 4;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = idx / (I / 4);
    int i = (int)(idx % (I / 4)) * 4;
    const uint16_t* g = gateup + t * 2 * I + i;
    const uint16_t* u = gateup + t * 2 * I + I + i;
    short4v gv = *reinterpret_cast<const short4v*>(g);
    short4v uv = *reinterpret_cast<const short4v*>(u);
    short4v o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf16_to_f32((uint16_t)gv[j]);
      float uf = bf16_to_f32((uint16_t)uv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = (short)f32_to_bf16(gf * sig * uf);
    }
    *reinterpret_cast<short4v*>(out + t * I + i) = o;
  }
}

__global__ void swiglu_bwd_kernel(
    const uint16_t* __restrict__ dout,   // [T, I]
    const uint16_t* __restrict__ gateup, // [T, 2I]
    uint16_t* __restrict__ dgateup,      // [T, 2I]
    int64_t T, int I) {
  const int64_t total = T * (int64_t)I / 4;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = idx / (I / 4);
    int i = (int)(idx % (I / 4)) * 4;
    short4v gv = *reinterpret_cast<const short4v*>(gateup + t * 2 * I + i);
    short4v uv = *reinterpret_cast<const short4v*>(gateup + t * 2 * I + I + i);
    short4v dv = *reinterpret_cast<const short4v*>(dout + t * I + i);
    short4v dg, du;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf16_to_f32((uint16_t)gv[j]);
      float uf = bf16_to_f32((uint16_t)uv[j]);
      float df = bf16_to_f32((uint16_t)dv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      float dsilu = sig * (1.f + gf * (1.f - sig));
      dg[j] = (short)f32_to_bf16(df * uf * dsilu);
      du[j] = (short)f32_to_bf16(df * silu);
    }
    *reinterpret_cast<short4v*>(dgateup + t * 2 * I + i) = dg;
    *reinterpret_cast<short4v*>(dgateup + t * 2 * I + I + i) = du;
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps, bool save_inv_rms) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const int64_t T = x.numel() / x.size(-1);
  const int H = (int)x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  auto y = torch::empty_like(x);
  torch::Tensor inv_rms;
  float* inv_ptr = nullptr;
  if (save_inv_rms) {
    inv_rms = torch::empty({T}, x.options().dtype(torch::kFloat32));
    inv_ptr = inv_rms.data_ptr<float>();
  }
  dim3 grid((unsigned)T);
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, grid, dim3(256), 0, cur_stream(),
                     (const uint16_t*)x.data_ptr(), (const uint16_t*)w.data_ptr(),
                     (uint16_t*)y.data_ptr(), inv_ptr, H, (float)eps);
  HIP_CHECK_KERNEL();
  if (save_inv_rms) return {y, inv_rms};
  return {y};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w, torch::Tensor inv_rms) {
  const int64_t T = x.numel() / x.size(-1);
  const int H = (int)x.size(-1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(rmsnorm_bwd_dx_kernel, dim3((unsigned)T), dim3(256), 0, cur_stream(),
                     (const uint16_t*)dy.data_ptr(), (const uint16_t*)x.data_ptr(),
                     (const uint16_t*)w.data_ptr(), inv_rms.data_ptr<float>(),
                     (uint16_t*)dx.data_ptr(), H);
  HIP_CHECK_KERNEL();
  dim3 grid2((H + 255) / 256, 128);
  hipLaunchKernelGGL(rmsnorm_bwd_dw_kernel, grid2, dim3(256), 0, cur_stream(),
                     (const uint16_t*)dy.data_ptr(), (const uint16_t*)x.data_ptr(),
                     inv_rms.data_ptr<float>(), dw.data_ptr<float>(), T, H);
  HIP_CHECK_KERNEL();
  return {dx, dw};
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_tab, torch::Tensor sin_tab,
                  torch::Tensor positions, bool backward) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(k.is_cuda() && k.dtype() == torch::kBFloat16 && k.is_contiguous());
  TORCH_CHECK(positions.dtype() == torch::kInt32);
  const int64_t T = q.size(0);
  const int Hq = (int)q.size(1), Hk = (int)k.size(1), D = (int)q.size(2);
  TORCH_CHECK(k.size(0) == T && k.size(2) == D);
  const int64_t total = T * (int64_t)(Hq + Hk) * (D / 2);
  hipLaunchKernelGGL(rope_kernel, dim3(grid_for(total, 256)), dim3(256), 0, cur_stream(),
                     (uint16_t*)q.data_ptr(), (uint16_t*)k.data_ptr(),
                     cos_tab.data_ptr<float>(), sin_tab.data_ptr<float>(),
                     positions.data_ptr<int32_t>(), T, Hq, Hk, D, backward ? 1 : 0);
  HIP_CHECK_KERNEL();
}

torch::Tensor swiglu_fwd(torch::Tensor gateup) {
  TORCH_CHECK(gateup.is_cuda() && gateup.dtype() == torch::kBFloat16 && gateup.is_contiguous());
  const int64_t T = gateup.numel() / gateup.size(-1);
  const int I2 = (int)gateup.size(-1);
  TORCH_CHECK(I2 % 8 == 0);
  const int I = I2 / 2;
  auto sizes = gateup.sizes().vec();
  sizes.back() = I;
  auto out = torch::empty(sizes, gateup.options());
  if (I % 8 == 0) {
    hipLaunchKernelGGL(swiglu_fwd_kernel8, dim3(grid_for(T * I / 8, 256)), dim3(256), 0, cur_stream(),
                       (const uint16_t*)gateup.data_ptr(), (uint16_t*)out.data_ptr(), T, I);
  } else {
    hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid_for(T * I / 4, 256)), dim3(256), 0, cur_stream(),
                       (const uint16_t*)gateup.data_ptr(), (uint16_t*)out.data_ptr(), T, I);
  }
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor swiglu_bwd(torch::Tensor dout, torch::Tensor gateup) {
  const int64_t T = gateup.numel() / gateup.size(-1);
  const int I = (int)gateup.size(-1) / 2;
  auto dgateup = torch::empty_like(gateup);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid_for(T * I / 4, 256)), dim3(256), 0, cur_stream(),
                     (const uint16_t*)dout.data_ptr(), (const uint16_t*)gateup.data_ptr(),
                     (uint16_t*)dgateup.data_ptr(), T, I);
  HIP_CHECK_KERNEL();
  return dgateup;
}

// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm (decode hot path): h += delta; y = rmsnorm(h).
// Removes one full read+write of the hidden stream per layer-norm site.
// ---------------------------------------------------------------------------
__global__ void add_rmsnorm_fwd_kernel(
    uint16_t* __restrict__ h,        // [T, H] residual stream, updated in place
    const uint16_t* __restrict__ delta, // [T, H] (may be null: plain rmsnorm)
    const uint16_t* __restrict__ w,  // [H]
    uint16_t* __restrict__ y,        // [T, H]
    int H, float eps) {
  __shared__ float scratch[16];
  const int64_t row = blockIdx.x;
  uint16_t* hr = h + row * (int64_t)H;
  const uint16_t* dr = delta ? delta + row * (int64_t)H : nullptr;
  uint16_t* yr = y + row * (int64_t)H;

  float ss = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(hr + i);
    short4v b = *reinterpret_cast<const short4v*>(hr + i + 4);
    if (dr) {
      short4v da = *reinterpret_cast<const short4v*>(dr + i);
      short4v db = *reinterpret_cast<const short4v*>(dr + i + 4);
      short4v oa, ob;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float va = bf16_to_f32((uint16_t)a[j]) + bf16_to_f32((uint16_t)da[j]);
        float vb = bf16_to_f32((uint16_t)b[j]) + bf16_to_f32((uint16_t)db[j]);
        oa[j] = (short)f32_to_bf16(va);
        ob[j] = (short)f32_to_bf16(vb);
        ss += va * va + vb * vb;
      }
      *reinterpret_cast<short4v*>(hr + i) = oa;
      *reinterpret_cast<short4v*>(hr + i + 4) = ob;
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float va = bf16_to_f32((uint16_t)a[j]);
        float vb = bf16_to_f32((uint16_t)b[j]);
        ss += va * va + vb * vb;
      }
    }
  }
  ss = block_reduce_sum(ss, scratch);
  const float inv_rms = rsqrtf(ss / (float)H + eps);
  for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
    short4v a = *reinterpret_cast<const short4v*>(hr + i);
    short4v b = *reinterpret_cast<const short4v*>(hr + i + 4);
    short4v wa = *reinterpret_cast<const short4v*>(w + i);
    short4v wb = *reinterpret_cast<const short4v*>(w + i + 4);
    short4v oa, ob;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      oa[j] = (short)f32_to_bf16(bf16_to_f32((uint16_t)a[j]) * inv_rms * bf16_to_f32((uint16_t)wa[j]));
      ob[j] = (short)f32_to_bf16(bf16_to_f32((uint16_t)b[j]) * inv_rms * bf16_to_f32((uint16_t)wb[j]));
    }
    *reinterpret_cast<short4v*>(yr + i) = oa;
    *reinterpret_cast<short4v*>(yr + i + 4) = ob;
  }
}

// Wave-per-row variant: 4 independent waves per block, wave-local shfl
// reduction only (no block barrier) — the block-per-row version is
// latency-bound at decode sizes (T=256 rows of H=1536: two barrier phases
// per tiny row dominate).
__global__ void add_rmsnorm_fwd_wave_kernel(
    uint16_t* __restrict__ h,
    const uint16_t* __restrict__ delta,
    const uint16_t* __restrict__ w,
    uint16_t* __restrict__ y,
    int64_t T, int H, float eps) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * 4 + wid;
  if (row >= T) return;
  uint16_t* hr = h + row * (int64_t)H;
  const uint16_t* dr = delta ? delta + row * (int64_t)H : nullptr;
  uint16_t* yr = y + row * (int64_t)H;

  float ss = 0.f;
  for (int i = lane * 8; i < H; i += WAVE_SIZE * 8) {
    short8v a = *reinterpret_cast<const short8v*>(hr + i);
    if (dr) {
      short8v da = *reinterpret_cast<const short8v*>(dr + i);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_to_f32((uint16_t)a[j]) + bf16_to_f32((uint16_t)da[j]);
        o[j] = (short)f32_to_bf16(v);
        ss += v * v;
      }
      *reinterpret_cast<short8v*>(hr + i) = o;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_to_f32((uint16_t)a[j]);
        ss += v * v;
      }
    }
  }
#pragma unroll
  for (int off = 1; off < WAVE_SIZE; off <<= 1) ss += __shfl_xor(ss, off, WAVE_SIZE);
  const float inv_rms = rsqrtf(ss / (float)H + eps);
  for (int i = lane * 8; i < H; i += WAVE_SIZE * 8) {
    short8v a = *reinterpret_cast<const short8v*>(hr + i);
    short8v wv = *reinterpret_cast<const short8v*>(w + i);
    short8v o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      o[j] = (short)f32_to_bf16(bf16_to_f32((uint16_t)a[j]) * inv_rms * bf16_to_f32((uint16_t)wv[j]));
    }
    *reinterpret_cast<short8v*>(yr + i) = o;
  }
}

torch::Tensor add_rmsnorm_(torch::Tensor h, c10::optional<torch::Tensor> delta,
                           torch::Tensor w, double eps) {
  TORCH_CHECK(h.is_cuda() && h.dtype() == torch::kBFloat16 && h.is_contiguous());
  const int64_t T = h.numel() / h.size(-1);
  const int H = (int)h.size(-1);
  auto y = torch::empty_like(h);
  const uint16_t* dptr = nullptr;
  if (delta.has_value()) {
    TORCH_CHECK(delta->is_contiguous() && delta->sizes() == h.sizes());
    dptr = (const uint16_t*)delta->data_ptr();
  }
  if (H % 8 == 0) {
    hipLaunchKernelGGL(add_rmsnorm_fwd_wave_kernel, dim3((unsigned)((T + 3) / 4)), dim3(256),
                       0, cur_stream(), (uint16_t*)h.data_ptr(), dptr,
                       (const uint16_t*)w.data_ptr(), (uint16_t*)y.data_ptr(), T, H, (float)eps);
    HIP_CHECK_KERNEL();
    return y;
  }
  hipLaunchKernelGGL(add_rmsnorm_fwd_kernel, dim3((unsigned)T), dim3(256), 0, cur_stream(),
                     (uint16_t*)h.data_ptr(), dptr, (const uint16_t*)w.data_ptr(),
                     (uint16_t*)y.data_ptr(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return y;
}

// ---------------------------------------------------------------------------
// fp8 rollout: fused DELAYED-SCALING activation quantization.
// Quantizes bf16 x with the scale computed from the PREVIOUS call's amax
// (the standard fp8 delayed-scaling recipe) while accumulating this call's
// amax — one pass, no fp32 materialization, no reduction dependency before
// the GEMM. A 1-block epilogue kernel folds amax into the scale AFTER the
// GEMM is enqueued (stream-ordered), ready for the next call.
// ---------------------------------------------------------------------------

#include <hip/hip_fp8.h>

__global__ void fp8_quant_delayed_kernel(
    const uint16_t* __restrict__ x,   // [n] bf16
    uint8_t* __restrict__ y,          // [n] e4m3
    const float* __restrict__ scale,  // [1] current scale (prev amax / 448)
    float* __restrict__ amax_next,    // [1] running amax accumulator
    int64_t n) {
  __shared__ float red[16];
  const float inv_s = 1.f / scale[0];
  float local_max = 0.f;
  // 4 elements/thread/iter; native v_cvt_pk_fp8_f32 packs pairs
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4; i0 < n;
       i0 += (int64_t)gridDim.x * blockDim.x * 4) {
    short4v xv = *reinterpret_cast<const short4v*>(x + i0);
    float v0 = bf16_to_f32((uint16_t)xv[0]);
    float v1 = bf16_to_f32((uint16_t)xv[1]);
    float v2 = bf16_to_f32((uint16_t)xv[2]);
    float v3 = bf16_to_f32((uint16_t)xv[3]);
    local_max = fmaxf(local_max, fmaxf(fmaxf(fabsf(v0), fabsf(v1)), fmaxf(fabsf(v2), fabsf(v3))));
    // clamp to e4m3 max-normal (448): OCP e4m3 has no inf to saturate to
    auto cl = [](float v) { return fminf(fmaxf(v, -448.f), 448.f); };
    int packed = 0;
    packed = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v0 * inv_s), cl(v1 * inv_s), packed, false);
    packed = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v2 * inv_s), cl(v3 * inv_s), packed, true);
    *reinterpret_cast<uint32_t*>(y + i0) = (uint32_t)packed;
  }
  // wave max -> LDS -> block max -> ONE atomic per block (same-address
  // atomics from every wave serialize badly)
#pragma unroll
  for (int off = 1; off < WAVE_SIZE; off <<= 1)
    local_max = fmaxf(local_max, __shfl_xor(local_max, off, WAVE_SIZE));
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & (WAVE_SIZE - 1)) == 0) red[wid] = local_max;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = red[0];
    for (int w = 1; w < (int)(blockDim.x / WAVE_SIZE); ++w) m = fmaxf(m, red[w]);
    if (m > 0.f)
      atomicMax(reinterpret_cast<unsigned int*>(amax_next), __float_as_uint(m));
  }
}

// fold accumulated amax into scales for n call sites (one launch per
// decode step, not per GEMM)
__global__ void fp8_scale_update_kernel(float* __restrict__ scale,
                                        float* __restrict__ amax_next, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float a = amax_next[i];
  if (a > 0.f) scale[i] = fmaxf(a / 448.f, 1e-8f);
  amax_next[i] = 0.f;
}

void fp8_quant_delayed(torch::Tensor x, torch::Tensor y, torch::Tensor scale,
                       torch::Tensor amax_next) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(y.dtype() == torch::kFloat8_e4m3fn && y.numel() == x.numel());
  TORCH_CHECK(scale.dtype() == torch::kFloat32 && amax_next.dtype() == torch::kFloat32);
  const int64_t n = x.numel();
  TORCH_CHECK(n % 4 == 0);
  hipLaunchKernelGGL(fp8_quant_delayed_kernel, dim3(grid_for(n / 4, 256)), dim3(256), 0,
                     cur_stream(), (const uint16_t*)x.data_ptr(), (uint8_t*)y.data_ptr(),
                     scale.data_ptr<float>(), amax_next.data_ptr<float>(), n);
  HIP_CHECK_KERNEL();
}

// ---------------------------------------------------------------------------
// FUSED fp8 producers (round-2, TODO #12): the kernel that PRODUCES a GEMM
// activation emits e4m3 directly (same delayed-scaling recipe as
// fp8_quant_delayed) — erases one full read+write of the activation stream
// and one kernel launch per GEMM site. Used on the decode hot path for the
// qkv / gate_up / down inputs.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void store_fp8_8(uint8_t* dst, const float* v,
                                            float inv_s, float& local_max) {
  auto cl = [](float x) { return fminf(fmaxf(x, -448.f), 448.f); };
  int p0 = 0, p1 = 0;
#pragma unroll
  for (int j = 0; j < 8; ++j) local_max = fmaxf(local_max, fabsf(v[j]));
  p0 = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v[0] * inv_s), cl(v[1] * inv_s), p0, false);
  p0 = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v[2] * inv_s), cl(v[3] * inv_s), p0, true);
  p1 = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v[4] * inv_s), cl(v[5] * inv_s), p1, false);
  p1 = __builtin_amdgcn_cvt_pk_fp8_f32(cl(v[6] * inv_s), cl(v[7] * inv_s), p1, true);
  uint2 packed = make_uint2((uint32_t)p0, (uint32_t)p1);
  *reinterpret_cast<uint2*>(dst) = packed;
}

// add_rmsnorm with e4m3 output: h += delta (bf16, in place); y8 = e4m3 of
// rmsnorm(h)*w. One wave per row (H % 8 == 0).
__global__ void add_rmsnorm_fp8_wave_kernel(
    uint16_t* __restrict__ h,
    const uint16_t* __restrict__ delta,
    const uint16_t* __restrict__ w,
    uint8_t* __restrict__ y8,
    const float* __restrict__ scale,
    float* __restrict__ amax_next,
    int64_t T, int H, float eps) {
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * 4 + wid;
  if (row >= T) return;
  uint16_t* hr = h + row * (int64_t)H;
  const uint16_t* dr = delta ? delta + row * (int64_t)H : nullptr;
  uint8_t* yr = y8 + row * (int64_t)H;
  const float inv_s = 1.f / scale[0];

  float ss = 0.f;
  for (int i = lane * 8; i < H; i += WAVE_SIZE * 8) {
    short8v a = *reinterpret_cast<const short8v*>(hr + i);
    if (dr) {
      short8v da = *reinterpret_cast<const short8v*>(dr + i);
      short8v o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_to_f32((uint16_t)a[j]) + bf16_to_f32((uint16_t)da[j]);
        o[j] = (short)f32_to_bf16(v);
        ss += v * v;
      }
      *reinterpret_cast<short8v*>(hr + i) = o;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_to_f32((uint16_t)a[j]);
        ss += v * v;
      }
    }
  }
#pragma unroll
  for (int off = 1; off < WAVE_SIZE; off <<= 1) ss += __shfl_xor(ss, off, WAVE_SIZE);
  const float inv_rms = rsqrtf(ss / (float)H + eps);
  float local_max = 0.f;
  for (int i = lane * 8; i < H; i += WAVE_SIZE * 8) {
    short8v a = *reinterpret_cast<const short8v*>(hr + i);
    short8v wv = *reinterpret_cast<const short8v*>(w + i);
    float v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = bf16_to_f32((uint16_t)a[j]) * inv_rms * bf16_to_f32((uint16_t)wv[j]);
    store_fp8_8(yr + i, v, inv_s, local_max);
  }
#pragma unroll
  for (int off = 1; off < WAVE_SIZE; off <<= 1)
    local_max = fmaxf(local_max, __shfl_xor(local_max, off, WAVE_SIZE));
  if (lane == 0 && local_max > 0.f)
    atomicMax(reinterpret_cast<unsigned int*>(amax_next), __float_as_uint(local_max));
}

torch::Tensor add_rmsnorm_fp8_(torch::Tensor h, c10::optional<torch::Tensor> delta,
                               torch::Tensor w, double eps,
                               torch::Tensor scale, torch::Tensor amax_next) {
  TORCH_CHECK(h.is_cuda() && h.dtype() == torch::kBFloat16 && h.is_contiguous());
  const int64_t T = h.numel() / h.size(-1);
  const int H = (int)h.size(-1);
  TORCH_CHECK(H % 8 == 0, "add_rmsnorm_fp8_ needs H % 8 == 0");
  auto y = torch::empty(h.sizes(), h.options().dtype(torch::kFloat8_e4m3fn));
  const uint16_t* dptr = nullptr;
  if (delta.has_value()) {
    TORCH_CHECK(delta->is_contiguous() && delta->sizes() == h.sizes());
    dptr = (const uint16_t*)delta->data_ptr();
  }
  hipLaunchKernelGGL(add_rmsnorm_fp8_wave_kernel, dim3((unsigned)((T + 3) / 4)), dim3(256),
                     0, cur_stream(), (uint16_t*)h.data_ptr(), dptr,
                     (const uint16_t*)w.data_ptr(), (uint8_t*)y.data_ptr(),
                     scale.data_ptr<float>(), amax_next.data_ptr<float>(), T, H, (float)eps);
  HIP_CHECK_KERNEL();
  return y;
}

// SwiGLU with e4m3 output (gateup [T, 2I] bf16 -> y8 [T, I] e4m3)
__global__ void swiglu_fp8_kernel8(
    const uint16_t* __restrict__ gateup,
    uint8_t* __restrict__ out,
    const float* __restrict__ scale,
    float* __restrict__ amax_next,
    int64_t T, int I) {
  __shared__ float red[16];
  const float inv_s = 1.f / scale[0];
  float local_max = 0.f;
  const int64_t total = T * (int64_t)I / 8;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = idx / (I / 8);
    int i = (int)(idx % (I / 8)) * 8;
    short8v gv = *reinterpret_cast<const short8v*>(gateup + t * 2 * I + i);
    short8v uv = *reinterpret_cast<const short8v*>(gateup + t * 2 * I + I + i);
    float v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32((uint16_t)gv[j]);
      float uf = bf16_to_f32((uint16_t)uv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      v[j] = gf * sig * uf;
    }
    store_fp8_8(out + t * I + i, v, inv_s, local_max);
  }
#pragma unroll
  for (int off = 1; off < WAVE_SIZE; off <<= 1)
    local_max = fmaxf(local_max, __shfl_xor(local_max, off, WAVE_SIZE));
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & (WAVE_SIZE - 1)) == 0) red[wid] = local_max;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = red[0];
    for (int w2 = 1; w2 < (int)(blockDim.x / WAVE_SIZE); ++w2) m = fmaxf(m, red[w2]);
    if (m > 0.f)
      atomicMax(reinterpret_cast<unsigned int*>(amax_next), __float_as_uint(m));
  }
}

torch::Tensor swiglu_fp8(torch::Tensor gateup, torch::Tensor scale, torch::Tensor amax_next) {
  TORCH_CHECK(gateup.is_cuda() && gateup.dtype() == torch::kBFloat16 && gateup.is_contiguous());
  const int64_t T = gateup.numel() / gateup.size(-1);
  const int I = (int)gateup.size(-1) / 2;
  TORCH_CHECK(I % 8 == 0, "swiglu_fp8 needs I % 8 == 0");
  auto sizes = gateup.sizes().vec();
  sizes.back() = I;
  auto y = torch::empty(sizes, gateup.options().dtype(torch::kFloat8_e4m3fn));
  hipLaunchKernelGGL(swiglu_fp8_kernel8, dim3(grid_for(T * I / 8, 256)), dim3(256), 0,
                     cur_stream(), (const uint16_t*)gateup.data_ptr(), (uint8_t*)y.data_ptr(),
                     scale.data_ptr<float>(), amax_next.data_ptr<float>(), T, I);
  HIP_CHECK_KERNEL();
  return y;
}

void fp8_scale_update(torch::Tensor scale, torch::Tensor amax_next) {
  const int n = (int)scale.numel();
  TORCH_CHECK(amax_next.numel() == n && scale.is_contiguous() && amax_next.is_contiguous());
  hipLaunchKernelGGL(fp8_scale_update_kernel, dim3((n + 255) / 256), dim3(256), 0, cur_stream(),
                     scale.data_ptr<float>(), amax_next.data_ptr<float>(), n);
  HIP_CHECK_KERNEL();
}
