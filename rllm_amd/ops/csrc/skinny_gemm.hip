// Skinny-M GEMM for the decode path: C[M,N] = A[M,K] @ W[N,K]^T (+bias).
//
// At decode batch sizes (M <= 512) these GEMMs are weight-stream bound:
// the whole W matrix is read once per step while A (<1 MB) lives in L2.
// hipBLASLt's tile choices for these shapes reach ~30% of HBM bandwidth;
// this kernel targets the stream: each block owns a 16-column slice of W,
// stages W tiles into XOR-swizzled LDS with coalesced loads, keeps A in
// L2 (per-lane 16 B loads), and accumulates C[M,16] in AGPRs via MFMA
// 16x16x32. Small-N shapes add a K-split dimension (fp32 atomics + a
// finalize pass) so the chip stays full.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define SK_KCHUNK 512   // K tile staged in LDS (16 rows x 512 x 2B = 16 KiB)

__device__ __forceinline__ int sk_swz(int row, int byte_off) {
  return (row * (SK_KCHUNK * 2) + byte_off) ^ ((row & 7) << 4);
}

// grid: (ceil(N/16), ksplits); block: 256 threads (4 waves, each wave owns
// 4 m-subtiles of 16 rows => block covers M<=256 rows x 16 cols).
// For M in (256, 512]: grid.z = 2 m-blocks.
template <bool SPLIT>
__global__ __launch_bounds__(256, 4) void skinny_gemm_kernel(
    const __bf16* __restrict__ A,   // [M, K]
    const __bf16* __restrict__ W,   // [N, K]
    const __bf16* __restrict__ bias,  // [N] or null
    __bf16* __restrict__ C,         // [M, N] (direct path)
    float* __restrict__ C32,        // [M, N] fp32 (split path, pre-zeroed)
    int M, int N, int K) {
  const int n0 = blockIdx.x * 16;
  const int ksplit = blockIdx.y;
  const int nsplits = gridDim.y;
  const int m_blk = blockIdx.z * 256;
  if (n0 >= N) return;

  const int k_per = ((K + nsplits - 1) / nsplits + SK_KCHUNK - 1) / SK_KCHUNK * SK_KCHUNK;
  const int k_lo = ksplit * k_per;
  const int k_hi = min(K, k_lo + k_per);

  __shared__ char w_lds[2][16 * SK_KCHUNK * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int a_k0 = (lane >> 4) * 8;     // k offset within a 32-slice
  const int col = lane & 15;            // n within tile / m row selector

  f32x4_t acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc[i] = f32x4_t{};

  int buf = 0;
  // prologue: stage first W chunk (coalesced along K)
  {
    const int kc0 = k_lo;
    for (int idx = threadIdx.x; idx < 16 * (SK_KCHUNK / 8); idx += 256) {
      const int r = idx / (SK_KCHUNK / 8);
      const int c8 = (idx % (SK_KCHUNK / 8)) * 8;
      bf16x8_t wv{};
      if (n0 + r < N && kc0 + c8 < k_hi) {
        wv = *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + r) * K + kc0 + c8);
      }
      *reinterpret_cast<bf16x8_t*>(&w_lds[0][0] + sk_swz(r, c8 * 2)) = wv;
    }
  }
  __syncthreads();

  for (int kc = k_lo; kc < k_hi; kc += SK_KCHUNK, buf ^= 1) {
    // prefetch next chunk into the other buffer
    const int kn = kc + SK_KCHUNK;
    if (kn < k_hi) {
      for (int idx = threadIdx.x; idx < 16 * (SK_KCHUNK / 8); idx += 256) {
        const int r = idx / (SK_KCHUNK / 8);
        const int c8 = (idx % (SK_KCHUNK / 8)) * 8;
        bf16x8_t wv{};
        if (n0 + r < N && kn + c8 < k_hi) {
          wv = *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + r) * K + kn + c8);
        }
        *reinterpret_cast<bf16x8_t*>(&w_lds[buf ^ 1][0] + sk_swz(r, c8 * 2)) = wv;
      }
    }

    const int kend = min(SK_KCHUNK, k_hi - kc);
    for (int ks = 0; ks < kend; ks += 32) {
      bf16x8_t b_frag = *reinterpret_cast<const bf16x8_t*>(
          &w_lds[buf][0] + sk_swz(col, (ks + a_k0) * 2));
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        const int m = m_blk + wid * 64 + mt * 16 + col;
        bf16x8_t a_frag{};
        if (m < M && kc + ks + a_k0 < K) {
          a_frag = *reinterpret_cast<const bf16x8_t*>(A + (int64_t)m * K + kc + ks + a_k0);
        }
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[mt], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: C layout row=(lane>>4)*4+r within each m-subtile, col=lane&15
  const float bias_v = (bias && n0 + col < N) ? (float)bias[n0 + col] : 0.f;
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m_blk + wid * 64 + mt * 16 + (lane >> 4) * 4 + r;
      const int n = n0 + col;
      if (m >= M || n >= N) continue;
      if (SPLIT) {
        atomicAdd(&C32[(int64_t)m * N + n], acc[mt][r]);
      } else {
        C[(int64_t)m * N + n] = (__bf16)(acc[mt][r] + bias_v);
      }
    }
  }
}

// finalize split path: C = bf16(C32 + bias)
__global__ void skinny_gemm_finalize_kernel(
    const float* __restrict__ C32, const __bf16* __restrict__ bias,
    __bf16* __restrict__ C, int64_t M, int N) {
  const int64_t total = M * N;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int n = (int)(i % N);
    float v = C32[i];
    if (bias) v += (float)bias[n];
    C[i] = (__bf16)v;
  }
}

static inline hipStream_t sg_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w, c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K && M <= 512 && K % 32 == 0);
  auto c = torch::empty({M, N}, a.options());
  const __bf16* bptr = bias.has_value() ? (const __bf16*)bias->data_ptr() : nullptr;

  const int n_tiles = (N + 15) / 16;
  const int m_blocks = (M + 255) / 256;
  // enough blocks to fill the chip; round K-splits to chunk granularity
  int ksplits = 1;
  const int kchunks = (K + SK_KCHUNK - 1) / SK_KCHUNK;
  while (ksplits < kchunks && n_tiles * m_blocks * ksplits < 512) ksplits *= 2;
  ksplits = std::min(ksplits, kchunks);

  dim3 grid(n_tiles, ksplits, m_blocks);
  if (ksplits == 1) {
    hipLaunchKernelGGL((skinny_gemm_kernel<false>), grid, dim3(256), 0, sg_stream(),
                       (const __bf16*)a.data_ptr(), (const __bf16*)w.data_ptr(), bptr,
                       (__bf16*)c.data_ptr(), nullptr, M, N, K);
    HIP_CHECK_KERNEL();
  } else {
    auto c32 = torch::zeros({M, N}, a.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL((skinny_gemm_kernel<true>), grid, dim3(256), 0, sg_stream(),
                       (const __bf16*)a.data_ptr(), (const __bf16*)w.data_ptr(), nullptr,
                       nullptr, c32.data_ptr<float>(), M, N, K);
    HIP_CHECK_KERNEL();
    hipLaunchKernelGGL(skinny_gemm_finalize_kernel, dim3(grid_for((int64_t)M * N, 256)), dim3(256),
                       0, sg_stream(), c32.data_ptr<float>(), bptr, (__bf16*)c.data_ptr(),
                       (int64_t)M, N);
    HIP_CHECK_KERNEL();
  }
  return c;
}
