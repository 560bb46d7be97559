// Skinny-M GEMM for the decode path: C[M,N] = A[M,K] @ W[N,K]^T (+bias).
//
// Decode GEMMs (M <= 512) are weight-stream bound, but hipBLASLt shows a
// ~19-25 us fixed floor at these shapes (vs a ~1-9 us stream floor for
// qkv/o/gate_up). Design (v2 — v1's 16-col tiles over-read A from L2 and
// starved the MFMAs):
//   * block = 4 waves, BN = 64 (wave w owns n-subtile w*16..w*16+15)
//   * A chunk [256, 128] staged in LDS once per K-chunk, SHARED by all 4
//     waves (XOR-swizzled for conflict-free ds_read_b128)
//   * W chunk [64, 128] staged coalesced; per B-fragment each wave issues
//     16 MFMAs (one per m-subtile) -> MFMA:ds_read ~ 1:1
//   * small-N shapes K-split (fp32 atomics + finalize) to fill the chip
// LDS: A 64K + W 16K = 80 KiB -> 2 blocks/CU.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define SK_BM 256      // rows covered per block (m-block)
#define SK_BN 64       // cols per block (16 per wave)
#define SK_BK 128      // K chunk staged in LDS

// A tile [256][128] bf16, row stride 256B: XOR-swizzle for the 16-row
// column-slice reads of the A fragments.
__device__ __forceinline__ int a_swz(int row, int byte_off) {
  return (row * (SK_BK * 2) + byte_off) ^ ((row & 7) << 4);
}
// W tile [64][128] bf16 — same geometry.
__device__ __forceinline__ int w_swz(int row, int byte_off) {
  return (row * (SK_BK * 2) + byte_off) ^ ((row & 7) << 4);
}

template <bool SPLIT>
__global__ __launch_bounds__(256, 2) void skinny_gemm_kernel(
    const __bf16* __restrict__ A,    // [M, K]
    const __bf16* __restrict__ W,    // [N, K]
    const __bf16* __restrict__ bias, // [N] or null
    __bf16* __restrict__ C,          // [M, N] (direct)
    float* __restrict__ C32,         // [M, N] (split, pre-zeroed)
    int M, int N, int K) {
  const int n0 = blockIdx.x * SK_BN;
  const int ksplit = blockIdx.y;
  const int nsplits = gridDim.y;
  const int m_blk = blockIdx.z * SK_BM;

  const int k_per = ((K + nsplits - 1) / nsplits + SK_BK - 1) / SK_BK * SK_BK;
  const int k_lo = ksplit * k_per;
  const int k_hi = min(K, k_lo + k_per);

  __shared__ char a_lds[SK_BM * SK_BK * 2];  // 64 KiB
  __shared__ char w_lds[SK_BN * SK_BK * 2];  // 16 KiB

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int a_k0 = (lane >> 4) * 8;
  const int col = lane & 15;

  f32x4_t acc[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) acc[i] = f32x4_t{};

  for (int kc = k_lo; kc < k_hi; kc += SK_BK) {
    __syncthreads();
    // stage A [256, 128]: 256x128/8 = 4096 vec8 slots, 16 per thread
    for (int idx = threadIdx.x; idx < SK_BM * (SK_BK / 8); idx += 256) {
      const int r = idx / (SK_BK / 8);
      const int c8 = (idx % (SK_BK / 8)) * 8;
      bf16x8_t av{};
      if (m_blk + r < M && kc + c8 < K) {
        av = *reinterpret_cast<const bf16x8_t*>(A + (int64_t)(m_blk + r) * K + kc + c8);
      }
      *reinterpret_cast<bf16x8_t*>(&a_lds[0] + a_swz(r, c8 * 2)) = av;
    }
    // stage W [64, 128]: 1024 vec8 slots, 4 per thread
    for (int idx = threadIdx.x; idx < SK_BN * (SK_BK / 8); idx += 256) {
      const int r = idx / (SK_BK / 8);
      const int c8 = (idx % (SK_BK / 8)) * 8;
      bf16x8_t wv{};
      if (n0 + r < N && kc + c8 < K) {
        wv = *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + r) * K + kc + c8);
      }
      *reinterpret_cast<bf16x8_t*>(&w_lds[0] + w_swz(r, c8 * 2)) = wv;
    }
    __syncthreads();

    const int kend = min(SK_BK, k_hi - kc);
#pragma unroll
    for (int ks = 0; ks < SK_BK; ks += 32) {
      if (ks >= kend) break;
      // this wave's B fragment: W rows wid*16 + col, k slice ks..ks+32
      bf16x8_t b_frag = *reinterpret_cast<const bf16x8_t*>(
          &w_lds[0] + w_swz(wid * 16 + col, (ks + a_k0) * 2));
#pragma unroll
      for (int mt = 0; mt < 16; ++mt) {
        bf16x8_t a_frag = *reinterpret_cast<const bf16x8_t*>(
            &a_lds[0] + a_swz(mt * 16 + col, (ks + a_k0) * 2));
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[mt], 0, 0, 0);
      }
    }
  }

  // epilogue: C row = mt*16 + (lane>>4)*4 + r, col = n0 + wid*16 + (lane&15)
  const int n = n0 + wid * 16 + col;
  if (n >= N) return;
  const float bias_v = bias ? (float)bias[n] : 0.f;
#pragma unroll
  for (int mt = 0; mt < 16; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m_blk + mt * 16 + (lane >> 4) * 4 + r;
      if (m >= M) continue;
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

// ---------------------------------------------------------------------------
// v3: 64x64 wave tiles + double-buffered LDS + partials split-K.
// v2 autopsy: (a) wave tile 256x16 -> one ds_read per MFMA (LDS-bound);
// (b) split-K via 4M fp32 atomicAdds; (c) stage->compute serialized by a
// full-block barrier. v3: wave owns 64x64 (4x4 MFMA tiles, 8 ds_reads per
// 16 MFMAs), global loads for chunk i+1 issue before computing chunk i
// (latency hidden behind the MFMA loop), split-K writes disjoint fp32
// partials reduced by a tiny second kernel.
// Block: 4 waves; tile M=256 (wave w: rows 64w..64w+63) x N=64; BK=64.
// LDS: (A 32K + W 8K) x 2 buffers = 80 KiB -> 2 blocks/CU.
// ---------------------------------------------------------------------------

#define V3_BM 256
#define V3_BN 64
#define V3_BK 64

// row stride 128 B (64 bf16); XOR swizzle spreads the 16-row fragment reads
__device__ __forceinline__ int v3_swz(int row, int byte_off) {
  return (row * (V3_BK * 2) + byte_off) ^ ((row & 7) << 4);
}

template <bool SPLIT>
__global__ __launch_bounds__(256, 2) void skinny_gemm_v3_kernel(
    const __bf16* __restrict__ A,    // [M, K]
    const __bf16* __restrict__ W,    // [N, K]
    const __bf16* __restrict__ bias, // [N] or null
    __bf16* __restrict__ C,          // [M, N] (direct)
    float* __restrict__ P,           // [S, M, N] partials (split)
    int M, int N, int K) {
  const int n0 = blockIdx.x * V3_BN;
  const int ks_id = blockIdx.y;
  const int nsplits = gridDim.y;

  const int kchunks = (K + V3_BK - 1) / V3_BK;
  const int per = (kchunks + nsplits - 1) / nsplits;
  const int c_lo = ks_id * per;
  const int c_hi = min(kchunks, c_lo + per);
  if (c_lo >= c_hi) {
    // nothing to accumulate: still must write zeros in SPLIT mode
    if (SPLIT) {
      const int lane = threadIdx.x & 63;
      const int wid = threadIdx.x >> 6;
      for (int mt = 0; mt < 4; ++mt)
        for (int nt = 0; nt < 4; ++nt)
          for (int r = 0; r < 4; ++r) {
            const int m = wid * 64 + mt * 16 + (lane >> 4) * 4 + r;
            const int n = n0 + nt * 16 + (lane & 15);
            if (m < M && n < N)
              P[((int64_t)ks_id * M + m) * N + n] = 0.f;
          }
    }
    return;
  }

  __shared__ char lds[2 * (V3_BM * V3_BK * 2 + V3_BN * V3_BK * 2)];
  auto a_buf = [&](int b) -> char* { return lds + b * (V3_BM * V3_BK * 2 + V3_BN * V3_BK * 2); };
  auto w_buf = [&](int b) -> char* { return a_buf(b) + V3_BM * V3_BK * 2; };

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int frag_k0 = (lane >> 4) * 8;   // k offset within a 32-slice
  const int frag_col = lane & 15;

  // staging ownership: A 256x64 = 2048 vec8 -> 8/thread; W 64x64 = 512 -> 2/thread
  bf16x8_t a_reg[8];
  bf16x8_t w_reg[2];

  auto load_chunk = [&](int chunk) {
    const int kc = chunk * V3_BK;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = threadIdx.x + i * 256;      // 0..2047
      const int r = idx >> 3;                     // A row
      const int c8 = (idx & 7) * 8;
      a_reg[i] = bf16x8_t{};
      if (r < M && kc + c8 < K)
        a_reg[i] = *reinterpret_cast<const bf16x8_t*>(A + (int64_t)r * K + kc + c8);
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int idx = threadIdx.x + i * 256;      // 0..511
      const int r = idx >> 3;
      const int c8 = (idx & 7) * 8;
      w_reg[i] = bf16x8_t{};
      if (n0 + r < N && kc + c8 < K)
        w_reg[i] = *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + r) * K + kc + c8);
    }
  };

  auto store_chunk = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = threadIdx.x + i * 256;
      const int r = idx >> 3;
      const int c8 = (idx & 7) * 8;
      *reinterpret_cast<bf16x8_t*>(a_buf(buf) + v3_swz(r, c8 * 2)) = a_reg[i];
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int idx = threadIdx.x + i * 256;
      const int r = idx >> 3;
      const int c8 = (idx & 7) * 8;
      *reinterpret_cast<bf16x8_t*>(w_buf(buf) + v3_swz(r, c8 * 2)) = w_reg[i];
    }
  };

  f32x4_t acc[4][4];  // [m-subtile][n-subtile]
#pragma unroll
  for (int mt = 0; mt < 4; ++mt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[mt][nt] = f32x4_t{};

  load_chunk(c_lo);
  store_chunk(0);
  __syncthreads();

  for (int chunk = c_lo; chunk < c_hi; ++chunk) {
    const int cur = (chunk - c_lo) & 1;
    if (chunk + 1 < c_hi) load_chunk(chunk + 1);  // in flight during MFMAs

#pragma unroll
    for (int ks = 0; ks < V3_BK; ks += 32) {
      bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        a_frag[t] = *reinterpret_cast<const bf16x8_t*>(
            a_buf(cur) + v3_swz(wid * 64 + t * 16 + frag_col, (ks + frag_k0) * 2));
        b_frag[t] = *reinterpret_cast<const bf16x8_t*>(
            w_buf(cur) + v3_swz(t * 16 + frag_col, (ks + frag_k0) * 2));
      }
#pragma unroll
      for (int mt = 0; mt < 4; ++mt)
#pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mt], b_frag[nt], acc[mt][nt], 0, 0, 0);
    }

    if (chunk + 1 < c_hi) {
      __syncthreads();            // everyone done reading buf[cur]
      store_chunk(cur ^ 1);
      __syncthreads();            // buf[nxt] visible
    }
  }

  // epilogue: C row = wid*64 + mt*16 + (lane>>4)*4 + r, col = n0 + nt*16 + (lane&15)
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int n = n0 + nt * 16 + frag_col;
      if (n >= N) continue;
      const float bias_v = (!SPLIT && bias) ? (float)bias[n] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = wid * 64 + mt * 16 + (lane >> 4) * 4 + r;
        if (m >= M) continue;
        if (SPLIT) {
          P[((int64_t)ks_id * M + m) * N + n] = acc[mt][nt][r];
        } else {
          C[(int64_t)m * N + n] = (__bf16)(acc[mt][nt][r] + bias_v);
        }
      }
    }
  }
}

// reduce S fp32 partials -> bf16 (+bias)
__global__ void skinny_v3_reduce_kernel(
    const float* __restrict__ P, const __bf16* __restrict__ bias,
    __bf16* __restrict__ C, int64_t MN, int N, int S) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < MN;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = 0.f;
    for (int s = 0; s < S; ++s) v += P[(int64_t)s * MN + i];
    if (bias) v += (float)bias[(int)(i % N)];
    C[i] = (__bf16)v;
  }
}

static inline hipStream_t sg_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor skinny_gemm_v3(torch::Tensor a, torch::Tensor w, c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K && M <= V3_BM);
  auto c = torch::empty({M, N}, a.options());
  const __bf16* bptr = bias.has_value() ? (const __bf16*)bias->data_ptr() : nullptr;

  const int n_tiles = (N + V3_BN - 1) / V3_BN;
  const int kchunks = (K + V3_BK - 1) / V3_BK;
  int S = 1;
  while (S * 2 <= kchunks && n_tiles * S < 384) S *= 2;

  if (S == 1) {
    hipLaunchKernelGGL((skinny_gemm_v3_kernel<false>), dim3(n_tiles, 1), dim3(256), 0, sg_stream(),
                       (const __bf16*)a.data_ptr(), (const __bf16*)w.data_ptr(), bptr,
                       (__bf16*)c.data_ptr(), nullptr, M, N, K);
    HIP_CHECK_KERNEL();
  } else {
    auto part = torch::empty({S, M, N}, a.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL((skinny_gemm_v3_kernel<true>), dim3(n_tiles, S), dim3(256), 0, sg_stream(),
                       (const __bf16*)a.data_ptr(), (const __bf16*)w.data_ptr(), nullptr,
                       nullptr, part.data_ptr<float>(), M, N, K);
    HIP_CHECK_KERNEL();
    const int64_t MN = (int64_t)M * N;
    hipLaunchKernelGGL(skinny_v3_reduce_kernel, dim3(grid_for(MN, 256)), dim3(256), 0, sg_stream(),
                       part.data_ptr<float>(), bptr, (__bf16*)c.data_ptr(), MN, N, S);
    HIP_CHECK_KERNEL();
  }
  return c;
}

torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w, c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K && M <= 512 && K % 32 == 0);
  auto c = torch::empty({M, N}, a.options());
  const __bf16* bptr = bias.has_value() ? (const __bf16*)bias->data_ptr() : nullptr;

  const int n_tiles = (N + SK_BN - 1) / SK_BN;
  const int m_blocks = (M + SK_BM - 1) / SK_BM;
  int ksplits = 1;
  const int kchunks = (K + SK_BK - 1) / SK_BK;
  while (ksplits * 2 <= kchunks && n_tiles * m_blocks * ksplits < 512) ksplits *= 2;

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
