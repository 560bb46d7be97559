// Attention kernels for the rollout engine (K1/K2 in SURVEY.md §2.E):
//
//  * flash_prefill: flash-style causal GQA prefill over packed varlen
//    sequences. MFMA 16x16x32 bf16 tiles; K/V staged in XOR-swizzled LDS
//    (guide §6 G4: row-major [*][128] bf16 tiles are a 32-way bank conflict
//    on ds_read_b128 — byte ^= ((row&7)<<4) fixes it); online softmax with
//    per-wave register state.
//  * paged_decode: one-token-per-seq decode over the paged KV cache.
//    Bandwidth-bound: VALU dot products, vectorized 16B KV loads,
//    flash-decoding style slot/wave merges.
//  * reshape_and_cache: scatter freshly computed K/V into KV pages.
//
// MFMA fragment layouts (gfx950, from the CDNA4 guide §3, HW-verified there):
//   mfma_f32_16x16x32_bf16: A[16,32] lane l holds row (l&15), k (l>>4)*8+j;
//   B[32,16] lane l holds col (l&15), k (l>>4)*8+j;
//   C/D: col = lane&15, row = (lane>>4)*4 + reg.

#include "common.hpp"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define HEAD_DIM 128

// fwd declarations (training-attention backward kernels defined at file end)
__global__ void attn_bwd_preprocess_kernel(const __bf16*, const __bf16*, float*, int64_t, int);
__global__ void flash_bwd_dq_kernel(const __bf16*, const __bf16*, const __bf16*, const __bf16*,
                                    const float*, const float*, __bf16*,
                                    const int32_t*, const int32_t*, const int32_t*, int, int, float);
__global__ void flash_bwd_dkv_kernel(const __bf16*, const __bf16*, const __bf16*, const __bf16*,
                                     const float*, const float*, __bf16*, __bf16*,
                                     const int32_t*, const int32_t*, const int32_t*, int, int, float);
__global__ void paged_decode_mfma_kernel(const __bf16*, const __bf16*, const __bf16*, __bf16*,
                                         float*, float*, const int32_t*, const int32_t*,
                                         int, int, int, int, float);
#define QTILE 64        // q rows per block (4 waves x 16)
#define KVTILE 64       // kv tokens per inner tile
#define PAGE_SIZE 16

// XOR swizzle for [rows][128] bf16 LDS tiles (and the transposed V tile):
// spreads the 16 rows a wave reads at one col-range across banks.
__device__ __forceinline__ int swz(int row, int byte_off) {
  return (row * (HEAD_DIM * 2) + byte_off) ^ ((row & 7) << 4);
}
// For P tile [rows][KVTILE] bf16 (row stride 128B)
__device__ __forceinline__ int swz_p(int row, int byte_off) {
  return (row * (KVTILE * 2) + byte_off) ^ ((row & 7) << 4);
}

// ---------------------------------------------------------------------------
// Prefill
// ---------------------------------------------------------------------------
// Q,K,V: [T, H*, 128] bf16 packed varlen. tile_seq_start/tile_row0: per q-tile
// the packed start index of its sequence and the tile's first row (packed
// index); tile_seq_len: the sequence's length.
__global__ __launch_bounds__(256, 2) void flash_prefill_kernel(
    const __bf16* __restrict__ Q,
    const __bf16* __restrict__ K,
    const __bf16* __restrict__ V,
    __bf16* __restrict__ O,
    const int32_t* __restrict__ tile_seq_start,
    const int32_t* __restrict__ tile_row0,
    const int32_t* __restrict__ tile_seq_len,
    float* __restrict__ lse_out,              // [T, Hq] fp32 or null (training fwd)
    int Hq, int Hk, float scale) {
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hk);

  const int seq_start = tile_seq_start[tile];
  const int row0 = tile_row0[tile];          // packed index of first q row
  const int seq_len = tile_seq_len[tile];
  const int q_local0 = row0 - seq_start;     // position of first row within seq

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;          // wave id: 0..3
  const int wrow0 = q_local0 + wid * 16;     // first local q row of this wave

  __shared__ char lds[/*K*/ KVTILE * HEAD_DIM * 2 + /*VT*/ HEAD_DIM * KVTILE * 2 + /*P*/ 4 * 16 * KVTILE * 2];
  char* k_lds = lds;
  char* vt_lds = lds + KVTILE * HEAD_DIM * 2;
  char* p_lds = vt_lds + HEAD_DIM * KVTILE * 2;

  // ---- load Q fragments (4 d-slices of 32) into registers ----
  // a-frag element: Q[wrow0 + (lane&15)][ds*32 + (lane>>4)*8 + j]
  bf16x8 q_frag[4];
  const int a_row = wrow0 + (lane & 15);
  const int a_k0 = (lane >> 4) * 8;
  const bool row_valid = a_row < seq_len;
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
    if (row_valid) {
      const __bf16* src = Q + ((int64_t)(seq_start + a_row) * Hq + qh) * HEAD_DIM + ds * 32 + a_k0;
      q_frag[ds] = *reinterpret_cast<const bf16x8*>(src);
    } else {
      q_frag[ds] = bf16x8{};
    }
  }

  // ---- online softmax state: 4 rows per lane (C layout rows) ----
  float m_st[4], l_st[4];
  f32x4 o_acc[8];  // 8 d-tiles of 16 cols; C layout
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_st[r] = -INFINITY; l_st[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) o_acc[dt] = f32x4{};

  // causal limit: last kv token needed by this block
  const int q_hi = min(q_local0 + QTILE, seq_len);   // exclusive
  const int kv_limit = q_hi;                          // causal

  for (int kv0 = 0; kv0 < kv_limit; kv0 += KVTILE) {
    const int kv_n = min(KVTILE, seq_len - kv0);
    // ---- stage K tile (swizzled) and V tile (transposed + swizzled) ----
    // 256 threads, K tile 64x128: each thread stages 2 rows' halves: 64*8=512
    // 16B chunks -> 2 per thread.
    __syncthreads();
    for (int idx = threadIdx.x; idx < KVTILE * (HEAD_DIM / 8); idx += 256) {
      const int r = idx / (HEAD_DIM / 8);
      const int c8 = (idx % (HEAD_DIM / 8)) * 8;
      bf16x8 kv;
      if (r < kv_n) {
        kv = *reinterpret_cast<const bf16x8*>(K + ((int64_t)(seq_start + kv0 + r) * Hk + kvh) * HEAD_DIM + c8);
      } else {
        kv = bf16x8{};
      }
      *reinterpret_cast<bf16x8*>(k_lds + swz(r, c8 * 2)) = kv;
      // V transposed: VT[d][kv]
      bf16x8 vv;
      if (r < kv_n) {
        vv = *reinterpret_cast<const bf16x8*>(V + ((int64_t)(seq_start + kv0 + r) * Hk + kvh) * HEAD_DIM + c8);
      } else {
        vv = bf16x8{};
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<__bf16*>(vt_lds + swz_p(c8 + j, r * 2)) = vv[j];
      }
    }
    __syncthreads();

    // ---- QK^T: 4 n-tiles of 16 kv cols ----
    f32x4 s_acc[4];
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      s_acc[nt] = f32x4{};
      const int b_col = nt * 16 + (lane & 15);  // kv token within tile
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(k_lds + swz(b_col, (ds * 32 + a_k0) * 2));
        s_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ds], b_frag, s_acc[nt], 0, 0, 0);
      }
    }

    // ---- mask + online softmax ----
    // lane's element (nt, r): q row = wrow0 + (lane>>4)*4 + r,
    //                         kv col = kv0 + nt*16 + (lane&15)
    const int my_row = wrow0 + (lane >> 4) * 4;  // +r
    float p[4][4];
    float row_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) row_max[r] = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int col = kv0 + nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = s_acc[nt][r] * scale;
        const int qrow = my_row + r;
        if (col > qrow || qrow >= seq_len || col >= seq_len) s = -INFINITY;
        p[nt][r] = s;
        row_max[r] = fmaxf(row_max[r], s);
      }
    }
    // reduce row max over the 16 lanes holding this row's cols
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) row_max[r] = fmaxf(row_max[r], __shfl_xor(row_max[r], off, 64));
    }

    float rescale[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_st[r], row_max[r]);
      rescale[r] = (m_st[r] == -INFINITY) ? 0.f : __expf(m_st[r] - m_new);
      m_st[r] = m_new;
      float row_sum = 0.f;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        float e = (p[nt][r] == -INFINITY) ? 0.f : __expf(p[nt][r] - m_new);
        p[nt][r] = e;
        row_sum += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) row_sum += __shfl_xor(row_sum, off, 64);
      l_st[r] = l_st[r] * rescale[r] + row_sum;
    }

    // ---- write P to LDS in A-frag layout (per wave region) ----
    char* pw = p_lds + wid * 16 * KVTILE * 2;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int prow = (lane >> 4) * 4 + r;      // 0..15 within wave tile
        const int pcol = nt * 16 + (lane & 15);    // 0..63
        *reinterpret_cast<__bf16*>(pw + swz_p(prow, pcol * 2)) = (__bf16)(float)p[nt][r];
      }
    }
    // rescale o accumulators (cols of same row share rescale; C-layout row = (lane>>4)*4+r)
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dt][r] *= rescale[r];
    }
    __syncthreads();  // P visible to own wave only — but LDS ops need wave-local order only; syncthreads for V staging reuse safety below

    // ---- PV: A = P[16, 64], B = VT -> O[16, 128] ----
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(pw + swz_p(lane & 15, (ks * 32 + a_k0) * 2));
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(vt_lds + swz_p(dt * 16 + (lane & 15), (ks * 32 + a_k0) * 2));
        o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, o_acc[dt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: normalize and store ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = wrow0 + (lane >> 4) * 4 + r;
    if (qrow >= seq_len || qrow < q_local0) continue;
    const float inv_l = (l_st[r] > 0.f) ? 1.f / l_st[r] : 0.f;
    __bf16* orow = O + ((int64_t)(seq_start + qrow) * Hq + qh) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      orow[dt * 16 + (lane & 15)] = (__bf16)(o_acc[dt][r] * inv_l);
    }
    if (lse_out != nullptr && (lane & 15) == 0) {
      lse_out[(int64_t)(seq_start + qrow) * Hq + qh] = m_st[r] + __logf(l_st[r]);
    }
  }
}

// ---------------------------------------------------------------------------
// Paged decode (flash-decoding style seq splits for occupancy: B*Hk blocks
// alone cannot fill 256 CUs at modest batch sizes)
// ---------------------------------------------------------------------------
// Q: [B, Hq, 128] (one new token per seq). K/V pages: [n_pages, Hk, 16, 128].
// block_tables: [B, max_pages] int32. seq_lens: [B] int32 (INCLUDING the new
// token, whose k/v must already be in the cache).
// If n_splits > 1, each (b, kvh, split) block writes partial (o, m, l) into
// ws_o [B, Hq, S, 128] / ws_ml [B, Hq, S, 2]; decode_merge_kernel combines.
template <int G>
__global__ __launch_bounds__(256, 2) void paged_decode_kernel(
    const __bf16* __restrict__ Q,
    const __bf16* __restrict__ Kp,
    const __bf16* __restrict__ Vp,
    __bf16* __restrict__ O,          // used when gridDim.z == 1
    float* __restrict__ ws_o,        // used when gridDim.z > 1
    float* __restrict__ ws_ml,
    const int32_t* __restrict__ block_tables,
    const int32_t* __restrict__ seq_lens,
    int Hq, int Hk, int max_pages, float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int n_splits = gridDim.z;
  const int seq_len = seq_lens[b];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int slot = lane >> 4;       // 4 token slots per wave
  const int li = lane & 15;         // 16 lanes per slot, 8 dims each
  const int d0 = li * 8;

  // this split's token range
  const int chunk = (seq_len + n_splits - 1) / n_splits;
  const int tok_lo = split * chunk;
  const int tok_hi = min(seq_len, tok_lo + chunk);

  __shared__ float q_sh[G][HEAD_DIM];
  __shared__ float merge_buf[4][G][HEAD_DIM];  // cross-wave merge
  __shared__ float merge_ml[4][G][2];

  // load q for this kv-head's group into LDS (fp32, pre-scaled)
  for (int idx = threadIdx.x; idx < G * HEAD_DIM; idx += 256) {
    const int g = idx / HEAD_DIM;
    const int d = idx % HEAD_DIM;
    q_sh[g][d] = (float)Q[((int64_t)b * Hq + kvh * G + g) * HEAD_DIM + d] * scale;
  }
  __syncthreads();

  // per-lane online state for this lane's token stream (slot-strided)
  float m_st[G], l_st[G], o_st[G][8];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m_st[g] = -INFINITY; l_st[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) o_st[g][j] = 0.f;
  }

  const int32_t* bt = block_tables + (int64_t)b * max_pages;

  for (int tok = tok_lo + wid * 4 + slot; tok < tok_hi; tok += 16) {
    const int page = bt[tok >> 4];
    const int off = tok & 15;
    const __bf16* krow = Kp + (((int64_t)page * Hk + kvh) * PAGE_SIZE + off) * HEAD_DIM + d0;
    const __bf16* vrow = Vp + (((int64_t)page * Hk + kvh) * PAGE_SIZE + off) * HEAD_DIM + d0;
    bf16x8 kv = *reinterpret_cast<const bf16x8*>(krow);
    bf16x8 vv = *reinterpret_cast<const bf16x8*>(vrow);

    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float acc = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += q_sh[g][d0 + j] * (float)kv[j];
      // reduce over the 16 lanes of this slot
#pragma unroll
      for (int offx = 1; offx < 16; offx <<= 1) acc += __shfl_xor(acc, offx, 64);
      s[g] = acc;
    }
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float m_new = fmaxf(m_st[g], s[g]);
      const float rs = (m_st[g] == -INFINITY) ? 0.f : __expf(m_st[g] - m_new);
      const float p = __expf(s[g] - m_new);
      l_st[g] = l_st[g] * rs + p;
#pragma unroll
      for (int j = 0; j < 8; ++j) o_st[g][j] = o_st[g][j] * rs + p * (float)vv[j];
      m_st[g] = m_new;
    }
  }

  // ---- merge the 4 slots within each wave (shfl by 16, 32) ----
#pragma unroll
  for (int half = 16; half <= 32; half <<= 1) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float m_o = __shfl_xor(m_st[g], half, 64);
      const float l_o = __shfl_xor(l_st[g], half, 64);
      const float m_new = fmaxf(m_st[g], m_o);
      const float r_a = (m_st[g] == -INFINITY) ? 0.f : __expf(m_st[g] - m_new);
      const float r_b = (m_o == -INFINITY) ? 0.f : __expf(m_o - m_new);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float o_o = __shfl_xor(o_st[g][j], half, 64);
        o_st[g][j] = o_st[g][j] * r_a + o_o * r_b;
      }
      l_st[g] = l_st[g] * r_a + l_o * r_b;
      m_st[g] = m_new;
    }
  }

  // ---- merge the 4 waves via LDS ----
  if (slot == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
#pragma unroll
      for (int j = 0; j < 8; ++j) merge_buf[wid][g][d0 + j] = o_st[g][j];
      if (li == 0) { merge_ml[wid][g][0] = m_st[g]; merge_ml[wid][g][1] = l_st[g]; }
    }
  }
  __syncthreads();
  if (wid == 0 && slot == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float m_tot = -INFINITY, l_tot = 0.f;
      for (int w2 = 0; w2 < 4; ++w2) m_tot = fmaxf(m_tot, merge_ml[w2][g][0]);
      float o_tot[8] = {0, 0, 0, 0, 0, 0, 0, 0};
      for (int w2 = 0; w2 < 4; ++w2) {
        const float mw = merge_ml[w2][g][0];
        const float r = (mw == -INFINITY) ? 0.f : __expf(mw - m_tot);
        l_tot += merge_ml[w2][g][1] * r;
#pragma unroll
        for (int j = 0; j < 8; ++j) o_tot[j] += merge_buf[w2][g][d0 + j] * r;
      }
      const int qh = kvh * G + g;
      if (n_splits == 1) {
        const float inv_l = (l_tot > 0.f) ? 1.f / l_tot : 0.f;
        __bf16* orow = O + ((int64_t)b * Hq + qh) * HEAD_DIM + d0;
#pragma unroll
        for (int j = 0; j < 8; ++j) orow[j] = (__bf16)(o_tot[j] * inv_l);
      } else {
        float* wrow = ws_o + (((int64_t)b * Hq + qh) * n_splits + split) * HEAD_DIM + d0;
#pragma unroll
        for (int j = 0; j < 8; ++j) wrow[j] = o_tot[j];
        if (li == 0) {
          float* mlrow = ws_ml + (((int64_t)b * Hq + qh) * n_splits + split) * 2;
          mlrow[0] = m_tot; mlrow[1] = l_tot;
        }
      }
    }
  }
}

// Combine split partials: one wave per (b, qh); 16 lanes x 8 dims.
__global__ void decode_merge_kernel(
    const float* __restrict__ ws_o,   // [B, Hq, S, 128]
    const float* __restrict__ ws_ml,  // [B, Hq, S, 2]
    __bf16* __restrict__ O,           // [B, Hq, 128]
    int Hq, int n_splits) {
  const int64_t bh = blockIdx.x;   // b * Hq + qh
  const int lane = threadIdx.x & 63;
  const int d0 = (lane & 15) * 8;
  if (lane >= 16) return;

  float m_tot = -INFINITY;
  for (int s = 0; s < n_splits; ++s) m_tot = fmaxf(m_tot, ws_ml[(bh * n_splits + s) * 2]);
  float l_tot = 0.f;
  float o_tot[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int s = 0; s < n_splits; ++s) {
    const float m = ws_ml[(bh * n_splits + s) * 2];
    const float l = ws_ml[(bh * n_splits + s) * 2 + 1];
    const float r = (m == -INFINITY) ? 0.f : __expf(m - m_tot);
    l_tot += l * r;
    const float* orow = ws_o + (bh * n_splits + s) * HEAD_DIM + d0;
#pragma unroll
    for (int j = 0; j < 8; ++j) o_tot[j] += orow[j] * r;
  }
  const float inv_l = (l_tot > 0.f) ? 1.f / l_tot : 0.f;
  __bf16* out = O + bh * HEAD_DIM + d0;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = (__bf16)(o_tot[j] * inv_l);
}

// ---------------------------------------------------------------------------
// reshape_and_cache: scatter K/V [T, Hk, 128] into pages via slot_mapping [T]
// ---------------------------------------------------------------------------
__global__ void reshape_and_cache_kernel(
    const __bf16* __restrict__ K,
    const __bf16* __restrict__ V,
    __bf16* __restrict__ Kp,
    __bf16* __restrict__ Vp,
    const int32_t* __restrict__ slot_mapping, // [T] global slot = page*16+off
    int64_t T, int Hk) {
  const int64_t total = T * Hk * (HEAD_DIM / 8);
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int c8 = (int)(idx % (HEAD_DIM / 8)) * 8;
    const int64_t th = idx / (HEAD_DIM / 8);
    const int h = (int)(th % Hk);
    const int64_t t = th / Hk;
    const int32_t slot = slot_mapping[t];
    if (slot < 0) continue;
    const int page = slot >> 4, off = slot & 15;
    const int64_t src = ((int64_t)t * Hk + h) * HEAD_DIM + c8;
    const int64_t dst = (((int64_t)page * Hk + h) * PAGE_SIZE + off) * HEAD_DIM + c8;
    *reinterpret_cast<bf16x8*>(Kp + dst) = *reinterpret_cast<const bf16x8*>(K + src);
    *reinterpret_cast<bf16x8*>(Vp + dst) = *reinterpret_cast<const bf16x8*>(V + src);
  }
}

// gather_cache: inverse of reshape_and_cache — collect K/V rows for a run of
// slots into contiguous [T, Hk, 128] buffers (prefix-cached prefill: the
// suffix attends against full-length K/V rebuilt from pages).
__global__ void gather_cache_kernel(
    const __bf16* __restrict__ Kp,
    const __bf16* __restrict__ Vp,
    __bf16* __restrict__ K,
    __bf16* __restrict__ V,
    const int32_t* __restrict__ slot_mapping, // [T]
    int64_t T, int Hk) {
  const int64_t total = T * Hk * (HEAD_DIM / 8);
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int c8 = (int)(idx % (HEAD_DIM / 8)) * 8;
    const int64_t th = idx / (HEAD_DIM / 8);
    const int h = (int)(th % Hk);
    const int64_t t = th / Hk;
    const int32_t slot = slot_mapping[t];
    if (slot < 0) continue;
    const int page = slot >> 4, off = slot & 15;
    const int64_t dst = ((int64_t)t * Hk + h) * HEAD_DIM + c8;
    const int64_t src = (((int64_t)page * Hk + h) * PAGE_SIZE + off) * HEAD_DIM + c8;
    *reinterpret_cast<bf16x8*>(K + dst) = *reinterpret_cast<const bf16x8*>(Kp + src);
    *reinterpret_cast<bf16x8*>(V + dst) = *reinterpret_cast<const bf16x8*>(Vp + src);
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

static inline hipStream_t at_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor flash_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                            torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                            torch::Tensor tile_seq_len, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(q.size(2) == HEAD_DIM, "head_dim must be 128");
  const int Hq = (int)q.size(1), Hk = (int)k.size(1);
  TORCH_CHECK(Hq % Hk == 0);
  const int n_tiles = (int)tile_row0.size(0);
  auto o = torch::empty_like(q);
  dim3 grid(n_tiles, Hq);
  hipLaunchKernelGGL(flash_prefill_kernel, grid, dim3(256), 0, at_stream(),
                     (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(),
                     (const __bf16*)v.data_ptr(), (__bf16*)o.data_ptr(),
                     tile_seq_start.data_ptr<int32_t>(), tile_row0.data_ptr<int32_t>(),
                     tile_seq_len.data_ptr<int32_t>(), nullptr, Hq, Hk, (float)scale);
  HIP_CHECK_KERNEL();
  return o;
}

std::vector<torch::Tensor> flash_train_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                           torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                                           torch::Tensor tile_seq_len, double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(q.size(2) == HEAD_DIM, "head_dim must be 128");
  const int Hq = (int)q.size(1), Hk = (int)k.size(1);
  const int n_tiles = (int)tile_row0.size(0);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({q.size(0), (int64_t)Hq}, q.options().dtype(torch::kFloat32));
  dim3 grid(n_tiles, Hq);
  hipLaunchKernelGGL(flash_prefill_kernel, grid, dim3(256), 0, at_stream(),
                     (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(),
                     (const __bf16*)v.data_ptr(), (__bf16*)o.data_ptr(),
                     tile_seq_start.data_ptr<int32_t>(), tile_row0.data_ptr<int32_t>(),
                     tile_seq_len.data_ptr<int32_t>(), lse.data_ptr<float>(),
                     Hq, Hk, (float)scale);
  HIP_CHECK_KERNEL();
  return {o, lse};
}

std::vector<torch::Tensor> flash_train_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                           torch::Tensor o, torch::Tensor dout, torch::Tensor lse,
                                           torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                                           torch::Tensor tile_seq_len, double scale) {
  const int64_t T = q.size(0);
  const int Hq = (int)q.size(1), Hk = (int)k.size(1);
  const int n_tiles = (int)tile_row0.size(0);
  auto dout_c = dout.contiguous();
  auto Dsum = torch::empty({T, (int64_t)Hq}, q.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(attn_bwd_preprocess_kernel, dim3((unsigned)(T * Hq)), dim3(64), 0, at_stream(),
                     (const __bf16*)dout_c.data_ptr(), (const __bf16*)o.data_ptr(),
                     Dsum.data_ptr<float>(), T, Hq);
  HIP_CHECK_KERNEL();

  auto dq = torch::empty_like(q);
  auto dkh = torch::empty({T, (int64_t)Hq, (int64_t)HEAD_DIM}, q.options());
  auto dvh = torch::empty({T, (int64_t)Hq, (int64_t)HEAD_DIM}, q.options());
  dim3 grid(n_tiles, Hq);
  hipLaunchKernelGGL(flash_bwd_dq_kernel, grid, dim3(256), 0, at_stream(),
                     (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(),
                     (const __bf16*)v.data_ptr(), (const __bf16*)dout_c.data_ptr(),
                     lse.data_ptr<float>(), Dsum.data_ptr<float>(), (__bf16*)dq.data_ptr(),
                     tile_seq_start.data_ptr<int32_t>(), tile_row0.data_ptr<int32_t>(),
                     tile_seq_len.data_ptr<int32_t>(), Hq, Hk, (float)scale);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL(flash_bwd_dkv_kernel, grid, dim3(256), 0, at_stream(),
                     (const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(),
                     (const __bf16*)v.data_ptr(), (const __bf16*)dout_c.data_ptr(),
                     lse.data_ptr<float>(), Dsum.data_ptr<float>(),
                     (__bf16*)dkh.data_ptr(), (__bf16*)dvh.data_ptr(),
                     tile_seq_start.data_ptr<int32_t>(), tile_row0.data_ptr<int32_t>(),
                     tile_seq_len.data_ptr<int32_t>(), Hq, Hk, (float)scale);
  HIP_CHECK_KERNEL();
  // GQA: sum per-q-head partials over each group (fp32 accumulate)
  const int G = Hq / Hk;
  torch::Tensor dk, dv;
  if (G > 1) {
    dk = dkh.view({T, (int64_t)Hk, (int64_t)G, (int64_t)HEAD_DIM}).to(torch::kFloat32).sum(2).to(torch::kBFloat16);
    dv = dvh.view({T, (int64_t)Hk, (int64_t)G, (int64_t)HEAD_DIM}).to(torch::kFloat32).sum(2).to(torch::kBFloat16);
  } else {
    dk = dkh; dv = dvh;
  }
  return {dq, dk, dv};
}

torch::Tensor paged_decode(torch::Tensor q, torch::Tensor k_pages, torch::Tensor v_pages,
                           torch::Tensor block_tables, torch::Tensor seq_lens, double scale,
                           int64_t n_splits) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  const int B = (int)q.size(0);
  const int Hq = (int)q.size(1), Hk = (int)k_pages.size(1);
  const int G = Hq / Hk;
  const int max_pages = (int)block_tables.size(1);
  auto o = torch::empty_like(q);

  if (n_splits <= 0) {
    // measured (profiles/paged_sweep.log, B=256 Hk=2, S=256..8192): once
    // B*Hk blocks already cover the 256 CUs ~2x, splitting only adds the
    // fp32 workspace round-trip + merge kernel (-5us/call). Split ONLY to
    // fill the chip at small batch, capped at 16.
    int target = (2 * 256) / std::max(1, B * Hk);
    n_splits = std::min(16, std::max(1, target));
  }
  dim3 grid(B, Hk, (unsigned)n_splits);

  torch::Tensor ws_o, ws_ml;
  float *ws_o_ptr = nullptr, *ws_ml_ptr = nullptr;
  if (n_splits > 1) {
    ws_o = torch::empty({B, Hq, n_splits, HEAD_DIM}, q.options().dtype(torch::kFloat32));
    ws_ml = torch::empty({B, Hq, n_splits, 2}, q.options().dtype(torch::kFloat32));
    ws_o_ptr = ws_o.data_ptr<float>();
    ws_ml_ptr = ws_ml.data_ptr<float>();
  }

  if (G <= 16) {
    hipLaunchKernelGGL(paged_decode_mfma_kernel, grid, dim3(256), 0, at_stream(),
                       (const __bf16*)q.data_ptr(), (const __bf16*)k_pages.data_ptr(),
                       (const __bf16*)v_pages.data_ptr(), (__bf16*)o.data_ptr(),
                       ws_o_ptr, ws_ml_ptr,
                       block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
                       Hq, Hk, G, max_pages, (float)scale);
    HIP_CHECK_KERNEL();
    if (n_splits > 1) {
      hipLaunchKernelGGL(decode_merge_kernel, dim3(B * Hq), dim3(64), 0, at_stream(),
                         ws_o_ptr, ws_ml_ptr, (__bf16*)o.data_ptr(), Hq, (int)n_splits);
      HIP_CHECK_KERNEL();
    }
    return o;
  }
#define LAUNCH_G(GV) \
  hipLaunchKernelGGL(paged_decode_kernel<GV>, grid, dim3(256), 0, at_stream(), \
                     (const __bf16*)q.data_ptr(), (const __bf16*)k_pages.data_ptr(), \
                     (const __bf16*)v_pages.data_ptr(), (__bf16*)o.data_ptr(), \
                     ws_o_ptr, ws_ml_ptr, \
                     block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(), \
                     Hq, Hk, max_pages, (float)scale)
  switch (G) {
    case 1: LAUNCH_G(1); break;
    case 2: LAUNCH_G(2); break;
    case 3: LAUNCH_G(3); break;
    case 4: LAUNCH_G(4); break;
    case 5: LAUNCH_G(5); break;
    case 6: LAUNCH_G(6); break;
    case 7: LAUNCH_G(7); break;
    case 8: LAUNCH_G(8); break;
    default: TORCH_CHECK(false, "GQA group size must be 1..8, got ", G);
  }
#undef LAUNCH_G
  HIP_CHECK_KERNEL();

  if (n_splits > 1) {
    hipLaunchKernelGGL(decode_merge_kernel, dim3(B * Hq), dim3(64), 0, at_stream(),
                       ws_o_ptr, ws_ml_ptr, (__bf16*)o.data_ptr(), Hq, (int)n_splits);
    HIP_CHECK_KERNEL();
  }
  return o;
}

std::vector<torch::Tensor> gather_cache(torch::Tensor k_pages, torch::Tensor v_pages,
                                        torch::Tensor slot_mapping) {
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt32 && slot_mapping.is_cuda());
  const int64_t T = slot_mapping.size(0);
  const int Hk = (int)k_pages.size(1);
  auto k = torch::empty({T, Hk, HEAD_DIM}, k_pages.options());
  auto v = torch::empty({T, Hk, HEAD_DIM}, v_pages.options());
  const int64_t total = T * Hk * (HEAD_DIM / 8);
  hipLaunchKernelGGL(gather_cache_kernel, dim3(grid_for(total, 256)), dim3(256), 0, at_stream(),
                     (const __bf16*)k_pages.data_ptr(), (const __bf16*)v_pages.data_ptr(),
                     (__bf16*)k.data_ptr(), (__bf16*)v.data_ptr(),
                     slot_mapping.data_ptr<int32_t>(), T, Hk);
  HIP_CHECK_KERNEL();
  return {k, v};
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_pages,
                       torch::Tensor v_pages, torch::Tensor slot_mapping) {
  const int64_t T = k.size(0);
  const int Hk = (int)k.size(1);
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt32);
  const int64_t total = T * Hk * (HEAD_DIM / 8);
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3(grid_for(total, 256)), dim3(256), 0, at_stream(),
                     (const __bf16*)k.data_ptr(), (const __bf16*)v.data_ptr(),
                     (__bf16*)k_pages.data_ptr(), (__bf16*)v_pages.data_ptr(),
                     slot_mapping.data_ptr<int32_t>(), T, Hk);
  HIP_CHECK_KERNEL();
}

// ---------------------------------------------------------------------------
// Fused QKV postprocess (decode/prefill rollout path): from the fused QKV
// GEMM output, apply bias + rotary embedding and scatter K/V into the
// paged cache, emitting contiguous Q — replaces split+clone+rope+
// reshape_and_cache (4 kernels + 3 extra memory round-trips).
// ---------------------------------------------------------------------------
__global__ void qkv_rope_cache_kernel(
    const __bf16* __restrict__ qkv,     // [T, (Hq+2Hk)*D]
    const __bf16* __restrict__ bias,    // [(Hq+2Hk)*D] or null
    __bf16* __restrict__ q_out,         // [T, Hq, D]
    __bf16* __restrict__ Kp,            // pages [n_pages, Hk, 16, D]
    __bf16* __restrict__ Vp,
    const float* __restrict__ cos_tab,  // [P, D/2]
    const float* __restrict__ sin_tab,
    const int32_t* __restrict__ positions,    // [T]
    const int32_t* __restrict__ slot_mapping, // [T]
    int64_t T, int Hq, int Hk, int D) {
  const int half = D / 2;
  const int H_all = Hq + 2 * Hk;
  const int64_t total = T * (int64_t)H_all * half;
  const int row_w = H_all * D;

  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int i = (int)(idx % half);
    const int64_t th = idx / half;
    const int h = (int)(th % H_all);
    const int64_t t = th / H_all;

    const int col_lo = h * D + i;
    const int col_hi = col_lo + half;
    float a = (float)qkv[t * row_w + col_lo];
    float b = (float)qkv[t * row_w + col_hi];
    if (bias) {
      a += (float)bias[col_lo];
      b += (float)bias[col_hi];
    }

    if (h < Hq + Hk) {  // q and k heads get rope
      const int pos = positions[t];
      const float c = cos_tab[(int64_t)pos * half + i];
      const float s = sin_tab[(int64_t)pos * half + i];
      const float ra = a * c - b * s;
      const float rb = b * c + a * s;
      a = ra; b = rb;
    }

    if (h < Hq) {
      q_out[(t * Hq + h) * D + i] = (__bf16)a;
      q_out[(t * Hq + h) * D + i + half] = (__bf16)b;
    } else {
      const int32_t slot = slot_mapping[t];
      if (slot < 0) continue;
      const int page = slot >> 4, off = slot & 15;
      if (h < Hq + Hk) {  // k head
        const int kh = h - Hq;
        __bf16* dst = Kp + (((int64_t)page * Hk + kh) * PAGE_SIZE + off) * HEAD_DIM;
        dst[i] = (__bf16)a;
        dst[i + half] = (__bf16)b;
      } else {            // v head (no rope)
        const int vh = h - Hq - Hk;
        __bf16* dst = Vp + (((int64_t)page * Hk + vh) * PAGE_SIZE + off) * HEAD_DIM;
        dst[i] = (__bf16)a;
        dst[i + half] = (__bf16)b;
      }
    }
  }
}

torch::Tensor qkv_rope_cache(torch::Tensor qkv, c10::optional<torch::Tensor> bias,
                             torch::Tensor k_pages, torch::Tensor v_pages,
                             torch::Tensor cos_tab, torch::Tensor sin_tab,
                             torch::Tensor positions, torch::Tensor slot_mapping,
                             int64_t Hq, int64_t Hk) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16 && qkv.is_contiguous());
  const int64_t T = qkv.size(0);
  const int D = HEAD_DIM;
  TORCH_CHECK(qkv.size(1) == (Hq + 2 * Hk) * D);
  auto q_out = torch::empty({T, Hq, D}, qkv.options());
  const __bf16* bptr = bias.has_value() ? (const __bf16*)bias->data_ptr() : nullptr;
  const int64_t total = T * (Hq + 2 * Hk) * (D / 2);
  hipLaunchKernelGGL(qkv_rope_cache_kernel, dim3(grid_for(total, 256)), dim3(256), 0, at_stream(),
                     (const __bf16*)qkv.data_ptr(), bptr, (__bf16*)q_out.data_ptr(),
                     (__bf16*)k_pages.data_ptr(), (__bf16*)v_pages.data_ptr(),
                     cos_tab.data_ptr<float>(), sin_tab.data_ptr<float>(),
                     positions.data_ptr<int32_t>(), slot_mapping.data_ptr<int32_t>(),
                     T, (int)Hq, (int)Hk, D);
  HIP_CHECK_KERNEL();
  return q_out;
}

// ===========================================================================
// Training flash attention (fwd + bwd) — the update-path attention
// (replaces the chunked eager softmax composition; K5/K12).
//
// Forward reuses the prefill structure but also emits per-row LSE
// (m + log l, fp32) for the backward recompute. Backward is FA2-style:
//   D[t,h]  = rowsum(dO * O)
//   dQ pass: per q-tile, iterate kv<=q tiles: S=QK^T, P=exp(S-lse),
//            dP=dO V^T, dS=P*(dP-D), dQ += dS K * scale
//   dKV pass: per kv-tile, iterate q>=kv tiles: S^T=K Q^T,
//            P^T=exp(S^T-lse_col), dV += P^T dO,
//            dS^T = P^T*(dP^T - D_col), dK += dS^T Q * scale
// Both passes tile exactly like the forward (MFMA 16x16x32, XOR-swizzled
// LDS); no atomics — each output tile has one writer.
// ===========================================================================

// rowsum(dO * O) -> D [T, Hq] fp32
__global__ void attn_bwd_preprocess_kernel(
    const __bf16* __restrict__ dO, const __bf16* __restrict__ O,
    float* __restrict__ D, int64_t T, int Hq) {
  const int64_t row = blockIdx.x;  // t * Hq + h
  if (row >= T * Hq) return;
  const __bf16* d = dO + row * HEAD_DIM;
  const __bf16* o = O + row * HEAD_DIM;
  float acc = 0.f;
  for (int i = threadIdx.x * 2; i < HEAD_DIM; i += blockDim.x * 2) {
    acc += (float)d[i] * (float)o[i] + (float)d[i + 1] * (float)o[i + 1];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
  if (threadIdx.x == 0) D[row] = acc;
}

// dQ: one block per (q-tile, q-head); 4 waves x 16 q rows.
__global__ __launch_bounds__(256, 2) void flash_bwd_dq_kernel(
    const __bf16* __restrict__ Q, const __bf16* __restrict__ K, const __bf16* __restrict__ V,
    const __bf16* __restrict__ dO, const float* __restrict__ lse, const float* __restrict__ Dsum,
    __bf16* __restrict__ dQ,
    const int32_t* __restrict__ tile_seq_start, const int32_t* __restrict__ tile_row0,
    const int32_t* __restrict__ tile_seq_len, int Hq, int Hk, float scale) {
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hk);
  const int seq_start = tile_seq_start[tile];
  const int q_local0 = tile_row0[tile] - seq_start;
  const int seq_len = tile_seq_len[tile];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wrow0 = q_local0 + wid * 16;

  // LDS: K tile (linear-swz, for B of QK^T), KT (transposed, for B of dS*K),
  //      V tile (linear-swz, for B of dO*V^T), per-wave dS staging
  __shared__ char lds[KVTILE * HEAD_DIM * 2 + HEAD_DIM * KVTILE * 2 + KVTILE * HEAD_DIM * 2 + 4 * 16 * KVTILE * 2];
  char* k_lds = lds;
  char* kt_lds = lds + KVTILE * HEAD_DIM * 2;
  char* v_lds = kt_lds + HEAD_DIM * KVTILE * 2;
  char* ds_lds = v_lds + KVTILE * HEAD_DIM * 2;

  const int a_row = wrow0 + (lane & 15);
  const int a_k0 = (lane >> 4) * 8;
  const bool row_valid = a_row < seq_len;

  bf16x8 q_frag[4], do_frag[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
    if (row_valid) {
      q_frag[ds] = *reinterpret_cast<const bf16x8*>(
          Q + ((int64_t)(seq_start + a_row) * Hq + qh) * HEAD_DIM + ds * 32 + a_k0);
      do_frag[ds] = *reinterpret_cast<const bf16x8*>(
          dO + ((int64_t)(seq_start + a_row) * Hq + qh) * HEAD_DIM + ds * 32 + a_k0);
    } else {
      q_frag[ds] = bf16x8{};
      do_frag[ds] = bf16x8{};
    }
  }
  // per-lane row stats for its 4 C-layout rows
  const int my_row0 = wrow0 + (lane >> 4) * 4;
  float lse_r[4], d_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = my_row0 + r;
    if (qr < seq_len) {
      lse_r[r] = lse[(int64_t)(seq_start + qr) * Hq + qh];
      d_r[r] = Dsum[(int64_t)(seq_start + qr) * Hq + qh];
    } else { lse_r[r] = 0.f; d_r[r] = 0.f; }
  }

  f32x4 dq_acc[8];
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) dq_acc[dt] = f32x4{};

  const int q_hi = min(q_local0 + QTILE, seq_len);
  for (int kv0 = 0; kv0 < q_hi; kv0 += KVTILE) {
    const int kv_n = min(KVTILE, seq_len - kv0);
    __syncthreads();
    for (int idx = threadIdx.x; idx < KVTILE * (HEAD_DIM / 8); idx += 256) {
      const int r = idx / (HEAD_DIM / 8);
      const int c8 = (idx % (HEAD_DIM / 8)) * 8;
      bf16x8 kv = (r < kv_n) ? *reinterpret_cast<const bf16x8*>(
          K + ((int64_t)(seq_start + kv0 + r) * Hk + kvh) * HEAD_DIM + c8) : bf16x8{};
      *reinterpret_cast<bf16x8*>(k_lds + swz(r, c8 * 2)) = kv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<__bf16*>(kt_lds + swz_p(c8 + j, r * 2)) = kv[j];
      bf16x8 vv = (r < kv_n) ? *reinterpret_cast<const bf16x8*>(
          V + ((int64_t)(seq_start + kv0 + r) * Hk + kvh) * HEAD_DIM + c8) : bf16x8{};
      *reinterpret_cast<bf16x8*>(v_lds + swz(r, c8 * 2)) = vv;
    }
    __syncthreads();

    // S = QK^T, dP = dO V^T per n-tile
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      f32x4 s_acc = f32x4{};
      f32x4 dp_acc = f32x4{};
      const int b_col = nt * 16 + (lane & 15);
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(k_lds + swz(b_col, (ds * 32 + a_k0) * 2));
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(v_lds + swz(b_col, (ds * 32 + a_k0) * 2));
        s_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ds], kb, s_acc, 0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(do_frag[ds], vb, dp_acc, 0, 0, 0);
      }
      // dS = P * (dP - D); write to per-wave LDS in A-frag layout
      char* dsw = ds_lds + wid * 16 * KVTILE * 2;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qr = my_row0 + r;
        const int col = kv0 + nt * 16 + (lane & 15);
        float dsv = 0.f;
        if (qr < seq_len && col <= qr && col < seq_len) {
          const float p = __expf(s_acc[r] * scale - lse_r[r]);
          dsv = p * (dp_acc[r] - d_r[r]);
        }
        const int prow = (lane >> 4) * 4 + r;
        *reinterpret_cast<__bf16*>(dsw + swz_p(prow, (nt * 16 + (lane & 15)) * 2)) = (__bf16)dsv;
      }
    }
    __syncthreads();  // actually per-wave, but keep LDS state coherent with staging loop

    // dQ += dS * K  (A = dS [16, 64kv], B = KT [kv, d])
    char* dsw = ds_lds + wid * 16 * KVTILE * 2;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(dsw + swz_p(lane & 15, (ks * 32 + a_k0) * 2));
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(kt_lds + swz_p(dt * 16 + (lane & 15), (ks * 32 + a_k0) * 2));
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, dq_acc[dt], 0, 0, 0);
      }
    }
  }

  // store dQ (scaled)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = my_row0 + r;
    if (qr >= seq_len || qr < q_local0) continue;
    __bf16* out = dQ + ((int64_t)(seq_start + qr) * Hq + qh) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
      out[dt * 16 + (lane & 15)] = (__bf16)(dq_acc[dt][r] * scale);
  }
}

// dK/dV: one block per (kv-tile, q-head); each wave owns 16 kv rows and
// accumulates over all q tiles >= its kv tile. Writes PER-Q-HEAD partials
// dKh/dVh [T, Hq, D]; the wrapper sums each GQA group (no atomics).
__global__ __launch_bounds__(256, 2) void flash_bwd_dkv_kernel(
    const __bf16* __restrict__ Q, const __bf16* __restrict__ K, const __bf16* __restrict__ V,
    const __bf16* __restrict__ dO, const float* __restrict__ lse, const float* __restrict__ Dsum,
    __bf16* __restrict__ dKh,   // [T, Hq, D]
    __bf16* __restrict__ dVh,   // [T, Hq, D]
    const int32_t* __restrict__ tile_seq_start, const int32_t* __restrict__ tile_row0,
    const int32_t* __restrict__ tile_seq_len, int Hq, int Hk, float scale) {
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hk);
  const int seq_start = tile_seq_start[tile];
  const int kv_local0 = tile_row0[tile] - seq_start;
  const int seq_len = tile_seq_len[tile];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wrow0 = kv_local0 + wid * 16;        // this wave's kv rows

  // LDS: QT 16K | Q 16K | dOT 16K | dO 16K | P^T 8K | dS^T 8K = 80 KiB
  __shared__ char lds[4 * KVTILE * HEAD_DIM * 2 + 2 * 4 * 16 * KVTILE * 2];
  char* qt_lds = lds;
  char* q_lds = qt_lds + HEAD_DIM * KVTILE * 2;
  char* dot_lds = q_lds + KVTILE * HEAD_DIM * 2;
  char* do_lds = dot_lds + HEAD_DIM * KVTILE * 2;
  char* pt_lds = do_lds + KVTILE * HEAD_DIM * 2;
  char* dst_lds = pt_lds + 4 * 16 * KVTILE * 2;

  const int a_row = wrow0 + (lane & 15);          // kv row for A frags
  const int a_k0 = (lane >> 4) * 8;
  const bool row_valid = a_row < seq_len;

  bf16x8 k_frag[4], v_frag[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
    if (row_valid) {
      k_frag[ds] = *reinterpret_cast<const bf16x8*>(
          K + ((int64_t)(seq_start + a_row) * Hk + kvh) * HEAD_DIM + ds * 32 + a_k0);
      v_frag[ds] = *reinterpret_cast<const bf16x8*>(
          V + ((int64_t)(seq_start + a_row) * Hk + kvh) * HEAD_DIM + ds * 32 + a_k0);
    } else { k_frag[ds] = bf16x8{}; v_frag[ds] = bf16x8{}; }
  }

  f32x4 dk_acc[8], dv_acc[8];
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) { dk_acc[dt] = f32x4{}; dv_acc[dt] = f32x4{}; }

  const int my_kv0 = wrow0 + (lane >> 4) * 4;

  // causal: only q tiles that can see this kv tile
  const int q_start_tile = (kv_local0 / KVTILE) * KVTILE;
  for (int q0 = q_start_tile; q0 < seq_len; q0 += KVTILE) {
    const int q_n = min(KVTILE, seq_len - q0);
    __syncthreads();
    for (int idx = threadIdx.x; idx < KVTILE * (HEAD_DIM / 8); idx += 256) {
      const int r = idx / (HEAD_DIM / 8);
      const int c8 = (idx % (HEAD_DIM / 8)) * 8;
      bf16x8 qv = (r < q_n) ? *reinterpret_cast<const bf16x8*>(
          Q + ((int64_t)(seq_start + q0 + r) * Hq + qh) * HEAD_DIM + c8) : bf16x8{};
      *reinterpret_cast<bf16x8*>(q_lds + swz(r, c8 * 2)) = qv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<__bf16*>(qt_lds + swz_p(c8 + j, r * 2)) = qv[j];
      bf16x8 dv = (r < q_n) ? *reinterpret_cast<const bf16x8*>(
          dO + ((int64_t)(seq_start + q0 + r) * Hq + qh) * HEAD_DIM + c8) : bf16x8{};
      *reinterpret_cast<bf16x8*>(do_lds + swz(r, c8 * 2)) = dv;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<__bf16*>(dot_lds + swz_p(c8 + j, r * 2)) = dv[j];
    }
    __syncthreads();

    // per n-tile of q cols: S^T = K Q^T, dP^T = V dO^T, then stage
    // P^T and dS^T in per-wave LDS (A-frag layout for the next MFMAs)
    char* ptw = pt_lds + wid * 16 * KVTILE * 2;
    char* dstw = dst_lds + wid * 16 * KVTILE * 2;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      f32x4 st_acc = f32x4{};
      f32x4 dpt_acc = f32x4{};
      const int b_col = nt * 16 + (lane & 15);  // q col within tile
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(q_lds + swz(b_col, (ds * 32 + a_k0) * 2));
        bf16x8 dob = *reinterpret_cast<const bf16x8*>(do_lds + swz(b_col, (ds * 32 + a_k0) * 2));
        st_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[ds], qb, st_acc, 0, 0, 0);
        dpt_acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[ds], dob, dpt_acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kvr = my_kv0 + r;
        const int qc = q0 + nt * 16 + (lane & 15);
        float pt = 0.f, dst = 0.f;
        if (kvr < seq_len && qc < seq_len && qc >= kvr) {
          const float l = lse[(int64_t)(seq_start + qc) * Hq + qh];
          const float dd = Dsum[(int64_t)(seq_start + qc) * Hq + qh];
          pt = __expf(st_acc[r] * scale - l);
          dst = pt * (dpt_acc[r] - dd);
        }
        const int prow = (lane >> 4) * 4 + r;
        *reinterpret_cast<__bf16*>(ptw + swz_p(prow, (nt * 16 + (lane & 15)) * 2)) = (__bf16)pt;
        *reinterpret_cast<__bf16*>(dstw + swz_p(prow, (nt * 16 + (lane & 15)) * 2)) = (__bf16)dst;
      }
    }

    // dV += P^T dO ; dK += dS^T Q   (A staged per-wave; B from transposed tiles)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(ptw + swz_p(lane & 15, (ks * 32 + a_k0) * 2));
      bf16x8 da = *reinterpret_cast<const bf16x8*>(dstw + swz_p(lane & 15, (ks * 32 + a_k0) * 2));
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        bf16x8 dob = *reinterpret_cast<const bf16x8*>(dot_lds + swz_p(dt * 16 + (lane & 15), (ks * 32 + a_k0) * 2));
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(qt_lds + swz_p(dt * 16 + (lane & 15), (ks * 32 + a_k0) * 2));
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, dob, dv_acc[dt], 0, 0, 0);
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, qb, dk_acc[dt], 0, 0, 0);
      }
    }
  }

  // store per-q-head partials (C layout rows = kv rows)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvr = my_kv0 + r;
    if (kvr >= seq_len || kvr < kv_local0) continue;
    __bf16* dko = dKh + ((int64_t)(seq_start + kvr) * Hq + qh) * HEAD_DIM;
    __bf16* dvo = dVh + ((int64_t)(seq_start + kvr) * Hq + qh) * HEAD_DIM;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      dko[dt * 16 + (lane & 15)] = (__bf16)(dk_acc[dt][r] * scale);
      dvo[dt * 16 + (lane & 15)] = (__bf16)dv_acc[dt][r];
    }
  }
}

// ---------------------------------------------------------------------------
// Paged decode v2 (MFMA): the VALU version spends ~200 VALU ops per KV token
// on GQA dot products; here the GQA group (G<=16 q heads, padded to the
// MFMA M-tile) attends to page PAIRS (2x16 tokens = one 16-col MFMA B tile
// each) — QK^T 8 MFMAs + PV 8 MFMAs per 32 tokens, K read direct from
// pages (one 16 B load per lane), V staged transposed per wave.
// Grid (B, Hk, splits); block = 4 waves, each wave owns a token quarter;
// two-level merge (wave LDS + split workspace) as in v1.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void paged_decode_mfma_kernel(
    const __bf16* __restrict__ Q,
    const __bf16* __restrict__ Kp,
    const __bf16* __restrict__ Vp,
    __bf16* __restrict__ O,
    float* __restrict__ ws_o,
    float* __restrict__ ws_ml,
    const int32_t* __restrict__ block_tables,
    const int32_t* __restrict__ seq_lens,
    int Hq, int Hk, int G, int max_pages, float scale) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int n_splits = gridDim.z;
  const int seq_len = seq_lens[b];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  // token range for this (split, wave): page-pair (32-token) granularity
  const int pairs_total = (seq_len + 31) / 32;
  const int pairs_per_split = (pairs_total + n_splits - 1) / n_splits;
  const int pair_lo_split = split * pairs_per_split;
  const int pair_hi_split = min(pairs_total, pair_lo_split + pairs_per_split);
  const int pairs_per_wave = (pair_hi_split - pair_lo_split + 3) / 4;
  const int pair_lo = pair_lo_split + wid * pairs_per_wave;
  const int pair_hi = min(pair_hi_split, pair_lo + pairs_per_wave);

  const int32_t* bt = block_tables + (int64_t)b * max_pages;

  // per-wave LDS: VT [128][32] bf16 (8 KiB) + P [16][32] bf16 (1 KiB)
  __shared__ char vt_lds_all[4][128 * 32 * 2];
  __shared__ char p_lds_all[4][16 * 32 * 2];
  __shared__ float merge_o[4][16][HEAD_DIM];   // per-wave o partials (16 KiB... fp32 = 32 KiB)
  __shared__ float merge_ml[4][16][2];
  char* vt_lds = vt_lds_all[wid];
  char* p_lds = p_lds_all[wid];

  const int a_k0 = (lane >> 4) * 8;
  const int col = lane & 15;

  // Q fragments: A[16 rows = q heads (padded), 32 dims per slice]
  bf16x8 q_frag[4];
  const int qh = kvh * G + col;  // row 'col' of the M tile
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) {
    if (col < G) {
      q_frag[ds] = *reinterpret_cast<const bf16x8*>(
          Q + ((int64_t)b * Hq + qh) * HEAD_DIM + ds * 32 + a_k0);
    } else {
      q_frag[ds] = bf16x8{};
    }
  }

  float m_st[4], l_st[4];
  f32x4 o_acc[8];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_st[r] = -INFINITY; l_st[r] = 0.f; }
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) o_acc[dt] = f32x4{};

  for (int pp = pair_lo; pp < pair_hi; ++pp) {
    const int t0 = pp * 32;
    // ---- QK^T over the two pages ----
    f32x4 s_acc[2];
#pragma unroll
    for (int pg = 0; pg < 2; ++pg) {
      s_acc[pg] = f32x4{};
      const int tok = t0 + pg * 16 + col;       // this lane's B column
      const int tok_c = min(tok, seq_len - 1);
      const int page = bt[tok_c >> 4];
      const __bf16* krow = Kp + (((int64_t)page * Hk + kvh) * PAGE_SIZE + (tok_c & 15)) * HEAD_DIM;
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(krow + ds * 32 + a_k0);
        s_acc[pg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ds], kb, s_acc[pg], 0, 0, 0);
      }
    }
    // ---- stage V^T for the pair (each lane: 2 tokens x 8 dims -> 16 scalar writes) ----
    {
      const int tok_base = t0 + (lane >> 4) * 4;  // 4 waves of 16 lanes... use 64 lanes = 32 tok x 2 halves
      // simpler mapping: lane covers token (lane & 31), dim half (lane >> 5)
      const int tok = t0 + (lane & 31);
      const int half = (lane >> 5) * 64;          // dims 0..63 or 64..127
      const int tok_c = min(tok, seq_len - 1);
      const int page = bt[tok_c >> 4];
      const __bf16* vrow = Vp + (((int64_t)page * Hk + kvh) * PAGE_SIZE + (tok_c & 15)) * HEAD_DIM + half;
#pragma unroll
      for (int c8 = 0; c8 < 64; c8 += 8) {
        bf16x8 vv = *reinterpret_cast<const bf16x8*>(vrow + c8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          // VT[dim][tok], row stride 32 tokens * 2B = 64B, swizzled
          const int d = half + c8 + j;
          const int byte = (d * 32 + (tok - t0)) * 2;
          *reinterpret_cast<__bf16*>(vt_lds + (byte ^ ((d & 7) << 4))) = vv[j];
        }
      }
    }
    // ---- softmax (online) ----
    const int my_r0 = (lane >> 4) * 4;
    float p_val[2][4];
    float row_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) row_max[r] = -INFINITY;
#pragma unroll
    for (int pg = 0; pg < 2; ++pg) {
      const int tok = t0 + pg * 16 + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = s_acc[pg][r] * scale;
        if (tok >= seq_len) s = -INFINITY;
        p_val[pg][r] = s;
        row_max[r] = fmaxf(row_max[r], s);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        row_max[r] = fmaxf(row_max[r], __shfl_xor(row_max[r], off, 64));
      const float m_new = fmaxf(m_st[r], row_max[r]);
      const float rs = (m_st[r] == -INFINITY) ? 0.f : __expf(m_st[r] - m_new);
      float row_sum = 0.f;
#pragma unroll
      for (int pg = 0; pg < 2; ++pg) {
        float e = (p_val[pg][r] == -INFINITY) ? 0.f : __expf(p_val[pg][r] - m_new);
        p_val[pg][r] = e;
        row_sum += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) row_sum += __shfl_xor(row_sum, off, 64);
      l_st[r] = l_st[r] * rs + row_sum;
      m_st[r] = m_new;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) o_acc[dt][r] *= rs;
    }
    // ---- write P to LDS in A-frag layout [q=16][tok=32] ----
#pragma unroll
    for (int pg = 0; pg < 2; ++pg) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int prow = my_r0 + r;
        const int pcol = pg * 16 + col;
        const int byte = (prow * 32 + pcol) * 2;
        *reinterpret_cast<__bf16*>(p_lds + (byte ^ ((prow & 7) << 4))) = (__bf16)p_val[pg][r];
      }
    }
    // ---- PV: A = P [16, 32], B = VT (dim tiles) ----
    bf16x8 a_frag;
    {
      const int prow = col;
      const int byte = (prow * 32 + a_k0) * 2;
      a_frag = *reinterpret_cast<const bf16x8*>(p_lds + (byte ^ ((prow & 7) << 4)));
    }
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      const int d = dt * 16 + col;
      const int byte = (d * 32 + a_k0) * 2;
      bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(vt_lds + (byte ^ ((d & 7) << 4)));
      o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, o_acc[dt], 0, 0, 0);
    }
  }

  // ---- merge 4 waves via LDS (rows = q heads 0..15 in C layout) ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = (lane >> 4) * 4 + r;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      merge_o[wid][qrow][dt * 16 + col] = o_acc[dt][r];
    }
    if (col == 0) {
      merge_ml[wid][qrow][0] = m_st[r];
      merge_ml[wid][qrow][1] = l_st[r];
    }
  }
  __syncthreads();
  if (wid == 0) {
    // 64 lanes: lane covers (q row = lane>>2 & 15? ) — use 16 rows x 4 dim-blocks
    const int qrow = lane >> 2;          // 0..15
    const int dblk = (lane & 3) * 32;    // 4 x 32 dims
    if (qrow < 16) {
      float m_tot = -INFINITY;
#pragma unroll
      for (int w = 0; w < 4; ++w) m_tot = fmaxf(m_tot, merge_ml[w][qrow][0]);
      float l_tot = 0.f;
      float o_tot[32];
#pragma unroll
      for (int j = 0; j < 32; ++j) o_tot[j] = 0.f;
#pragma unroll
      for (int w = 0; w < 4; ++w) {
        const float mw = merge_ml[w][qrow][0];
        const float rr = (mw == -INFINITY) ? 0.f : __expf(mw - m_tot);
        l_tot += merge_ml[w][qrow][1] * rr;
#pragma unroll
        for (int j = 0; j < 32; ++j) o_tot[j] += merge_o[w][qrow][dblk + j] * rr;
      }
      if (qrow < G) {
        const int qh_out = kvh * G + qrow;
        if (n_splits == 1) {
          const float inv_l = (l_tot > 0.f) ? 1.f / l_tot : 0.f;
          __bf16* orow = O + ((int64_t)b * Hq + qh_out) * HEAD_DIM + dblk;
#pragma unroll
          for (int j = 0; j < 32; ++j) orow[j] = (__bf16)(o_tot[j] * inv_l);
        } else {
          float* wrow = ws_o + (((int64_t)b * Hq + qh_out) * n_splits + split) * HEAD_DIM + dblk;
#pragma unroll
          for (int j = 0; j < 32; ++j) wrow[j] = o_tot[j];
          if (dblk == 0) {
            float* mlrow = ws_ml + (((int64_t)b * Hq + qh_out) * n_splits + split) * 2;
            mlrow[0] = m_tot;
            mlrow[1] = l_tot;
          }
        }
      }
    }
  }
}
