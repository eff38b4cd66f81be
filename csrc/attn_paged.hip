// Unified paged attention for gfx950 — serves BOTH the decode step (one
// query token per sequence) and prefill (every prompt row attends over the
// cache slots the fused RoPE+append kernel just wrote, so no separate
// varlen kernel is needed; GEMMs dominate prefill and the K/V re-reads hit
// the 256 MiB Infinity Cache).
//
// Geometry (memory-bound design, wave64-native):
//   grid = (num_query_rows, num_kv_heads); block = 256 threads = 4 waves.
//   A workgroup owns one (row, kv_head): it processes the whole GQA group
//   (GROUP query heads) so K/V stream from HBM exactly once per group.
//   Waves split the KV blocks round-robin (flash-decode style) and keep
//   per-head online-softmax state (m, l, acc); a final LDS combine merges
//   the four waves.
//   K tile read: 4 lanes per token x 32B-slices -> 1 KiB coalesced per
//   16-token page; V read: lane owns a D/64-element slice of every row.
//
// BLOCK_SIZE is fixed at 16 tokens (one KV page = 16 x 128 x 2B = 4 KiB per
// head — the pool unit sized for 288 GB HBM3E).
#include <float.h>

#include "common.h"

#define ATTN_BS 16   // KV page size in tokens

template <int GROUP, int D>
__global__ __launch_bounds__(256) void paged_attn_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ q,
    const bf16* __restrict__ k_cache, const bf16* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ ctx_lens,
    const int* __restrict__ row_seq, int max_blocks, int num_kv_heads,
    float scale, long q_stride) {
  constexpr int SLICE = D / 4;        // K elems per lane (4 lanes/token)
  constexpr int VPL = D / 64;         // V elems per lane (lane owns a slice)
  const int row = blockIdx.x;
  const int kvh = blockIdx.y;
  const int H = num_kv_heads * GROUP;
  const int ctx = ctx_lens[row];
  const int table = row_seq ? row_seq[row] : row;
  const int* bt = block_tables + (long)table * max_blocks;
  const int nblocks = (ctx + ATTN_BS - 1) / ATTN_BS;

  __shared__ float q_lds[GROUP][D];
  __shared__ float sc[4][GROUP][ATTN_BS];
  __shared__ float comb_o[4][GROUP][D];
  __shared__ float comb_m[4][GROUP];
  __shared__ float comb_l[4][GROUP];

  // stage the query group, pre-scaled
  for (int i = threadIdx.x; i < GROUP * D; i += 256) {
    const int h = i / D, d = i % D;
    q_lds[h][d] = bf2f(*(const unsigned short*)(
        q + (long)row * q_stride + (kvh * GROUP + h) * D + d)) * scale;
  }
  __syncthreads();

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tok = lane >> 2;          // 0..15
  const int sl = lane & 3;            // K slice index
  const int d0 = sl * SLICE;

  float m[GROUP], l[GROUP], acc[GROUP][VPL];
  #pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    m[h] = -FLT_MAX; l[h] = 0.f;
    #pragma unroll
    for (int i = 0; i < VPL; ++i) acc[h][i] = 0.f;
  }

  for (int b = wid; b < nblocks; b += 4) {
    const long blk = bt[b];
    const int nb = min(ATTN_BS, ctx - b * ATTN_BS);
    // ---- K dot: lane covers SLICE elems of its token's key ----
    float kf[SLICE];
    {
      const bf16* kp = k_cache +
          (((blk * num_kv_heads + kvh) * ATTN_BS + tok) * D + d0);
      #pragma unroll
      for (int i = 0; i < SLICE / 8; ++i) {
        bf16x8 kv8 = load_bf16x8(kp + i * 8);
        #pragma unroll
        for (int j = 0; j < 8; ++j) kf[i * 8 + j] = bf16x8_get(kv8, j);
      }
    }
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float p = 0.f;
      #pragma unroll
      for (int i = 0; i < SLICE; ++i) p += q_lds[h][d0 + i] * kf[i];
      p += __shfl_xor(p, 1);
      p += __shfl_xor(p, 2);
      if (sl == 0)
        sc[wid][h][tok] = (tok < nb) ? p : -FLT_MAX;
    }
    // wave-private LDS area; DS ops of one wave are in program order,
    // so no barrier is needed before re-reading sc.
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float bm = -FLT_MAX;
      #pragma unroll
      for (int t = 0; t < ATTN_BS; ++t) bm = fmaxf(bm, sc[wid][h][t]);
      const float mn = fmaxf(m[h], bm);
      const float alpha = (m[h] == -FLT_MAX) ? 0.f : __expf(m[h] - mn);
      #pragma unroll
      for (int i = 0; i < VPL; ++i) acc[h][i] *= alpha;
      l[h] *= alpha;
      m[h] = mn;
      if (sl == 0)  // one lane per token exponentiates it
        sc[wid][h][tok] = (tok < nb) ? __expf(sc[wid][h][tok] - mn) : 0.f;
    }
    // ---- V accumulate: lane owns elements [lane*VPL, lane*VPL+VPL) ----
    const bf16* vbase = v_cache +
        ((blk * num_kv_heads + kvh) * ATTN_BS) * D + lane * VPL;
    for (int t = 0; t < nb; ++t) {
      float vf[VPL];
      if constexpr (VPL == 2) {
        ushort2 vv = *reinterpret_cast<const ushort2*>(vbase + (long)t * D);
        vf[0] = bf2f(vv.x); vf[1] = bf2f(vv.y);
      } else {
        vf[0] = bf2f(*(const unsigned short*)(vbase + (long)t * D));
      }
      #pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        const float p = sc[wid][h][t];
        #pragma unroll
        for (int i = 0; i < VPL; ++i) acc[h][i] += p * vf[i];
      }
    }
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float s = 0.f;
      #pragma unroll
      for (int t = 0; t < ATTN_BS; ++t) s += sc[wid][h][t];
      l[h] += s;
    }
  }

  // ---- combine the 4 waves ----
  #pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    #pragma unroll
    for (int i = 0; i < VPL; ++i) comb_o[wid][h][lane * VPL + i] = acc[h][i];
    if (lane == 0) { comb_m[wid][h] = m[h]; comb_l[wid][h] = l[h]; }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < GROUP * D; i += 256) {
    const int h = i / D, d = i % D;
    float M = fmaxf(fmaxf(comb_m[0][h], comb_m[1][h]),
                    fmaxf(comb_m[2][h], comb_m[3][h]));
    float o = 0.f, L = 0.f;
    #pragma unroll
    for (int w = 0; w < 4; ++w) {
      const float f = (comb_m[w][h] == -FLT_MAX) ? 0.f
                                                 : __expf(comb_m[w][h] - M);
      o += f * comb_o[w][h][d];
      L += f * comb_l[w][h];
    }
    *(unsigned short*)(out + ((long)row * H + kvh * GROUP + h) * D + d) =
        f2bf(o / L);
  }
}

void launch_paged_attn(bf16* out, const bf16* q, const bf16* k_cache,
                       const bf16* v_cache, const int* block_tables,
                       const int* ctx_lens, const int* row_seq,
                       long num_rows, int num_kv_heads, int group, int D,
                       int max_blocks, int block_size, float scale,
                       long q_stride, hipStream_t stream) {
  if (block_size != ATTN_BS)
    throw std::runtime_error("paged_attn: block_size must be 16");
  dim3 grid((unsigned)num_rows, (unsigned)num_kv_heads);
  #define CASE(G, DD)                                                       \
    if (group == G && D == DD) {                                            \
      paged_attn_kernel<G, DD><<<grid, 256, 0, stream>>>(                   \
          out, q, k_cache, v_cache, block_tables, ctx_lens, row_seq,        \
          max_blocks, num_kv_heads, scale, q_stride);                       \
      return;                                                               \
    }
  CASE(1, 128) CASE(2, 128) CASE(4, 128) CASE(8, 128)
  CASE(1, 64) CASE(2, 64) CASE(4, 64) CASE(8, 64)
  #undef CASE
  throw std::runtime_error("paged_attn: unsupported (group, head_dim)");
}
