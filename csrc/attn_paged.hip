// Unified paged attention for gfx950 — serves the decode step (one query
// token per sequence) and short/odd-shape prefill fallback (the MFMA flash
// kernel covers head_dim-128 prefill).
//
// Geometry (memory-bound design, wave64-native):
//   grid = (num_query_rows, num_kv_heads); block = 256 threads = 4 waves.
//   A workgroup owns one (row, kv_head): it processes the whole GQA group
//   (GROUP query heads) so K/V stream from HBM exactly once per group.
//   Waves split the KV pages round-robin (flash-decode style) with
//   per-head online-softmax state; a final LDS combine merges the waves.
//
//   K page read: 4 lanes per token x 64B slices -> 4 KiB coalesced/page.
//   V page read: lane owns a D/64-element column slice of all 16 rows.
//   v2 (profiles/r01: v1 was LDS-issue-bound at 155us/layer, 14x off
//   roofline): the lane's q slice lives in REGISTERS as raw bf16 pairs
//   (GROUP<=4) or is read as float4 from LDS (GROUP=8); score/softmax
//   LDS traffic is float4; V rows are preloaded per page.
//
// BLOCK_SIZE is fixed at 16 tokens (one KV page = 16 x 128 x 2B = 4 KiB
// per head — the pool unit sized for 288 GB HBM3E).
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
  constexpr bool QREG = (GROUP <= 4); // q slice in registers vs LDS
  const int row = blockIdx.x;
  const int kvh = blockIdx.y;
  const int H = num_kv_heads * GROUP;
  const int ctx = ctx_lens[row];
  const int table = row_seq ? row_seq[row] : row;
  const int* bt = block_tables + (long)table * max_blocks;
  const int nblocks = (ctx + ATTN_BS - 1) / ATTN_BS;

  __shared__ __attribute__((aligned(16))) float q_lds[GROUP][D];
  __shared__ __attribute__((aligned(16))) float sc[4][GROUP][ATTN_BS];
  __shared__ __attribute__((aligned(16))) float comb_o[4][GROUP][D];
  __shared__ float comb_m[4][GROUP];
  __shared__ float comb_l[4][GROUP];

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tok = lane >> 2;          // 0..15
  const int sl = lane & 3;            // K slice index
  const int d0 = sl * SLICE;

  // ---- stage the query group ----
  unsigned q_pk[QREG ? GROUP : 1][QREG ? SLICE / 2 : 1];  // raw bf16 pairs
  if constexpr (QREG) {
    const bf16* qp = q + (long)row * q_stride + (long)kvh * GROUP * D + d0;
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      #pragma unroll
      for (int i = 0; i < SLICE / 8; ++i) {
        uint4 r4 = *reinterpret_cast<const uint4*>(qp + h * D + i * 8);
        q_pk[h][i * 4 + 0] = r4.x;
        q_pk[h][i * 4 + 1] = r4.y;
        q_pk[h][i * 4 + 2] = r4.z;
        q_pk[h][i * 4 + 3] = r4.w;
      }
    }
  } else {
    for (int i = threadIdx.x; i < GROUP * D; i += 256) {
      const int h = i / D, d = i % D;
      q_lds[h][d] = bf2f(*(const unsigned short*)(
          q + (long)row * q_stride + (kvh * GROUP + h) * D + d));
    }
    __syncthreads();
  }

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
    unsigned k_pk[SLICE / 2];
    {
      const bf16* kp = k_cache +
          (((blk * num_kv_heads + kvh) * ATTN_BS + tok) * D + d0);
      #pragma unroll
      for (int i = 0; i < SLICE / 8; ++i) {
        uint4 r4 = *reinterpret_cast<const uint4*>(kp + i * 8);
        k_pk[i * 4 + 0] = r4.x;
        k_pk[i * 4 + 1] = r4.y;
        k_pk[i * 4 + 2] = r4.z;
        k_pk[i * 4 + 3] = r4.w;
      }
    }
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float p = 0.f;
      if constexpr (QREG) {
        #pragma unroll
        for (int i = 0; i < SLICE / 2; ++i) {
          const unsigned qa = q_pk[h][i], ka = k_pk[i];
          p += bf2f((unsigned short)(qa & 0xffff)) *
               bf2f((unsigned short)(ka & 0xffff));
          p += bf2f((unsigned short)(qa >> 16)) *
               bf2f((unsigned short)(ka >> 16));
        }
      } else {
        #pragma unroll
        for (int i = 0; i < SLICE / 4; ++i) {
          const float4 qv = *reinterpret_cast<const float4*>(
              &q_lds[h][d0 + i * 4]);
          p += qv.x * bf2f((unsigned short)(k_pk[i * 2] & 0xffff));
          p += qv.y * bf2f((unsigned short)(k_pk[i * 2] >> 16));
          p += qv.z * bf2f((unsigned short)(k_pk[i * 2 + 1] & 0xffff));
          p += qv.w * bf2f((unsigned short)(k_pk[i * 2 + 1] >> 16));
        }
      }
      p += __shfl_xor(p, 1);
      p += __shfl_xor(p, 2);
      if (sl == 0)
        sc[wid][h][tok] = (tok < nb) ? p * scale : -FLT_MAX;
    }
    // wave-private LDS area; DS ops of one wave are in program order.
    // ---- online softmax update + exponentiate in LDS ----
    float mn_h[GROUP];
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float bm = -FLT_MAX;
      #pragma unroll
      for (int t4 = 0; t4 < ATTN_BS / 4; ++t4) {
        const float4 s4 = *reinterpret_cast<const float4*>(
            &sc[wid][h][t4 * 4]);
        bm = fmaxf(fmaxf(bm, fmaxf(s4.x, s4.y)), fmaxf(s4.z, s4.w));
      }
      const float mn = fmaxf(m[h], bm);
      mn_h[h] = mn;
      const float alpha = (m[h] == -FLT_MAX) ? 0.f : __expf(m[h] - mn);
      #pragma unroll
      for (int i = 0; i < VPL; ++i) acc[h][i] *= alpha;
      l[h] *= alpha;
      m[h] = mn;
      if (sl == 0)  // one lane per token exponentiates it
        sc[wid][h][tok] = (tok < nb) ? __expf(sc[wid][h][tok] - mn) : 0.f;
    }
    // ---- V: preload the lane's column slice of all 16 rows ----
    const bf16* vbase = v_cache +
        ((blk * num_kv_heads + kvh) * ATTN_BS) * D + lane * VPL;
    float vf[ATTN_BS][VPL];
    #pragma unroll
    for (int t = 0; t < ATTN_BS; ++t) {
      if (t < nb) {
        if constexpr (VPL == 2) {
          ushort2 vv = *reinterpret_cast<const ushort2*>(vbase + (long)t * D);
          vf[t][0] = bf2f(vv.x); vf[t][1] = bf2f(vv.y);
        } else {
          vf[t][0] = bf2f(*(const unsigned short*)(vbase + (long)t * D));
        }
      } else {
        #pragma unroll
        for (int i = 0; i < VPL; ++i) vf[t][i] = 0.f;
      }
    }
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float s = 0.f;
      #pragma unroll
      for (int t4 = 0; t4 < ATTN_BS / 4; ++t4) {
        const float4 p4 = *reinterpret_cast<const float4*>(
            &sc[wid][h][t4 * 4]);
        #pragma unroll
        for (int i = 0; i < VPL; ++i) {
          acc[h][i] += p4.x * vf[t4 * 4 + 0][i];
          acc[h][i] += p4.y * vf[t4 * 4 + 1][i];
          acc[h][i] += p4.z * vf[t4 * 4 + 2][i];
          acc[h][i] += p4.w * vf[t4 * 4 + 3][i];
        }
        s += p4.x + p4.y + p4.z + p4.w;
      }
      l[h] += s;
      (void)mn_h;
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

// ---------------------------------------------------------------------------
// v3 (D=128, GROUP<=4): latency-oriented redesign.  v2's q-in-register
// layout (4 lanes/token, 32-elem q slice) cost 244 VGPRs -> occupancy 2
// waves/SIMD, and its V read was 16x 4-byte loads/page; with two dependent
// HBM round trips per page the kernel ran 5x off the KV-stream roofline
// (334us/layer at bs=256 ctx=512, profiles/r01).  v3:
//   - K phase: 8 lanes/token (16-elem slice) -> q_pk is GROUP*8 dwords
//     (32 VGPRs at GROUP=4); dot via v_dot2_f32_bf16 (guide: packed bf16
//     dot at f32 accumulate, 6x fewer VALU ops than scalar cvt+fma).
//   - V phase: lane owns an 8-elem dim slice (lane&15) of 4 consecutive
//     tokens (lane>>4) -> 4x dwordx4 loads/page; PV accumulate in packed
//     f32 FMAs (SLP-packed float2 math), P stays f32 (no bf16 round).
//   - All K+V loads for the page are issued together at the top of the
//     loop so one HBM round trip covers both phases.
//   - Softmax is shuffle-only (score all-reduce over slice lanes, max and
//     denominator over token lanes); LDS is used once per page to move
//     P from K-lane-layout to V-lane-layout (wave-private, DS ops of one
//     wave are in program order).
// ---------------------------------------------------------------------------

DEV float dot2_bf16(unsigned a, unsigned b, float c) {
  asm("v_dot2_f32_bf16 %0, %1, %2, %0" : "+v"(c) : "v"(a), "v"(b));
  return c;
}

// fp8-KV support: unpack 4 e4m3fn bytes -> 2 packed-bf16 dwords (for the
// dot2 K path) or 2 float2 (for the PV path).  gfx950-native converts.
typedef __attribute__((ext_vector_type(2))) float float2v;

DEV unsigned pk_bf16(float a, float b) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// SPLIT > 1 (flash-decode): grid.z WGs share one (row, kvh), each owning
// pages [4*z+wid :: 4*SPLIT]; per-WG partials (m, l, o) land in `partial`
// [rows, kvh, SPLIT, GROUP, D+2] fp32 and a second kernel merges them.
// Used when rows*kvh alone cannot fill the 256 CUs (small batch, or the
// kvh=1 GQA shard of Llama-70B at TP=8).
template <int GROUP, bool KV8>
__global__ __launch_bounds__(256) void paged_attn_v3_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ q,
    const bf16* __restrict__ k_cache, const bf16* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ ctx_lens,
    const int* __restrict__ row_seq, int max_blocks, int num_kv_heads,
    float scale, long q_stride, int h_total, int g_stride, int h_off,
    float* __restrict__ partial) {
  constexpr int D = 128;
  const int row = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = gridDim.z;
  const int H = h_total;
  const int ctx = ctx_lens[row];
  const int table = row_seq ? row_seq[row] : row;
  const int* bt = block_tables + (long)table * max_blocks;
  const int nblocks = (ctx + ATTN_BS - 1) / ATTN_BS;

  __shared__ __attribute__((aligned(16))) float sc[4][GROUP][ATTN_BS];
  __shared__ __attribute__((aligned(16))) float comb_o[4][GROUP][D];
  __shared__ float comb_m[4][GROUP];
  __shared__ float comb_l[4][GROUP];

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  // K-phase role: token t_loc = lane>>3 (8 per pass, 2 passes/page),
  // slice sl = lane&7 -> elems [sl*16, sl*16+16).
  const int t_loc = lane >> 3;
  const int sl = lane & 7;
  // V-phase role: dim slice ds = lane&15 -> elems [ds*8, ds*8+8),
  // token group g = lane>>4 -> tokens [4g, 4g+4).
  const int ds = lane & 15;
  const int g = lane >> 4;

  // ---- query group: raw packed-bf16 slice, GROUP*8 dwords ----
  unsigned q_pk[GROUP][8];
  {
    const bf16* qp = q + (long)row * q_stride
                     + ((long)kvh * g_stride + h_off) * D + sl * 16;
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      uint4 a = *reinterpret_cast<const uint4*>(qp + h * D);
      uint4 b = *reinterpret_cast<const uint4*>(qp + h * D + 8);
      q_pk[h][0] = a.x; q_pk[h][1] = a.y; q_pk[h][2] = a.z; q_pk[h][3] = a.w;
      q_pk[h][4] = b.x; q_pk[h][5] = b.y; q_pk[h][6] = b.z; q_pk[h][7] = b.w;
    }
  }

  float m[GROUP], l[GROUP];
  float2 acc[GROUP][4];
  #pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    m[h] = -FLT_MAX; l[h] = 0.f;
    #pragma unroll
    for (int j = 0; j < 4; ++j) acc[h][j] = make_float2(0.f, 0.f);
  }

  for (int b = 4 * blockIdx.z + wid; b < nblocks; b += 4 * split) {
    const long blk = bt[b];
    const int nb = min(ATTN_BS, ctx - b * ATTN_BS);
    const long page_off = ((blk * num_kv_heads + kvh) * ATTN_BS) * D;
    const bf16* pbase = k_cache + page_off;
    const bf16* vbase = v_cache + page_off;
    const unsigned char* pbase8 =
        reinterpret_cast<const unsigned char*>(k_cache) + page_off;
    const unsigned char* vbase8 =
        reinterpret_cast<const unsigned char*>(v_cache) + page_off;
    // ---- issue every load for this page up front (one round trip) ----
    uint4 kr[2][KV8 ? 1 : 2];
    uint4 vr4[KV8 ? 1 : 4];
    uint2 vr2[KV8 ? 4 : 1];
    #pragma unroll
    for (int p = 0; p < 2; ++p) {
      if constexpr (KV8) {
        kr[p][0] = *reinterpret_cast<const uint4*>(
            pbase8 + (p * 8 + t_loc) * D + sl * 16);
      } else {
        const bf16* kp = pbase + (p * 8 + t_loc) * D + sl * 16;
        kr[p][0] = *reinterpret_cast<const uint4*>(kp);
        kr[p][1] = *reinterpret_cast<const uint4*>(kp + 8);
      }
    }
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      if constexpr (KV8)
        vr2[i] = *reinterpret_cast<const uint2*>(vbase8 + (4 * g + i) * D
                                                 + ds * 8);
      else
        vr4[i] = *reinterpret_cast<const uint4*>(vbase + (4 * g + i) * D
                                                 + ds * 8);
    }
    // ---- scores: dot2 over the lane's 16-elem slice, then all-reduce
    // over the 8 slice lanes (bits 0..2) ----
    float s[GROUP][2];
    #pragma unroll
    for (int p = 0; p < 2; ++p) {
      unsigned kw[8];
      if constexpr (KV8) {
        const unsigned* kq = reinterpret_cast<const unsigned*>(&kr[p][0]);
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
          float2v lo = __builtin_amdgcn_cvt_pk_f32_fp8(kq[i], false);
          float2v hi = __builtin_amdgcn_cvt_pk_f32_fp8(kq[i], true);
          kw[2 * i] = pk_bf16(lo.x, lo.y);
          kw[2 * i + 1] = pk_bf16(hi.x, hi.y);
        }
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          kw[j] = reinterpret_cast<const unsigned*>(&kr[p][0])[j];
      }
      #pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        float d = 0.f;
        #pragma unroll
        for (int j = 0; j < 8; ++j) d = dot2_bf16(q_pk[h][j], kw[j], d);
        d += __shfl_xor(d, 1);
        d += __shfl_xor(d, 2);
        d += __shfl_xor(d, 4);
        const int tok = p * 8 + t_loc;
        s[h][p] = (tok < nb) ? d * scale : -FLT_MAX;
      }
    }
    // ---- online softmax, shuffle-only (token index lives in bits 3..5) --
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      float bm = fmaxf(s[h][0], s[h][1]);
      bm = fmaxf(bm, __shfl_xor(bm, 8));
      bm = fmaxf(bm, __shfl_xor(bm, 16));
      bm = fmaxf(bm, __shfl_xor(bm, 32));
      const float mn = fmaxf(m[h], bm);
      const float alpha = (m[h] == -FLT_MAX) ? 0.f : __expf(m[h] - mn);
      const float p0 = __expf(s[h][0] - mn);  // masked lanes underflow to 0
      const float p1 = __expf(s[h][1] - mn);
      float ps = p0 + p1;
      ps += __shfl_xor(ps, 8);
      ps += __shfl_xor(ps, 16);
      ps += __shfl_xor(ps, 32);
      l[h] = l[h] * alpha + ps;
      m[h] = mn;
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        acc[h][j].x *= alpha;
        acc[h][j].y *= alpha;
      }
      if (sl == 0) {
        sc[wid][h][t_loc] = p0;
        sc[wid][h][8 + t_loc] = p1;
      }
    }
    // ---- PV: lane accumulates its dim slice over its 4 tokens ----
    float4 p4[GROUP];
    #pragma unroll
    for (int h = 0; h < GROUP; ++h)
      p4[h] = *reinterpret_cast<const float4*>(&sc[wid][h][4 * g]);
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      float2 vf[4];
      if constexpr (KV8) {
        const unsigned* vw = reinterpret_cast<const unsigned*>(&vr2[i]);
        #pragma unroll
        for (int jj = 0; jj < 2; ++jj) {
          float2v lo = __builtin_amdgcn_cvt_pk_f32_fp8(vw[jj], false);
          float2v hi = __builtin_amdgcn_cvt_pk_f32_fp8(vw[jj], true);
          vf[2 * jj].x = lo.x; vf[2 * jj].y = lo.y;
          vf[2 * jj + 1].x = hi.x; vf[2 * jj + 1].y = hi.y;
        }
      } else {
        const unsigned* vw = reinterpret_cast<const unsigned*>(&vr4[i]);
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          vf[j].x = bf2f((unsigned short)(vw[j] & 0xffff));
          vf[j].y = bf2f((unsigned short)(vw[j] >> 16));
        }
      }
      #pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        const float p = (&p4[h].x)[i];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc[h][j].x += p * vf[j].x;
          acc[h][j].y += p * vf[j].y;
        }
      }
    }
  }

  // ---- reduce the 4 token groups (bits 4..5); m,l are wave-uniform ----
  #pragma unroll
  for (int h = 0; h < GROUP; ++h) {
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      acc[h][j].x += __shfl_xor(acc[h][j].x, 16);
      acc[h][j].x += __shfl_xor(acc[h][j].x, 32);
      acc[h][j].y += __shfl_xor(acc[h][j].y, 16);
      acc[h][j].y += __shfl_xor(acc[h][j].y, 32);
    }
  }
  // ---- combine the 4 waves via LDS ----
  if (lane < 16) {
    #pragma unroll
    for (int h = 0; h < GROUP; ++h)
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        *reinterpret_cast<float2*>(&comb_o[wid][h][ds * 8 + j * 2]) =
            acc[h][j];
  }
  if (lane == 0) {
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) {
      comb_m[wid][h] = m[h];
      comb_l[wid][h] = l[h];
    }
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
    if (split == 1) {
      *(unsigned short*)(out +
          ((long)row * H + kvh * g_stride + h + h_off) * D + d) =
          f2bf(o / L);
    } else {
      // partial layout: [row][kvh][z][h][D+2] fp32 (o | m | l)
      float* pr = partial +
          ((((long)row * gridDim.y + kvh) * split + blockIdx.z) * GROUP + h)
          * (D + 2);
      pr[d] = o;
      if (d == 0) { pr[D] = M; pr[D + 1] = L; }
    }
  }
}

// merge the z-split partials: grid (rows, kvh); 256 threads cover GROUP*D
template <int GROUP>
__global__ __launch_bounds__(256) void paged_attn_v3_merge_kernel(
    bf16* __restrict__ out, const float* __restrict__ partial, int split,
    int h_total, int g_stride, int h_off) {
  constexpr int D = 128;
  const int row = blockIdx.x;
  const int kvh = blockIdx.y;
  for (int i = threadIdx.x; i < GROUP * D; i += 256) {
    const int h = i / D, d = i % D;
    const float* base = partial +
        (((long)row * gridDim.y + kvh) * split * GROUP + h) * (D + 2);
    float M = -FLT_MAX;
    for (int z = 0; z < split; ++z)
      M = fmaxf(M, base[(long)z * GROUP * (D + 2) + D]);
    float o = 0.f, L = 0.f;
    for (int z = 0; z < split; ++z) {
      const float* pz = base + (long)z * GROUP * (D + 2);
      const float mz = pz[D];
      const float f = (mz == -FLT_MAX) ? 0.f : __expf(mz - M);
      o += f * pz[d];
      L += f * pz[D + 1];
    }
    *(unsigned short*)(out +
        ((long)row * h_total + kvh * g_stride + h + h_off) * D + d) =
        f2bf(o / L);
  }
}

void launch_paged_attn(bf16* out, const bf16* q, const bf16* k_cache,
                       const bf16* v_cache, const int* block_tables,
                       const int* ctx_lens, const int* row_seq,
                       long num_rows, int num_kv_heads, int group, int D,
                       int max_blocks, int block_size, float scale,
                       long q_stride, float* split_ws, int split,
                       bool kv_fp8, hipStream_t stream) {
  if (block_size != ATTN_BS)
    throw std::runtime_error("paged_attn: block_size must be 16");
  // v3 (D=128): GROUP<=4 direct; GROUP=8 as two GROUP=4 half-calls over
  // the same KV stream (the kvh=1 70B-TP8 shard: 2x a small KV re-read
  // buys the v3 structure + occupancy).  split>1 = flash-decode page
  // split with an fp32 partial workspace + merge kernel.
  if (D == 128 && (group <= 4 || group == 8)) {
    const int calls = (group == 8) ? 2 : 1;
    const int g = (group == 8) ? 4 : group;
    const int h_total = num_kv_heads * group;
    for (int c = 0; c < calls; ++c) {
      const int h_off = c * 4;
      dim3 grid((unsigned)num_rows, (unsigned)num_kv_heads,
                (unsigned)split);
      dim3 mgrid((unsigned)num_rows, (unsigned)num_kv_heads);
      #define CASE3(G)                                                      \
        if (g == G) {                                                       \
          if (kv_fp8)                                                       \
            paged_attn_v3_kernel<G, true><<<grid, 256, 0, stream>>>(        \
                out, q, k_cache, v_cache, block_tables, ctx_lens, row_seq,  \
                max_blocks, num_kv_heads, scale, q_stride, h_total, group,  \
                h_off, split_ws);                                           \
          else                                                              \
            paged_attn_v3_kernel<G, false><<<grid, 256, 0, stream>>>(       \
                out, q, k_cache, v_cache, block_tables, ctx_lens, row_seq,  \
                max_blocks, num_kv_heads, scale, q_stride, h_total, group,  \
                h_off, split_ws);                                           \
          if (split > 1)                                                    \
            paged_attn_v3_merge_kernel<G><<<mgrid, 256, 0, stream>>>(       \
                out, split_ws, split, h_total, group, h_off);               \
        }
      CASE3(1) CASE3(2) CASE3(4)
      #undef CASE3
    }
    return;
  }
  if (split != 1)
    throw std::runtime_error("paged_attn: split needs the v3 path");
  if (kv_fp8)
    throw std::runtime_error("paged_attn: fp8 KV needs head_dim 128");
  dim3 grid((unsigned)num_rows, (unsigned)num_kv_heads);
  #define CASE(G, DD)                                                       \
    if (group == G && D == DD) {                                            \
      paged_attn_kernel<G, DD><<<grid, 256, 0, stream>>>(                   \
          out, q, k_cache, v_cache, block_tables, ctx_lens, row_seq,        \
          max_blocks, num_kv_heads, scale, q_stride);                       \
      return;                                                               \
    }
  CASE(1, 64) CASE(2, 64) CASE(4, 64) CASE(8, 64)
  #undef CASE
  throw std::runtime_error("paged_attn: unsupported (group, head_dim)");
}
