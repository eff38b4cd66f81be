// Flash-style varlen causal prefill attention on MFMA (gfx950), bf16,
// head_dim 128.
//
// Replaces the per-row prefill path (each query re-reading its context —
// measured 1.34 ms/layer at 8k prefill tokens, profiles/r01): K/V stream
// through LDS once per 256-row q tile and the matmuls run on the matrix
// cores (v_mfma_f32_32x32x16_bf16).
//
// Geometry (per /opt/skills/guides/cdna_hip_programming.md §B attention):
//   workgroup = 8 waves (512 thr) = 256 q rows of ONE (seq, q-head);
//   wave owns 32 rows; KV tiles of 64 tokens staged in LDS
//   (K [64][128] row-major + XOR swizzle; V transposed to [128][64] at
//   stage so the PV B-fragment reads rows).
//   SWAPPED QK^T — mfma(A=K, B=Q) gives S^T with the q row in the lane
//   column, so the online-softmax row reduction is one shfl_xor(32), not a
//   cross-lane tree (guide T12 mechanism; P goes through a per-wave LDS
//   tile instead of permlane packing — v1).
#include <float.h>

#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#define QBLK 32       // q rows per wave
#define NWAVE 8
#define QTILE (QBLK * NWAVE)   // 256 q rows per workgroup
#define KVBLK 64
#define HD 128        // head_dim

// XOR swizzle: spread a row-major [rows][128] bf16 tile's column slices
// over LDS banks (guide G4: D=128 rows are a 16-way conflict unswizzled)
DEV int swz(int row, int col_byte) { return col_byte ^ ((row & 7) << 4); }

__global__ __launch_bounds__(512, 1) void attn_prefill_mfma_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ q,
    const bf16* __restrict__ k, const bf16* __restrict__ v,
    const int* __restrict__ seq_start, int num_heads, int num_kv_heads,
    float scale, long sq, long sk, long sv) {
  const int h = blockIdx.y;
  const int kvh = h / (num_heads / num_kv_heads);
  const int z = blockIdx.z;
  const long t0 = seq_start[z];
  const int n = seq_start[z + 1] - (int)t0;
  const int q0 = blockIdx.x * QTILE;
  if (q0 >= n) return;

  // LDS: K tile + V^T tile + per-wave P tiles + per-wave alpha/lsum
  __shared__ __attribute__((aligned(16))) short k_lds[KVBLK * HD];
  __shared__ __attribute__((aligned(16))) short vt_lds[HD * KVBLK];
  __shared__ __attribute__((aligned(16))) short p_lds[NWAVE][QBLK * KVBLK];
  __shared__ float alpha_lds[NWAVE][QBLK];
  __shared__ float lsum_lds[NWAVE][QBLK];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;     // q column this lane owns in S^T
  const int lhalf = lane >> 5;

  // ---- load this wave's Q as 8 B-fragments (held for the whole loop) ----
  // B[k=dim][j=q]: lane j = lcol, k = c*16 + lhalf*8 + e  (16B per chunk)
  const int qrow = q0 + wid * QBLK + lcol;
  short8 qfrag[8];
  {
    const bf16* qp = q + (t0 + qrow) * sq + (long)h * HD;
    #pragma unroll
    for (int c = 0; c < 8; ++c) {
      if (qrow < n) {
        qfrag[c] = *reinterpret_cast<const short8*>(
            qp + c * 16 + lhalf * 8);
      } else {
        qfrag[c] = short8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  f32x16 acc[4];      // O[q][d] per 32-wide d tile
  #pragma unroll
  for (int dt = 0; dt < 4; ++dt) acc[dt] = f32x16(0.f);
  float m_run = -FLT_MAX, l_run = 0.f;

  const int q_hi = min(q0 + QTILE, n);         // exclusive
  const int nkv_tiles = (q_hi + KVBLK - 1) / KVBLK;

  for (int kt = 0; kt < nkv_tiles; ++kt) {
    const int kv0 = kt * KVBLK;
    const int kvn = min(KVBLK, n - kv0);
    // ---- stage K [64][128] (swizzled) and V^T [128][64] ----
    {
      // K: 1024 16B units; unit u: row r = u/16, colb = (u%16)*16
      for (int u = tid; u < KVBLK * 16; u += 512) {
        const int r = u >> 4, cb = (u & 15) << 4;
        uint4 val{0, 0, 0, 0};
        if (r < kvn)
          val = *reinterpret_cast<const uint4*>(
              reinterpret_cast<const char*>(k + (t0 + kv0 + r) * sk +
                                            (long)kvh * HD) + cb);
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(k_lds) + r * 256 + swz(r, cb)) = val;
      }
      // V^T: unit u handles kv rows (2p, 2p+1) x 8 dims: 32 pairs x 16
      // dim-chunks = 512 units
      for (int u = tid; u < 512; u += 512) {
        const int p2 = u & 31, dc = u >> 5;          // pair, dim chunk
        const int r0 = 2 * p2;
        uint4 a{0, 0, 0, 0}, b{0, 0, 0, 0};
        if (r0 < kvn)
          a = *reinterpret_cast<const uint4*>(
              v + (t0 + kv0 + r0) * sv + (long)kvh * HD + dc * 8);
        if (r0 + 1 < kvn)
          b = *reinterpret_cast<const uint4*>(
              v + (t0 + kv0 + r0 + 1) * sv + (long)kvh * HD + dc * 8);
        const unsigned* au = reinterpret_cast<const unsigned*>(&a);
        const unsigned* bu = reinterpret_cast<const unsigned*>(&b);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const unsigned av = (au[j >> 1] >> ((j & 1) * 16)) & 0xffff;
          const unsigned bv = (bu[j >> 1] >> ((j & 1) * 16)) & 0xffff;
          const int d = dc * 8 + j;
          *reinterpret_cast<unsigned*>(
              reinterpret_cast<char*>(vt_lds) + d * (KVBLK * 2) +
              swz(d, r0 * 2)) = av | (bv << 16);
        }
      }
    }
    __syncthreads();

    // ---- QK^T (swapped): S^T[kv][q] in two 32-row tiles ----
    f32x16 s[2];
    #pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      s[rt] = f32x16(0.f);
      const int arow = rt * 32 + lcol;         // K row this lane feeds
      #pragma unroll
      for (int c = 0; c < 8; ++c) {
        short8 kf = *reinterpret_cast<const short8*>(
            reinterpret_cast<char*>(k_lds) + arow * 256 +
            swz(arow, c * 32 + lhalf * 16));
        s[rt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfrag[c],
                                                        s[rt], 0, 0, 0);
      }
    }
    // ---- online softmax on the lane's 32 scores (one q column) ----
    const int qg = q0 + wid * QBLK + lcol;     // this lane's q row
    float p[2][16];
    float tile_max = -FLT_MAX;
    #pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvr = kv0 + rt * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhalf;
        float sv_ = s[rt][r] * scale;
        if (kvr > qg || kvr >= n || qg >= n) sv_ = -FLT_MAX;
        p[rt][r] = sv_;
        tile_max = fmaxf(tile_max, sv_);
      }
    }
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32));
    const float m_new = fmaxf(m_run, tile_max);
    const float alpha = (m_run == -FLT_MAX || m_new == -FLT_MAX)
        ? 0.f : __expf(m_run - m_new);
    float lsum = 0.f;
    #pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const float e = (p[rt][r] == -FLT_MAX || m_new == -FLT_MAX)
            ? 0.f : __expf(p[rt][r] - m_new);
        p[rt][r] = e;
        lsum += e;
      }
    }
    lsum += __shfl_xor(lsum, 32);
    l_run = l_run * alpha + lsum;
    m_run = m_new;
    if (lhalf == 0) alpha_lds[wid][lcol] = alpha;
    // ---- P -> per-wave LDS [32 q][64 kv] bf16 (packed pair writes) ----
    #pragma unroll
    for (int rt = 0; rt < 2; ++rt) {
      #pragma unroll
      for (int r = 0; r < 16; r += 2) {
        const int kvr = rt * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhalf;
        const unsigned lo = f2bf(p[rt][r]);
        const unsigned hi = f2bf(p[rt][r + 1]);
        *reinterpret_cast<unsigned*>(
            reinterpret_cast<char*>(p_lds[wid]) + lcol * (KVBLK * 2) +
            swz(lcol, kvr * 2)) = lo | (hi << 16);
      }
    }
    // ---- rescale O by alpha of each C row (alphas via wave LDS) ----
    #pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow_c = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
        acc[dt][r] *= alpha_lds[wid][qrow_c];
      }
    }
    // ---- PV: O[q][d] += P[q][kv] * V[kv][d] ----
    #pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      #pragma unroll
      for (int c = 0; c < 4; ++c) {
        // A = P: i=q=lcol, k = c*16 + lhalf*8 (+e)
        short8 pf = *reinterpret_cast<const short8*>(
            reinterpret_cast<char*>(p_lds[wid]) + lcol * (KVBLK * 2) +
            swz(lcol, c * 32 + lhalf * 16));
        // B = V: j=d=dt*32+lcol, k = kv chunk -> V^T row j
        const int vrow = dt * 32 + lcol;
        short8 vf = *reinterpret_cast<const short8*>(
            reinterpret_cast<char*>(vt_lds) + vrow * (KVBLK * 2) +
            swz(vrow, c * 32 + lhalf * 16));
        acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf, vf, acc[dt],
                                                          0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O /= l, store masked rows ----
  // (lsum_lds is wave-private; in-wave LDS ops are program-ordered)
  if (lhalf == 0) lsum_lds[wid][lcol] = l_run;
  #pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow_c = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
      const int qg = q0 + wid * QBLK + qrow_c;
      if (qg >= n) continue;
      const float l = lsum_lds[wid][qrow_c];
      const float o = l > 0.f ? acc[dt][r] / l : 0.f;
      *(unsigned short*)(out + ((t0 + qg) * (long)num_heads + h) * HD +
                         dt * 32 + lcol) = f2bf(o);
    }
  }
}

}  // namespace

void launch_attn_prefill_mfma(bf16* out, const bf16* q, const bf16* k,
                              const bf16* v, const int* seq_start,
                              int num_seqs, int max_seqlen, int num_heads,
                              int num_kv_heads, int head_dim, float scale,
                              long sq, long sk, long sv,
                              hipStream_t stream) {
  if (head_dim != HD)
    throw std::runtime_error("attn_prefill_mfma: head_dim must be 128");
  dim3 grid((unsigned)((max_seqlen + QTILE - 1) / QTILE),
            (unsigned)num_heads, (unsigned)num_seqs);
  attn_prefill_mfma_kernel<<<grid, 512, 0, stream>>>(
      out, q, k, v, seq_start, num_heads, num_kv_heads, scale, sq, sk, sv);
}
