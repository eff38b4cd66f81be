// Fused NeoX-RoPE (q,k in place) + paged-KV append — one pass per token.
//
// The reference has no kernels (SURVEY.md §2.9 derives this list from the
// serving contract): rotating q/k and scattering k/v into the 288 GB-HBM3E
// paged pool in one kernel saves two extra passes over the qkv tensor every
// layer.  One 256-thread workgroup per token; cos/sin row staged in LDS.
#include "common.h"

// q: [T, H, D] bf16 (in place), k: [T, KV, D] (in place), v: [T, KV, D]
// cos_sin: [max_pos, D] float (cos | sin halves), positions: [T] i64
// slot_mapping: [T] i64 (slot = block*BS + off; <0 = no append)
// k_cache/v_cache: [NB, KV, BS, D] bf16
// q/k/v may be strided views into one fused qkv GEMM output buffer
// (token stride qs/ks/vs in elements; head dim contiguous).
// KV8: the paged cache stores OCP e4m3fn bytes (fp8 KV mode — halves the
// decode-attention HBM stream; scale fixed at 1.0, k/v magnitudes sit in
// the e4m3 normal range).
template <int D, bool KV8>
__global__ __launch_bounds__(256) void rope_kv_append_kernel(
    bf16* __restrict__ q, bf16* __restrict__ k, const bf16* __restrict__ v,
    const long* __restrict__ positions, const float* __restrict__ cos_sin,
    const long* __restrict__ slot_mapping, bf16* __restrict__ k_cache,
    bf16* __restrict__ v_cache, int H, int KV, int block_size,
    long qs_stride, long ks_stride, long vs_stride) {
  constexpr int HALF = D / 2;
  __shared__ float cs[D];
  const long t = blockIdx.x;
  const long pos = positions[t];
  const long slot = slot_mapping[t];
  const long blk = slot >= 0 ? slot / block_size : 0;
  const long off = slot >= 0 ? slot % block_size : 0;

  for (int i = threadIdx.x; i < D; i += 256)
    cs[i] = cos_sin[pos * D + i];
  __syncthreads();

  // rotate: thread u handles (head h, 2 consecutive pairs starting at 2j)
  // elements [2j, 2j+1] and [HALF+2j, HALF+2j+1] -> two ushort2 accesses
  const int PAIR2 = HALF / 2;               // u-slots per head
  const int total_q = H * PAIR2;
  for (int u = threadIdx.x; u < total_q; u += 256) {
    const int h = u / PAIR2, j = (u % PAIR2) * 2;
    bf16* base = q + t * qs_stride + h * D;
    ushort2 lo = *reinterpret_cast<ushort2*>(base + j);
    ushort2 hi = *reinterpret_cast<ushort2*>(base + HALF + j);
    float c0 = cs[j], c1 = cs[j + 1];
    float s0 = cs[HALF + j], s1 = cs[HALF + j + 1];
    float x0 = bf2f(lo.x), x1 = bf2f(lo.y);
    float y0 = bf2f(hi.x), y1 = bf2f(hi.y);
    lo.x = f2bf(x0 * c0 - y0 * s0); lo.y = f2bf(x1 * c1 - y1 * s1);
    hi.x = f2bf(y0 * c0 + x0 * s0); hi.y = f2bf(y1 * c1 + x1 * s1);
    *reinterpret_cast<ushort2*>(base + j) = lo;
    *reinterpret_cast<ushort2*>(base + HALF + j) = hi;
  }
  const int total_k = KV * PAIR2;
  for (int u = threadIdx.x; u < total_k; u += 256) {
    const int h = u / PAIR2, j = (u % PAIR2) * 2;
    bf16* base = k + t * ks_stride + h * D;
    ushort2 lo = *reinterpret_cast<ushort2*>(base + j);
    ushort2 hi = *reinterpret_cast<ushort2*>(base + HALF + j);
    float c0 = cs[j], c1 = cs[j + 1];
    float s0 = cs[HALF + j], s1 = cs[HALF + j + 1];
    float x0 = bf2f(lo.x), x1 = bf2f(lo.y);
    float y0 = bf2f(hi.x), y1 = bf2f(hi.y);
    const float r0 = x0 * c0 - y0 * s0, r1 = x1 * c1 - y1 * s1;
    const float r2 = y0 * c0 + x0 * s0, r3 = y1 * c1 + x1 * s1;
    lo.x = f2bf(r0); lo.y = f2bf(r1);
    hi.x = f2bf(r2); hi.y = f2bf(r3);
    *reinterpret_cast<ushort2*>(base + j) = lo;
    *reinterpret_cast<ushort2*>(base + HALF + j) = hi;
    if (slot >= 0) {
      if constexpr (KV8) {
        unsigned char* kc = reinterpret_cast<unsigned char*>(k_cache) +
            (((blk * KV + h) * block_size + off) * D);
        unsigned plo = __builtin_amdgcn_cvt_pk_fp8_f32(r0, r1, 0u, false);
        unsigned phi = __builtin_amdgcn_cvt_pk_fp8_f32(r2, r3, 0u, false);
        *reinterpret_cast<unsigned short*>(kc + j) = (unsigned short)plo;
        *reinterpret_cast<unsigned short*>(kc + HALF + j) =
            (unsigned short)phi;
      } else {
        bf16* kc = k_cache + (((blk * KV + h) * block_size + off) * D);
        *reinterpret_cast<ushort2*>(kc + j) = lo;
        *reinterpret_cast<ushort2*>(kc + HALF + j) = hi;
      }
    }
  }
  // v append: straight copy (bf16) or pack-to-fp8, 8-elem vectors
  if (slot >= 0) {
    const int total_v = KV * D / 8;
    for (int u = threadIdx.x; u < total_v; u += 256) {
      const int h = u / (D / 8), j = (u % (D / 8)) * 8;
      const bf16* src = v + t * vs_stride + h * D + j;
      if constexpr (KV8) {
        bf16x8 vv = load_bf16x8(src);
        unsigned w0 = 0, w1 = 0;
        w0 = __builtin_amdgcn_cvt_pk_fp8_f32(bf16x8_get(vv, 0),
                                             bf16x8_get(vv, 1), w0, false);
        w0 = __builtin_amdgcn_cvt_pk_fp8_f32(bf16x8_get(vv, 2),
                                             bf16x8_get(vv, 3), w0, true);
        w1 = __builtin_amdgcn_cvt_pk_fp8_f32(bf16x8_get(vv, 4),
                                             bf16x8_get(vv, 5), w1, false);
        w1 = __builtin_amdgcn_cvt_pk_fp8_f32(bf16x8_get(vv, 6),
                                             bf16x8_get(vv, 7), w1, true);
        uint2 pk{w0, w1};
        *reinterpret_cast<uint2*>(reinterpret_cast<unsigned char*>(v_cache)
            + (((blk * KV + h) * block_size + off) * D) + j) = pk;
      } else {
        bf16* vc = v_cache + (((blk * KV + h) * block_size + off) * D) + j;
        *reinterpret_cast<uint4*>(vc) =
            *reinterpret_cast<const uint4*>(src);
      }
    }
  }
}

void launch_rope_kv_append(bf16* q, bf16* k, const bf16* v,
                           const long* positions, const float* cos_sin,
                           const long* slot_mapping, bf16* k_cache,
                           bf16* v_cache, long T, int H, int KV, int D,
                           int block_size, long qs, long ks, long vs,
                           bool kv_fp8, hipStream_t stream) {
  dim3 grid((unsigned)T);
  if (D == 128 && kv_fp8)
    rope_kv_append_kernel<128, true><<<grid, 256, 0, stream>>>(
        q, k, v, positions, cos_sin, slot_mapping, k_cache, v_cache,
        H, KV, block_size, qs, ks, vs);
  else if (D == 128)
    rope_kv_append_kernel<128, false><<<grid, 256, 0, stream>>>(
        q, k, v, positions, cos_sin, slot_mapping, k_cache, v_cache,
        H, KV, block_size, qs, ks, vs);
  else if (D == 64 && !kv_fp8)
    rope_kv_append_kernel<64, false><<<grid, 256, 0, stream>>>(
        q, k, v, positions, cos_sin, slot_mapping, k_cache, v_cache,
        H, KV, block_size, qs, ks, vs);
  else
    throw std::runtime_error("rope_kv_append: unsupported head_dim/dtype");
}
