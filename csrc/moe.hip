// MoE kernels for gfx950: token alignment, grouped expert GEMM (MFMA),
// weighted combine.  SURVEY.md §2.9 kernel obligations for Mixtral (EP).
//
// Capture-safety is the design driver: decode steps replay as hipGraphs,
// so every grid here is a function of (T, K, E) only — per-expert token
// counts live in device memory and never reach the host.
//
//   moe_align:   top-k pair ids -> per-expert sorted order, padded to
//                BM=32-row tiles; emits sorted_ids (pair idx | -1 pad),
//                tile_expert (expert per 32-row tile | -1 unused) and
//                inv_pos (pair -> sorted position) in ONE workgroup.
//   moe_gemm:    out[p, n] = x[row(p)] @ W[expert(tile(p))]^T over the
//                sorted space; 32x128 tile per workgroup, 4 waves, LDS
//                XOR-swizzled A/B, v_mfma_f32_32x32x16_bf16 (fragment
//                conventions validated by csrc/attn_prefill_mfma.hip).
//   moe_combine: out[t] = sum_k w[t,k] * y[inv_pos[t*K+k]]  (gather, no
//                atomics).
#include "common.h"

#define MOE_BM 32   // sorted rows per expert tile (the align granularity)

// ---------------------------------------------------------------- align
__global__ __launch_bounds__(256) void moe_align_kernel(
    int* __restrict__ sorted_ids, int* __restrict__ tile_expert,
    int* __restrict__ inv_pos, const int* __restrict__ flat_ids,
    int num_pairs, int num_experts, int ntiles_max) {
  extern __shared__ int lds[];   // [E] counts | [E+1] offsets | [E] cursor
  int* cnt = lds;
  int* off = lds + num_experts;
  int* cur = lds + 2 * num_experts + 1;
  for (int e = threadIdx.x; e < num_experts; e += 256) {
    cnt[e] = 0;
    cur[e] = 0;
  }
  __syncthreads();
  for (int p = threadIdx.x; p < num_pairs; p += 256)
    atomicAdd(&cnt[flat_ids[p]], 1);
  __syncthreads();
  if (threadIdx.x == 0) {        // serial prefix over E (E <= 64)
    int acc = 0;
    for (int e = 0; e < num_experts; ++e) {
      off[e] = acc;
      acc += (cnt[e] + MOE_BM - 1) / MOE_BM * MOE_BM;  // pad to tile
    }
    off[num_experts] = acc;
  }
  __syncthreads();
  const int P = ntiles_max * MOE_BM;
  for (int i = threadIdx.x; i < P; i += 256) sorted_ids[i] = -1;
  for (int t = threadIdx.x; t < ntiles_max; t += 256) {
    int e_of = -1;
    const int r0 = t * MOE_BM;
    for (int e = 0; e < num_experts; ++e)
      if (r0 >= off[e] && r0 < off[e + 1]) { e_of = e; break; }
    tile_expert[t] = e_of;
  }
  __syncthreads();
  for (int p = threadIdx.x; p < num_pairs; p += 256) {
    const int e = flat_ids[p];
    const int pos = off[e] + atomicAdd(&cur[e], 1);
    sorted_ids[pos] = p;
    inv_pos[p] = pos;
  }
}

// ------------------------------------------------------------- grouped GEMM
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// XOR swizzle: 128-B LDS rows, spread the 16-B slots over banks
DEV int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

// out[P, N] (sorted space) = gather(x) @ W[e]^T ; W is [E, N, K] row-major.
// gather_div > 0: A row p comes from x[sorted_ids[p] / gather_div]
// gather_div == 0: A row p is x[p] (identity; padded rows are junk and the
// combine step drops them).
__global__ __launch_bounds__(256) void moe_gemm_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x,
    const bf16* __restrict__ w, const int* __restrict__ sorted_ids,
    const int* __restrict__ tile_expert, int N, int K, int gather_div) {
  const int e = tile_expert[blockIdx.x];
  if (e < 0) return;
  const int m0 = blockIdx.x * MOE_BM;
  const int n0 = blockIdx.y * 128;

  __shared__ __attribute__((aligned(16))) unsigned char As[MOE_BM * 128];
  __shared__ __attribute__((aligned(16))) unsigned char Bs[128 * 128];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;
  const int lhalf = lane >> 5;

  // A loader role: one 16-B chunk per thread (32 rows x 8 chunks)
  const int ar = tid >> 3, ac = tid & 7;
  int a_src = -1;
  if (gather_div > 0) {
    const int pair = sorted_ids[m0 + ar];
    if (pair >= 0) a_src = pair / gather_div;
  } else {
    a_src = m0 + ar;
  }

  f32x16 acc;
  #pragma unroll
  for (int i = 0; i < 16; ++i) acc[i] = 0.f;

  for (int k0 = 0; k0 < K; k0 += 64) {
    // ---- stage A (32x64) and B (128x64) ----
    {
      uint4 av = uint4{0, 0, 0, 0};
      if (a_src >= 0)
        av = *reinterpret_cast<const uint4*>(x + (long)a_src * K + k0
                                             + ac * 8);
      *reinterpret_cast<uint4*>(&As[swz(ar, ac * 16)]) = av;
      #pragma unroll
      for (int c = 0; c < 4; ++c) {
        const int lin = tid * 4 + c;
        const int br = lin >> 3, bc = lin & 7;
        uint4 bv = *reinterpret_cast<const uint4*>(
            w + ((long)e * N + n0 + br) * K + k0 + bc * 8);
        *reinterpret_cast<uint4*>(&Bs[swz(br, bc * 16)]) = bv;
      }
    }
    __syncthreads();
    // ---- 4 MFMAs: C(32x32) += A(32x16) * B(16x32) per k-chunk ----
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      short8 af = *reinterpret_cast<const short8*>(
          &As[swz(lcol, kk * 32 + lhalf * 16)]);
      short8 bf = *reinterpret_cast<const short8*>(
          &Bs[swz(wid * 32 + lcol, kk * 32 + lhalf * 16)]);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
    }
    __syncthreads();
  }
  // ---- epilogue: C row = (reg&3) + 8*(reg>>2) + 4*lhalf, col = lcol ----
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int m = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
    *(unsigned short*)(out + (long)(m0 + m) * N + n0 + wid * 32 + lcol) =
        f2bf(acc[r]);
  }
}

// ------------------------------------------------------------- combine
__global__ __launch_bounds__(256) void moe_combine_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ y,
    const float* __restrict__ wts, const int* __restrict__ inv_pos,
    int topk, int hidden) {
  const long t = blockIdx.x;
  for (int j = threadIdx.x * 8; j < hidden; j += 256 * 8) {
    float accv[8];
    #pragma unroll
    for (int i = 0; i < 8; ++i) accv[i] = 0.f;
    for (int k = 0; k < topk; ++k) {
      const float wk = wts[t * topk + k];
      const long pos = inv_pos[t * topk + k];
      bf16x8 v = load_bf16x8(y + pos * hidden + j);
      #pragma unroll
      for (int i = 0; i < 8; ++i) accv[i] += wk * bf16x8_get(v, i);
    }
    bf16x8 o;
    #pragma unroll
    for (int i = 0; i < 8; ++i) bf16x8_set(o, i, accv[i]);
    store_bf16x8(out + t * hidden + j, o);
  }
}

void launch_moe_align(int* sorted_ids, int* tile_expert, int* inv_pos,
                      const int* flat_ids, int num_pairs, int num_experts,
                      int ntiles_max, hipStream_t stream) {
  if (num_experts > 64)
    throw std::runtime_error("moe_align: num_experts > 64");
  const int lds_bytes = (2 * num_experts + 1 + num_experts) * 4;
  moe_align_kernel<<<1, 256, lds_bytes, stream>>>(
      sorted_ids, tile_expert, inv_pos, flat_ids, num_pairs, num_experts,
      ntiles_max);
}

void launch_moe_gemm(bf16* out, const bf16* x, const bf16* w,
                     const int* sorted_ids, const int* tile_expert,
                     int ntiles_max, int N, int K, int gather_div,
                     hipStream_t stream) {
  if (N % 128 || K % 64)
    throw std::runtime_error("moe_gemm: N%128 or K%64 != 0");
  dim3 grid((unsigned)ntiles_max, (unsigned)(N / 128));
  moe_gemm_kernel<<<grid, 256, 0, stream>>>(out, x, w, sorted_ids,
                                            tile_expert, N, K, gather_div);
}

void launch_moe_combine(bf16* out, const bf16* y, const float* wts,
                        const int* inv_pos, long T, int topk, int hidden,
                        hipStream_t stream) {
  if (hidden % 8)
    throw std::runtime_error("moe_combine: hidden % 8 != 0");
  moe_combine_kernel<<<dim3((unsigned)T), 256, 0, stream>>>(
      out, y, wts, inv_pos, topk, hidden);
}
