// MoE kernels for gfx950: token alignment, grouped expert GEMM (MFMA),
// weighted combine.  SURVEY.md §2.9 kernel obligations for Mixtral (EP).
//
// Capture-safety is the design driver: decode steps replay as hipGraphs,
// so every grid here is a function of (T, K, E) only — per-expert token
// counts live in device memory and never reach the host.
//
//   moe_align:   top-k pair ids -> per-expert sorted order, padded to
//                MOE_BM-row tiles; emits sorted_ids (pair idx | -1 pad),
//                tile_expert (expert per 32-row tile | -1 unused) and
//                inv_pos (pair -> sorted position) in ONE workgroup.
//   moe_gemm:    out[p, n] = x[row(p)] @ W[expert(tile(p))]^T over the
//                sorted space; 128x128 tile per workgroup (8 waves, each
//                32x64 via 2x v_mfma_f32_32x32x16_bf16 per k-chunk), LDS
//                XOR-swizzled A/B.  BM=128 is a WEIGHT-TRAFFIC choice:
//                with 32-row tiles each expert's [N,K] matrix was
//                re-streamed once per m-tile (~9x at batch 1024 -> the
//                measured 139 ms Mixtral step); 128-row tiles cut the
//                re-read to ceil(tokens_e/128) at the price of padding
//                compute, and padding is cheap where HBM is the bound.
//   moe_combine: out[t] = sum_k w[t,k] * y[inv_pos[t*K+k]]  (gather, no
//                atomics).
#include "common.h"

#define MOE_BM 128  // sorted rows per expert tile (the align granularity)

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
__global__ __launch_bounds__(512) void moe_gemm_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x,
    const bf16* __restrict__ w, const int* __restrict__ sorted_ids,
    const int* __restrict__ tile_expert, int N, int K, int gather_div) {
  // grid is (n_tiles, m_tiles): with n_tiles % 8 == 0 every m-tile of one
  // (expert, n0) pair lands on the SAME XCD (dispatch round-robins
  // blockIdx linearly over the 8 XCDs), so the expert's B-slice is read
  // from that XCD's L2 instead of HBM for the 2nd..kth m-tile
  const int e = tile_expert[blockIdx.y];
  if (e < 0) return;
  const int m0 = blockIdx.y * MOE_BM;
  const int n0 = blockIdx.x * 128;

  // single-buffered tiles: a 2-deep LDS ring (64 KiB) was measured 10%
  // SLOWER here — it halves WG residency (guide: explicit dbuf at HIP
  // source is not a lever; occupancy is)
  __shared__ __attribute__((aligned(16))) unsigned char As[128 * 128];
  __shared__ __attribute__((aligned(16))) unsigned char Bs[128 * 128];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;
  const int lhalf = lane >> 5;
  const int wave_m = wid & 3;   // 4 row-blocks of 32
  const int wave_n = wid >> 2;  // 2 col-blocks of 64

  // loader role: thread covers 32 B of one row (4 threads per 128-row tile
  // side, 64-elem K slice = 128 B per row)
  const int ar = tid >> 2;        // 0..127
  const int ac32 = (tid & 3) * 32;  // byte offset of this thread's 32 B
  int a_src = -1;
  if (gather_div > 0) {
    const int pair = sorted_ids[m0 + ar];
    if (pair >= 0) a_src = pair / gather_div;
  } else {
    a_src = m0 + ar;
  }

  f32x16 acc[2];
  #pragma unroll
  for (int nb = 0; nb < 2; ++nb)
    #pragma unroll
    for (int i = 0; i < 16; ++i) acc[nb][i] = 0.f;

  const bf16* ap0 = a_src >= 0 ? x + (long)a_src * K + ac32 / 2 : nullptr;
  const bf16* bp0 = w + ((long)e * N + n0 + ar) * K + ac32 / 2;
  for (int k0 = 0; k0 < K; k0 += 64) {
    uint4 av0 = uint4{0, 0, 0, 0}, av1 = uint4{0, 0, 0, 0};
    if (ap0) {
      av0 = *reinterpret_cast<const uint4*>(ap0 + k0);
      av1 = *reinterpret_cast<const uint4*>(ap0 + k0 + 8);
    }
    uint4 bv0 = *reinterpret_cast<const uint4*>(bp0 + k0);
    uint4 bv1 = *reinterpret_cast<const uint4*>(bp0 + k0 + 8);
    *reinterpret_cast<uint4*>(&As[swz(ar, ac32)]) = av0;
    *reinterpret_cast<uint4*>(&As[swz(ar, ac32 + 16)]) = av1;
    *reinterpret_cast<uint4*>(&Bs[swz(ar, ac32)]) = bv0;
    *reinterpret_cast<uint4*>(&Bs[swz(ar, ac32 + 16)]) = bv1;
    __syncthreads();
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      short8 af = *reinterpret_cast<const short8*>(
          &As[swz(wave_m * 32 + lcol, kk * 32 + lhalf * 16)]);
      #pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        short8 bf = *reinterpret_cast<const short8*>(
            &Bs[swz(wave_n * 64 + nb * 32 + lcol,
                    kk * 32 + lhalf * 16)]);
        acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc[nb],
                                                          0, 0, 0);
      }
    }
    __syncthreads();
  }
  // ---- epilogue: C row = (reg&3) + 8*(reg>>2) + 4*lhalf, col = lcol ----
  #pragma unroll
  for (int nb = 0; nb < 2; ++nb)
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int m = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
      *(unsigned short*)(out + (long)(m0 + wave_m * 32 + m) * N + n0 +
                         wave_n * 64 + nb * 32 + lcol) = f2bf(acc[nb][r]);
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
  dim3 grid((unsigned)(N / 128), (unsigned)ntiles_max);
  moe_gemm_kernel<<<grid, 512, 0, stream>>>(out, x, w, sorted_ids,
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

// ----------------------------------------------------- fp8 grouped GEMM
// Same 128x128 tiling as moe_gemm_kernel, e4m3fn operands:
// out[p,n] = (sum_k A8[p,k] * B8[e,n,k]) * a_scale[row(p)] * b_scale[e,n]
// via v_mfma_f32_32x32x16_fp8_fp8 (identical C layout to the bf16 tile;
// each lane feeds 8 fp8 bytes per operand).  Halved LDS/HBM per tile —
// the MoE step is expert-weight-stream bound, so fp8 is ~2x weight BW.
typedef long long i64;

// fp8 tiles stage BK=128 (rows of 128 B): half the barriers of the bf16
// BK=64 tile at the same LDS footprint — with 64-B rows the kernel was
// measured BARRIER-bound (86 ms vs 69 ms bf16 Mixtral step).  XOR 16-B
// slots over 8 positions; b64 fragment reads stay 8-B aligned.
DEV int swz8(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

__global__ __launch_bounds__(512) void moe_gemm_fp8_kernel(
    bf16* __restrict__ out, const unsigned char* __restrict__ xq,
    const float* __restrict__ xs, const unsigned char* __restrict__ wq,
    const float* __restrict__ ws, const int* __restrict__ sorted_ids,
    const int* __restrict__ tile_expert, int N, int K, int gather_div) {
  const int e = tile_expert[blockIdx.y];      // (n, m) grid — see bf16 note
  if (e < 0) return;
  const int m0 = blockIdx.y * MOE_BM;
  const int n0 = blockIdx.x * 128;

  __shared__ __attribute__((aligned(16))) unsigned char As[128 * 128];
  __shared__ __attribute__((aligned(16))) unsigned char Bs[128 * 128];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;
  const int lhalf = lane >> 5;
  const int wave_m = wid & 3;
  const int wave_n = wid >> 2;

  // loaders: 2x16 B per thread cover a 128x128B tile per K step
  const int ar = tid >> 2;
  const int ac16 = (tid & 3) * 16;
  int a_src = -1;
  if (gather_div > 0) {
    const int pair = sorted_ids[m0 + ar];
    if (pair >= 0) a_src = pair / gather_div;
  } else {
    a_src = m0 + ar;
  }

  f32x16 acc[2];
  #pragma unroll
  for (int nb = 0; nb < 2; ++nb)
    #pragma unroll
    for (int i = 0; i < 16; ++i) acc[nb][i] = 0.f;

  const unsigned char* ap0 = a_src >= 0 ? xq + (long)a_src * K + ac16
                                        : nullptr;
  const unsigned char* bp0 = wq + ((long)e * N + n0 + ar) * K + ac16;

  for (int k0 = 0; k0 < K; k0 += 128) {
    uint4 av0 = uint4{0, 0, 0, 0}, av1 = uint4{0, 0, 0, 0};
    if (ap0) {
      av0 = *reinterpret_cast<const uint4*>(ap0 + k0);
      av1 = *reinterpret_cast<const uint4*>(ap0 + k0 + 64);
    }
    uint4 bv0 = *reinterpret_cast<const uint4*>(bp0 + k0);
    uint4 bv1 = *reinterpret_cast<const uint4*>(bp0 + k0 + 64);
    *reinterpret_cast<uint4*>(&As[swz8(ar, ac16)]) = av0;
    *reinterpret_cast<uint4*>(&As[swz8(ar, ac16 + 64)]) = av1;
    *reinterpret_cast<uint4*>(&Bs[swz8(ar, ac16)]) = bv0;
    *reinterpret_cast<uint4*>(&Bs[swz8(ar, ac16 + 64)]) = bv1;
    __syncthreads();
    #pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      i64 af = *reinterpret_cast<const i64*>(
          &As[swz8(wave_m * 32 + lcol, kk * 16 + lhalf * 8)]);
      #pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        i64 bf = *reinterpret_cast<const i64*>(
            &Bs[swz8(wave_n * 64 + nb * 32 + lcol,
                     kk * 16 + lhalf * 8)]);
        acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(
            af, bf, acc[nb], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  // epilogue: scale by a_scale[row] * b_scale[e, col], store bf16
  #pragma unroll
  for (int nb = 0; nb < 2; ++nb) {
    const int col = n0 + wave_n * 64 + nb * 32 + lcol;
    const float bs = ws[(long)e * N + col];
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int m = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
      const int p = m0 + wave_m * 32 + m;
      float as = 1.f;
      if (gather_div > 0) {
        const int pair = sorted_ids[p];
        as = pair >= 0 ? xs[pair / gather_div] : 0.f;
      } else {
        as = xs[p];
      }
      *(unsigned short*)(out + (long)p * N + col) =
          f2bf(acc[nb][r] * as * bs);
    }
  }
}

void launch_moe_gemm_fp8(bf16* out, const unsigned char* xq,
                         const float* xs, const unsigned char* wq,
                         const float* ws, const int* sorted_ids,
                         const int* tile_expert, int ntiles_max, int N,
                         int K, int gather_div, hipStream_t stream) {
  if (N % 128 || K % 128)
    throw std::runtime_error("moe_gemm_fp8: N%128 or K%128 != 0");
  dim3 grid((unsigned)(N / 128), (unsigned)ntiles_max);
  moe_gemm_fp8_kernel<<<grid, 512, 0, stream>>>(
      out, xq, xs, wq, ws, sorted_ids, tile_expert, N, K, gather_div);
}
