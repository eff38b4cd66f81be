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

#define MOE_BM 256  // sorted rows per expert tile (the align granularity)

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

// glds helper: each lane supplies a per-lane GLOBAL address; the LDS
// destination is wave-uniform base + lane*16 (guide §5) — swizzled LDS
// images are made by pre-swizzling the SOURCE address, LDS stays linear.
DEV void glds16(const void* gp, void* lp) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gp,
      (__attribute__((address_space(3))) unsigned int*)lp, 16, 0, 0);
}

// out[P, N] (sorted space) = gather(x) @ W[e]^T ; W is [E, N, K] row-major.
//
// v3 (guide §5 glds table, "3-buf span" row): 256(M) x 128(N) tile per
// 512-thread block, BK = 64 bf16 (128-B staged rows), async global->LDS
// (global_load_lds_dwordx4) into a 3-DEEP LDS ring with COUNTED
// s_waitcnt vmcnt(N) + raw s_barrier, so one whole chunk stays in
// flight across each barrier.  6 glds per wave per chunk (4 A-pieces +
// 2 B-pieces of 8 rows x 128 B, source-address XOR swizzle).  The
// binding constraint is the per-expert weight stream
// (profiles/r01_decode_v3.md): B traffic = ntiles * N * K * 2 is
// INVARIANT to the column-tile width, so BN=128 (acc 64 VGPR) buys the
// LDS headroom for depth 3: 3 x (32 KiB A + 16 KiB B) = 144 KiB.
// 256-row tiles halve the expert re-read vs the 128-row v1 kernel.
// XCD-grouped block decode: consecutive blockIdx round-robin the 8
// XCDs (guide: workgroup dispatch), so id%8 selects an XCD and the 32
// ids sharing it form one CU-sized group.  All n-blocks of ONE m-tile
// are placed on the SAME XCD: the tile's A rows (w2: 7.3 MB — bigger
// than the 4 MiB per-XCD L2 if spread over all XCDs) are streamed from
// HBM once per XCD-group instead of once per block.
struct TileMap {
  int t, j;      // m-tile index, n-block index (-1 = out of range)
};
DEV TileMap xcd_tile_map(int id, int ntiles, int nx, int use_xcd) {
  if (!use_xcd) {
    const int t = id / nx;
    const int j = id - t * nx;
    if (t >= ntiles) return {-1, -1};
    return {t, j};
  }
  const int xcd = id & 7;
  const int s = id >> 3;
  const int jg = s & 31;             // position in the 32-CU group
  const int grp = s >> 5;
  const int tpx = (ntiles + 7) >> 3; // tiles owned per XCD (padded)
  const int G = (nx + 31) >> 5;      // n-groups per tile
  const int t = xcd + 8 * (grp % tpx);
  const int g = grp / tpx;
  const int j = g * 32 + jg;
  if (t >= ntiles || g >= G || j >= nx) return {-1, -1};
  return {t, j};
}

// Split-K (ws != nullptr, splitk > 1): the w2 GEMM has only
// ntiles * N/128 ~ 288 blocks at Mixtral decode shapes — barely one
// round on 256 CUs, so the ragged second round idles ~7/8 of the chip.
// Each of `splitk` segments owns nchunks/splitk k-chunks and writes an
// fp32 partial tile to ws[s, P, N] (disjoint — no atomics, so no
// cross-XCD L2 coherence traffic); moe_sk_reduce sums the segments.
__global__ __launch_bounds__(512) void moe_gemm_v3_kernel(
    bf16* __restrict__ out, float* __restrict__ ws,
    const bf16* __restrict__ x,
    const bf16* __restrict__ w, const int* __restrict__ sorted_ids,
    const int* __restrict__ tile_expert, int N, int K, int gather_div,
    int ntiles, int use_xcd, int splitk) {
  constexpr int BN = 128;
  const int nb_base = (int)(gridDim.x / (unsigned)splitk);
  const int seg = (int)blockIdx.x / nb_base;    // k-segment index
  const TileMap tm =
      xcd_tile_map((int)blockIdx.x % nb_base, ntiles, N / BN, use_xcd);
  if (tm.t < 0) return;
  const int e = tile_expert[tm.t];
  if (e < 0) return;
  const int m0 = tm.t * MOE_BM;
  const int n0 = tm.j * BN;

  // dynamic LDS addressed by INTEGER offsets (pointer indirection decays
  // to generic AS => flat_load + vmcnt stalls in the MFMA stream)
  // ring layout: A0 A1 A2 | B0 B1 B2
  extern __shared__ int lds[];     // shared decl w/ moe_align
  unsigned char* dyn_lds = reinterpret_cast<unsigned char*>(lds);

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;
  const int lhalf = lane >> 5;
  const int wave_m = wid & 3;   // 4 row-blocks of 64
  const int wave_n = wid >> 2;  // 2 col-blocks of 64

  const int lrow = lane >> 3;           // 0..7
  const int gbyte = ((lane & 7) * 16) ^ (lrow << 4);
  const unsigned char* a_src[4];
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int r = wid * 32 + j * 8 + lrow;
    long row = 0;
    if (gather_div > 0) {
      const int pair = sorted_ids[m0 + r];
      if (pair >= 0) row = pair / gather_div;
    } else {
      row = m0 + r;
    }
    a_src[j] = reinterpret_cast<const unsigned char*>(x) +
               row * (long)K * 2 + gbyte;
  }
  const unsigned char* b_src[2];
  #pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int r = wid * 16 + j * 8 + lrow;
    b_src[j] = reinterpret_cast<const unsigned char*>(w) +
               ((long)e * N + n0 + r) * (long)K * 2 + gbyte;
  }

  f32x16 acc[2][2];
  #pragma unroll
  for (int sm = 0; sm < 2; ++sm)
    #pragma unroll
    for (int nb = 0; nb < 2; ++nb)
      #pragma unroll
      for (int i = 0; i < 16; ++i) acc[sm][nb][i] = 0.f;

  // padding skip: a wave whose 64 output rows are ALL padding keeps its
  // acc at zero and skips fragment reads + MFMAs (it still issues its
  // glds share and hits every barrier — the tile stage is cooperative).
  // ~1/3 of tile rows are padding at bs1024 (256-row align granularity)
  bool wave_live = true;
  if (gather_div > 0) {
    wave_live = false;
    #pragma unroll
    for (int r = 0; r < 64; r += 16)        // 4 probes/lane cover 64 rows
      if (sorted_ids[m0 + wave_m * 64 + r + (lane & 15)] >= 0)
        wave_live = true;
    wave_live = __any(wave_live);
  }

  auto stage = [&](int slot, int k0) {
    const long cb = (long)k0 * 2;
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      glds16(a_src[j] + cb,
             &dyn_lds[slot * 32768 + wid * 4096 + j * 1024]);
    #pragma unroll
    for (int j = 0; j < 2; ++j)
      glds16(b_src[j] + cb,
             &dyn_lds[98304 + slot * 16384 + wid * 2048 + j * 1024]);
  };

  auto compute = [&](int slot) {
    if (!wave_live) return;
    const int a_base = slot * 32768;
    const int b_base = 98304 + slot * 16384;
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      short8 af[2];
      #pragma unroll
      for (int sm = 0; sm < 2; ++sm)
        af[sm] = *reinterpret_cast<const short8*>(
            &dyn_lds[a_base + swz(wave_m * 64 + sm * 32 + lcol,
                                  kk * 32 + lhalf * 16)]);
      #pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        short8 bf = *reinterpret_cast<const short8*>(
            &dyn_lds[b_base + swz(wave_n * 64 + nb * 32 + lcol,
                                  kk * 32 + lhalf * 16)]);
        #pragma unroll
        for (int sm = 0; sm < 2; ++sm)
          acc[sm][nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[sm], bf, acc[sm][nb], 0, 0, 0);
      }
    }
  };

  const int nchunks = (K >> 6) / splitk;  // chunks in THIS k-segment
  const int c0 = seg * nchunks;           // launch checks divisibility
  // 3-slot software pipeline, one barrier per chunk:
  //   [compute d | issue d+2 | wait vmcnt(6) (d+1 landed) | barrier]
  // The issue of d+2 reuses slot (d-1)%3, whose chunk was computed by
  // every wave BEFORE the previous barrier — no read/overwrite race —
  // and one whole chunk (d+2) stays in flight across each barrier
  // (guide's 3-buf span: counted vmcnt + raw s_barrier, never
  // __syncthreads while a glds is outstanding).
  stage(0, c0 << 6);
  stage(1, (c0 + 1) << 6);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int d = 0;
  for (; d + 2 < nchunks; ++d) {
    compute(d % 3);
    stage((d + 2) % 3, (c0 + d + 2) << 6);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
  compute(d % 3);                      // d == nchunks-2
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  ++d;
  compute(d % 3);                      // d == nchunks-1

  // ---- epilogue: C row = (reg&3) + 8*(reg>>2) + 4*lhalf, col = lcol ----
  const long P = (long)ntiles * MOE_BM;
  #pragma unroll
  for (int sm = 0; sm < 2; ++sm)
    #pragma unroll
    for (int nb = 0; nb < 2; ++nb)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int m = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
        const long p = m0 + wave_m * 64 + sm * 32 + m;
        const long col = n0 + wave_n * 64 + nb * 32 + lcol;
        if (splitk == 1)
          *(unsigned short*)(out + p * N + col) = f2bf(acc[sm][nb][r]);
        else
          ws[((long)seg * P + p) * N + col] = acc[sm][nb][r];
      }
}

// sum split-K fp32 partials into bf16 (rows of unused tiles carry
// garbage in ws AND out alike; the combine gather never reads them)
__global__ __launch_bounds__(256) void moe_sk_reduce_kernel(
    bf16* __restrict__ out, const float* __restrict__ ws, long PN,
    int splitk) {
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 + 3 >= PN) {
    for (long i = i0; i < PN; ++i) {
      float a = 0.f;
      for (int s = 0; s < splitk; ++s) a += ws[s * PN + i];
      *(unsigned short*)(out + i) = f2bf(a);
    }
    return;
  }
  float a[4] = {0.f, 0.f, 0.f, 0.f};
  for (int s = 0; s < splitk; ++s) {
    const float4 v = *reinterpret_cast<const float4*>(&ws[s * PN + i0]);
    a[0] += v.x; a[1] += v.y; a[2] += v.z; a[3] += v.w;
  }
  unsigned short o[4];
  #pragma unroll
  for (int j = 0; j < 4; ++j) o[j] = f2bf(a[j]);
  *reinterpret_cast<ushort4*>(
      reinterpret_cast<unsigned short*>(out) + i0) =
      *reinterpret_cast<ushort4*>(o);
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

// splitk segments (>1 needs ws = fp32 [splitk, ntiles*MOE_BM, N]);
// nchunks must divide evenly — the python op picks splitk accordingly
void launch_moe_gemm(bf16* out, float* ws, const bf16* x, const bf16* w,
                     const int* sorted_ids, const int* tile_expert,
                     int ntiles_max, int N, int K, int gather_div,
                     int splitk, hipStream_t stream) {
  if (K % 128 || N % 128)
    throw std::runtime_error("moe_gemm: K%128 or N%128 != 0");
  // each k-segment needs >= 2 chunks (the 2-stage pipeline prologue)
  if (splitk < 1 || (K >> 6) % splitk || ((K >> 6) / splitk) < 2 ||
      (splitk > 1 && !ws))
    throw std::runtime_error("moe_gemm: bad splitk");
  const int lds_bytes = 3 * (32 * 1024 + 16 * 1024);   // 144 KiB ring
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&moe_gemm_v3_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
    attr_set = true;
  }
  static const int use_xcd = [] {
    const char* v = getenv("HS_MOE_XCD");
    return v ? atoi(v) : 0;
  }();
  const int nx = N / 128;
  const long nblocks = use_xcd
      ? 8L * 32 * ((ntiles_max + 7) / 8) * ((nx + 31) / 32)
      : (long)nx * ntiles_max;
  moe_gemm_v3_kernel<<<dim3((unsigned)(nblocks * splitk)), 512,
                       lds_bytes, stream>>>(
      out, ws, x, w, sorted_ids, tile_expert, N, K, gather_div,
      ntiles_max, use_xcd, splitk);
  if (splitk > 1) {
    const long PN = (long)ntiles_max * MOE_BM * N;
    moe_sk_reduce_kernel<<<dim3((unsigned)((PN / 4 + 255) / 256)), 256,
                           0, stream>>>(out, ws, PN, splitk);
  }
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
// Same 256x128 3-deep-ring structure as moe_gemm_v3_kernel, e4m3fn:
// out[p,n] = (sum_k A8[p,k] * B8[e,n,k]) * a_scale[row(p)] * b_scale[e,n]
// via v_mfma_f32_32x32x16_fp8_fp8.  128-B staged rows = BK 128 fp8
// elements (byte-identical loader geometry; kk 0..7 per chunk).
typedef long long i64;

__global__ __launch_bounds__(512) void moe_gemm_fp8_v3_kernel(
    bf16* __restrict__ out, float* __restrict__ skw,
    const unsigned char* __restrict__ xq,
    const float* __restrict__ xs, const unsigned char* __restrict__ wq,
    const float* __restrict__ ws, const int* __restrict__ sorted_ids,
    const int* __restrict__ tile_expert, int N, int K, int gather_div,
    int ntiles, int use_xcd, int splitk) {
  constexpr int BN = 128;
  const int nb_base = (int)(gridDim.x / (unsigned)splitk);
  const int seg = (int)blockIdx.x / nb_base;
  const TileMap tm =
      xcd_tile_map((int)blockIdx.x % nb_base, ntiles, N / BN, use_xcd);
  if (tm.t < 0) return;
  const int e = tile_expert[tm.t];
  if (e < 0) return;
  const int m0 = tm.t * MOE_BM;
  const int n0 = tm.j * BN;

  extern __shared__ int lds[];     // shared decl w/ moe_align
  unsigned char* dyn_lds = reinterpret_cast<unsigned char*>(lds);

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lcol = lane & 31;
  const int lhalf = lane >> 5;
  const int wave_m = wid & 3;
  const int wave_n = wid >> 2;

  const int lrow = lane >> 3;
  const int gbyte = ((lane & 7) * 16) ^ (lrow << 4);
  const unsigned char* a_src[4];
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int r = wid * 32 + j * 8 + lrow;
    long row = 0;
    if (gather_div > 0) {
      const int pair = sorted_ids[m0 + r];
      if (pair >= 0) row = pair / gather_div;
    } else {
      row = m0 + r;
    }
    a_src[j] = xq + row * (long)K + gbyte;
  }
  const unsigned char* b_src[2];
  #pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int r = wid * 16 + j * 8 + lrow;
    b_src[j] = wq + ((long)e * N + n0 + r) * (long)K + gbyte;
  }

  f32x16 acc[2][2];
  #pragma unroll
  for (int sm = 0; sm < 2; ++sm)
    #pragma unroll
    for (int nb = 0; nb < 2; ++nb)
      #pragma unroll
      for (int i = 0; i < 16; ++i) acc[sm][nb][i] = 0.f;

  // padding skip: a wave whose 64 output rows are ALL padding keeps its
  // acc at zero and skips fragment reads + MFMAs (it still issues its
  // glds share and hits every barrier — the tile stage is cooperative).
  // ~1/3 of tile rows are padding at bs1024 (256-row align granularity)
  bool wave_live = true;
  if (gather_div > 0) {
    wave_live = false;
    #pragma unroll
    for (int r = 0; r < 64; r += 16)        // 4 probes/lane cover 64 rows
      if (sorted_ids[m0 + wave_m * 64 + r + (lane & 15)] >= 0)
        wave_live = true;
    wave_live = __any(wave_live);
  }

  auto stage = [&](int slot, int k0) {
    const long cb = (long)k0;          // 1 B per fp8 element
    #pragma unroll
    for (int j = 0; j < 4; ++j)
      glds16(a_src[j] + cb,
             &dyn_lds[slot * 32768 + wid * 4096 + j * 1024]);
    #pragma unroll
    for (int j = 0; j < 2; ++j)
      glds16(b_src[j] + cb,
             &dyn_lds[98304 + slot * 16384 + wid * 2048 + j * 1024]);
  };

  auto compute = [&](int slot) {
    if (!wave_live) return;
    const int a_base = slot * 32768;
    const int b_base = 98304 + slot * 16384;
    #pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      i64 af[2];
      #pragma unroll
      for (int sm = 0; sm < 2; ++sm)
        af[sm] = *reinterpret_cast<const i64*>(
            &dyn_lds[a_base + swz(wave_m * 64 + sm * 32 + lcol,
                                  kk * 16 + lhalf * 8)]);
      #pragma unroll
      for (int nb = 0; nb < 2; ++nb) {
        i64 bf = *reinterpret_cast<const i64*>(
            &dyn_lds[b_base + swz(wave_n * 64 + nb * 32 + lcol,
                                  kk * 16 + lhalf * 8)]);
        #pragma unroll
        for (int sm = 0; sm < 2; ++sm)
          acc[sm][nb] = __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(
              af[sm], bf, acc[sm][nb], 0, 0, 0);
      }
    }
  };

  const int nchunks = (K >> 7) / splitk;
  const int c0 = seg * nchunks;
  // same race-free 3-slot pipeline as the bf16 kernel
  stage(0, c0 << 7);
  stage(1, (c0 + 1) << 7);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int d = 0;
  for (; d + 2 < nchunks; ++d) {
    compute(d % 3);
    stage((d + 2) % 3, (c0 + d + 2) << 7);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }
  compute(d % 3);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  ++d;
  compute(d % 3);

  // epilogue: scale by a_scale[row] * b_scale[e, col], store bf16
  // (split-K partials are stored ALREADY scaled, so the reduce is a sum)
  const long P = (long)ntiles * MOE_BM;
  #pragma unroll
  for (int sm = 0; sm < 2; ++sm)
    #pragma unroll
    for (int nb = 0; nb < 2; ++nb) {
      const int col = n0 + wave_n * 64 + nb * 32 + lcol;
      const float bs = ws[(long)e * N + col];
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int m = (r & 3) + 8 * (r >> 2) + 4 * lhalf;
        const int p = m0 + wave_m * 64 + sm * 32 + m;
        float as = 1.f;
        if (gather_div > 0) {
          const int pair = sorted_ids[p];
          as = pair >= 0 ? xs[pair / gather_div] : 0.f;
        } else {
          as = xs[p];
        }
        const float v = acc[sm][nb][r] * as * bs;
        if (splitk == 1)
          *(unsigned short*)(out + (long)p * N + col) = f2bf(v);
        else
          skw[((long)seg * P + p) * N + col] = v;
      }
    }
}

void launch_moe_gemm_fp8(bf16* out, float* skw, const unsigned char* xq,
                         const float* xs, const unsigned char* wq,
                         const float* ws, const int* sorted_ids,
                         const int* tile_expert, int ntiles_max, int N,
                         int K, int gather_div, int splitk,
                         hipStream_t stream) {
  if (K % 128 || K < 256 || N % 128)
    throw std::runtime_error("moe_gemm_fp8: bad K/N alignment");
  if (splitk < 1 || (K >> 7) % splitk || ((K >> 7) / splitk) < 2 ||
      (splitk > 1 && !skw))
    throw std::runtime_error("moe_gemm_fp8: bad splitk");
  const int lds_bytes = 3 * (32 * 1024 + 16 * 1024);
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&moe_gemm_fp8_v3_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
    attr_set = true;
  }
  static const int use_xcd = [] {
    const char* v = getenv("HS_MOE_XCD");
    return v ? atoi(v) : 0;
  }();
  const int nx = N / 128;
  const long nblocks = use_xcd
      ? 8L * 32 * ((ntiles_max + 7) / 8) * ((nx + 31) / 32)
      : (long)nx * ntiles_max;
  moe_gemm_fp8_v3_kernel<<<dim3((unsigned)(nblocks * splitk)), 512,
                           lds_bytes, stream>>>(
      out, skw, xq, xs, wq, ws, sorted_ids, tile_expert, N, K,
      gather_div, ntiles_max, use_xcd, splitk);
  if (splitk > 1) {
    const long PN = (long)ntiles_max * MOE_BM * N;
    moe_sk_reduce_kernel<<<dim3((unsigned)((PN / 4 + 255) / 256)), 256,
                           0, stream>>>(out, skw, PN, splitk);
  }
}
