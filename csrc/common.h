// Common helpers for the gfx950 (CDNA4, MI355X) kernels.
//
// These kernels are written natively for CDNA4 — wave64, 4 SIMD-32/CU,
// 160 KiB LDS, HBM3E — per /opt/skills/guides/cdna_hip_programming.md.
// No CUDA compatibility layer, no hipify, gfx950 only.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

using bf16 = __hip_bfloat16;

// ---- bf16 <-> f32 bit tricks (one ushort per value) --------------------
DEV float bf2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}

DEV unsigned short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even
  unsigned int lsb = (c.i >> 16) & 1;
  c.i += 0x7fffu + lsb;
  return (unsigned short)(c.i >> 16);
}

// 8 bf16 loaded as one 16-byte vector (guideline 13: always vectorize bf16)
struct bf16x8 { uint4 raw; };

DEV bf16x8 load_bf16x8(const bf16* p) {
  bf16x8 r;
  r.raw = *reinterpret_cast<const uint4*>(p);
  return r;
}

DEV void store_bf16x8(bf16* p, const bf16x8& v) {
  *reinterpret_cast<uint4*>(p) = v.raw;
}

DEV float bf16x8_get(const bf16x8& v, int i) {
  unsigned int w = (&v.raw.x)[i >> 1];
  return bf2f((unsigned short)(i & 1 ? w >> 16 : w & 0xffff));
}

DEV void bf16x8_set(bf16x8& v, int i, float f) {
  unsigned int* w = &(&v.raw.x)[i >> 1];
  unsigned short b = f2bf(f);
  if (i & 1) *w = (*w & 0x0000ffffu) | ((unsigned int)b << 16);
  else       *w = (*w & 0xffff0000u) | b;
}

// ---- reductions ---------------------------------------------------------
DEV float wave_reduce_sum(float v) {
  #pragma unroll
  for (int o = 32; o > 0; o >>= 1) v += __shfl_xor(v, o);
  return v;
}

DEV float wave_reduce_max(float v) {
  #pragma unroll
  for (int o = 32; o > 0; o >>= 1) v = fmaxf(v, __shfl_xor(v, o));
  return v;
}

// 256-thread block reduce; scratch must hold >= 4 floats; result valid in
// every thread.  Caller owns barrier hygiene around reuse of scratch.
DEV float block_reduce_sum_256(float v, float* scratch) {
  v = wave_reduce_sum(v);
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) scratch[wid] = v;
  __syncthreads();
  float total = scratch[0] + scratch[1] + scratch[2] + scratch[3];
  __syncthreads();
  return total;
}

DEV float block_reduce_max_256(float v, float* scratch) {
  v = wave_reduce_max(v);
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) scratch[wid] = v;
  __syncthreads();
  float total = fmaxf(fmaxf(scratch[0], scratch[1]),
                      fmaxf(scratch[2], scratch[3]));
  __syncthreads();
  return total;
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess)                                                   \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(_e));                      \
  } while (0)
