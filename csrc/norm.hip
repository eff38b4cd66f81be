// RMSNorm + fused residual-add RMSNorm, bf16, row-per-workgroup.
//
// Memory-bound (hidden <= 16k): one 256-thread workgroup per token row,
// bf16x8 vector loads (scalar bf16 loads are ~2x slower on hipcc — guide
// Common-mistake #2), fp32 accumulation, row held in registers between the
// sumsq pass and the scale pass so each byte is read once.
#include "common.h"

// MAX_CHUNKS * 256 threads * 8 elems = 16384 max hidden size
#define NORM_MAX_CHUNKS 8

__global__ __launch_bounds__(256) void rmsnorm_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x,
    const bf16* __restrict__ w, float eps, int hidden) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)hidden;
  bf16* orow = out + row * (long)hidden;

  bf16x8 v[NORM_MAX_CHUNKS];
  float ss = 0.f;
  #pragma unroll
  for (int c = 0; c < NORM_MAX_CHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      v[c] = load_bf16x8(xr + base);
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        float f = bf16x8_get(v[c], i);
        ss += f * f;
      }
    }
  }
  const float total = block_reduce_sum_256(ss, scratch);
  const float rs = rsqrtf(total / hidden + eps);
  #pragma unroll
  for (int c = 0; c < NORM_MAX_CHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      bf16x8 wv = load_bf16x8(w + base);
      bf16x8 o;
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        bf16x8_set(o, i, bf16x8_get(v[c], i) * rs * bf16x8_get(wv, i));
      store_bf16x8(orow + base, o);
    }
  }
}

// x <- rmsnorm(residual + x) * w ; residual <- residual + x   (both in place)
__global__ __launch_bounds__(256) void fused_add_rmsnorm_kernel(
    bf16* __restrict__ x, bf16* __restrict__ residual,
    const bf16* __restrict__ w, float eps, int hidden) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  bf16* xr = x + row * (long)hidden;
  bf16* rr = residual + row * (long)hidden;

  float r[NORM_MAX_CHUNKS][8];
  float ss = 0.f;
  #pragma unroll
  for (int c = 0; c < NORM_MAX_CHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      bf16x8 xv = load_bf16x8(xr + base);
      bf16x8 rv = load_bf16x8(rr + base);
      bf16x8 o;
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        float f = bf16x8_get(xv, i) + bf16x8_get(rv, i);
        r[c][i] = f;
        ss += f * f;
        bf16x8_set(o, i, f);
      }
      store_bf16x8(rr + base, o);   // new residual
    }
  }
  const float total = block_reduce_sum_256(ss, scratch);
  const float rs = rsqrtf(total / hidden + eps);
  #pragma unroll
  for (int c = 0; c < NORM_MAX_CHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      bf16x8 wv = load_bf16x8(w + base);
      bf16x8 o;
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        bf16x8_set(o, i, r[c][i] * rs * bf16x8_get(wv, i));
      store_bf16x8(xr + base, o);
    }
  }
}

void launch_rmsnorm(bf16* out, const bf16* x, const bf16* w, float eps,
                    long rows, int hidden, hipStream_t stream) {
  rmsnorm_kernel<<<dim3((unsigned)rows), dim3(256), 0, stream>>>(
      out, x, w, eps, hidden);
}

void launch_fused_add_rmsnorm(bf16* x, bf16* residual, const bf16* w,
                              float eps, long rows, int hidden,
                              hipStream_t stream) {
  fused_add_rmsnorm_kernel<<<dim3((unsigned)rows), dim3(256), 0, stream>>>(
      x, residual, w, eps, hidden);
}
