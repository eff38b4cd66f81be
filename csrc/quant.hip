// fp8 (OCP e4m3fn) activation-quant kernels for gfx950.
//
// The fp8 GEMM path (torch._scaled_mm -> hipBLASLt fp8 MFMA) needs
// per-token-scaled fp8 activations; doing amax+scale+cast as separate
// torch ops costs an extra read+write of every activation and erased the
// fp8 GEMM win (bench_kernels.py fp8: mm-only 1.7-1.8x faster than bf16,
// with eager quant overall slower).  These kernels produce fp8 directly:
//   - quant_fp8:             x bf16 -> (fp8, row scale)    [o_proj input]
//   - fused_add_rmsnorm_fp8: residual+=x; norm -> fp8      [qkv/gate_up in]
//   - silu_mul_fp8:          silu(g)*u -> fp8              [down_proj in]
// All row-per-workgroup, row held in registers between the stat pass and
// the store pass; fp8 packing via v_cvt_pk_fp8_f32 (native on gfx950).
#include "common.h"

#define QCHUNKS 8  // * 256 threads * 8 = 16384 max row width

// pack 8 f32 (already scaled) -> 8 fp8 bytes (word_sel is an imm operand)
DEV uint2 pack_fp8x8(const float* v) {
  uint2 r{0u, 0u};
  r.x = __builtin_amdgcn_cvt_pk_fp8_f32(v[0], v[1], r.x, false);
  r.x = __builtin_amdgcn_cvt_pk_fp8_f32(v[2], v[3], r.x, true);
  r.y = __builtin_amdgcn_cvt_pk_fp8_f32(v[4], v[5], r.y, false);
  r.y = __builtin_amdgcn_cvt_pk_fp8_f32(v[6], v[7], r.y, true);
  return r;
}

#define FP8_MAXN 448.f

__global__ __launch_bounds__(256) void quant_fp8_kernel(
    unsigned char* __restrict__ out, float* __restrict__ scales,
    const bf16* __restrict__ x, int hidden, long x_row_stride) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  const bf16* xr = x + row * x_row_stride;

  float v[QCHUNKS][8];
  float amax = 0.f;
  #pragma unroll
  for (int c = 0; c < QCHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      bf16x8 xv = load_bf16x8(xr + base);
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        v[c][i] = bf16x8_get(xv, i);
        amax = fmaxf(amax, fabsf(v[c][i]));
      }
    }
  }
  const float m = block_reduce_max_256(amax, scratch);
  const float s = fmaxf(m / FP8_MAXN, 1e-8f);
  const float inv = 1.f / s;
  if (threadIdx.x == 0) scales[row] = s;
  #pragma unroll
  for (int c = 0; c < QCHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      float t[8];
      #pragma unroll
      for (int i = 0; i < 8; ++i) t[i] = v[c][i] * inv;
      *reinterpret_cast<uint2*>(out + row * (long)hidden + base) =
          pack_fp8x8(t);
    }
  }
}

// residual <- residual + x (bf16, in place); out <- fp8(rmsnorm(residual)*w)
__global__ __launch_bounds__(256) void fused_add_rmsnorm_fp8_kernel(
    unsigned char* __restrict__ out, float* __restrict__ scales,
    const bf16* __restrict__ x, bf16* __restrict__ residual,
    const bf16* __restrict__ w, float eps, int hidden) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  const bf16* xr = x + row * (long)hidden;
  bf16* rr = residual + row * (long)hidden;

  float r[QCHUNKS][8];
  float ss = 0.f;
  #pragma unroll
  for (int c = 0; c < QCHUNKS; ++c) {
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
      store_bf16x8(rr + base, o);
    }
  }
  const float total = block_reduce_sum_256(ss, scratch);
  const float rs = rsqrtf(total / hidden + eps);
  // normalized values + row amax
  float amax = 0.f;
  #pragma unroll
  for (int c = 0; c < QCHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      bf16x8 wv = load_bf16x8(w + base);
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        r[c][i] = r[c][i] * rs * bf16x8_get(wv, i);
        amax = fmaxf(amax, fabsf(r[c][i]));
      }
    }
  }
  const float m = block_reduce_max_256(amax, scratch);
  const float s = fmaxf(m / FP8_MAXN, 1e-8f);
  const float inv = 1.f / s;
  if (threadIdx.x == 0) scales[row] = s;
  #pragma unroll
  for (int c = 0; c < QCHUNKS; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < hidden) {
      float t[8];
      #pragma unroll
      for (int i = 0; i < 8; ++i) t[i] = r[c][i] * inv;
      *reinterpret_cast<uint2*>(out + row * (long)hidden + base) =
          pack_fp8x8(t);
    }
  }
}

// out <- fp8(silu(x[:, :inter]) * x[:, inter:]) with per-row scale.
// CH=16 covers inter up to 32768 (Llama-70B MLP: 28672) at the price of
// a bigger register file for the row.
template <int CH>
__global__ __launch_bounds__(256) void silu_mul_fp8_kernel(
    unsigned char* __restrict__ out, float* __restrict__ scales,
    const bf16* __restrict__ x, int inter) {
  __shared__ float scratch[4];
  const long row = blockIdx.x;
  const bf16* xr = x + row * 2ll * inter;

  float v[CH][8];
  float amax = 0.f;
  #pragma unroll
  for (int c = 0; c < CH; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < inter) {
      bf16x8 g = load_bf16x8(xr + base);
      bf16x8 u = load_bf16x8(xr + inter + base);
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float gv = bf16x8_get(g, i);
        const float f = gv / (1.f + __expf(-gv)) * bf16x8_get(u, i);
        v[c][i] = f;
        amax = fmaxf(amax, fabsf(f));
      }
    }
  }
  const float m = block_reduce_max_256(amax, scratch);
  const float s = fmaxf(m / FP8_MAXN, 1e-8f);
  const float inv = 1.f / s;
  if (threadIdx.x == 0) scales[row] = s;
  #pragma unroll
  for (int c = 0; c < CH; ++c) {
    const int base = (c * 256 + threadIdx.x) * 8;
    if (base < inter) {
      float t[8];
      #pragma unroll
      for (int i = 0; i < 8; ++i) t[i] = v[c][i] * inv;
      *reinterpret_cast<uint2*>(out + row * (long)inter + base) =
          pack_fp8x8(t);
    }
  }
}

void launch_quant_fp8(unsigned char* out, float* scales, const bf16* x,
                      long rows, int hidden, long x_row_stride,
                      hipStream_t stream) {
  if (hidden % 8 || hidden > QCHUNKS * 256 * 8)
    throw std::runtime_error("quant_fp8: bad hidden");
  quant_fp8_kernel<<<dim3((unsigned)rows), 256, 0, stream>>>(
      out, scales, x, hidden, x_row_stride);
}

void launch_fused_add_rmsnorm_fp8(unsigned char* out, float* scales,
                                  const bf16* x, bf16* residual,
                                  const bf16* w, float eps, long rows,
                                  int hidden, hipStream_t stream) {
  if (hidden % 8 || hidden > QCHUNKS * 256 * 8)
    throw std::runtime_error("fused_add_rmsnorm_fp8: bad hidden");
  fused_add_rmsnorm_fp8_kernel<<<dim3((unsigned)rows), 256, 0, stream>>>(
      out, scales, x, residual, w, eps, hidden);
}

void launch_silu_mul_fp8(unsigned char* out, float* scales, const bf16* x,
                         long rows, int inter, hipStream_t stream) {
  if (inter % 8 || inter > 2 * QCHUNKS * 256 * 8)
    throw std::runtime_error("silu_mul_fp8: bad inter");
  if (inter <= QCHUNKS * 256 * 8)
    silu_mul_fp8_kernel<QCHUNKS><<<dim3((unsigned)rows), 256, 0, stream>>>(
        out, scales, x, inter);
  else
    silu_mul_fp8_kernel<2 * QCHUNKS>
        <<<dim3((unsigned)rows), 256, 0, stream>>>(out, scales, x, inter);
}
