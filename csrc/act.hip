// Fused SwiGLU: out[t, i] = silu(x[t, i]) * x[t, I + i] — one pass,
// bf16x8-vectorized, grid-stride (guide G11: cap the grid, stride the rest).
#include "common.h"

__global__ __launch_bounds__(256) void silu_mul_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x, long rows,
    int inter) {
  const long nvec = rows * (inter / 8);
  for (long u = blockIdx.x * 256ll + threadIdx.x; u < nvec;
       u += gridDim.x * 256ll) {
    const long t = u / (inter / 8);
    const int j = (int)(u % (inter / 8)) * 8;
    bf16x8 g = load_bf16x8(x + t * 2 * inter + j);
    bf16x8 p = load_bf16x8(x + t * 2 * inter + inter + j);
    bf16x8 o;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float gv = bf16x8_get(g, i);
      const float s = gv / (1.f + __expf(-gv));
      bf16x8_set(o, i, s * bf16x8_get(p, i));
    }
    store_bf16x8(out + t * inter + j, o);
  }
}

void launch_silu_mul(bf16* out, const bf16* x, long rows, int inter,
                     hipStream_t stream) {
  const long nvec = rows * (inter / 8);
  const unsigned blocks =
      (unsigned)min((nvec + 255) / 256, (long)2048);
  silu_mul_kernel<<<dim3(blocks), 256, 0, stream>>>(out, x, rows, inter);
}
