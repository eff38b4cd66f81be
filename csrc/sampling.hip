// Sampling kernels: greedy argmax and inverse-CDF categorical draw.
//
// One 256-thread workgroup per batch row over vocab-size logits
// (128256 for Llama-3).  Vectorized bf16/f32 reads; fp32 math.
#include <float.h>

#include "common.h"

// ---- greedy argmax (bf16 logits) ---------------------------------------
__global__ __launch_bounds__(256) void greedy_sample_kernel(
    long* __restrict__ out, const bf16* __restrict__ logits, int vocab) {
  __shared__ float smax[4];
  __shared__ int sidx[4];
  const bf16* row = logits + (long)blockIdx.x * vocab;
  float best = -FLT_MAX;
  int besti = 0;
  const int nvec = vocab / 8;
  for (int u = threadIdx.x; u < nvec; u += 256) {
    bf16x8 v = load_bf16x8(row + u * 8);
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float f = bf16x8_get(v, i);
      if (f > best) { best = f; besti = u * 8 + i; }
    }
  }
  for (int t = vocab - vocab % 8 + threadIdx.x; t < vocab; t += 256) {
    const float f = bf2f(*(const unsigned short*)(row + t));
    if (f > best) { best = f; besti = t; }
  }
  // wave reduce (value, index) — ties resolve to the smaller index like
  // torch.argmax on contiguous fp32
  #pragma unroll
  for (int o = 32; o > 0; o >>= 1) {
    const float ob = __shfl_xor(best, o);
    const int oi = __shfl_xor(besti, o);
    if (ob > best || (ob == best && oi < besti)) { best = ob; besti = oi; }
  }
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) { smax[wid] = best; sidx[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < 4; ++w)
      if (smax[w] > best || (smax[w] == best && sidx[w] < besti)) {
        best = smax[w]; besti = sidx[w];
      }
    out[blockIdx.x] = besti;
  }
}

// ---- inverse-CDF categorical draw (fp32 filtered logits) ----------------
// logits already divided by temperature and masked (-inf) by top-k/top-p.
// Exact semantics of the torch reference: j = first index with
// cumsum(softmax(row)) > u (searchsorted right=False on the running CDF).
__global__ __launch_bounds__(256) void inv_cdf_sample_kernel(
    long* __restrict__ out, const float* __restrict__ logits,
    const float* __restrict__ uniform, int vocab) {
  __shared__ float scratch[4];
  __shared__ float chunk_scan[256];
  __shared__ int found;
  const float* row = logits + (long)blockIdx.x * vocab;

  // pass 1: max & normalizer
  float mx = -FLT_MAX;
  for (int t = threadIdx.x; t < vocab; t += 256)
    mx = fmaxf(mx, row[t]);
  mx = block_reduce_max_256(mx, scratch);
  float z = 0.f;
  for (int t = threadIdx.x; t < vocab; t += 256)
    z += __expf(row[t] - mx);
  z = block_reduce_sum_256(z, scratch);
  const float target = uniform[blockIdx.x] * z;

  // pass 2: walk 2048-element chunks; exclusive block scan over the 256
  // per-thread partials finds the crossing chunk, then the owning thread
  // walks its 8 elements.
  if (threadIdx.x == 0) found = -1;
  __syncthreads();
  float running = 0.f;
  for (int base = 0; base < vocab; base += 2048) {
    float part = 0.f;
    float e[8];
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int t = base + threadIdx.x * 8 + i;
      e[i] = (t < vocab) ? __expf(row[t] - mx) : 0.f;
      part += e[i];
    }
    // inclusive scan over 256 thread partials (Hillis-Steele in LDS)
    chunk_scan[threadIdx.x] = part;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
      const float prev = (threadIdx.x >= off)
          ? chunk_scan[threadIdx.x - off] : 0.f;
      __syncthreads();
      chunk_scan[threadIdx.x] += prev;
      __syncthreads();
    }
    const float total = chunk_scan[255];
    if (found < 0 && running + total > target) {
      const float before = running +
          (threadIdx.x ? chunk_scan[threadIdx.x - 1] : 0.f);
      if (before <= target && before + part > target) {
        // this thread's 8-run contains the crossing
        float c = before;
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
          c += e[i];
          if (c > target) { found = base + threadIdx.x * 8 + i; break; }
        }
      }
    }
    __syncthreads();
    if (found >= 0) break;
    running += total;
    __syncthreads();
  }
  if (threadIdx.x == 0)
    out[blockIdx.x] = (found >= 0) ? min(found, vocab - 1) : vocab - 1;
}

void launch_greedy_sample(long* out, const bf16* logits, long rows,
                          int vocab, hipStream_t stream) {
  greedy_sample_kernel<<<dim3((unsigned)rows), 256, 0, stream>>>(
      out, logits, vocab);
}

void launch_inv_cdf_sample(long* out, const float* logits,
                           const float* uniform, long rows, int vocab,
                           hipStream_t stream) {
  inv_cdf_sample_kernel<<<dim3((unsigned)rows), 256, 0, stream>>>(
      out, logits, uniform, vocab);
}
