// Python bindings for the gfx950 kernel set (hyperspot._C).
//
// Host-side glue only — every kernel lives in a .hip translation unit and
// is reached through a plain launcher so this file needs no device code.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>

#include <stdexcept>

using bf16 = __hip_bfloat16;

void launch_rmsnorm(bf16*, const bf16*, const bf16*, float, long, int,
                    hipStream_t);
void launch_fused_add_rmsnorm(bf16*, bf16*, const bf16*, float, long, int,
                              hipStream_t);
void launch_rope_kv_append(bf16*, bf16*, const bf16*, const long*,
                           const float*, const long*, bf16*, bf16*, long,
                           int, int, int, int, long, long, long, bool,
                           hipStream_t);
void launch_paged_attn(bf16*, const bf16*, const bf16*, const bf16*,
                       const int*, const int*, const int*, long, int, int,
                       int, int, int, float, long, float*, int, bool,
                       hipStream_t);
void launch_attn_prefill_mfma(bf16*, const bf16*, const bf16*,
                              const bf16*, const int*, int, int, int, int,
                              int, float, long, long, long, hipStream_t);
void launch_silu_mul(bf16*, const bf16*, long, int, hipStream_t);
void launch_quant_fp8(unsigned char*, float*, const bf16*, long, int, long,
                      hipStream_t);
void launch_fused_add_rmsnorm_fp8(unsigned char*, float*, const bf16*,
                                  bf16*, const bf16*, float, long, int,
                                  hipStream_t);
void launch_silu_mul_fp8(unsigned char*, float*, const bf16*, long, int,
                         hipStream_t);
void launch_moe_align(int*, int*, int*, const int*, int, int, int,
                      hipStream_t);
void launch_moe_gemm(bf16*, float*, const bf16*, const bf16*, const int*,
                     const int*, int, int, int, int, int, hipStream_t);
void launch_moe_combine(bf16*, const bf16*, const float*, const int*, long,
                        int, int, hipStream_t);
void launch_moe_gemm_fp8(bf16*, float*, const unsigned char*, const float*,
                         const unsigned char*, const float*, const int*,
                         const int*, int, int, int, int, int, hipStream_t);
void launch_greedy_sample(long*, const bf16*, long, int, hipStream_t);
void launch_inv_cdf_sample(long*, const float*, const float*, long, int,
                           hipStream_t);

namespace {

hipStream_t stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check(const torch::Tensor& t, torch::ScalarType dt, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype");
}

bf16* bf(torch::Tensor& t) { return reinterpret_cast<bf16*>(t.data_ptr()); }
const bf16* cbf(const torch::Tensor& t) {
  return reinterpret_cast<const bf16*>(t.data_ptr());
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
             double eps) {
  check(out, torch::kBFloat16, "out");
  check(x, torch::kBFloat16, "x");
  check(w, torch::kBFloat16, "w");
  const int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden % 8 == 0 && hidden <= 16384, "hidden size");
  launch_rmsnorm(bf(out), cbf(x), cbf(w), (float)eps, x.numel() / hidden,
                 hidden, stream());
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor w, double eps) {
  check(x, torch::kBFloat16, "x");
  check(residual, torch::kBFloat16, "residual");
  const int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden % 8 == 0 && hidden <= 16384, "hidden size");
  launch_fused_add_rmsnorm(bf(x), bf(residual), cbf(w), (float)eps,
                           x.numel() / hidden, hidden, stream());
}

// q/k/v may be strided views into one fused qkv buffer: require only the
// head dim contiguous (stride(2)==1) and head stride == D.
void check_qkv_view(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 3 && t.stride(2) == 1 &&
                  t.stride(1) == t.size(2),
              name, " must be [T, H, D] with contiguous heads");
}

void rope_kv_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor positions, torch::Tensor cos_sin,
                    torch::Tensor slot_mapping, torch::Tensor k_cache,
                    torch::Tensor v_cache) {
  check_qkv_view(q, "q");
  check_qkv_view(k, "k");
  check_qkv_view(v, "v");
  check(cos_sin, torch::kFloat32, "cos_sin");
  check(positions, torch::kInt64, "positions");
  check(slot_mapping, torch::kInt64, "slot_mapping");
  const long T = q.size(0);
  const int H = (int)q.size(1), KV = (int)k.size(1), D = (int)q.size(2);
  const int BS = (int)k_cache.size(2);
  const bool kv_fp8 =
      k_cache.scalar_type() == torch::kFloat8_e4m3fn;
  launch_rope_kv_append(bf(q), bf(k), cbf(v),
                        positions.data_ptr<long>(),
                        cos_sin.data_ptr<float>(),
                        slot_mapping.data_ptr<long>(), bf(k_cache),
                        bf(v_cache), T, H, KV, D, BS, q.stride(0),
                        k.stride(0), v.stride(0), kv_fp8, stream());
}

void paged_attn(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                torch::Tensor v_cache, torch::Tensor block_tables,
                torch::Tensor ctx_lens,
                c10::optional<torch::Tensor> row_seq, double scale,
                c10::optional<torch::Tensor> split_ws, long split) {
  check(out, torch::kBFloat16, "out");
  check_qkv_view(q, "q");
  check(block_tables, torch::kInt32, "block_tables");
  check(ctx_lens, torch::kInt32, "ctx_lens");
  const long R = q.size(0);
  const int H = (int)q.size(1), D = (int)q.size(2);
  const int KV = (int)k_cache.size(1), BS = (int)k_cache.size(2);
  TORCH_CHECK(H % KV == 0, "GQA group");
  const int* rs = nullptr;
  if (row_seq.has_value()) {
    check(*row_seq, torch::kInt32, "row_seq");
    rs = row_seq->data_ptr<int>();
  }
  float* ws = nullptr;
  if (split > 1) {
    TORCH_CHECK(split_ws.has_value(), "split>1 needs a workspace");
    check(*split_ws, torch::kFloat, "split_ws");
    TORCH_CHECK(split_ws->numel() >= R * KV * split * (H / KV) * (D + 2),
                "split_ws too small");
    ws = split_ws->data_ptr<float>();
  }
  const bool kv_fp8 =
      k_cache.scalar_type() == torch::kFloat8_e4m3fn;
  TORCH_CHECK(v_cache.scalar_type() == k_cache.scalar_type(),
              "k/v cache dtype mismatch");
  launch_paged_attn(bf(out), cbf(q), cbf(k_cache), cbf(v_cache),
                    block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                    rs, R, KV, H / KV, D, (int)block_tables.size(1), BS,
                    (float)scale, q.stride(0), ws, (int)split, kv_fp8,
                    stream());
}

void attn_prefill_mfma(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor seq_start,
                       long max_seqlen, double scale) {
  check(out, torch::kBFloat16, "out");
  check_qkv_view(q, "q");
  check_qkv_view(k, "k");
  check_qkv_view(v, "v");
  check(seq_start, torch::kInt32, "seq_start");
  const int H = (int)q.size(1), D = (int)q.size(2);
  const int KV = (int)k.size(1);
  launch_attn_prefill_mfma(
      bf(out), cbf(q), cbf(k), cbf(v), seq_start.data_ptr<int>(),
      (int)seq_start.numel() - 1, (int)max_seqlen, H, KV, D, (float)scale,
      q.stride(0), k.stride(0), v.stride(0), stream());
}

void silu_mul(torch::Tensor out, torch::Tensor x) {
  check(out, torch::kBFloat16, "out");
  check(x, torch::kBFloat16, "x");
  const int inter = (int)out.size(-1);
  TORCH_CHECK(x.size(-1) == 2 * inter && inter % 8 == 0, "shape");
  launch_silu_mul(bf(out), cbf(x), out.numel() / inter, inter, stream());
}

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  check(out, torch::kInt64, "out");
  check(logits, torch::kBFloat16, "logits");
  launch_greedy_sample(out.data_ptr<long>(), cbf(logits), logits.size(0),
                       (int)logits.size(1), stream());
}

void inv_cdf_sample(torch::Tensor out, torch::Tensor logits,
                    torch::Tensor uniform) {
  check(out, torch::kInt64, "out");
  check(logits, torch::kFloat32, "logits");
  check(uniform, torch::kFloat32, "uniform");
  launch_inv_cdf_sample(out.data_ptr<long>(), logits.data_ptr<float>(),
                        uniform.data_ptr<float>(), logits.size(0),
                        (int)logits.size(1), stream());
}

}  // namespace


unsigned char* u8(torch::Tensor& t) {
  return reinterpret_cast<unsigned char*>(t.data_ptr());
}

void quant_fp8(torch::Tensor out, torch::Tensor scales, torch::Tensor x) {
  check(out, torch::kFloat8_e4m3fn, "out");
  check(scales, torch::kFloat, "scales");
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16, "x");
  const int hidden = (int)x.size(-1);
  TORCH_CHECK(x.stride(-1) == 1, "x inner dim must be contiguous");
  const long rows = x.numel() / hidden;
  const long stride = x.dim() > 1 ? x.stride(0) : hidden;
  launch_quant_fp8(u8(out), scales.data_ptr<float>(), cbf(x), rows, hidden,
                   stride, stream());
}

void fused_add_rmsnorm_fp8(torch::Tensor out, torch::Tensor scales,
                           torch::Tensor x, torch::Tensor residual,
                           torch::Tensor w, double eps) {
  check(out, torch::kFloat8_e4m3fn, "out");
  check(scales, torch::kFloat, "scales");
  check(x, torch::kBFloat16, "x");
  check(residual, torch::kBFloat16, "residual");
  check(w, torch::kBFloat16, "w");
  const int hidden = (int)x.size(-1);
  launch_fused_add_rmsnorm_fp8(u8(out), scales.data_ptr<float>(), cbf(x),
                               bf(residual), cbf(w), (float)eps,
                               x.numel() / hidden, hidden, stream());
}

void silu_mul_fp8(torch::Tensor out, torch::Tensor scales, torch::Tensor x) {
  check(out, torch::kFloat8_e4m3fn, "out");
  check(scales, torch::kFloat, "scales");
  check(x, torch::kBFloat16, "x");
  const int inter = (int)out.size(-1);
  TORCH_CHECK(x.size(-1) == 2 * inter, "x must be [rows, 2*inter]");
  launch_silu_mul_fp8(u8(out), scales.data_ptr<float>(), cbf(x),
                      out.numel() / inter, inter, stream());
}


void moe_align(torch::Tensor sorted_ids, torch::Tensor tile_expert,
               torch::Tensor inv_pos, torch::Tensor flat_ids,
               long num_experts) {
  check(sorted_ids, torch::kInt, "sorted_ids");
  check(tile_expert, torch::kInt, "tile_expert");
  check(inv_pos, torch::kInt, "inv_pos");
  check(flat_ids, torch::kInt, "flat_ids");
  launch_moe_align(sorted_ids.data_ptr<int>(), tile_expert.data_ptr<int>(),
                   inv_pos.data_ptr<int>(), flat_ids.data_ptr<int>(),
                   (int)flat_ids.numel(), (int)num_experts,
                   (int)tile_expert.numel(), stream());
}

void moe_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
              torch::Tensor sorted_ids, torch::Tensor tile_expert,
              long gather_div, long splitk,
              c10::optional<torch::Tensor> ws) {
  check(out, torch::kBFloat16, "out");
  check(x, torch::kBFloat16, "x");
  check(w, torch::kBFloat16, "w");
  check(sorted_ids, torch::kInt, "sorted_ids");
  check(tile_expert, torch::kInt, "tile_expert");
  TORCH_CHECK(w.dim() == 3, "w must be [E, N, K]");
  float* wsp = nullptr;
  if (splitk > 1) {
    TORCH_CHECK(ws.has_value(), "moe_gemm: splitk>1 needs a workspace");
    check(*ws, torch::kFloat, "ws");
    TORCH_CHECK(ws->numel() >= splitk * out.numel(),
                "moe_gemm: workspace too small");
    wsp = ws->data_ptr<float>();
  }
  launch_moe_gemm(bf(out), wsp, cbf(x), cbf(w),
                  sorted_ids.data_ptr<int>(),
                  tile_expert.data_ptr<int>(), (int)tile_expert.numel(),
                  (int)w.size(1), (int)w.size(2), (int)gather_div,
                  (int)splitk, stream());
}

void moe_combine(torch::Tensor out, torch::Tensor y, torch::Tensor wts,
                 torch::Tensor inv_pos, long topk) {
  check(out, torch::kBFloat16, "out");
  check(y, torch::kBFloat16, "y");
  check(wts, torch::kFloat, "wts");
  check(inv_pos, torch::kInt, "inv_pos");
  launch_moe_combine(bf(out), cbf(y), wts.data_ptr<float>(),
                     inv_pos.data_ptr<int>(), out.size(0), (int)topk,
                     (int)out.size(-1), stream());
}


void moe_gemm_fp8(torch::Tensor out, torch::Tensor xq, torch::Tensor xs,
                  torch::Tensor wq, torch::Tensor ws,
                  torch::Tensor sorted_ids, torch::Tensor tile_expert,
                  long gather_div, long splitk,
                  c10::optional<torch::Tensor> skw) {
  check(out, torch::kBFloat16, "out");
  check(xq, torch::kFloat8_e4m3fn, "xq");
  check(xs, torch::kFloat, "xs");
  check(wq, torch::kFloat8_e4m3fn, "wq");
  check(ws, torch::kFloat, "ws");
  check(sorted_ids, torch::kInt, "sorted_ids");
  check(tile_expert, torch::kInt, "tile_expert");
  TORCH_CHECK(wq.dim() == 3, "wq must be [E, N, K]");
  float* skp = nullptr;
  if (splitk > 1) {
    TORCH_CHECK(skw.has_value(),
                "moe_gemm_fp8: splitk>1 needs a workspace");
    check(*skw, torch::kFloat, "skw");
    TORCH_CHECK(skw->numel() >= splitk * out.numel(),
                "moe_gemm_fp8: workspace too small");
    skp = skw->data_ptr<float>();
  }
  launch_moe_gemm_fp8(bf(out), skp, u8(xq), xs.data_ptr<float>(), u8(wq),
                      ws.data_ptr<float>(), sorted_ids.data_ptr<int>(),
                      tile_expert.data_ptr<int>(),
                      (int)tile_expert.numel(), (int)wq.size(1),
                      (int)wq.size(2), (int)gather_div, (int)splitk,
                      stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "HyperSpot-AMD gfx950 (CDNA4) kernels";
  m.def("rmsnorm", &rmsnorm);
  m.def("quant_fp8", &quant_fp8);
  m.def("fused_add_rmsnorm_fp8", &fused_add_rmsnorm_fp8);
  m.def("silu_mul_fp8", &silu_mul_fp8);
  m.def("moe_align", &moe_align);
  m.def("moe_gemm", &moe_gemm);
  m.def("moe_combine", &moe_combine);
  m.def("moe_gemm_fp8", &moe_gemm_fp8);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("rope_kv_append", &rope_kv_append);
  m.def("paged_attn", &paged_attn);
  m.def("attn_prefill_mfma", &attn_prefill_mfma);
  m.def("silu_mul", &silu_mul);
  m.def("greedy_sample", &greedy_sample);
  m.def("inv_cdf_sample", &inv_cdf_sample);
}
