// Python bindings for the petals_amd CDNA4 kernel suite.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor w, double eps);
torch::Tensor rms_norm_f32out(torch::Tensor x, torch::Tensor w, double eps);
torch::Tensor layer_norm(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps);
torch::Tensor layer_norm_f32out(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps);
torch::Tensor swiglu(torch::Tensor gate, torch::Tensor up);
void warm_spin(int64_t microseconds);
std::vector<torch::Tensor> apply_rope(
    torch::Tensor q, torch::Tensor k, torch::Tensor cos_t, torch::Tensor sin_t, torch::Tensor pos);
void rope_cache_write(
    torch::Tensor qkv, torch::Tensor cos_t, torch::Tensor sin_t, torch::Tensor pos,
    torch::Tensor k_cache, torch::Tensor v_cache, int64_t qh, int64_t kh);
void kv_cache_write(
    torch::Tensor qkv, torch::Tensor pos,
    torch::Tensor k_cache, torch::Tensor v_cache, int64_t qh, int64_t kh);
torch::Tensor qkv_rope_reduce(
    torch::Tensor partials, c10::optional<torch::Tensor> cos_t, c10::optional<torch::Tensor> sin_t,
    torch::Tensor pos, torch::Tensor k_cache, torch::Tensor v_cache,
    int64_t qh, int64_t kh, bool rope, c10::optional<torch::Tensor> bias);
torch::Tensor gemv_bf16(
    torch::Tensor wt, torch::Tensor x, torch::Tensor workspace,
    c10::optional<torch::Tensor> residual, int64_t epilogue, int64_t splits_override,
    c10::optional<torch::Tensor> bias);
torch::Tensor attn_decode_fused(
    torch::Tensor q, torch::Tensor k_cache, torch::Tensor v_cache, torch::Tensor kv_len,
    int64_t gq, int64_t n_splits, torch::Tensor part_o, torch::Tensor part_ml, double scale,
    c10::optional<torch::Tensor> alibi_slopes);
torch::Tensor attn_prefill_fused(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    int64_t kv_len, int64_t kv_offset, double scale, bool causal,
    c10::optional<torch::Tensor> alibi_slopes);
std::vector<torch::Tensor> nf4_quantize(torch::Tensor w);
torch::Tensor gemv_int8(
    torch::Tensor q, torch::Tensor scale, torch::Tensor x, torch::Tensor workspace,
    c10::optional<torch::Tensor> residual, int64_t epilogue, int64_t splits_override,
    c10::optional<torch::Tensor> bias);
torch::Tensor nf4_dequantize(torch::Tensor packed, torch::Tensor absmax);
torch::Tensor gemv_nf4(
    torch::Tensor packed, torch::Tensor absmax, torch::Tensor x, torch::Tensor workspace,
    c10::optional<torch::Tensor> residual, int64_t epilogue, int64_t splits_override,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> absmax_t,
    c10::optional<torch::Tensor> x_parts, double fold_eps,
    c10::optional<torch::Tensor> sumsq_out);
torch::Tensor sumsq_rows(torch::Tensor x);
torch::Tensor gemv_bf16_moe(
    torch::Tensor wt_all, torch::Tensor x, torch::Tensor sel, int64_t k_per_tok,
    torch::Tensor workspace, int64_t epilogue, int64_t splits_override);
torch::Tensor gemv_nf4_moe(
    torch::Tensor packed_all, torch::Tensor absmax_all, torch::Tensor x, torch::Tensor sel,
    int64_t k_per_tok, torch::Tensor workspace, int64_t epilogue, int64_t splits_override);
torch::Tensor moe_gemm(
    c10::optional<torch::Tensor> wt_all, c10::optional<torch::Tensor> packed_all,
    c10::optional<torch::Tensor> absmax_all, torch::Tensor x, torch::Tensor sorted_pairs,
    torch::Tensor tile_expert, int64_t k_per_tok, int64_t n_rows);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm, "RMSNorm (bf16 -> bf16)");
  m.def("rms_norm_f32out", &rms_norm_f32out, "RMSNorm (bf16 -> f32)");
  m.def("layer_norm", &layer_norm, "LayerNorm with weight+bias (bf16 -> bf16)");
  m.def("layer_norm_f32out", &layer_norm_f32out, "LayerNorm with weight+bias (bf16 -> f32)");
  m.def("swiglu", &swiglu, "silu(gate) * up (bf16)");
  m.def("warm_spin", &warm_spin, "occupy the device queue for ~N us (keep-warm)");
  m.def("apply_rope", &apply_rope, "rotate q,k by positions (bf16)");
  m.def("rope_cache_write", &rope_cache_write, "fused decode rope + kv cache write");
  m.def("kv_cache_write", &kv_cache_write, "decode kv cache write without rope (ALiBi families)");
  m.def("qkv_rope_reduce", &qkv_rope_reduce,
        "fused qkv split-K reduce + rope + kv-cache write (consumes RAW gemv partials)",
        py::arg("partials"), py::arg("cos_t"), py::arg("sin_t"), py::arg("pos"),
        py::arg("k_cache"), py::arg("v_cache"), py::arg("qh"), py::arg("kh"), py::arg("rope") = true,
        py::arg("bias") = py::none());
  m.def("gemv_bf16", &gemv_bf16, "split-K bf16 gemv with fused epilogue",
        py::arg("wt"), py::arg("x"), py::arg("workspace"), py::arg("residual"), py::arg("epilogue"),
        py::arg("splits") = 0, py::arg("bias") = py::none());
  m.def("attn_decode_fused", &attn_decode_fused, "GQA decode attention (flash-decoding)",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"), py::arg("kv_len"), py::arg("gq"),
        py::arg("n_splits"), py::arg("part_o"), py::arg("part_ml"), py::arg("scale"),
        py::arg("alibi_slopes") = py::none());
  m.def("attn_prefill_fused", &attn_prefill_fused, "MFMA flash prefill attention (bf16, causal, GQA)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("kv_len"), py::arg("kv_offset"),
        py::arg("scale"), py::arg("causal"), py::arg("alibi_slopes") = py::none());
  m.def("gemv_int8", &gemv_int8, "split-K weight-only int8 gemv (per-column scales) with epilogue",
        py::arg("q"), py::arg("scale"), py::arg("x"), py::arg("workspace"),
        py::arg("residual"), py::arg("epilogue"), py::arg("splits") = 0, py::arg("bias") = py::none());
  m.def("nf4_quantize", &nf4_quantize, "blockwise NF4 quantize [in,out] bf16 -> (packed u8, absmax bf16)");
  m.def("nf4_dequantize", &nf4_dequantize, "NF4 -> bf16");
  m.def("gemv_nf4", &gemv_nf4, "split-K NF4 gemv (dequant fused) with epilogue; "
        "x_parts switches to the folded-RMSNorm mode (bf16 x + sum(h^2) partials)",
        py::arg("packed"), py::arg("absmax"), py::arg("x"), py::arg("workspace"),
        py::arg("residual"), py::arg("epilogue"), py::arg("splits") = 0, py::arg("bias") = py::none(),
        py::arg("absmax_t") = py::none(), py::arg("x_parts") = py::none(),
        py::arg("fold_eps") = 0.0, py::arg("sumsq_out") = py::none());
  m.def("sumsq_rows", &sumsq_rows, "per-row sum(x^2) of a bf16 matrix -> [rows, 1] f32");
  m.def("gemv_bf16_moe", &gemv_bf16_moe,
        "device-routed MoE gemv: stacked bf16 expert weights, expert ids from a device tensor",
        py::arg("wt_all"), py::arg("x"), py::arg("sel"), py::arg("k_per_tok"),
        py::arg("workspace"), py::arg("epilogue"), py::arg("splits") = 0);
  m.def("gemv_nf4_moe", &gemv_nf4_moe,
        "device-routed MoE gemv: stacked NF4 expert weights, expert ids from a device tensor",
        py::arg("packed_all"), py::arg("absmax_all"), py::arg("x"), py::arg("sel"),
        py::arg("k_per_tok"), py::arg("workspace"), py::arg("epilogue"), py::arg("splits") = 0);
  m.def("moe_gemm", &moe_gemm,
        "grouped MFMA GEMM over expert-sorted (token, slot) pairs (prefill; NF4 dequant fused "
        "into the LDS B-tile staging)",
        py::arg("wt_all"), py::arg("packed_all"), py::arg("absmax_all"), py::arg("x"),
        py::arg("sorted_pairs"), py::arg("tile_expert"), py::arg("k_per_tok"), py::arg("n_rows"));
}
