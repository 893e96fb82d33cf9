// Python bindings for the pipeedge-amd gfx950 kernel extension.
//
// The Python-side dispatch (pipeedge_amd/ops/__init__.py) calls these for
// every GPU tensor; there is no eager fallback on GPU (fail-loud contract).
#include <torch/extension.h>

std::tuple<torch::Tensor, torch::Tensor> gemm_bf16(
        torch::Tensor a2d, torch::Tensor b,
        c10::optional<torch::Tensor> bias, int64_t act,
        c10::optional<torch::Tensor> residual2d, int64_t tile = 0,
        c10::optional<torch::Tensor> ln_stats = c10::nullopt,
        c10::optional<torch::Tensor> ln_s = c10::nullopt,
        c10::optional<torch::Tensor> ln_c = c10::nullopt,
        double ln_eps = 1e-12, bool want_stats = false);
torch::Tensor ln_stats_bf16(torch::Tensor x);
torch::Tensor debug_mfma_tile(torch::Tensor a, torch::Tensor b);
torch::Tensor layernorm_bf16(torch::Tensor x, torch::Tensor weight,
                             torch::Tensor bias, double eps);
torch::Tensor add_layernorm_bf16(torch::Tensor x, torch::Tensor residual,
                                 torch::Tensor weight, torch::Tensor bias,
                                 double eps);
torch::Tensor qkv_attention_bf16(torch::Tensor qkv, int64_t num_heads,
                                 double scale,
                                 c10::optional<torch::Tensor> attn_mask);
torch::Tensor patchify_bf16(torch::Tensor pixels, int64_t patch_size);
torch::Tensor embed_finish_bf16(torch::Tensor x, torch::Tensor prefix,
                                torch::Tensor pos);
torch::Tensor debug_mfma32_raw(torch::Tensor a, torch::Tensor b);
torch::Tensor debug_tr16_raw();
torch::Tensor debug_permlane32_raw();
std::vector<torch::Tensor> quant_encode_bf16(torch::Tensor x, int64_t bit,
                                             int64_t pad_words);
torch::Tensor quant_decode_bf16(torch::Tensor comm, torch::Tensor scale,
                                torch::Tensor shift, int64_t bit,
                                std::vector<int64_t> image_shape);

namespace {

// reshape [.., K] -> [M, K] for the GEMM, restoring leading dims after
std::tuple<torch::Tensor, torch::Tensor> linear_op(
        torch::Tensor x, torch::Tensor w,
        c10::optional<torch::Tensor> bias, int64_t act,
        c10::optional<torch::Tensor> residual, int64_t tile = 0,
        c10::optional<torch::Tensor> ln_stats = c10::nullopt,
        c10::optional<torch::Tensor> ln_s = c10::nullopt,
        c10::optional<torch::Tensor> ln_c = c10::nullopt,
        double ln_eps = 1e-12, bool want_stats = false) {
    auto sizes = x.sizes().vec();
    const long K = sizes.back();
    auto x2d = x.reshape({-1, K});
    c10::optional<torch::Tensor> res2d;
    if (residual.has_value())
        res2d = residual->reshape({-1, w.size(0)});  // w is [N, K]
    auto [out, stats] = gemm_bf16(x2d, w, bias, act, res2d, tile, ln_stats,
                                  ln_s, ln_c, ln_eps, want_stats);
    sizes.back() = w.size(0);
    return {out.reshape(sizes), stats};
}

torch::Tensor qkv_attention_op(torch::Tensor qkv, int64_t num_heads,
                               double scale,
                               c10::optional<torch::Tensor> attn_mask) {
    // mask: [B, S] additive bias per kv position (BERT padding); broadcast
    // shapes [B,1,1,S] / [B,1,S] collapse to [B,S]
    if (attn_mask.has_value()) {
        auto m = attn_mask.value();
        while (m.dim() > 2 && m.size(1) == 1)
            m = m.squeeze(1);
        attn_mask = m.contiguous();
    }
    return qkv_attention_bf16(qkv, num_heads, scale, attn_mask);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("linear", &linear_op, "fused GEMM+bias+act+residual (bf16 MFMA)",
          py::arg("x"), py::arg("weight"), py::arg("bias") = py::none(),
          py::arg("act") = 0, py::arg("residual") = py::none(),
          py::arg("tile") = 0, py::arg("ln_stats") = py::none(),
          py::arg("ln_s") = py::none(), py::arg("ln_c") = py::none(),
          py::arg("ln_eps") = 1e-12, py::arg("want_stats") = false);
    m.def("ln_stats", &ln_stats_bf16, py::arg("x"));
    m.def("layernorm", &layernorm_bf16, py::arg("x"), py::arg("weight"),
          py::arg("bias"), py::arg("eps") = 1e-6);
    m.def("add_layernorm", &add_layernorm_bf16, py::arg("x"),
          py::arg("residual"), py::arg("weight"), py::arg("bias"),
          py::arg("eps") = 1e-6);
    m.def("qkv_attention", &qkv_attention_op, py::arg("qkv"),
          py::arg("num_heads"), py::arg("scale"),
          py::arg("attn_mask") = py::none());
    m.def("patchify", &patchify_bf16, py::arg("pixels"),
          py::arg("patch_size"));
    m.def("embed_finish", &embed_finish_bf16, py::arg("x"), py::arg("prefix"),
          py::arg("pos"));
    m.def("quant_encode", &quant_encode_bf16, py::arg("x"), py::arg("bit"),
          py::arg("pad_words") = 0);
    m.def("quant_decode", &quant_decode_bf16, py::arg("comm"),
          py::arg("scale"), py::arg("shift"), py::arg("bit"),
          py::arg("image_shape"));
    m.def("_debug_mfma_tile", &debug_mfma_tile);
    m.def("_debug_mfma32_raw", &debug_mfma32_raw);
    m.def("_debug_tr16_raw", &debug_tr16_raw);
    m.def("_debug_permlane32_raw", &debug_permlane32_raw);
}
