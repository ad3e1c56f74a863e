// Python bindings for the perceiver_amd CDNA4 (gfx950) kernel extension.
#include <torch/extension.h>

torch::Tensor gelu_bias_fwd(torch::Tensor x, c10::optional<torch::Tensor> bias);
torch::Tensor gelu_bias_bwd(torch::Tensor x, c10::optional<torch::Tensor> bias, torch::Tensor dy);

std::vector<torch::Tensor> flash_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                     c10::optional<torch::Tensor> pad_mask, bool causal,
                                     double dropout_p, int64_t seed);
std::vector<torch::Tensor> flash_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                     c10::optional<torch::Tensor> pad_mask, bool causal,
                                     double dropout_p, int64_t seed);
std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> b,
                                  double eps);
std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                  torch::Tensor mean, torch::Tensor rstd, bool needs_dwdb);
bool flash_supported_impl(long d_qk, long d_v, long needs_dropout);
void adamw_step(torch::Tensor master, torch::Tensor m, torch::Tensor v, torch::Tensor g,
                double lr, double beta1, double beta2, double eps, double weight_decay,
                int64_t step);
torch::Tensor rotary_apply(torch::Tensor t, torch::Tensor frq, int64_t rot, bool neg_sin);
torch::Tensor dropout_add_fwd(torch::Tensor x, torch::Tensor res, double p, int64_t seed);
torch::Tensor dropout_add_bwd(torch::Tensor dy, double p, int64_t seed);
torch::Tensor colsum_bf16(torch::Tensor x);
torch::Tensor gemm_bt_bf16(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> gemm_bt_gelu_bf16(torch::Tensor x, torch::Tensor w,
                                             c10::optional<torch::Tensor> bias);
bool gemm_bt_applicable(long M, long N, long K);
torch::Tensor transpose_bf16(torch::Tensor x);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("gelu_bias_fwd", &gelu_bias_fwd, "fused bias+GELU forward (bf16)");
    m.def("gelu_bias_bwd", &gelu_bias_bwd, "fused bias+GELU backward (bf16)");
    m.def("flash_fwd", &flash_fwd, "fused flash attention forward");
    m.def("flash_bwd", &flash_bwd, "fused flash attention backward");
    m.def("flash_supported", &flash_supported_impl, "shape gate for the flash kernel");
    m.def("ln_fwd", &ln_fwd, "fused bf16 LayerNorm forward");
    m.def("ln_bwd", &ln_bwd, "fused bf16 LayerNorm backward");
    m.def("adamw_step", &adamw_step, "single-pass fused AdamW on flat fp32 state");
    m.def("rotary_apply", &rotary_apply, "fused rotary embedding (bf16, fwd/bwd via neg_sin)");
    m.def("dropout_add_fwd", &dropout_add_fwd, "fused residual dropout-add forward (bf16)");
    m.def("dropout_add_bwd", &dropout_add_bwd, "fused residual dropout-add backward (bf16)");
    m.def("colsum_bf16", &colsum_bf16, "coalesced bf16 column sum (bias gradients)");
    m.def("gemm_bt", &gemm_bt_bf16, "deep-pipeline bf16 GEMM: x @ w^T (+ bias)");
    m.def("transpose_bf16", &transpose_bf16, "LDS-tiled bf16 2-D transpose");
    m.def("gemm_bt_gelu", &gemm_bt_gelu_bf16,
          "deep-pipeline bf16 GEMM with fused GELU epilogue: returns (pre, gelu(pre+bias))");
    m.def("gemm_bt_applicable", [](int64_t M, int64_t N, int64_t K) {
        return gemm_bt_applicable(M, N, K);
    }, "shape gate for the custom GEMM");
}
