// genrec_amd._C — single op library for the CDNA4 kernel layer.
#include <torch/extension.h>

namespace genrec {

std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w,
                                        double eps, bool t5_style);
std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, torch::Tensor inv_rms,
                                        bool t5_style);
std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x, double eps);
torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor x,
                         torch::Tensor inv_norm, double eps);

std::vector<torch::Tensor> attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> key_pad,
    c10::optional<torch::Tensor> add_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, bool causal, int64_t act, double dropout_p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev);
std::vector<torch::Tensor> attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor p_saved, torch::Tensor drop_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, int64_t act, double dropout_p, int64_t seed,
    bool bias_grad, int64_t bias_dim);

std::vector<torch::Tensor> attn_fwd_mfma(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> key_pad,
    c10::optional<torch::Tensor> add_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, bool causal, int64_t act, double dropout_p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev,
    c10::optional<torch::Tensor> bias_bucket);
std::vector<torch::Tensor> attn_bwd_mfma(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor p_saved, torch::Tensor drop_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, int64_t act, double dropout_p, int64_t seed,
    bool bias_grad, int64_t bias_dim,
    c10::optional<torch::Tensor> bias_bucket, int64_t n_buckets);

std::vector<torch::Tensor> hstu_attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor pos_bucket, torch::Tensor pos_table,
    c10::optional<torch::Tensor> time_table,
    c10::optional<torch::Tensor> timestamps,
    c10::optional<torch::Tensor> key_pad);
std::vector<torch::Tensor> hstu_attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor s_saved, torch::Tensor pos_bucket,
    c10::optional<torch::Tensor> timestamps,
    int64_t n_pos, int64_t n_time);

std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor targets,
                                          int64_t ignore_index);
torch::Tensor softmax_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor targets, torch::Tensor lse,
                             torch::Tensor n_valid, int64_t ignore_index);

std::vector<torch::Tensor> sqdist_argmin(torch::Tensor x,
                                         torch::Tensor codebook);
torch::Tensor embedding_fwd(torch::Tensor weight, torch::Tensor indices);
torch::Tensor embedding_bwd(torch::Tensor dy, torch::Tensor indices,
                            int64_t num_weights, int64_t padding_idx);
std::vector<torch::Tensor> dropout_add_fwd(
    torch::Tensor x, torch::Tensor residual, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev);
std::vector<torch::Tensor> relu_dropout_fwd(
    torch::Tensor x, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev);
std::vector<torch::Tensor> plain_dropout_fwd(
    torch::Tensor x, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev);
torch::Tensor colsum(torch::Tensor x);
torch::Tensor chunk_sum(torch::Tensor x);
std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor w,
                                          torch::Tensor b, double eps);
std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w, torch::Tensor mean,
                                          torch::Tensor rstd);
torch::Tensor dropout_fuse_bwd(torch::Tensor dy, torch::Tensor mask,
                               double p, bool relu);
torch::Tensor topk_hit_ranks(torch::Tensor actual, torch::Tensor topk);
void fused_adamw(torch::Tensor master, torch::Tensor grad, torch::Tensor m,
                 torch::Tensor v, torch::Tensor out_p, torch::Tensor lr,
                 torch::Tensor scale, torch::Tensor step, double beta1,
                 double beta2, double eps, double weight_decay);
torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias);
torch::Tensor skinny_gemm_tn(torch::Tensor x, torch::Tensor w,
                             c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> attn_fwd_flash(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> key_pad,
    c10::optional<torch::Tensor> add_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, bool causal, double dropout_p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev);
std::vector<torch::Tensor> attn_bwd_flash(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor s_saved, torch::Tensor ml,
    torch::Tensor drop_mask, c10::optional<torch::Tensor> query_mask,
    double scale, double dropout_p, bool bias_grad, int64_t bias_dim);

}  // namespace genrec

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm_fwd", &genrec::rms_norm_fwd, "fused RMSNorm forward");
  m.def("rms_norm_bwd", &genrec::rms_norm_bwd, "fused RMSNorm backward");
  m.def("l2norm_fwd", &genrec::l2norm_fwd, "fused L2Norm forward");
  m.def("l2norm_bwd", &genrec::l2norm_bwd, "fused L2Norm backward");
  m.def("attn_fwd", &genrec::attn_fwd, "fused attention forward");
  m.def("attn_bwd", &genrec::attn_bwd, "fused attention backward");
  m.def("attn_fwd_mfma", &genrec::attn_fwd_mfma, "MFMA attention forward",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("bias"),
        py::arg("key_pad"), py::arg("add_mask"), py::arg("query_mask"),
        py::arg("scale"), py::arg("causal"), py::arg("act"),
        py::arg("dropout_p"), py::arg("seed"), py::arg("seed_dev"),
        py::arg("bias_bucket") = py::none());
  m.def("attn_bwd_mfma", &genrec::attn_bwd_mfma, "MFMA attention backward",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("p_saved"), py::arg("drop_mask"), py::arg("query_mask"),
        py::arg("scale"), py::arg("act"), py::arg("dropout_p"),
        py::arg("seed"), py::arg("bias_grad"), py::arg("bias_dim"),
        py::arg("bias_bucket") = py::none(),
        py::arg("n_buckets") = 0);
  m.def("hstu_attn_fwd", &genrec::hstu_attn_fwd, "HSTU fused attention fwd");
  m.def("hstu_attn_bwd", &genrec::hstu_attn_bwd, "HSTU fused attention bwd");
  m.def("softmax_ce_fwd", &genrec::softmax_ce_fwd, "fused CE forward");
  m.def("softmax_ce_bwd", &genrec::softmax_ce_bwd, "fused CE backward");
  m.def("sqdist_argmin", &genrec::sqdist_argmin, "L2 dist + argmin");
  m.def("embedding_fwd", &genrec::embedding_fwd, "embedding gather");
  m.def("embedding_bwd", &genrec::embedding_bwd, "embedding scatter-add bwd");
  m.def("dropout_add_fwd", &genrec::dropout_add_fwd, "residual+dropout fwd");
  m.def("relu_dropout_fwd", &genrec::relu_dropout_fwd, "dropout(relu) fwd");
  m.def("plain_dropout_fwd", &genrec::plain_dropout_fwd,
        "graph-replay-safe plain dropout fwd");
  m.def("colsum", &genrec::colsum,
        "deterministic replay-safe column sum (bias grads)");
  m.def("chunk_sum", &genrec::chunk_sum,
        "small-row-count column sum (split-K dW partials)");
  m.def("layer_norm_fwd", &genrec::layer_norm_fwd, "fused LayerNorm fwd");
  m.def("layer_norm_bwd", &genrec::layer_norm_bwd, "fused LayerNorm bwd");
  m.def("dropout_fuse_bwd", &genrec::dropout_fuse_bwd, "fused dropout bwd");
  m.def("topk_hit_ranks", &genrec::topk_hit_ranks, "first-match ranks");
  m.def("fused_adamw", &genrec::fused_adamw,
        "fused flat AdamW step (device lr/scale/step scalars)");
  m.def("skinny_gemm", &genrec::skinny_gemm,
        "tall-skinny linear fwd GEMM (bf16, fp32 accum)");
  m.def("skinny_gemm_tn", &genrec::skinny_gemm_tn,
        "tall-skinny dX GEMM: x[M,K] @ w[K,N] (transposed-B staging)");
  m.def("attn_fwd_flash", &genrec::attn_fwd_flash,
        "flash-tiled attention fwd (staged; GENREC_ATTN_FLASH=1)");
  m.def("attn_bwd_flash", &genrec::attn_bwd_flash,
        "flash-tiled attention bwd (staged; GENREC_ATTN_FLASH=1)");
}
