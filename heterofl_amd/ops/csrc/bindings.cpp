// Python bindings for the gfx950 HIP kernels (_heterofl_hip).
#include <torch/extension.h>

std::vector<at::Tensor> bn_relu_fwd(at::Tensor x, at::Tensor gamma,
                                    at::Tensor beta, double eps);
std::vector<at::Tensor> bn_relu_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor gamma, at::Tensor beta,
                                    at::Tensor mean, at::Tensor invstd);
std::vector<at::Tensor> gn_relu_fwd(at::Tensor x, at::Tensor gamma,
                                    at::Tensor beta, int64_t G, double eps);
std::vector<at::Tensor> gn_relu_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor gamma, at::Tensor beta,
                                    at::Tensor mean, at::Tensor invstd,
                                    int64_t G);
std::vector<at::Tensor> masked_ce_fwd(at::Tensor scores, at::Tensor labels,
                                      at::Tensor mask, at::Tensor metrics);
at::Tensor masked_ce_bwd(at::Tensor scores, at::Tensor labels, at::Tensor mask,
                         at::Tensor up);
std::vector<at::Tensor> build_chunk_table(std::vector<at::Tensor> grads,
                                          std::vector<at::Tensor> params,
                                          std::vector<at::Tensor> bufs,
                                          int64_t R, int64_t chunk_elems,
                                          std::vector<at::Tensor> shadows);
void fill_chunk_table(at::Tensor blob, std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> params,
                      std::vector<at::Tensor> bufs, int64_t R,
                      int64_t chunk_elems, std::vector<at::Tensor> shadows);
void clip_sgd_step(at::Tensor table_blob, int64_t n_chunks,
                   at::Tensor chunk_client, at::Tensor partials,
                   at::Tensor normsq, double max_norm, double lr,
                   double momentum, double weight_decay);
at::Tensor conv_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                    at::Tensor residual, int64_t groups, int64_t stride,
                    int64_t pad, int64_t fp8);
at::Tensor conv_bwd_data(at::Tensor dy, at::Tensor w, int64_t groups,
                         int64_t stride, int64_t pad, int64_t H, int64_t W,
                         int64_t fp8);
at::Tensor conv_bwd_weight(at::Tensor dy, at::Tensor x, int64_t groups,
                           int64_t stride, int64_t pad, int64_t khw,
                           int64_t fp8);
at::Tensor mfma_probe(at::Tensor A, at::Tensor B);
std::vector<at::Tensor> head_fwd(at::Tensor feat, at::Tensor w, at::Tensor b,
                                 int64_t R);
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double temperature);
std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor p,
                                 double temperature);
std::vector<at::Tensor> attn_fwd_block(at::Tensor q, at::Tensor k,
                                       at::Tensor v, double temperature);
std::vector<at::Tensor> attn_bwd_block(at::Tensor dout, at::Tensor q,
                                       at::Tensor k, at::Tensor v,
                                       at::Tensor out, at::Tensor lse,
                                       double temperature);
std::vector<at::Tensor> ln_fwd(at::Tensor x, at::Tensor gamma,
                               at::Tensor beta, int64_t R, double eps);
std::vector<at::Tensor> ln_bwd(at::Tensor dy, at::Tensor x, at::Tensor gamma,
                               at::Tensor mean, at::Tensor invstd, int64_t R);
std::vector<at::Tensor> lm_ce_fwd(at::Tensor logits, at::Tensor labels,
                                  at::Tensor mask, int64_t R);
at::Tensor lm_ce_bwd(at::Tensor logits, at::Tensor labels, at::Tensor mask,
                     at::Tensor row_lse, at::Tensor up, int64_t R);
std::vector<at::Tensor> head_bwd(at::Tensor dscores, at::Tensor pooled,
                                 at::Tensor w, int64_t R, int64_t H,
                                 int64_t W, bool bf16_feat, bool want_db);
void rng_bump(at::Tensor seed);
std::vector<at::Tensor> gelu_drop_fwd(at::Tensor x, at::Tensor seed,
                                      int64_t salt, double rate, double p);
at::Tensor gelu_drop_bwd(at::Tensor dy, at::Tensor x, at::Tensor mask,
                         double rate, double p);
std::vector<at::Tensor> res_drop_fwd(at::Tensor src, at::Tensor h,
                                     at::Tensor seed, int64_t salt,
                                     double rate, double p);
at::Tensor drop_scale_bwd(at::Tensor dt, at::Tensor mask, double rate,
                          double p);
at::Tensor token_mask(at::Tensor tokens, at::Tensor seed, int64_t salt,
                      double rate, int64_t mask_id);
at::Tensor embed_pos_fwd(at::Tensor ids, at::Tensor table, at::Tensor pos,
                         double rate);
std::vector<at::Tensor> embed_pos_bwd(at::Tensor dy, at::Tensor ids,
                                      int64_t V, int64_t P, double rate);
std::vector<at::Tensor> maxpool2_fwd(at::Tensor x);
void pack_slices(at::Tensor blob, int64_t n, int64_t max_rows);
at::Tensor maxpool2_bwd(at::Tensor dy, at::Tensor arg, int64_t H,
                        int64_t W);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("bn_relu_fwd", &bn_relu_fwd, "fused sBN+ReLU forward");
    m.def("bn_relu_bwd", &bn_relu_bwd, "fused sBN+ReLU backward");
    m.def("gn_relu_fwd", &gn_relu_fwd, "fused GroupNorm+ReLU forward");
    m.def("gn_relu_bwd", &gn_relu_bwd, "fused GroupNorm+ReLU backward");
    m.def("masked_ce_fwd", &masked_ce_fwd, "batched masked CE forward");
    m.def("masked_ce_bwd", &masked_ce_bwd, "batched masked CE backward");
    m.def("build_chunk_table", &build_chunk_table,
          "build clip+SGD chunk table");
    m.def("clip_sgd_step", &clip_sgd_step, "fused per-client clip+SGD step");
    m.def("fill_chunk_table", &fill_chunk_table,
          "fill a preallocated chunk table post-capture");
    m.def("conv_fwd", &conv_fwd, "MFMA implicit-GEMM grouped conv forward");
    m.def("conv_bwd_data", &conv_bwd_data, "MFMA grouped conv backward-data");
    m.def("conv_bwd_weight", &conv_bwd_weight,
          "MFMA grouped conv backward-weight (fp32 out)");
    m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
    m.def("head_fwd", &head_fwd, "fused avgpool+per-client-linear forward");
    m.def("head_bwd", &head_bwd, "fused head backward");
    m.def("attn_fwd", &attn_fwd, "fused small-S attention forward");
    m.def("attn_bwd", &attn_bwd, "fused small-S attention backward");
    m.def("attn_fwd_block", &attn_fwd_block,
          "blockwise (flash-style) attention forward, any S");
    m.def("attn_bwd_block", &attn_bwd_block,
          "blockwise attention backward (deterministic two-pass)");
    m.def("ln_fwd", &ln_fwd, "fused per-client LayerNorm forward");
    m.def("ln_bwd", &ln_bwd, "fused per-client LayerNorm backward");
    m.def("lm_ce_fwd", &lm_ce_fwd, "fused vocab-masked LM CE forward");
    m.def("lm_ce_bwd", &lm_ce_bwd, "fused vocab-masked LM CE backward");
    m.def("rng_bump", &rng_bump, "advance the device RNG seed cell");
    m.def("gelu_drop_fwd", &gelu_drop_fwd, "fused scaler+GELU+dropout fwd");
    m.def("gelu_drop_bwd", &gelu_drop_bwd, "fused scaler+GELU+dropout bwd");
    m.def("res_drop_fwd", &res_drop_fwd,
          "fused scaler+dropout+residual-add fwd");
    m.def("drop_scale_bwd", &drop_scale_bwd,
          "dropout-mask * 1/rate backward");
    m.def("token_mask", &token_mask, "Bernoulli token masking (in-kernel RNG)");
    m.def("embed_pos_fwd", &embed_pos_fwd,
          "fused token+positional embedding gather with Scaler");
    m.def("embed_pos_bwd", &embed_pos_bwd,
          "deterministic embedding backward (fp32 master grads)");
    m.def("maxpool2_fwd", &maxpool2_fwd, "2x2/2 MaxPool forward (saves argmax)");
    m.def("pack_slices", &pack_slices,
          "one-launch distribute slice pack from a descriptor table");
    m.def("maxpool2_bwd", &maxpool2_bwd, "2x2/2 MaxPool backward");
}
