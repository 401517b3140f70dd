// pybind bindings for the lingvo_amd gfx950 HIP op library.
#include <torch/extension.h>

#include <vector>

// layer_norm.hip
std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x,
                                          torch::Tensor scale,
                                          torch::Tensor bias, double eps,
                                          bool rms);
std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor scale,
                                          torch::Tensor mean,
                                          torch::Tensor rstd, bool rms);

// mfma_probe.hip
torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bT);
torch::Tensor tr16_probe(torch::Tensor a, torch::Tensor b);

// flash_attn.hip
std::vector<torch::Tensor> fa_fwd(torch::Tensor q, torch::Tensor k,
                                  torch::Tensor v,
                                  c10::optional<torch::Tensor> klen,
                                  c10::optional<torch::Tensor> bias,
                                  c10::optional<torch::Tensor> qseg,
                                  c10::optional<torch::Tensor> kseg,
                                  int64_t win_l, int64_t win_r,
                                  int64_t bias_clip, double scale,
                                  int64_t chunk_size, int64_t left_chunks);
std::vector<torch::Tensor> fa_bwd(torch::Tensor dout, torch::Tensor q,
                                  torch::Tensor k, torch::Tensor v,
                                  torch::Tensor o, torch::Tensor lse,
                                  c10::optional<torch::Tensor> klen,
                                  c10::optional<torch::Tensor> bias,
                                  c10::optional<torch::Tensor> qseg,
                                  c10::optional<torch::Tensor> kseg,
                                  bool bias_grad, int64_t win_l,
                                  int64_t win_r, int64_t bias_clip,
                                  double scale, int64_t chunk_size,
                                  int64_t left_chunks);

// conv1d.hip
torch::Tensor dwconv1d_fwd(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, int64_t pad);
std::vector<torch::Tensor> dwconv1d_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, int64_t pad);

// softmax_xent.hip
std::vector<torch::Tensor> xent_fwd(torch::Tensor logits,
                                    torch::Tensor labels);
torch::Tensor xent_bwd(torch::Tensor logits, torch::Tensor labels,
                       torch::Tensor lse, torch::Tensor gout);

// dropout.hip
torch::Tensor dropout_fwd(torch::Tensor x, c10::optional<torch::Tensor> res,
                          int64_t seed,
                          c10::optional<torch::Tensor> step_seed,
                          double keep, int64_t act, double scale,
                          c10::optional<torch::Tensor> pad, int64_t d);
torch::Tensor dropout_bwd(torch::Tensor dy, c10::optional<torch::Tensor> x,
                          int64_t seed,
                          c10::optional<torch::Tensor> step_seed,
                          double keep, int64_t act, double scale,
                          c10::optional<torch::Tensor> pad, int64_t d);

// group_norm.hip
std::vector<torch::Tensor> group_norm_fwd(torch::Tensor x,
                                          torch::Tensor gamma,
                                          torch::Tensor beta,
                                          c10::optional<torch::Tensor> pad,
                                          int64_t groups, double eps,
                                          int64_t act);
std::vector<torch::Tensor> group_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor gamma,
                                          torch::Tensor beta,
                                          c10::optional<torch::Tensor> pad,
                                          torch::Tensor mean,
                                          torch::Tensor rstd,
                                          int64_t groups, int64_t act);

// lstm_gates.hip
std::vector<torch::Tensor> lstm_gates_fwd_op(torch::Tensor gates,
                                             torch::Tensor c0, double fgb,
                                             double cap);
std::vector<torch::Tensor> lstm_gates_bwd_op(
    torch::Tensor gates, torch::Tensor c0, torch::Tensor c1,
    torch::Tensor dm1, c10::optional<torch::Tensor> dc1, double fgb,
    double cap);

// input_pipeline.cpp
void RegisterInputPipeline(py::module_& m);

// wpm_tokenizer.cpp
void RegisterWpmTokenizer(py::module_& m);

// record_batcher.cpp
void RegisterRecordBatcher(py::module_& m);

// embedding.hip
torch::Tensor emb_gather(torch::Tensor table, torch::Tensor ids,
                         double scale);
torch::Tensor emb_scatter_add(torch::Tensor dy, torch::Tensor ids,
                              int64_t vocab, double scale);

// moe_gating.hip
std::vector<torch::Tensor> moe_positions(torch::Tensor top1,
                                         torch::Tensor top2,
                                         int64_t num_experts);

// s2d.hip
torch::Tensor s2d_fwd(torch::Tensor x);
torch::Tensor s2d_inv(torch::Tensor dX, int64_t C);
torch::Tensor pad_scatter(torch::Tensor dout);

// topk.hip
std::vector<torch::Tensor> topk_rows(torch::Tensor scores, int64_t k);

// las_decoder.hip
void smallm_gemm(torch::Tensor a, torch::Tensor wt,
                 c10::optional<torch::Tensor> pre, torch::Tensor out,
                 int64_t k, int64_t wt_col0, double alpha,
                 int64_t pre_mode);
std::vector<torch::Tensor> attend_fwd(torch::Tensor q, torch::Tensor enc,
                                      torch::Tensor pad, double scale);
std::vector<torch::Tensor> attend_bwd(torch::Tensor dctx,
                                      torch::Tensor probs, torch::Tensor q,
                                      torch::Tensor enc, torch::Tensor denc,
                                      double scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  RegisterInputPipeline(m);
  RegisterWpmTokenizer(m);
  RegisterRecordBatcher(m);
  m.def("group_norm_fwd", &group_norm_fwd, "Fused padded GroupNorm fwd");
  m.def("group_norm_bwd", &group_norm_bwd, "Fused padded GroupNorm bwd");
  m.def("lstm_gates_fwd", &lstm_gates_fwd_op, "Fused LSTM gates fwd");
  m.def("lstm_gates_bwd", &lstm_gates_bwd_op, "Fused LSTM gates bwd");
  m.def("dropout_fwd", &dropout_fwd, "Fused dropout fwd");
  m.def("dropout_bwd", &dropout_bwd, "Fused dropout bwd");
  m.def("xent_fwd", &xent_fwd, "Fused softmax-xent fwd");
  m.def("xent_bwd", &xent_bwd, "Fused softmax-xent bwd");
  m.def("dwconv1d_fwd", &dwconv1d_fwd, "Depthwise time conv fwd");
  m.def("dwconv1d_bwd", &dwconv1d_bwd, "Depthwise time conv bwd");
  m.def("layer_norm_fwd", &layer_norm_fwd, "Fused LayerNorm/RMSNorm fwd");
  m.def("layer_norm_bwd", &layer_norm_bwd, "Fused LayerNorm/RMSNorm bwd");
  m.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 layout probe");
  m.def("tr16_probe", &tr16_probe, "ds_read_tr16_b64 layout probe");
  m.def("fa_fwd", &fa_fwd, "Flash attention forward");
  m.def("fa_bwd", &fa_bwd, "Flash attention backward");
  m.def("emb_gather", &emb_gather, "Embedding gather fwd");
  m.def("emb_scatter_add", &emb_scatter_add,
        "Deterministic embedding scatter-add bwd");
  m.def("moe_positions", &moe_positions,
        "Deterministic MoE top-2 position scan");
  m.def("smallm_gemm", &smallm_gemm,
        "Small-M MFMA GEMM (decode-step projections)");
  m.def("attend_fwd", &attend_fwd, "Fused dot-attention step fwd");
  m.def("attend_bwd", &attend_bwd, "Fused dot-attention step bwd");
  m.def("s2d_fwd", &s2d_fwd, "Space-to-depth + pad (conv frontend)");
  m.def("topk_rows", &topk_rows, "Row-wise top-k (beam pruning, K12)");
  m.def("s2d_inv", &s2d_inv, "Inverse space-to-depth");
  m.def("pad_scatter", &pad_scatter, "Zero-border pad scatter");
}
