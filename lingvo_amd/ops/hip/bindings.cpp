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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layer_norm_fwd", &layer_norm_fwd, "Fused LayerNorm/RMSNorm fwd");
  m.def("layer_norm_bwd", &layer_norm_bwd, "Fused LayerNorm/RMSNorm bwd");
}
