// Python bindings for the nats_amd CDNA4 kernels.

#include <torch/extension.h>

std::vector<torch::Tensor> gru_scan_fwd(torch::Tensor xg, torch::Tensor xc,
                                        c10::optional<torch::Tensor> mask,
                                        torch::Tensor Upk,
                                        c10::optional<torch::Tensor> h0);
std::vector<torch::Tensor> gru_scan_bwd(torch::Tensor dh_out,
                                        torch::Tensor h_all,
                                        torch::Tensor saved, torch::Tensor xc,
                                        c10::optional<torch::Tensor> mask,
                                        torch::Tensor Ubwd,
                                        c10::optional<torch::Tensor> h0);
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor targets);
torch::Tensor softmax_ce_bwd(torch::Tensor logits, torch::Tensor targets,
                             torch::Tensor stats, torch::Tensor dnll);
torch::Tensor mfma_gemm_bt(torch::Tensor A, torch::Tensor Bt);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gru_scan_fwd", &gru_scan_fwd, "fused GRU scan forward");
  m.def("gru_scan_bwd", &gru_scan_bwd, "fused GRU scan backward");
  m.def("softmax_ce_fwd", &softmax_ce_fwd, "fused softmax+CE forward");
  m.def("softmax_ce_bwd", &softmax_ce_bwd, "fused softmax+CE backward");
  m.def("mfma_gemm_bt", &mfma_gemm_bt, "MFMA layout self-test GEMM");
}
