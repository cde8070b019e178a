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
std::vector<torch::Tensor> gru_scan_fwd_bidir(
    torch::Tensor xg0, torch::Tensor xc0, c10::optional<torch::Tensor> mask0,
    torch::Tensor Upk0, torch::Tensor xg1, torch::Tensor xc1,
    c10::optional<torch::Tensor> mask1, torch::Tensor Upk1);
std::vector<torch::Tensor> gru_scan_bwd_bidir(
    torch::Tensor dh_out0, torch::Tensor h_all0, torch::Tensor saved0,
    torch::Tensor xc0, c10::optional<torch::Tensor> mask0,
    torch::Tensor Ubwd0, torch::Tensor dh_out1, torch::Tensor h_all1,
    torch::Tensor saved1, torch::Tensor xc1,
    c10::optional<torch::Tensor> mask1, torch::Tensor Ubwd1);
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor targets);
torch::Tensor softmax_ce_bwd(torch::Tensor logits, torch::Tensor targets,
                             torch::Tensor stats, torch::Tensor dnll);
torch::Tensor mfma_gemm_bt(torch::Tensor A, torch::Tensor Bt);
double barrier_bench(int nwg_x, int nwg_y, int iters);
bool gru_persistent_ok(long H);
double barrier_bench_xcd(int nwg, int nxcd, int iters);
torch::Tensor rerank_penalties(torch::Tensor hist_a, torch::Tensor hist_c,
                               torch::Tensor hist_s, torch::Tensor cur_a,
                               torch::Tensor cur_c, torch::Tensor cur_s,
                               double kl_f, double ctx_f, double state_f);
void fused_adadelta_step(torch::Tensor ptrs, torch::Tensor sizes,
                         torch::Tensor chunk_tensor, torch::Tensor chunk_off,
                         torch::Tensor g2, double clip_c, double rho,
                         double eps);
std::vector<torch::Tensor> cond_gru_fwd(
    torch::Tensor yg, torch::Tensor yc, c10::optional<torch::Tensor> mask,
    torch::Tensor init_state, torch::Tensor ctx_bf,
    c10::optional<torch::Tensor> ctx_mask, torch::Tensor pctx,
    torch::Tensor Upk2, torch::Tensor W1pk, torch::Tensor WattPk,
    torch::Tensor b1, torch::Tensor bx1, torch::Tensor Uatt,
    torch::Tensor catt, torch::Tensor Dwei, torch::Tensor Wcon,
    torch::Tensor Ucon,
    c10::optional<torch::Tensor> accC0, c10::optional<torch::Tensor> accA0);
std::vector<torch::Tensor> cond_gru_bwd(
    torch::Tensor dh2_all, c10::optional<torch::Tensor> dctxs_all,
    c10::optional<torch::Tensor> dalphas_all,
    c10::optional<torch::Tensor> daccC_f, c10::optional<torch::Tensor> daccA_f,
    torch::Tensor yc, torch::Tensor h1_all, torch::Tensor h2_all,
    torch::Tensor ctxs_all, torch::Tensor alphas_all, torch::Tensor saved2,
    torch::Tensor saved1, torch::Tensor pstate_all, torch::Tensor ctxpre_all,
    torch::Tensor accA_used, torch::Tensor accC_used, torch::Tensor ctx_bf,
    torch::Tensor pctx, torch::Tensor init_state,
    c10::optional<torch::Tensor> mask, torch::Tensor U1cat,
    torch::Tensor W1cat, torch::Tensor U2cat, torch::Tensor WattB,
    torch::Tensor bx1, torch::Tensor Dwei, torch::Tensor Uatt,
    torch::Tensor Ucon, torch::Tensor Wcon);

torch::Tensor embed_gather(torch::Tensor Wemb, torch::Tensor ids,
                           long shift_rows);
torch::Tensor pack_fwd_weights_hip(torch::Tensor U, torch::Tensor Ux);
torch::Tensor pack_cat2_hip(torch::Tensor A, torch::Tensor B, long R, long K);
torch::Tensor pack_pad_hip(torch::Tensor src, long R, long K, bool transpose);
torch::Tensor pack_gru1_weights_hip(torch::Tensor U_1, torch::Tensor W_1,
                                    torch::Tensor Ux_1, torch::Tensor Wx_1,
                                    long Hpad, long Cpad);
torch::Tensor embed_scatter_add(torch::Tensor dout, torch::Tensor ids,
                                long V, long shift_rows);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("pack_fwd_weights", &pack_fwd_weights_hip,
        "fused forward-scan weight pack");
  m.def("pack_cat2", &pack_cat2_hip, "fused [A|B] zero-padded bf16 pack");
  m.def("pack_pad", &pack_pad_hip, "fused pad(+transpose) bf16 pack");
  m.def("pack_gru1_weights", &pack_gru1_weights_hip,
        "fused GRU_1 4-group operand pack");
  m.def("embed_gather", &embed_gather,
        "embedding gather (+fused decoder shift)");
  m.def("embed_scatter_add", &embed_scatter_add,
        "embedding backward scatter-add");
  m.def("gru_scan_fwd", &gru_scan_fwd, "fused GRU scan forward");
  m.def("gru_scan_bwd", &gru_scan_bwd, "fused GRU scan backward");
  m.def("gru_scan_fwd_bidir", &gru_scan_fwd_bidir,
        "bidirectional fused GRU scan forward");
  m.def("gru_scan_bwd_bidir", &gru_scan_bwd_bidir,
        "bidirectional fused GRU scan backward");
  m.def("softmax_ce_fwd", &softmax_ce_fwd, "fused softmax+CE forward");
  m.def("softmax_ce_bwd", &softmax_ce_bwd, "fused softmax+CE backward");
  m.def("mfma_gemm_bt", &mfma_gemm_bt, "MFMA layout self-test GEMM");
  m.def("barrier_bench", &barrier_bench, "grid barrier us/iteration");
  m.def("gru_persistent_ok", &gru_persistent_ok,
        "persistent scan available for this hidden size");
  m.def("barrier_bench_xcd", &barrier_bench_xcd,
        "XCD-constrained grid barrier us/iteration");
  m.def("rerank_penalties", &rerank_penalties,
        "fused decode-time distraction rerank reductions");
  m.def("fused_adadelta_step", &fused_adadelta_step,
        "fused multi-tensor global-norm clip + adadelta");
  m.def("cond_gru_fwd", &cond_gru_fwd, "fused cond-GRU decoder forward");
  m.def("cond_gru_bwd", &cond_gru_bwd, "fused cond-GRU decoder backward");
}
