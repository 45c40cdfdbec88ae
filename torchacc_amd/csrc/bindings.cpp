// Python bindings for the torchacc_amd CDNA4 kernel library.
#include <torch/extension.h>
#include <vector>

// elementwise.hip
std::vector<torch::Tensor> rmsnorm_forward(torch::Tensor x, torch::Tensor w,
                                           double eps);
std::vector<torch::Tensor> rmsnorm_backward(torch::Tensor dy, torch::Tensor x,
                                            torch::Tensor w,
                                            torch::Tensor inv_rms);
std::vector<torch::Tensor> rope_forward(torch::Tensor q, torch::Tensor k,
                                        torch::Tensor cos, torch::Tensor sin);
torch::Tensor swiglu_forward(torch::Tensor g, torch::Tensor u);
std::vector<torch::Tensor> add_rmsnorm_forward(torch::Tensor x,
                                               torch::Tensor resid,
                                               torch::Tensor w, double eps);
std::vector<torch::Tensor> add_rmsnorm_backward(torch::Tensor dy,
                                                torch::Tensor dresid,
                                                torch::Tensor r_saved,
                                                torch::Tensor w,
                                                torch::Tensor inv_rms);
std::vector<torch::Tensor> swiglu_backward(torch::Tensor dy, torch::Tensor g,
                                           torch::Tensor u);
// cross_entropy.hip
std::vector<torch::Tensor> cross_entropy_forward(torch::Tensor logits,
                                                 torch::Tensor target,
                                                 long ignore_index);
torch::Tensor cross_entropy_backward(torch::Tensor logits,
                                     torch::Tensor target, torch::Tensor lse,
                                     torch::Tensor scale, long ignore_index);
// adamw.hip
void fused_adamw(std::vector<torch::Tensor> params,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> masters, std::vector<double> steps,
                 torch::Tensor found_inf, double lr, double beta1,
                 double beta2, double eps, double wd);
// flash_attn_fwd.hip
std::vector<torch::Tensor> fa_forward(torch::Tensor q, torch::Tensor k,
                                      torch::Tensor v, double softmax_scale,
                                      bool causal, long wl, long wr,
                                      torch::Tensor q_lens,
                                      torch::Tensor k_lens,
                                      torch::Tensor alibi_slopes,
                                      double p_drop, long rng_seed);
// flash_attn_bwd.hip
std::vector<torch::Tensor> fa_backward(torch::Tensor dout, torch::Tensor q,
                                       torch::Tensor k, torch::Tensor v,
                                       torch::Tensor out, torch::Tensor lse,
                                       double softmax_scale, bool causal,
                                       long wl, long wr, torch::Tensor q_lens,
                                       torch::Tensor k_lens,
                                       torch::Tensor alibi_slopes,
                                       double p_drop, long rng_seed);

// flash_attn_varlen.hip
std::vector<torch::Tensor> fa_varlen_forward(torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v,
                                             torch::Tensor bounds,
                                             double softmax_scale,
                                             bool causal);
std::vector<torch::Tensor> fa_varlen_backward(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor lse, torch::Tensor bounds,
    double softmax_scale, bool causal);

// gemm.hip (hipBLASLt tuned GEMM)
std::vector<int64_t> lt_gemm_candidates(int64_t m, int64_t n, int64_t k,
                                        bool ta, bool tb,
                                        int64_t max_workspace_mb);
torch::Tensor lt_gemm(torch::Tensor a, torch::Tensor b, bool ta, bool tb,
                      int64_t algo_index,
                      c10::optional<torch::Tensor> out_opt);
std::string lt_gemm_algo_name(int64_t algo_index);
torch::Tensor lt_gemv(torch::Tensor x, torch::Tensor w);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "torchacc_amd CDNA4 (gfx950) kernels";
  m.def("rmsnorm_forward", &rmsnorm_forward);
  m.def("rmsnorm_backward", &rmsnorm_backward);
  m.def("rope_forward", &rope_forward);
  m.def("swiglu_forward", &swiglu_forward);
  m.def("add_rmsnorm_forward", &add_rmsnorm_forward);
  m.def("add_rmsnorm_backward", &add_rmsnorm_backward);
  m.def("swiglu_backward", &swiglu_backward);
  m.def("cross_entropy_forward", &cross_entropy_forward);
  m.def("cross_entropy_backward", &cross_entropy_backward);
  m.def("fused_adamw", &fused_adamw);
  m.def("fa_forward", &fa_forward);
  m.def("fa_backward", &fa_backward);
  m.def("fa_varlen_forward", &fa_varlen_forward);
  m.def("fa_varlen_backward", &fa_varlen_backward);
  m.def("lt_gemm_candidates", &lt_gemm_candidates);
  m.def("lt_gemm", &lt_gemm, pybind11::arg("a"), pybind11::arg("b"),
        pybind11::arg("ta"), pybind11::arg("tb"),
        pybind11::arg("algo_index") = -1,
        pybind11::arg("out") = pybind11::none());
  m.def("lt_gemm_algo_name", &lt_gemm_algo_name);
  m.def("lt_gemv", &lt_gemv);
}
