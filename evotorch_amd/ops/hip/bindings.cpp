// Python bindings for the gfx950 kernel extension `evotorch_amd._C`.
#include <torch/extension.h>

namespace ea {
void sample_gaussian(torch::Tensor out, torch::Tensor mu, torch::Tensor sigma, bool symmetric, int64_t seed, int64_t row_offset);
void sample_gaussian_graphsafe(torch::Tensor out, torch::Tensor mu, torch::Tensor sigma, bool symmetric,
                               torch::Tensor seed_buf);
void affine_from_noise(torch::Tensor out, torch::Tensor z, torch::Tensor mu, torch::Tensor sigma, bool symmetric);
std::vector<torch::Tensor> es_gradients(torch::Tensor samples, torch::Tensor mu, torch::Tensor sigma,
                                        torch::Tensor weights, bool symmetric);
std::vector<torch::Tensor> snes_gradients(torch::Tensor samples, torch::Tensor mu, torch::Tensor sigma,
                                          torch::Tensor weights);
void clipup_step(torch::Tensor velocity, torch::Tensor grad, double step_size, double max_speed, double momentum);
void adam_step(torch::Tensor step_out, torch::Tensor grad, torch::Tensor m, torch::Tensor v, int64_t step_count,
               double stepsize, double beta1, double beta2, double epsilon);
void adam_step_graphsafe(torch::Tensor step_out, torch::Tensor grad, torch::Tensor m, torch::Tensor v,
                         torch::Tensor t_buf, double stepsize, double beta1, double beta2, double epsilon);
torch::Tensor rollout_linear(torch::Tensor params, torch::Tensor env_blob, torch::Tensor obs_stats_out,
                             int64_t obs_dim, int64_t act_dim, int64_t rank, int64_t steps, double alive_bonus,
                             double act_cost, int64_t init_seed, int64_t member_offset, int64_t policy_hidden,
                             c10::optional<torch::Tensor> seed_buf);
void bump_seed(torch::Tensor seed_buf);
void cma_update_c(torch::Tensor C, torch::Tensor Y, torch::Tensor w, torch::Tensor pc, torch::Tensor hs_f,
                  torch::Tensor wsum, double c1, double cmu, double cc);
void potrf_tile(torch::Tensor A, torch::Tensor info);
torch::Tensor fused_rank(torch::Tensor fitnesses, int64_t method, bool higher_better);
torch::Tensor domination_counts(torch::Tensor utils);
torch::Tensor pareto_ranks(torch::Tensor utils, int64_t min_assigned);
std::vector<torch::Tensor> mapelites_assign(torch::Tensor grid, torch::Tensor feats, torch::Tensor utils);
}  // namespace ea

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "evotorch_amd gfx950 (MI355X / CDNA4) HIP kernels";
    m.def("sample_gaussian", &ea::sample_gaussian, "K1: philox Gaussian population sampling (plain/antithetic)",
          py::arg("out"), py::arg("mu"), py::arg("sigma"), py::arg("symmetric"), py::arg("seed"),
          py::arg("row_offset") = 0);
    m.def("sample_gaussian_graphsafe", &ea::sample_gaussian_graphsafe,
          "K1 (hipGraph-safe): seed read from and advanced in device memory");
    m.def("affine_from_noise", &ea::affine_from_noise,
          "K1 overlap: mu + sigma*z from pre-generated side-stream noise");
    m.def("es_gradients", &ea::es_gradients, "K3: fused (mu, sigma) ES gradient reduction");
    m.def("snes_gradients", &ea::snes_gradients, "K3: SNES raw-noise gradient reduction");
    m.def("clipup_step", &ea::clipup_step, "K4: fused ClipUp velocity update (no host sync)");
    m.def("adam_step", &ea::adam_step, "K4: fused Adam ascent step");
    m.def("adam_step_graphsafe", &ea::adam_step_graphsafe,
          "K4 (hipGraph-safe): Adam with device-side step counter");
    m.def("rollout_linear", &ea::rollout_linear,
          "K10+K11: fused policy episode rollout (linear or MLP-H, synthetic env)",
          pybind11::arg("params"), pybind11::arg("env_blob"), pybind11::arg("obs_stats_out"),
          pybind11::arg("obs_dim"), pybind11::arg("act_dim"), pybind11::arg("rank"), pybind11::arg("steps"),
          pybind11::arg("alive_bonus"), pybind11::arg("act_cost"), pybind11::arg("init_seed"),
          pybind11::arg("member_offset"), pybind11::arg("policy_hidden") = 0,
          pybind11::arg("seed_buf") = c10::optional<torch::Tensor>());
    m.def("bump_seed", &ea::bump_seed,
          "advance a device splitmix64 seed chain by one step (hipGraph-safe episode seeds)");
    m.def("cma_update_c", &ea::cma_update_c,
          "K5: fused CMA-ES covariance update (scale*C + c1*pc pc^T + cmu*Y^T diag(w) Y, exact symmetry)");
    m.def("potrf_tile", &ea::potrf_tile,
          "K5b: in-LDS Cholesky of one <=128x128 SPD diagonal panel (in place, device info flag)");
    m.def("fused_rank", &ea::fused_rank,
          "K2: fused bitonic ranking + utility map (centered/linear/nes) in one launch");
    m.def("domination_counts", &ea::domination_counts, "K7: NSGA-II domination counts (no N x N matrix)");
    m.def("pareto_ranks", &ea::pareto_ranks, "K7: NSGA-II non-dominated sorting by front peeling",
          py::arg("utils"), py::arg("min_assigned") = 0);
    m.def("mapelites_assign", &ea::mapelites_assign,
          "K9: MAPElites cell assignment, O(C+N) memory (streamed best-in-box argmax)");
}
