// Python bindings for the gfx950 HIP kernels.
#include <torch/extension.h>

torch::Tensor floyd_warshall_hip(torch::Tensor w,
                                 c10::optional<torch::Tensor> n_arr);
std::vector<torch::Tensor> decide_hip(
    torch::Tensor sp, torch::Tensor hop, torch::Tensor uds,
    torch::Tensor servers, torch::Tensor src, torch::Tensor mask,
    torch::Tensor ul, torch::Tensor dl,
    c10::optional<torch::Tensor> explore, c10::optional<torch::Tensor> rng,
    long prob);
std::vector<torch::Tensor> walk_eval_hip(
    torch::Tensor sp, torch::Tensor src, torch::Tensor dst,
    torch::Tensor mask, torch::Tensor rate, torch::Tensor ul,
    torch::Tensor dl, torch::Tensor adj_indptr, torch::Tensor adj_idx,
    torch::Tensor adj_link, torch::Tensor conf_indptr,
    torch::Tensor conf_base, torch::Tensor conf_cols, torch::Tensor rates,
    torch::Tensor bw, torch::Tensor edges, torch::Tensor T_arr,
    torch::Tensor E_arr, torch::Tensor n_arr, long H,
    long fp_iters);
std::vector<torch::Tensor> critic_hip(
    torch::Tensor route_links, torch::Tensor nhop, torch::Tensor vedge_dst,
    torch::Tensor mask, torch::Tensor rate, torch::Tensor ul,
    torch::Tensor dl, torch::Tensor conf_indptr, torch::Tensor conf_base,
    torch::Tensor conf_cols, torch::Tensor rates, torch::Tensor bw_comp,
    torch::Tensor E_arr, torch::Tensor T_arr, long Ee, long iters,
    double cap);
std::vector<torch::Tensor> actor_head_fwd_hip(
    torch::Tensor lam_ext, torch::Tensor conf_indptr,
    torch::Tensor conf_base, torch::Tensor conf_cols, torch::Tensor rates,
    torch::Tensor bw_comp, torch::Tensor edges, torch::Tensor node_vedge,
    torch::Tensor T_arr, torch::Tensor E_arr, long N, long iters,
    double cap);
torch::Tensor actor_head_bwd_hip(
    torch::Tensor grad_dist, torch::Tensor lam_ext, torch::Tensor mu_hist,
    torch::Tensor conf_indptr, torch::Tensor conf_base,
    torch::Tensor conf_cols, torch::Tensor rates, torch::Tensor bw_comp,
    torch::Tensor edges, torch::Tensor node_vedge, torch::Tensor T_arr,
    torch::Tensor E_arr, long iters, double cap);

std::vector<torch::Tensor> cheb_fwd_hip(
    torch::Tensor x, torch::Tensor W, torch::Tensor bias,
    torch::Tensor ext_indptr, torch::Tensor ext_base, torch::Tensor ext_cols,
    long max_nnz);
std::vector<torch::Tensor> cheb_bwd_hip(
    torch::Tensor dlam, torch::Tensor acts, torch::Tensor t1s,
    torch::Tensor W, torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols, long max_nnz);

std::vector<torch::Tensor> cheb_bwd_hip_mask(
    torch::Tensor dlam, torch::Tensor acts, torch::Tensor t1s,
    torch::Tensor W, torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols, long max_nnz, long stage_mask);
std::vector<torch::Tensor> cheb_kn_fwd_hip(
    torch::Tensor x, torch::Tensor W, torch::Tensor bias,
    torch::Tensor ext_indptr, torch::Tensor ext_base, torch::Tensor ext_cols,
    long max_nnz);
std::vector<torch::Tensor> cheb_kn_bwd_hip(
    torch::Tensor dlam, torch::Tensor acts, torch::Tensor tks,
    torch::Tensor W, torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols, long max_nnz);

std::vector<torch::Tensor> cheb_large_fwd_hip(
    torch::Tensor x, torch::Tensor W, torch::Tensor bias,
    torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols);
std::vector<torch::Tensor> cheb_large_bwd_hip(
    torch::Tensor dlam, torch::Tensor acts, torch::Tensor W,
    torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols);
void fused_adam_hip(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                    torch::Tensor v, torch::Tensor seg, torch::Tensor step,
                    double scale, double lr, double beta1, double beta2,
                    double eps, bool constraints);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("floyd_warshall", &floyd_warshall_hip);
    m.def("decide", &decide_hip);
    m.def("walk_eval", &walk_eval_hip);
    m.def("critic", &critic_hip);
    m.def("actor_head_fwd", &actor_head_fwd_hip);
    m.def("actor_head_bwd", &actor_head_bwd_hip);
    m.def("cheb_fwd", &cheb_fwd_hip);
    m.def("cheb_bwd", &cheb_bwd_hip);
    m.def("cheb_bwd_ablate", &cheb_bwd_hip_mask);
    m.def("cheb_kn_fwd", &cheb_kn_fwd_hip);
    m.def("cheb_kn_bwd", &cheb_kn_bwd_hip);
    m.def("fused_adam", &fused_adam_hip);
    m.def("cheb_large_fwd", &cheb_large_fwd_hip);
    m.def("cheb_large_bwd", &cheb_large_bwd_hip);
}
