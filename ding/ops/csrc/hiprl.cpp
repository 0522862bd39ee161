// Python bindings for the DI-engine MI355X HIP kernel library.
#include <torch/extension.h>

torch::Tensor reverse_scan(torch::Tensor delta, torch::Tensor factor);
torch::Tensor multistep_forward_view(
    torch::Tensor bootstrap, torch::Tensor rewards, torch::Tensor gammas, torch::Tensor lambdas, torch::Tensor done
);
torch::Tensor c51_project(
    torch::Tensor next_dist, torch::Tensor next_act, torch::Tensor reward, torch::Tensor done, double v_min,
    double v_max, double gamma_n
);
torch::Tensor scatter_connection(torch::Tensor x, torch::Tensor index, int64_t H, int64_t W, int64_t scatter_add);
std::vector<torch::Tensor> ppo_fwd(
    torch::Tensor logit_new, torch::Tensor logit_old, torch::Tensor action, torch::Tensor value_new,
    torch::Tensor value_old, torch::Tensor adv, torch::Tensor ret, torch::Tensor weight, double clip_ratio,
    int64_t use_value_clip
);
std::vector<torch::Tensor> lstm_cell_fwd(
    torch::Tensor gxn, torch::Tensor gh_raw, torch::Tensor gamma, torch::Tensor beta, torch::Tensor bias,
    torch::Tensor c_in
);
std::vector<torch::Tensor> lstm_cell_bwd(
    torch::Tensor dh, torch::Tensor dc_next, torch::Tensor acts, torch::Tensor xhat, torch::Tensor gamma,
    torch::Tensor c_in, torch::Tensor c_out, torch::Tensor rstd
);
torch::Tensor stem_conv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias);
torch::Tensor conv_wrw_nhwc(torch::Tensor x, torch::Tensor dy, int64_t K, int64_t S);
torch::Tensor stem_conv_wrw(torch::Tensor x, torch::Tensor dy, int64_t O);
std::vector<torch::Tensor> q_nstep_fwd(
    torch::Tensor q, torch::Tensor next_n_q, torch::Tensor action, torch::Tensor next_action,
    torch::Tensor reward, torch::Tensor done, torch::Tensor value_gamma, double gamma, int64_t nstep,
    int64_t rescale
);
std::vector<torch::Tensor> ppo_bwd(
    torch::Tensor logit_new, torch::Tensor action, torch::Tensor value_new, torch::Tensor value_old,
    torch::Tensor adv, torch::Tensor ret, torch::Tensor weight, torch::Tensor fwd_out, double clip_ratio,
    torch::Tensor grad_scales
);
std::vector<torch::Tensor> vtrace_fwd(
    torch::Tensor t_logit, torch::Tensor b_logit, torch::Tensor action, torch::Tensor value,
    torch::Tensor reward, torch::Tensor weight, double gamma, double lambda_, double rho_c, double c_c,
    double rho_pg_c
);
std::vector<torch::Tensor> vtrace_bwd(
    torch::Tensor t_logit, torch::Tensor row_out, torch::Tensor value, torch::Tensor ret, torch::Tensor adv,
    torch::Tensor action, torch::Tensor weight, torch::Tensor gscales
);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "DI-engine MI355X HIP kernels (gfx950)";
    m.def("reverse_scan", &reverse_scan, "reverse discounted scan over dim0");
    m.def("multistep_forward_view", &multistep_forward_view, "TD(lambda) forward-view reverse scan");
    m.def("c51_project", &c51_project, "C51 categorical projection");
    m.def("scatter_connection", &scatter_connection, "entity->spatial scatter");
    m.def("ppo_fwd", &ppo_fwd, "fused PPO loss forward");
    m.def("ppo_bwd", &ppo_bwd, "fused PPO loss backward");
    m.def("q_nstep_fwd", &q_nstep_fwd, "fused n-step Q TD forward (+value rescale)");
    m.def("stem_conv_fwd", &stem_conv_fwd, "direct 8x8s4 stem conv forward");
    m.def("stem_conv_wrw", &stem_conv_wrw, "direct 8x8s4 stem conv weight grad");
    m.def("lstm_cell_fwd", &lstm_cell_fwd, "fused LN-LSTM cell forward");
    m.def("lstm_cell_bwd", &lstm_cell_bwd, "fused LN-LSTM cell backward");
    m.def("conv_wrw_nhwc", &conv_wrw_nhwc, "NHWC conv2d backward-weights (fp32, register-tiled)");
    m.def("vtrace_fwd", &vtrace_fwd, "fused v-trace loss forward");
    m.def("vtrace_bwd", &vtrace_bwd, "fused v-trace loss backward");
}
