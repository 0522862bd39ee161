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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "DI-engine MI355X HIP kernels (gfx950)";
    m.def("reverse_scan", &reverse_scan, "reverse discounted scan over dim0");
    m.def("multistep_forward_view", &multistep_forward_view, "TD(lambda) forward-view reverse scan");
    m.def("c51_project", &c51_project, "C51 categorical projection");
    m.def("scatter_connection", &scatter_connection, "entity->spatial scatter");
}
