// Python bindings for the gfx950 HIP kernels.
#include <torch/extension.h>

torch::Tensor floyd_warshall_hip(torch::Tensor w);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("floyd_warshall", &floyd_warshall_hip,
          "Batched min-plus Floyd-Warshall APSP (gfx950)");
}
