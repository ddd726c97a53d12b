// Python bindings for the NornicDB-AMD native kernel library (_C).
#include <torch/extension.h>

// vector_ops.hip
void l2_normalize_(at::Tensor x);
void fill_random_unit_(at::Tensor x, long long row_base, long long seed);
std::tuple<at::Tensor, at::Tensor> knn_gemv(at::Tensor db, at::Tensor q,
                                            long long row_base, int k_out);

// knn_mfma.hip
std::tuple<at::Tensor, at::Tensor> knn_mfma(at::Tensor db, at::Tensor q,
                                            long long row_base, int k_out);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "NornicDB-AMD CDNA4 (gfx950) native kernels";
  m.def("l2_normalize_", &l2_normalize_, "In-place row L2 normalize (bf16/f32)");
  m.def("fill_random_unit_", &fill_random_unit_,
        "Fill with deterministic unit-norm pseudo-gaussian rows (bf16)",
        py::arg("x"), py::arg("row_base") = 0, py::arg("seed") = 0x6e6f726eLL);
  m.def("knn_gemv", &knn_gemv,
        "Fused cosine score + top-k for <=16 queries (bf16 db)",
        py::arg("db"), py::arg("q"), py::arg("row_base") = 0,
        py::arg("k_out") = 10);
  m.def("knn_mfma", &knn_mfma,
        "Fused MFMA cosine score + top-k, 256-query batches (bf16 db)",
        py::arg("db"), py::arg("q"), py::arg("row_base") = 0,
        py::arg("k_out") = 10);
}
