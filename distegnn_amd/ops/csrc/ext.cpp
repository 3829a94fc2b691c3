// Python bindings for the distegnn_amd gfx950 HIP kernels.
#include <torch/extension.h>

torch::Tensor segment_reduce_csr(torch::Tensor data, torch::Tensor rowptr,
                                 bool mean);
torch::Tensor segment_reduce_chunked(torch::Tensor data, torch::Tensor rowptr,
                                     torch::Tensor chunk_begin,
                                     torch::Tensor chunk_end,
                                     torch::Tensor seg_chunk_ptr, bool mean);
std::tuple<torch::Tensor, torch::Tensor> radius_graph_gpu(torch::Tensor pos,
                                                          double r);

torch::Tensor radius_graph(torch::Tensor pos, double r) {
  return std::get<0>(radius_graph_gpu(pos, r));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "distegnn_amd hand-written HIP/CDNA4 kernels (gfx950)";
  m.def("segment_reduce_csr", &segment_reduce_csr,
        "deterministic CSR segmented sum/mean", py::arg("data"),
        py::arg("rowptr"), py::arg("mean"));
  m.def("segment_reduce_chunked", &segment_reduce_chunked,
        "two-stage deterministic segmented reduce for huge segments",
        py::arg("data"), py::arg("rowptr"), py::arg("chunk_begin"),
        py::arg("chunk_end"), py::arg("seg_chunk_ptr"), py::arg("mean"));
  m.def("radius_graph_gpu", &radius_graph_gpu,
        "cell-list radius graph -> (edge_index, rowptr)", py::arg("pos"),
        py::arg("r"));
  m.def("radius_graph", &radius_graph, "cell-list radius graph edge_index",
        py::arg("pos"), py::arg("r"));
}
