// Python bindings for the distegnn_amd gfx950 HIP kernels.
#include <torch/extension.h>

torch::Tensor segment_reduce_csr(torch::Tensor data, torch::Tensor rowptr,
                                 bool mean);
torch::Tensor segment_reduce_csr_perm(torch::Tensor data,
                                      torch::Tensor rowptr,
                                      torch::Tensor perm, bool mean);
torch::Tensor gather_rows_fast(torch::Tensor data, torch::Tensor idx);
torch::Tensor mid_reduce(torch::Tensor x, double scale);
torch::Tensor mid_expand(torch::Tensor g, int64_t c, double scale);
torch::Tensor segment_reduce_chunked(torch::Tensor data, torch::Tensor rowptr,
                                     torch::Tensor chunk_begin,
                                     torch::Tensor chunk_end,
                                     torch::Tensor seg_chunk_ptr, bool mean);
std::tuple<torch::Tensor, torch::Tensor> radius_graph_gpu(torch::Tensor pos,
                                                          double r);
std::tuple<torch::Tensor, torch::Tensor> fused_edge_forward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor eattr,
    torch::Tensor row, torch::Tensor col, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    torch::Tensor w3v, bool normalize, double eps,
    std::vector<torch::Tensor> prepped = {});
torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bt);
torch::Tensor coord_update_forward(torch::Tensor coord, torch::Tensor agg,
                                   torch::Tensor trans_v,
                                   torch::Tensor phiv, torch::Tensor vel);
torch::Tensor coord_update_backward(torch::Tensor g, torch::Tensor vel);
torch::Tensor edge_softmax_fwd(torch::Tensor scores, torch::Tensor dstptr,
                               torch::Tensor perm);
torch::Tensor cfconv_forward(torch::Tensor xw1, torch::Tensor dist,
                             torch::Tensor row, torch::Tensor col,
                             torch::Tensor w1f, torch::Tensor b1,
                             torch::Tensor w2f, torch::Tensor b2,
                             torch::Tensor offsets, double coeff,
                             double cutoff);
torch::Tensor wgrad_splitk_launch(torch::Tensor g, torch::Tensor x);
torch::Tensor tall_linear(torch::Tensor x, torch::Tensor bmat,
                          c10::optional<torch::Tensor> bias, int64_t act);
std::vector<torch::Tensor> fused_virtual_forward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor vcoord,
    torch::Tensor vfeat, torch::Tensor gram, torch::Tensor batch,
    torch::Tensor w1, torch::Tensor b1, torch::Tensor w2, torch::Tensor b2,
    torch::Tensor wxv, torch::Tensor bxv, torch::Tensor wxvv,
    torch::Tensor wX, torch::Tensor bX, torch::Tensor wXv, bool train,
    std::vector<torch::Tensor> prepped = {});
std::vector<torch::Tensor> fused_virtual_backward(
    torch::Tensor coord, torch::Tensor vcoord, torch::Tensor batch,
    torch::Tensor dvmsg, torch::Tensor dtv, torch::Tensor dtx,
    torch::Tensor z1, torch::Tensor z2, torch::Tensor zxv, torch::Tensor zX,
    torch::Tensor p2, torch::Tensor w1, torch::Tensor w2, torch::Tensor wxv,
    torch::Tensor wX, torch::Tensor wxvv, torch::Tensor wXv,
    std::vector<torch::Tensor> prepped = {});
std::vector<torch::Tensor> fused_edge_backward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor eattr,
    torch::Tensor row, torch::Tensor col, torch::Tensor dmsg_n,
    torch::Tensor dtrans_n, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    torch::Tensor w3v, bool normalize, double eps,
    std::vector<torch::Tensor> prepped = {});
std::vector<torch::Tensor> fused_edge_backward_wg(
    torch::Tensor h, torch::Tensor coord, torch::Tensor eattr,
    torch::Tensor row, torch::Tensor col, torch::Tensor dmsg_n,
    torch::Tensor dtrans_n, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    torch::Tensor w3v, bool normalize, double eps,
    std::vector<torch::Tensor> prepped = {});

torch::Tensor radius_graph(torch::Tensor pos, double r) {
  return std::get<0>(radius_graph_gpu(pos, r));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "distegnn_amd hand-written HIP/CDNA4 kernels (gfx950)";
  m.def("segment_reduce_csr", &segment_reduce_csr,
        "deterministic CSR segmented sum/mean", py::arg("data"),
        py::arg("rowptr"), py::arg("mean"));
  m.def("segment_reduce_csr_perm", &segment_reduce_csr_perm,
        "CSR segmented sum/mean over permuted rows (data[perm[k]])",
        py::arg("data"), py::arg("rowptr"), py::arg("perm"), py::arg("mean"));
  m.def("gather_rows_fast", &gather_rows_fast,
        "vectorized dst[i] = src[idx[i]] row gather", py::arg("data"),
        py::arg("idx"));
  m.def("mid_reduce", &mid_reduce,
        "scale * sum over the middle dim of [N,C,F]", py::arg("x"),
        py::arg("scale"));
  m.def("mid_expand", &mid_expand,
        "broadcast [N,F] over a middle dim: [N,C,F] * scale", py::arg("g"),
        py::arg("c"), py::arg("scale"));
  m.def("segment_reduce_chunked", &segment_reduce_chunked,
        "two-stage deterministic segmented reduce for huge segments",
        py::arg("data"), py::arg("rowptr"), py::arg("chunk_begin"),
        py::arg("chunk_end"), py::arg("seg_chunk_ptr"), py::arg("mean"));
  m.def("radius_graph_gpu", &radius_graph_gpu,
        "cell-list radius graph -> (edge_index, rowptr)", py::arg("pos"),
        py::arg("r"));
  m.def("radius_graph", &radius_graph, "cell-list radius graph edge_index",
        py::arg("pos"), py::arg("r"));
  m.def("fused_edge_forward", &fused_edge_forward,
        "fused MFMA edge block: gather + phi_e MLP + phi_x head + trans",
        py::arg("h"), py::arg("coord"), py::arg("eattr"), py::arg("row"),
        py::arg("col"), py::arg("w1"), py::arg("b1"), py::arg("w2"),
        py::arg("b2"), py::arg("w3"), py::arg("b3"), py::arg("w3v"),
        py::arg("normalize"), py::arg("eps"),
        py::arg("prepped") = std::vector<torch::Tensor>{});
  m.def("fused_edge_backward", &fused_edge_backward,
        "fused edge-block backward: in-LDS recompute + per-edge grads",
        py::arg("h"), py::arg("coord"), py::arg("eattr"), py::arg("row"),
        py::arg("col"), py::arg("dmsg_n"), py::arg("dtrans_n"),
        py::arg("w1"), py::arg("b1"), py::arg("w2"), py::arg("b2"),
        py::arg("w3"), py::arg("b3"), py::arg("w3v"), py::arg("normalize"),
        py::arg("eps"),
        py::arg("prepped") = std::vector<torch::Tensor>{});
  m.def("fused_edge_backward_wg", &fused_edge_backward_wg,
        "edge-block backward with in-kernel MFMA weight gradients "
        "(no per-edge intermediates): {dhr, dhc, dcd, dw3v, gb, gw1, "
        "gw2, gw3}",
        py::arg("h"), py::arg("coord"), py::arg("eattr"), py::arg("row"),
        py::arg("col"), py::arg("dmsg_n"), py::arg("dtrans_n"),
        py::arg("w1"), py::arg("b1"), py::arg("w2"), py::arg("b2"),
        py::arg("w3"), py::arg("b3"), py::arg("w3v"), py::arg("normalize"),
        py::arg("eps"),
        py::arg("prepped") = std::vector<torch::Tensor>{});
  m.def("wgrad_splitk", &wgrad_splitk_launch,
        "split-K MFMA weight gradient: g^T @ x for tall activations",
        py::arg("g"), py::arg("x"));
  m.def("tall_linear", &tall_linear,
        "tall-skinny MFMA linear: act(x @ B^T + bias)",
        py::arg("x"), py::arg("bmat"), py::arg("bias") = c10::nullopt,
        py::arg("act") = 0);
  m.def("fused_virtual_forward", &fused_virtual_forward,
        "fused MFMA virtual-edge block forward");
  m.def("fused_virtual_backward", &fused_virtual_backward,
        "fused MFMA virtual-edge block backward");
  m.def("coord_update_forward", &coord_update_forward,
        "fused coord + agg + trans_v + phi_v*vel");
  m.def("coord_update_backward", &coord_update_backward,
        "dphiv = sum_d g*vel for the fused coordinate update");
  m.def("edge_softmax_fwd", &edge_softmax_fwd,
        "CSR edge softmax over destination segments (SE(3) attention)",
        py::arg("scores"), py::arg("dstptr"), py::arg("perm"));
  m.def("cfconv_forward", &cfconv_forward,
        "fused SchNet CFConv messages: smearing + filter MLP + cutoff + "
        "gathered multiply");
  m.def("mfma_probe", &mfma_probe,
        "16x16x32 bf16 MFMA layout probe: D = A @ B (bt = B^T)",
        py::arg("a"), py::arg("bt"));
}
