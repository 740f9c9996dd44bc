// pybind bindings for the dgl_operator_amd native extension (_C.so).

#include <torch/extension.h>

namespace doa {

at::Tensor spmm(at::Tensor indptr, at::Tensor indices, at::Tensor feat,
                c10::optional<at::Tensor> eweight, bool mean);
at::Tensor spmm_scatter(at::Tensor indptr, at::Tensor indices, at::Tensor grad,
                        c10::optional<at::Tensor> eweight, int64_t num_src);
at::Tensor sddmm_dot(at::Tensor src, at::Tensor dst, at::Tensor feat_u,
                     at::Tensor feat_v);
at::Tensor edge_softmax_fwd(at::Tensor indptr, at::Tensor scores);
at::Tensor edge_softmax_bwd(at::Tensor indptr, at::Tensor out,
                            at::Tensor grad_out);
at::Tensor segment_reduce(at::Tensor offsets, at::Tensor feat, bool mean);
at::Tensor gather_rows(at::Tensor feat, at::Tensor gids,
                       c10::optional<at::Tensor> map, int64_t offset);
at::Tensor gather_mm(at::Tensor feat, at::Tensor rows, at::Tensor weight,
                     c10::optional<at::Tensor> bias);
at::Tensor gat_score_fwd(at::Tensor src, at::Tensor dst, at::Tensor el,
                         at::Tensor er, double slope);
std::tuple<at::Tensor, at::Tensor> gat_score_bwd(at::Tensor src,
                                                 at::Tensor dst,
                                                 at::Tensor el, at::Tensor er,
                                                 at::Tensor gout,
                                                 double slope);
std::tuple<at::Tensor, at::Tensor> sample_neighbors(at::Tensor indptr,
                                                    at::Tensor indices,
                                                    at::Tensor seeds,
                                                    int64_t fanout, bool replace,
                                                    int64_t seed);
std::tuple<at::Tensor, at::Tensor> compact_ids(at::Tensor table,
                                               at::Tensor seeds,
                                               at::Tensor neighbors);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> sample_block(
    at::Tensor indptr, at::Tensor indices, at::Tensor table, at::Tensor seeds,
    int64_t fanout, bool replace, int64_t seed,
    c10::optional<at::Tensor> seed_dev, c10::optional<at::Tensor> rows_opt);
at::Tensor pack_padded(at::Tensor padded, at::Tensor counts,
                       at::Tensor offsets, int64_t total);
void sparse_adagrad(at::Tensor emb, at::Tensor state, at::Tensor ids,
                    at::Tensor grad, double lr, double eps);
at::Tensor ldg_partition(at::Tensor indptr, at::Tensor indices,
                         at::Tensor cindptr, at::Tensor cindices,
                         int64_t num_parts,
                         c10::optional<at::Tensor> train_mask,
                         bool balance_edges);
at::Tensor pdist_neg_fwd(at::Tensor base, at::Tensor neg, int64_t p,
                         double gamma);
std::tuple<at::Tensor, at::Tensor> pdist_neg_bwd(at::Tensor base,
                                                 at::Tensor neg,
                                                 at::Tensor out,
                                                 at::Tensor gout, int64_t p,
                                                 double gamma);
at::Tensor cpdist_neg_fwd(at::Tensor base_r, at::Tensor base_i, at::Tensor neg,
                          double gamma);
std::tuple<at::Tensor, at::Tensor, at::Tensor> cpdist_neg_bwd(
    at::Tensor base_r, at::Tensor base_i, at::Tensor neg, at::Tensor gout);

}  // namespace doa

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dgl_operator_amd native HIP/CDNA4 kernels (gfx950)";
  m.def("spmm", &doa::spmm, "generalized SpMM (copy_u/u_mul_e x sum/mean)",
        py::arg("indptr"), py::arg("indices"), py::arg("feat"),
        py::arg("eweight") = py::none(), py::arg("mean") = false);
  m.def("spmm_scatter", &doa::spmm_scatter,
        "transposed SpMM via atomic scatter (block backward)",
        py::arg("indptr"), py::arg("indices"), py::arg("grad"),
        py::arg("eweight") = py::none(), py::arg("num_src") = 0);
  m.def("sddmm_dot", &doa::sddmm_dot, "per-edge u dot v");
  m.def("edge_softmax_fwd", &doa::edge_softmax_fwd);
  m.def("edge_softmax_bwd", &doa::edge_softmax_bwd);
  m.def("segment_reduce", &doa::segment_reduce);
  m.def("gather_rows", &doa::gather_rows, py::arg("feat"), py::arg("gids"),
        py::arg("map") = py::none(), py::arg("offset") = 0);
  m.def("gather_mm", &doa::gather_mm, py::arg("feat"), py::arg("rows"),
        py::arg("weight"), py::arg("bias") = py::none());
  m.def("gat_score_fwd", &doa::gat_score_fwd);
  m.def("gat_score_bwd", &doa::gat_score_bwd);
  m.def("sample_neighbors", &doa::sample_neighbors);
  m.def("compact_ids", &doa::compact_ids);
  m.def("sample_block", &doa::sample_block, py::arg("indptr"),
        py::arg("indices"), py::arg("table"), py::arg("seeds"),
        py::arg("fanout"), py::arg("replace"), py::arg("seed"),
        py::arg("seed_dev") = py::none(), py::arg("rows") = py::none());
  m.def("pack_padded", &doa::pack_padded);
  m.def("sparse_adagrad", &doa::sparse_adagrad);
  m.def("ldg_partition", &doa::ldg_partition, py::arg("indptr"),
        py::arg("indices"), py::arg("cindptr"), py::arg("cindices"),
        py::arg("num_parts"), py::arg("train_mask") = py::none(),
        py::arg("balance_edges") = false);
  m.def("pdist_neg_fwd", &doa::pdist_neg_fwd);
  m.def("pdist_neg_bwd", &doa::pdist_neg_bwd);
  m.def("cpdist_neg_fwd", &doa::cpdist_neg_fwd);
  m.def("cpdist_neg_bwd", &doa::cpdist_neg_bwd);
}
