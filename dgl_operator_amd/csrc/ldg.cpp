// Linear Deterministic Greedy (LDG) streaming graph partitioner — native
// replacement for the reference's METIS call
// (dgl.distributed.partition_graph at
//  /root/reference/examples/GraphSAGE_dist/code/load_and_partition_graph.py:124-127;
// METIS itself is not available in this image). Stanton & Kliot LDG: nodes
// stream in a deterministic shuffled order; each is placed on the part with
// the most already-placed neighbors, damped by a capacity penalty.

#include <torch/extension.h>

#include <cstdint>
#include <vector>

namespace doa {

static inline uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

at::Tensor ldg_partition(at::Tensor indptr, at::Tensor indices,
                         at::Tensor cindptr, at::Tensor cindices,
                         int64_t num_parts,
                         c10::optional<at::Tensor> train_mask,
                         bool balance_edges) {
  TORCH_CHECK(!indptr.is_cuda(), "ldg_partition runs on CPU tensors");
  const int64_t n = indptr.numel() - 1;
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t* ix = indices.data_ptr<int64_t>();
  const int64_t* cip = cindptr.data_ptr<int64_t>();
  const int64_t* cix = cindices.data_ptr<int64_t>();
  auto out = at::empty({n}, indptr.options());
  int64_t* assign = out.data_ptr<int64_t>();
  std::fill(assign, assign + n, -1);

  // deterministic pseudo-shuffle order
  std::vector<int64_t> order(n);
  for (int64_t i = 0; i < n; ++i) order[i] = i;
  for (int64_t i = n - 1; i > 0; --i) {
    const int64_t j = (int64_t)(mix64((uint64_t)i) % (uint64_t)(i + 1));
    std::swap(order[i], order[j]);
  }

  // balance dimensions (the reference's METIS balance_train/balance_edges,
  // load_and_partition_graph.py:124-127): node count always; optionally the
  // train-node count and the in+out edge count
  const bool* tm = nullptr;
  at::Tensor tmc;
  if (train_mask.has_value()) {
    tmc = train_mask->to(at::kBool).contiguous();
    tm = tmc.data_ptr<bool>();
  }
  const double cap = (double)(n + num_parts - 1) / num_parts * 1.05 + 1.0;
  double train_total = 0.0, edge_total = 0.0;
  if (tm) for (int64_t i = 0; i < n; ++i) train_total += tm[i] ? 1.0 : 0.0;
  if (balance_edges)
    edge_total = (double)(indices.numel() + cindices.numel());
  const double tcap = train_total / num_parts * 1.05 + 1.0;
  const double ecap = edge_total / num_parts * 1.05 + 1.0;
  std::vector<int64_t> sizes(num_parts, 0);
  std::vector<double> tsizes(num_parts, 0.0), esizes(num_parts, 0.0);
  std::vector<int64_t> counts(num_parts);
  for (int64_t t = 0; t < n; ++t) {
    const int64_t v = order[t];
    std::fill(counts.begin(), counts.end(), 0);
    for (int64_t p = ip[v]; p < ip[v + 1]; ++p) {
      const int64_t a = assign[ix[p]];
      if (a >= 0) counts[a]++;
    }
    for (int64_t p = cip[v]; p < cip[v + 1]; ++p) {
      const int64_t a = assign[cix[p]];
      if (a >= 0) counts[a]++;
    }
    const double vdeg =
        (double)((ip[v + 1] - ip[v]) + (cip[v + 1] - cip[v]));
    int best = 0;
    double best_score = -1.0;
    for (int64_t q = 0; q < num_parts; ++q) {
      double penalty = 1.0 - sizes[q] / cap;
      if (tm) penalty *= (1.0 - tsizes[q] / tcap);
      if (balance_edges) penalty *= (1.0 - esizes[q] / ecap);
      const double score = (counts[q] + 1e-9) * std::max(penalty, 0.0);
      if (score > best_score) {
        best_score = score;
        best = (int)q;
      }
    }
    assign[v] = best;
    sizes[best]++;
    if (tm && tm[v]) tsizes[best] += 1.0;
    if (balance_edges) esizes[best] += vdeg;
  }
  return out;
}

}  // namespace doa
