// Native batch collator (reference K17 — the PyG DataLoader concat +
// edge-offset + batch-vector build of pert_gnn.py:196-210, re-designed as a
// single C++ pass producing the CSR/CSC layout the HIP kernels consume).
//
// One call concatenates all per-sample tensors (parallel memcpy via
// at::parallel_for), offsets edge indices, counting-sorts edges by
// destination (O(E), stable), builds row_ptr/col_ptr + the CSC permutation,
// and permutes edge_attr into CSR order.  Optionally writes into pinned
// host memory so the H2D copy of the whole batch is a single async DMA.

#include <torch/extension.h>

#include <ATen/Parallel.h>
#include <cstring>

#include "collate_core.h"

namespace {

int64_t total_rows(const std::vector<torch::Tensor>& ts, int dim = 0) {
  int64_t n = 0;
  for (const auto& t : ts) n += t.size(dim);
  return n;
}

torch::TensorOptions opts(torch::ScalarType t, bool pin) {
  auto o = torch::TensorOptions().dtype(t);
  return pin ? o.pinned_memory(true) : o;
}

// concat float/long 2-D tensors along dim 0 (parallel over samples)
torch::Tensor cat_rows(const std::vector<torch::Tensor>& ts, bool pin) {
  const int64_t rows = total_rows(ts);
  const int64_t cols = ts[0].dim() > 1 ? ts[0].size(1) : 1;
  auto out = torch::empty({rows, cols}, opts(ts[0].scalar_type(), pin));
  std::vector<int64_t> offs(ts.size() + 1, 0);
  for (size_t i = 0; i < ts.size(); ++i) offs[i + 1] = offs[i] + ts[i].size(0);
  const size_t esz = ts[0].element_size();
  char* dst = (char*)out.data_ptr();
  at::parallel_for(0, (int64_t)ts.size(), 1, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) {
      auto src = ts[i].contiguous();
      std::memcpy(dst + offs[i] * cols * esz, src.data_ptr(),
                  (size_t)src.size(0) * cols * esz);
    }
  });
  return out;
}

}  // namespace

// returns {x, cat_X, edge_index(csr order), edge_attr(csr order), rt_probs,
//          pattern_num_nodes, node_depth, entry_id, batch, batch_ptr, y,
//          row_ptr, csr_src, col_ptr, csc_dst, csc_eid}
std::vector<torch::Tensor> collate_native(
    std::vector<torch::Tensor> xs, std::vector<torch::Tensor> edge_indices,
    std::vector<torch::Tensor> edge_attrs, std::vector<torch::Tensor> cat_xs,
    std::vector<torch::Tensor> rt_probs, std::vector<torch::Tensor> pnns,
    std::vector<torch::Tensor> node_depths,
    std::vector<torch::Tensor> entry_ids, std::vector<torch::Tensor> ys,
    bool pin) {
  const int64_t bsz = (int64_t)xs.size();
  TORCH_CHECK(bsz > 0, "empty batch");
  const int64_t n_nodes = total_rows(xs);
  const int64_t n_edges = total_rows(edge_indices, 1);

  auto x = cat_rows(xs, pin);
  auto cat_X = cat_rows(cat_xs, pin);
  auto probs = cat_rows(rt_probs, pin);
  auto pnn = cat_rows(pnns, pin);
  auto nd = cat_rows(node_depths, pin);

  auto entry = torch::empty({bsz}, opts(torch::kLong, pin));
  auto y = torch::empty({bsz}, opts(torch::kFloat, pin));
  auto batch = torch::empty({n_nodes}, opts(torch::kLong, pin));
  auto batch_ptr = torch::empty({bsz + 1}, opts(torch::kInt, pin));
  long* batch_p = batch.data_ptr<long>();
  int* bptr = batch_ptr.data_ptr<int>();
  {
    int64_t off = 0;
    bptr[0] = 0;
    for (int64_t i = 0; i < bsz; ++i) {
      entry.data_ptr<long>()[i] = entry_ids[i].item<long>();
      y.data_ptr<float>()[i] = ys[i].item<float>();
      const int64_t nn = xs[i].size(0);
      for (int64_t r = 0; r < nn; ++r) batch_p[off + r] = i;
      off += nn;
      bptr[i + 1] = (int)off;
    }
  }

  // gather offset edges (original order)
  std::vector<int64_t> eoffs(bsz + 1, 0), noffs(bsz + 1, 0);
  for (int64_t i = 0; i < bsz; ++i) {
    eoffs[i + 1] = eoffs[i] + edge_indices[i].size(1);
    noffs[i + 1] = noffs[i] + xs[i].size(0);
  }
  std::vector<long> src0(n_edges), dst0(n_edges);
  at::parallel_for(0, bsz, 1, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) {
      auto ei = edge_indices[i].contiguous();
      const long* s = ei.data_ptr<long>();
      const long* d = s + ei.size(1);
      for (int64_t j = 0; j < ei.size(1); ++j) {
        src0[eoffs[i] + j] = s[j] + noffs[i];
        dst0[eoffs[i] + j] = d[j] + noffs[i];
      }
    }
  });

  // stable counting sort by dst -> CSR (core shared with the sanitizer
  // harness, csrc/collate_core.h)
  auto row_ptr = torch::empty({n_nodes + 1}, opts(torch::kInt, pin));
  auto perm = torch::empty({n_edges}, torch::kLong);  // original -> csr slot
  long* pm = perm.data_ptr<long>();
  auto csr_src = torch::empty({n_edges}, opts(torch::kInt, pin));
  int* cs = csr_src.data_ptr<int>();
  auto edge_index = torch::empty({2, n_edges}, opts(torch::kLong, pin));
  long* ei_s = edge_index.data_ptr<long>();
  long* ei_d = ei_s + n_edges;
  pertgnn_core::build_csr(src0.data(), dst0.data(), n_edges, n_nodes,
                          row_ptr.data_ptr<int>(), pm, cs, ei_s, ei_d);

  // permute edge_attr into CSR order
  const int64_t acols = edge_attrs[0].size(1);
  auto edge_attr = torch::empty({n_edges, acols}, opts(torch::kLong, pin));
  long* ea = edge_attr.data_ptr<long>();
  at::parallel_for(0, bsz, 1, [&](int64_t b, int64_t e) {
    for (int64_t i = b; i < e; ++i) {
      auto a = edge_attrs[i].contiguous();
      pertgnn_core::permute_attrs(a.data_ptr<long>(), a.size(0), acols,
                                  pm + eoffs[i], ea);
    }
  });

  // CSC over the CSR-ordered edges (stable counting sort by src)
  auto col_ptr = torch::empty({n_nodes + 1}, opts(torch::kInt, pin));
  auto csc_dst = torch::empty({n_edges}, opts(torch::kInt, pin));
  auto csc_eid = torch::empty({n_edges}, opts(torch::kInt, pin));
  pertgnn_core::build_csc(cs, ei_d, n_edges, n_nodes,
                          col_ptr.data_ptr<int>(), csc_dst.data_ptr<int>(),
                          csc_eid.data_ptr<int>());

  return {x, cat_X, edge_index, edge_attr, probs, pnn, nd,
          entry, batch, batch_ptr, y, row_ptr, csr_src, col_ptr,
          csc_dst, csc_eid};
}
