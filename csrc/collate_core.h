// Torch-free core of the native collator (reference K17): stable counting
// sorts building the CSR/CSC edge layout.  Kept free of torch/ATen so the
// sanitizer harness (csrc/sanitize_main.cpp, SURVEY.md §5 "race detection /
// sanitizers") can compile EXACTLY this code under
// -fsanitize=address,undefined and hammer it with adversarial shapes; the
// extension (csrc/collate.cpp) includes and calls the same functions.
#pragma once

#include <cstdint>
#include <vector>

namespace pertgnn_core {

// Stable counting sort of (src0, dst0) edge lists by dst -> CSR layout.
//   rp   [n_nodes+1]  row_ptr (output, zero-initialized by this function)
//   pm   [n_edges]    original edge position -> CSR slot
//   cs   [n_edges]    csr_src: source node per CSR-ordered edge
//   ei_s/ei_d [n_edges] CSR-ordered src/dst (int64 copies for edge_index)
inline void build_csr(const int64_t* src0, const int64_t* dst0,
                      int64_t n_edges, int64_t n_nodes, int* rp, int64_t* pm,
                      int* cs, int64_t* ei_s, int64_t* ei_d) {
  for (int64_t i = 0; i <= n_nodes; ++i) rp[i] = 0;
  for (int64_t e = 0; e < n_edges; ++e) rp[dst0[e] + 1]++;
  for (int64_t i = 0; i < n_nodes; ++i) rp[i + 1] += rp[i];
  std::vector<int> cursor(rp, rp + n_nodes);
  for (int64_t e = 0; e < n_edges; ++e) {
    const int slot = cursor[dst0[e]]++;
    pm[e] = slot;
    cs[slot] = (int)src0[e];
    ei_s[slot] = src0[e];
    ei_d[slot] = dst0[e];
  }
}

// Stable counting sort of the CSR-ordered edges by src -> CSC layout.
//   cp [n_nodes+1]  col_ptr (output, zero-initialized here)
//   cd [n_edges]    csc_dst: destination node per CSC-ordered edge
//   ce [n_edges]    csc_eid: CSR edge id per CSC-ordered edge
inline void build_csc(const int* cs, const int64_t* ei_d, int64_t n_edges,
                      int64_t n_nodes, int* cp, int* cd, int* ce) {
  for (int64_t i = 0; i <= n_nodes; ++i) cp[i] = 0;
  for (int64_t e = 0; e < n_edges; ++e) cp[cs[e] + 1]++;
  for (int64_t i = 0; i < n_nodes; ++i) cp[i + 1] += cp[i];
  std::vector<int> cursor(cp, cp + n_nodes);
  for (int64_t e = 0; e < n_edges; ++e) {
    const int slot = cursor[cs[e]]++;
    cd[slot] = (int)ei_d[e];
    ce[slot] = (int)e;
  }
}

// Permute per-sample edge attributes into CSR order.
//   ap [sample_edges * acols] one sample's attrs; pm offset by the sample's
//   first edge position; ea [n_edges * acols] output.
inline void permute_attrs(const int64_t* ap, int64_t sample_edges,
                          int64_t acols, const int64_t* pm_off, int64_t* ea) {
  for (int64_t j = 0; j < sample_edges; ++j) {
    const int64_t slot = pm_off[j];
    for (int64_t c = 0; c < acols; ++c)
      ea[slot * acols + c] = ap[j * acols + c];
  }
}

}  // namespace pertgnn_core
