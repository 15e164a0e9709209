// Sanitizer harness for the native collator core (SURVEY.md §5 "race
// detection / sanitizers"): compiled by tests/test_sanitizer.py with
//   g++ -std=c++17 -g -O1 -fsanitize=address,undefined -fno-sanitize-recover=all
// and run over adversarial shapes.  Any heap overflow / UB in the counting
// sorts aborts the process; the harness additionally checks the structural
// invariants (monotone ptrs, bijective permutation, CSR/CSC consistency).
#include <cassert>
#include <cstdio>
#include <cstdlib>
#include <random>
#include <vector>

#include "collate_core.h"

static void check_case(const std::vector<int64_t>& src,
                       const std::vector<int64_t>& dst, int64_t n_nodes) {
  const int64_t E = (int64_t)src.size();
  std::vector<int> rp(n_nodes + 1), cs(E), cp(n_nodes + 1), cd(E), ce(E);
  std::vector<int64_t> pm(E), ei_s(E), ei_d(E);
  pertgnn_core::build_csr(src.data(), dst.data(), E, n_nodes, rp.data(),
                          pm.data(), cs.data(), ei_s.data(), ei_d.data());
  pertgnn_core::build_csc(cs.data(), ei_d.data(), E, n_nodes, cp.data(),
                          cd.data(), ce.data());

  // invariants
  assert(rp[0] == 0 && rp[n_nodes] == E);
  assert(cp[0] == 0 && cp[n_nodes] == E);
  for (int64_t i = 0; i < n_nodes; ++i) {
    assert(rp[i] <= rp[i + 1]);
    assert(cp[i] <= cp[i + 1]);
  }
  std::vector<char> seen(E, 0);
  for (int64_t e = 0; e < E; ++e) {
    assert(pm[e] >= 0 && pm[e] < E && !seen[pm[e]]);
    seen[pm[e]] = 1;
  }
  // CSR rows hold exactly the edges whose dst is the row
  for (int64_t r = 0; r < n_nodes; ++r)
    for (int e = rp[r]; e < rp[r + 1]; ++e) assert(ei_d[e] == r);
  // CSC cols hold exactly the edges whose src is the col, eids valid
  for (int64_t c = 0; c < n_nodes; ++c)
    for (int e = cp[c]; e < cp[c + 1]; ++e) {
      assert(ce[e] >= 0 && ce[e] < E);
      assert(cs[ce[e]] == c);
      assert(cd[e] == ei_d[ce[e]]);
    }

  // attr permutation round-trip (2 columns)
  std::vector<int64_t> ap(E * 2), ea(E * 2);
  for (int64_t e = 0; e < E * 2; ++e) ap[e] = e;
  pertgnn_core::permute_attrs(ap.data(), E, 2, pm.data(), ea.data());
  for (int64_t e = 0; e < E; ++e) {
    assert(ea[pm[e] * 2] == e * 2);
    assert(ea[pm[e] * 2 + 1] == e * 2 + 1);
  }
}

int main() {
  // adversarial shapes: empty, single self-loop, all-into-one hub (the PERT
  // root fan-in), duplicate edges, dense clique, randomized fuzz
  check_case({}, {}, 0);
  check_case({}, {}, 5);
  check_case({0}, {0}, 1);
  {
    std::vector<int64_t> s, d;
    for (int i = 0; i < 1000; ++i) { s.push_back(i % 37); d.push_back(0); }
    check_case(s, d, 37);
  }
  {
    std::vector<int64_t> s, d;  // duplicates
    for (int i = 0; i < 64; ++i) { s.push_back(3); d.push_back(4); }
    check_case(s, d, 8);
  }
  {
    std::vector<int64_t> s, d;  // clique
    for (int a = 0; a < 20; ++a)
      for (int b = 0; b < 20; ++b) { s.push_back(a); d.push_back(b); }
    check_case(s, d, 20);
  }
  std::mt19937_64 rng(7);
  for (int it = 0; it < 50; ++it) {
    const int64_t n = 1 + (int64_t)(rng() % 300);
    const int64_t e = rng() % 2000;
    std::vector<int64_t> s(e), d(e);
    for (int64_t j = 0; j < e; ++j) {
      s[j] = (int64_t)(rng() % n);
      d[j] = (int64_t)(rng() % n);
    }
    check_case(s, d, n);
  }
  std::puts("sanitize: all cases clean");
  return 0;
}
