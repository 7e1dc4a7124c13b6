/* CPU random negative sampler + uniform random walk.
 *
 * Negative sampling parity: reference csrc/cpu/random_negative_sampler.cc —
 * draw (row, col) uniformly; in strict mode reject pairs present in CSR
 * (binary search within the row — requires indices sorted per row, which
 * glt_amd.data.Topology guarantees) with a bounded trial count; in padded
 * mode top up with unchecked pairs so exactly req_num edges return.
 *
 * Random walk: the reference declares SamplingType.RANDOM_WALK
 * (python/sampler/base.py:329-335) but never implements it; we provide a
 * real uniform random walk (node2vec-style p=q=1) as the missing capability.
 */
#include "../include/common.h"

#include <ATen/Parallel.h>

#include <algorithm>
#include <vector>

namespace glt {

namespace {
inline bool edge_in_csr(const int64_t* indptr, const int64_t* indices,
                        int64_t num_rows, int64_t r, int64_t c) {
  if (r < 0 || r >= num_rows) return false;
  const int64_t* lo = indices + indptr[r];
  const int64_t* hi = indices + indptr[r + 1];
  return std::binary_search(lo, hi, c);
}
}  // namespace

// Returns edge_index [2, m] with m <= req_num (== req_num when padded).
torch::Tensor cpu_sample_negative(const torch::Tensor& indptr,
                                  const torch::Tensor& indices,
                                  int64_t num_cols, int64_t req_num,
                                  int64_t trials, bool padding) {
  check_int64_1d(indptr, "indptr");
  check_int64_1d(indices, "indices");
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t* ci = indices.data_ptr<int64_t>();
  const int64_t num_rows = indptr.size(0) - 1;
  TORCH_CHECK(num_cols > 0 && num_rows > 0, "empty graph");

  const uint64_t call_seed = SeedManager::instance().next_call_seed();
  std::vector<int64_t> rows(req_num, -1), cols(req_num, -1);
  std::atomic<int64_t> found{0};
  at::parallel_for(0, req_num, 256, [&](int64_t s, int64_t e) {
    for (int64_t i = s; i < e; ++i) {
      Rng64 rng(splitmix64(call_seed ^ (uint64_t)i * 0xA24BAED4963EE407ull));
      for (int64_t t = 0; t < trials; ++t) {
        int64_t r = (int64_t)rng.uniform((uint64_t)num_rows);
        int64_t c = (int64_t)rng.uniform((uint64_t)num_cols);
        if (!edge_in_csr(ip, ci, num_rows, r, c)) {
          rows[i] = r;
          cols[i] = c;
          found.fetch_add(1, std::memory_order_relaxed);
          break;
        }
      }
      if (rows[i] < 0 && padding) {
        rows[i] = (int64_t)rng.uniform((uint64_t)num_rows);
        cols[i] = (int64_t)rng.uniform((uint64_t)num_cols);
      }
    }
  });
  // Compact successes to the front (order-preserving).
  int64_t m = 0;
  std::vector<int64_t> out_r, out_c;
  out_r.reserve(req_num);
  out_c.reserve(req_num);
  for (int64_t i = 0; i < req_num; ++i) {
    if (rows[i] >= 0) {
      out_r.push_back(rows[i]);
      out_c.push_back(cols[i]);
      ++m;
    }
  }
  auto out = torch::empty({2, m}, torch::dtype(torch::kInt64));
  std::memcpy(out[0].data_ptr<int64_t>(), out_r.data(), m * sizeof(int64_t));
  std::memcpy(out[1].data_ptr<int64_t>(), out_c.data(), m * sizeof(int64_t));
  return out;
}

// Uniform random walk: returns [n, walk_len + 1]; walks that hit a node with
// no outgoing edge stay there (self-padding), matching common PyG semantics.
torch::Tensor cpu_random_walk(const torch::Tensor& indptr,
                              const torch::Tensor& indices,
                              const torch::Tensor& seeds, int64_t walk_len) {
  check_int64_1d(seeds, "seeds");
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t* ci = indices.data_ptr<int64_t>();
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t n = seeds.size(0);
  const int64_t* sp = seeds.data_ptr<int64_t>();
  auto out = torch::empty({n, walk_len + 1}, seeds.options());
  int64_t* op = out.data_ptr<int64_t>();
  const uint64_t call_seed = SeedManager::instance().next_call_seed();
  at::parallel_for(0, n, 128, [&](int64_t s, int64_t e) {
    for (int64_t i = s; i < e; ++i) {
      Rng64 rng(splitmix64(call_seed ^ (uint64_t)i * 0x9FB21C651E98DF25ull));
      int64_t cur = sp[i];
      int64_t* row = op + i * (walk_len + 1);
      row[0] = cur;
      for (int64_t step = 1; step <= walk_len; ++step) {
        if (cur >= 0 && cur < num_rows && ip[cur + 1] > ip[cur]) {
          const int64_t deg = ip[cur + 1] - ip[cur];
          cur = ci[ip[cur] + (int64_t)rng.uniform((uint64_t)deg)];
        }
        row[step] = cur;
      }
    }
  });
  return out;
}

}  // namespace glt
