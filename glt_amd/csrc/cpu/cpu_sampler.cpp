/* CPU neighbor samplers.
 *
 * Uniform sampling is WITHOUT replacement (Floyd's algorithm), matching the
 * GPU reservoir kernel's semantics.  The reference diverges here: its CPU
 * path draws with replacement (reference csrc/cpu/random_sampler.cc:140-153)
 * while its CUDA kernel is a reservoir (csrc/cuda/random_sampler.cu:59-109);
 * we fix the divergence in favor of without-replacement on both devices.
 *
 * Weighted sampling draws WITH replacement proportional to edge weight when
 * degree > k, else copies all neighbors (parity:
 * reference csrc/cpu/weighted_sampler.cc:146-165).
 */
#include "../include/common.h"

#include <ATen/Parallel.h>

#include <algorithm>
#include <cmath>
#include <vector>

namespace glt {

namespace {

// Sample k distinct positions from [0, deg) using Floyd's algorithm.
// O(k) expected; the chosen set is kept sorted-insert in a small array
// (k is the fan-out: typically <= 32).
inline void floyd_sample(Rng64& rng, int64_t deg, int64_t k, int64_t* out) {
  int64_t n = 0;
  for (int64_t j = deg - k; j < deg; ++j) {
    int64_t t = (int64_t)rng.uniform((uint64_t)(j + 1));
    bool dup = false;
    for (int64_t i = 0; i < n; ++i) {
      if (out[i] == t) { dup = true; break; }
    }
    out[n++] = dup ? j : t;
  }
}

struct CsrView {
  const int64_t* indptr;
  const int64_t* indices;
  const int64_t* eids;     // may be null
  const float* weights;    // may be null
  int64_t num_rows;
  inline int64_t degree(int64_t v) const {
    return (v >= 0 && v < num_rows) ? indptr[v + 1] - indptr[v] : 0;
  }
};

CsrView make_view(const torch::Tensor& indptr, const torch::Tensor& indices,
                  const c10::optional<torch::Tensor>& edge_ids,
                  const c10::optional<torch::Tensor>& edge_weights) {
  CsrView v;
  v.indptr = indptr.data_ptr<int64_t>();
  v.indices = indices.data_ptr<int64_t>();
  v.eids = edge_ids.has_value() ? edge_ids->data_ptr<int64_t>() : nullptr;
  v.weights =
      edge_weights.has_value() ? edge_weights->data_ptr<float>() : nullptr;
  v.num_rows = indptr.size(0) - 1;
  return v;
}

}  // namespace

// Returns (nbrs, nbrs_num, eids?) — concatenated per-seed neighbor samples in
// seed order; nbrs_num[i] = number of neighbors emitted for seeds[i].
std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
cpu_sample_neighbors(const torch::Tensor& indptr, const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const c10::optional<torch::Tensor>& edge_weights,
                     const torch::Tensor& seeds, int64_t k, bool with_edge,
                     bool weighted, bool replace) {
  check_int64_1d(indptr, "indptr");
  check_int64_1d(indices, "indices");
  check_int64_1d(seeds, "seeds");
  TORCH_CHECK(!with_edge || edge_ids.has_value(),
              "with_edge requires edge_ids");
  TORCH_CHECK(!weighted || edge_weights.has_value(),
              "weighted sampling requires edge_weights");
  const auto view = make_view(indptr, indices, edge_ids, edge_weights);
  const int64_t bs = seeds.size(0);
  const int64_t* seed_ptr = seeds.data_ptr<int64_t>();
  const int64_t kk = k < 0 ? std::numeric_limits<int64_t>::max() : k;

  auto opts = seeds.options();
  auto nbrs_num = torch::empty({bs}, opts);
  int64_t* num_ptr = nbrs_num.data_ptr<int64_t>();
  std::vector<int64_t> offset(bs + 1);
  offset[0] = 0;
  for (int64_t i = 0; i < bs; ++i) {
    num_ptr[i] = std::min<int64_t>(kk, view.degree(seed_ptr[i]));
    offset[i + 1] = offset[i] + num_ptr[i];
  }
  const int64_t total = offset[bs];
  auto nbrs = torch::empty({total}, opts);
  auto out_eids =
      with_edge ? torch::empty({total}, opts) : torch::Tensor();
  int64_t* nbrs_ptr = nbrs.data_ptr<int64_t>();
  int64_t* out_eid_ptr = with_edge ? out_eids.data_ptr<int64_t>() : nullptr;

  const uint64_t call_seed = SeedManager::instance().next_call_seed();

  at::parallel_for(0, bs, 64, [&](int64_t start, int64_t end) {
    std::vector<int64_t> pick;
    for (int64_t i = start; i < end; ++i) {
      const int64_t v = seed_ptr[i];
      const int64_t deg = view.degree(v);
      if (deg == 0) continue;
      const int64_t base = view.indptr[v];
      int64_t* out = nbrs_ptr + offset[i];
      int64_t* oe = with_edge ? out_eid_ptr + offset[i] : nullptr;
      if (deg <= kk) {
        for (int64_t j = 0; j < deg; ++j) {
          out[j] = view.indices[base + j];
          if (oe) oe[j] = view.eids[base + j];
        }
        continue;
      }
      Rng64 rng(splitmix64(call_seed ^ (uint64_t)i * 0xD6E8FEB86659FD93ull));
      if (!weighted) {
        pick.resize(kk);
        floyd_sample(rng, deg, kk, pick.data());
        for (int64_t j = 0; j < kk; ++j) {
          out[j] = view.indices[base + pick[j]];
          if (oe) oe[j] = view.eids[base + pick[j]];
        }
      } else if (replace) {
        // CDF + binary search, with replacement.
        pick.resize(deg);
        double acc = 0.0;
        std::vector<double> cdf(deg);
        for (int64_t j = 0; j < deg; ++j) {
          acc += std::max(0.0f, view.weights[base + j]);
          cdf[j] = acc;
        }
        for (int64_t j = 0; j < kk; ++j) {
          double r = (double)rng.uniform_float() * acc;
          int64_t idx = std::lower_bound(cdf.begin(), cdf.end(), r) -
                        cdf.begin();
          if (idx >= deg) idx = deg - 1;
          out[j] = view.indices[base + idx];
          if (oe) oe[j] = view.eids[base + idx];
        }
      } else {
        // Weighted WITHOUT replacement: Efraimidis-Spirakis exponential
        // race — the kk smallest keys -log(u_i)/w_i (matches the GPU
        // kernel; zero/negative weights never selected, uniform pad if
        // fewer than kk positive weights).
        std::vector<std::pair<float, int64_t>> keys;
        keys.reserve(deg);
        for (int64_t j = 0; j < deg; ++j) {
          const float w = view.weights[base + j];
          if (!(w > 0.f)) continue;
          float u = rng.uniform_float();
          u = u < 1e-12f ? 1e-12f : u;
          keys.emplace_back(-std::log(u) / w, j);
        }
        const int64_t npos = (int64_t)keys.size();
        const int64_t take = std::min(npos, kk);
        std::partial_sort(keys.begin(), keys.begin() + take, keys.end());
        for (int64_t j = 0; j < take; ++j) {
          out[j] = view.indices[base + keys[j].second];
          if (oe) oe[j] = view.eids[base + keys[j].second];
        }
        for (int64_t j = take; j < kk; ++j) {
          const int64_t idx = (int64_t)rng.uniform((uint64_t)deg);
          out[j] = view.indices[base + idx];
          if (oe) oe[j] = view.eids[base + idx];
        }
      }
    }
  });

  return {nbrs, nbrs_num,
          with_edge ? c10::optional<torch::Tensor>(out_eids) : c10::nullopt};
}

// Per-node degrees for a node list (parity: reference graph.cu:30-48 /
// LookupDegree).
torch::Tensor cpu_lookup_degree(const torch::Tensor& indptr,
                                const torch::Tensor& nodes) {
  check_int64_1d(indptr, "indptr");
  check_int64_1d(nodes, "nodes");
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t n = nodes.size(0);
  const int64_t* np = nodes.data_ptr<int64_t>();
  auto out = torch::empty({n}, nodes.options());
  int64_t* op = out.data_ptr<int64_t>();
  at::parallel_for(0, n, 1024, [&](int64_t s, int64_t e) {
    for (int64_t i = s; i < e; ++i) {
      const int64_t v = np[i];
      op[i] = (v >= 0 && v < num_rows) ? ip[v + 1] - ip[v] : 0;
    }
  });
  return out;
}

// Importance-probability propagation ("trim" training support).
// Given last_prob over all nodes and the sampled one-hop fan-out k, computes
// cur_prob[u] = 1 - (1 - last_prob[u]) * prod_{v : u in N(v)} (1 - p_keep)
// approximated exactly as the reference kernel does per edge walk:
// for every node v with degree d, each neighbor u gets selected with
// probability min(1, k/d) when v itself is sampled (prob last_prob[v]).
// Parity: reference csrc/cuda/random_sampler.cu:167-209 (CalNbrProbKernel).
torch::Tensor cpu_cal_nbr_prob(const torch::Tensor& indptr,
                               const torch::Tensor& indices,
                               const torch::Tensor& last_prob,
                               const torch::Tensor& nodes, int64_t k) {
  check_int64_1d(indptr, "indptr");
  check_int64_1d(indices, "indices");
  check_int64_1d(nodes, "nodes");
  TORCH_CHECK(last_prob.scalar_type() == torch::kFloat32,
              "last_prob must be float32");
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t* ci = indices.data_ptr<int64_t>();
  const float* lp = last_prob.data_ptr<float>();
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t n = nodes.size(0);
  const int64_t* np = nodes.data_ptr<int64_t>();

  // cur_prob accumulates log(1 - p_v * min(1, k/deg)) over incident seeds.
  auto cur = torch::zeros({last_prob.size(0)}, last_prob.options());
  float* cp = cur.data_ptr<float>();
  // Serial over seeds (atomic-free); n is one batch hop — small.
  for (int64_t i = 0; i < n; ++i) {
    const int64_t v = np[i];
    if (v < 0 || v >= num_rows) continue;
    const int64_t deg = ip[v + 1] - ip[v];
    if (deg == 0) continue;
    const float p_edge =
        std::min(1.0f, (float)k / (float)deg) * lp[v];
    if (p_edge <= 0.f) continue;
    const float log1m = std::log(std::max(1e-20f, 1.0f - p_edge));
    for (int64_t e = ip[v]; e < ip[v + 1]; ++e) cp[ci[e]] += log1m;
  }
  at::parallel_for(0, cur.size(0), 8192, [&](int64_t s, int64_t e) {
    for (int64_t i = s; i < e; ++i) {
      const float keep = 1.0f - std::exp(cp[i]);
      cp[i] = 1.0f - (1.0f - lp[i]) * (1.0f - keep);
    }
  });
  return cur;
}

}  // namespace glt
