/* CPU inducers (header-only; bound directly from bindings.cpp).
 * Semantics: see cpu_inducer.cpp file comment.
 */
#pragma once

#include "common.h"

#include <cstring>
#include <string>
#include <unordered_map>
#include <vector>

namespace glt {

class CPUInducer {
 public:
  explicit CPUInducer(int64_t reserve = 1024) { map_.reserve(reserve); }

  torch::Tensor init_node(const torch::Tensor& seeds) {
    reset();
    check_int64_1d(seeds, "seeds");
    const int64_t n = seeds.size(0);
    const int64_t* sp = seeds.data_ptr<int64_t>();
    std::vector<int64_t> uniq;
    uniq.reserve(n);
    for (int64_t i = 0; i < n; ++i) {
      if (map_.emplace(sp[i], (int64_t)map_.size()).second) uniq.push_back(sp[i]);
    }
    auto out = torch::empty({(int64_t)uniq.size()}, seeds.options());
    std::memcpy(out.data_ptr<int64_t>(), uniq.data(),
                uniq.size() * sizeof(int64_t));
    return out;
  }

  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> induce_next(
      const torch::Tensor& srcs, const torch::Tensor& nbrs,
      const torch::Tensor& nbrs_num) {
    check_int64_1d(srcs, "srcs");
    check_int64_1d(nbrs, "nbrs");
    check_int64_1d(nbrs_num, "nbrs_num");
    const int64_t ns = srcs.size(0);
    const int64_t ne = nbrs.size(0);
    const int64_t* src_p = srcs.data_ptr<int64_t>();
    const int64_t* nbr_p = nbrs.data_ptr<int64_t>();
    const int64_t* num_p = nbrs_num.data_ptr<int64_t>();

    std::vector<int64_t> fresh;
    fresh.reserve(ne);
    for (int64_t i = 0; i < ne; ++i) {
      if (map_.emplace(nbr_p[i], (int64_t)map_.size()).second)
        fresh.push_back(nbr_p[i]);
    }
    auto rows = torch::empty({ne}, srcs.options());
    auto cols = torch::empty({ne}, srcs.options());
    int64_t* rp = rows.data_ptr<int64_t>();
    int64_t* cp = cols.data_ptr<int64_t>();
    int64_t e = 0;
    for (int64_t i = 0; i < ns; ++i) {
      const int64_t src_local = map_.at(src_p[i]);
      for (int64_t j = 0; j < num_p[i]; ++j, ++e) {
        rp[e] = src_local;
        cp[e] = map_.at(nbr_p[e]);
      }
    }
    auto nodes = torch::empty({(int64_t)fresh.size()}, srcs.options());
    std::memcpy(nodes.data_ptr<int64_t>(), fresh.data(),
                fresh.size() * sizeof(int64_t));
    return {nodes, rows, cols};
  }

  void reset() { map_.clear(); }

 private:
  std::unordered_map<int64_t, int64_t> map_;
};

// Hetero inducer: per-node-type incremental maps; edges keyed by
// "src_type__edge_type__dst_type" strings on the Python side — here we take
// parallel lists to stay pybind-friendly.
class CPUHeteroInducer {
 public:
  explicit CPUHeteroInducer(int64_t reserve = 1024) { (void)reserve; }

  std::unordered_map<std::string, torch::Tensor> init_node(
      const std::unordered_map<std::string, torch::Tensor>& seeds) {
    reset();
    std::unordered_map<std::string, torch::Tensor> out;
    for (const auto& kv : seeds) {
      auto& m = maps_[kv.first];
      const int64_t n = kv.second.size(0);
      const int64_t* sp = kv.second.data_ptr<int64_t>();
      std::vector<int64_t> uniq;
      uniq.reserve(n);
      for (int64_t i = 0; i < n; ++i) {
        if (m.emplace(sp[i], (int64_t)m.size()).second) uniq.push_back(sp[i]);
      }
      auto t = torch::empty({(int64_t)uniq.size()}, kv.second.options());
      std::memcpy(t.data_ptr<int64_t>(), uniq.data(),
                  uniq.size() * sizeof(int64_t));
      out.emplace(kv.first, std::move(t));
    }
    return out;
  }

  // One hop over multiple edge types.  Inputs are parallel vectors:
  // for edge i: src_types[i], dst_types[i], srcs[i], nbrs[i], nbrs_num[i].
  // Returns (nodes_dict: new nodes per dst type, rows list, cols list)
  // with rows/cols index-aligned to the input edge order.
  std::tuple<std::unordered_map<std::string, torch::Tensor>,
             std::vector<torch::Tensor>, std::vector<torch::Tensor>>
  induce_next(const std::vector<std::string>& src_types,
              const std::vector<std::string>& dst_types,
              const std::vector<torch::Tensor>& srcs,
              const std::vector<torch::Tensor>& nbrs,
              const std::vector<torch::Tensor>& nbrs_num) {
    const size_t n_rel = srcs.size();
    std::unordered_map<std::string, std::vector<int64_t>> fresh;
    // Phase 1: insert new dst nodes (in relation order, matching reference
    // determinism per type).
    for (size_t r = 0; r < n_rel; ++r) {
      auto& m = maps_[dst_types[r]];
      auto& f = fresh[dst_types[r]];
      const int64_t ne = nbrs[r].size(0);
      const int64_t* np = nbrs[r].data_ptr<int64_t>();
      for (int64_t i = 0; i < ne; ++i) {
        if (m.emplace(np[i], (int64_t)m.size()).second) f.push_back(np[i]);
      }
    }
    // Phase 2: relabel edges.
    std::vector<torch::Tensor> rows_out, cols_out;
    rows_out.reserve(n_rel);
    cols_out.reserve(n_rel);
    for (size_t r = 0; r < n_rel; ++r) {
      const auto& src_m = maps_[src_types[r]];
      const auto& dst_m = maps_[dst_types[r]];
      const int64_t ns = srcs[r].size(0);
      const int64_t ne = nbrs[r].size(0);
      const int64_t* sp = srcs[r].data_ptr<int64_t>();
      const int64_t* np = nbrs[r].data_ptr<int64_t>();
      const int64_t* cnt = nbrs_num[r].data_ptr<int64_t>();
      auto rows = torch::empty({ne}, srcs[r].options());
      auto cols = torch::empty({ne}, srcs[r].options());
      int64_t* rp = rows.data_ptr<int64_t>();
      int64_t* cp = cols.data_ptr<int64_t>();
      int64_t e = 0;
      for (int64_t i = 0; i < ns; ++i) {
        const int64_t sl = src_m.at(sp[i]);
        for (int64_t j = 0; j < cnt[i]; ++j, ++e) {
          rp[e] = sl;
          cp[e] = dst_m.at(np[e]);
        }
      }
      rows_out.push_back(std::move(rows));
      cols_out.push_back(std::move(cols));
    }
    std::unordered_map<std::string, torch::Tensor> nodes_out;
    for (auto& kv : fresh) {
      auto t = torch::empty({(int64_t)kv.second.size()},
                            torch::dtype(torch::kInt64));
      std::memcpy(t.data_ptr<int64_t>(), kv.second.data(),
                  kv.second.size() * sizeof(int64_t));
      nodes_out.emplace(kv.first, std::move(t));
    }
    return {nodes_out, rows_out, cols_out};
  }

  void reset() { maps_.clear(); }

 private:
  std::unordered_map<std::string, std::unordered_map<int64_t, int64_t>> maps_;
};


}  // namespace glt
