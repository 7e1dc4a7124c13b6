/* CPU subgraph inducers, node-set subgraph op and distributed-stitch.
 *
 * Semantics parity (reference csrc/cpu/inducer.cc, subgraph_op.cc,
 * stitch_sample_results.cc):
 *  - Inducer keeps an incremental global-id -> local-index map across hops;
 *    init_node(seeds) resets it and returns the deduped seed list;
 *    induce_next(srcs, nbrs, nbrs_num) inserts newly-seen neighbor ids and
 *    returns (new_nodes, rows, cols) with rows/cols relabeled to local ids.
 *  - node_subgraph(nodes) induces every edge among `nodes` (relabeled),
 *    deduping nodes first.
 *  - stitch merges per-partition one-hop results back into seed order.
 */
#include "../include/common.h"
#include "../include/cpu_inducer.h"

#include <ATen/Parallel.h>

#include <cstring>
#include <unordered_map>
#include <vector>

namespace glt {

// Induce the full edge set among `nodes` from CSR.
// Returns (unique_nodes, rows, cols, eids?) with rows/cols relabeled into
// unique_nodes positions.  Parity: reference csrc/cpu/subgraph_op.cc.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           c10::optional<torch::Tensor>>
cpu_node_subgraph(const torch::Tensor& indptr, const torch::Tensor& indices,
                  const c10::optional<torch::Tensor>& edge_ids,
                  const torch::Tensor& nodes, bool with_edge) {
  check_int64_1d(nodes, "nodes");
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int64_t* ci = indices.data_ptr<int64_t>();
  const int64_t* ei =
      edge_ids.has_value() ? edge_ids->data_ptr<int64_t>() : nullptr;
  TORCH_CHECK(!with_edge || ei, "with_edge requires edge_ids");
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t n = nodes.size(0);
  const int64_t* np = nodes.data_ptr<int64_t>();

  std::unordered_map<int64_t, int64_t> map;
  map.reserve(n * 2);
  std::vector<int64_t> uniq;
  uniq.reserve(n);
  for (int64_t i = 0; i < n; ++i) {
    if (map.emplace(np[i], (int64_t)map.size()).second) uniq.push_back(np[i]);
  }
  std::vector<int64_t> rows, cols, eids;
  for (size_t u = 0; u < uniq.size(); ++u) {
    const int64_t v = uniq[u];
    if (v < 0 || v >= num_rows) continue;
    for (int64_t e = ip[v]; e < ip[v + 1]; ++e) {
      auto it = map.find(ci[e]);
      if (it != map.end()) {
        rows.push_back((int64_t)u);
        cols.push_back(it->second);
        if (with_edge) eids.push_back(ei[e]);
      }
    }
  }
  auto opts = nodes.options();
  auto nodes_t = torch::empty({(int64_t)uniq.size()}, opts);
  std::memcpy(nodes_t.data_ptr<int64_t>(), uniq.data(),
              uniq.size() * sizeof(int64_t));
  auto rows_t = torch::empty({(int64_t)rows.size()}, opts);
  std::memcpy(rows_t.data_ptr<int64_t>(), rows.data(),
              rows.size() * sizeof(int64_t));
  auto cols_t = torch::empty({(int64_t)cols.size()}, opts);
  std::memcpy(cols_t.data_ptr<int64_t>(), cols.data(),
              cols.size() * sizeof(int64_t));
  c10::optional<torch::Tensor> eids_t = c10::nullopt;
  if (with_edge) {
    auto t = torch::empty({(int64_t)eids.size()}, opts);
    std::memcpy(t.data_ptr<int64_t>(), eids.data(),
                eids.size() * sizeof(int64_t));
    eids_t = t;
  }
  return {nodes_t, rows_t, cols_t, eids_t};
}

// Merge per-partition partial one-hop results back into seed order.
// idx_list[p][i] gives the position (in the original seed batch) of the i-th
// seed served by partition p.  Parity: reference cpu/stitch_sample_results.cc.
std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
cpu_stitch_sample_results(int64_t ids_count,
                          const std::vector<torch::Tensor>& idx_list,
                          const std::vector<torch::Tensor>& nbrs_list,
                          const std::vector<torch::Tensor>& nbrs_num_list,
                          const std::vector<torch::Tensor>& eids_list) {
  const size_t P = idx_list.size();
  const bool with_edge = !eids_list.empty();
  auto opts = torch::dtype(torch::kInt64);

  std::vector<int64_t> num(ids_count, 0);
  for (size_t p = 0; p < P; ++p) {
    const int64_t* idx = idx_list[p].data_ptr<int64_t>();
    const int64_t* cnt = nbrs_num_list[p].data_ptr<int64_t>();
    const int64_t m = idx_list[p].size(0);
    for (int64_t i = 0; i < m; ++i) num[idx[i]] = cnt[i];
  }
  std::vector<int64_t> off(ids_count + 1, 0);
  for (int64_t i = 0; i < ids_count; ++i) off[i + 1] = off[i] + num[i];
  const int64_t total = off[ids_count];

  auto nbrs_num = torch::empty({ids_count}, opts);
  std::memcpy(nbrs_num.data_ptr<int64_t>(), num.data(),
              ids_count * sizeof(int64_t));
  auto nbrs = torch::zeros({total}, opts);
  auto eids = with_edge ? torch::zeros({total}, opts) : torch::Tensor();
  int64_t* nb = nbrs.data_ptr<int64_t>();
  int64_t* eb = with_edge ? eids.data_ptr<int64_t>() : nullptr;

  for (size_t p = 0; p < P; ++p) {
    const int64_t* idx = idx_list[p].data_ptr<int64_t>();
    const int64_t* cnt = nbrs_num_list[p].data_ptr<int64_t>();
    const int64_t* src = nbrs_list[p].data_ptr<int64_t>();
    const int64_t* se = with_edge ? eids_list[p].data_ptr<int64_t>() : nullptr;
    const int64_t m = idx_list[p].size(0);
    int64_t local = 0;
    for (int64_t i = 0; i < m; ++i) {
      const int64_t c = cnt[i];
      if (c > 0) {
        std::memcpy(nb + off[idx[i]], src + local, c * sizeof(int64_t));
        if (with_edge)
          std::memcpy(eb + off[idx[i]], se + local, c * sizeof(int64_t));
      }
      local += c;
    }
  }
  return {nbrs, nbrs_num,
          with_edge ? c10::optional<torch::Tensor>(eids) : c10::nullopt};
}

}  // namespace glt
