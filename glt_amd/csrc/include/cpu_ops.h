#pragma once

#include <torch/extension.h>

#include <string>
#include <unordered_map>
#include <vector>

namespace glt {

std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
cpu_sample_neighbors(const torch::Tensor& indptr, const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const c10::optional<torch::Tensor>& edge_weights,
                     const torch::Tensor& seeds, int64_t k, bool with_edge,
                     bool weighted, bool replace = true);
torch::Tensor cpu_lookup_degree(const torch::Tensor& indptr,
                                const torch::Tensor& nodes);
torch::Tensor cpu_cal_nbr_prob(const torch::Tensor& indptr,
                               const torch::Tensor& indices,
                               const torch::Tensor& last_prob,
                               const torch::Tensor& nodes, int64_t k);
torch::Tensor cpu_sample_negative(const torch::Tensor& indptr,
                                  const torch::Tensor& indices,
                                  int64_t num_cols, int64_t req_num,
                                  int64_t trials, bool padding);
torch::Tensor cpu_random_walk(const torch::Tensor& indptr,
                              const torch::Tensor& indices,
                              const torch::Tensor& seeds, int64_t walk_len);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           c10::optional<torch::Tensor>>
cpu_node_subgraph(const torch::Tensor& indptr, const torch::Tensor& indices,
                  const c10::optional<torch::Tensor>& edge_ids,
                  const torch::Tensor& nodes, bool with_edge);
std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
cpu_stitch_sample_results(int64_t ids_count,
                          const std::vector<torch::Tensor>& idx_list,
                          const std::vector<torch::Tensor>& nbrs_list,
                          const std::vector<torch::Tensor>& nbrs_num_list,
                          const std::vector<torch::Tensor>& eids_list);

}  // namespace glt
