/* Host-callable declarations for the HIP (gfx950) kernels.
 *
 * bindings.cpp is compiled by the host compiler, so everything HIP-side is
 * exposed as plain functions + opaque handles here; implementations live in
 * csrc/hip/*.hip.
 */
#pragma once

#include <torch/extension.h>

#include <memory>
#include <string>
#include <vector>

namespace glt {

// --- samplers (hip_sampler.hip) -------------------------------------------
std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
hip_sample_neighbors(const torch::Tensor& indptr, const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const c10::optional<torch::Tensor>& edge_weights,
                     const torch::Tensor& seeds, int64_t k, bool with_edge,
                     bool weighted, bool replace = true);
// Staged sampling: stage 1 (counts+offsets, no host sync) lets callers
// batch the totals sync across edge types; stage 2 gathers.
std::tuple<torch::Tensor, torch::Tensor> hip_sample_neighbors_offsets(
    const torch::Tensor& indptr, const torch::Tensor& seeds, int64_t k);
std::tuple<torch::Tensor, c10::optional<torch::Tensor>>
hip_sample_neighbors_gather(const torch::Tensor& indptr,
                            const torch::Tensor& indices,
                            const c10::optional<torch::Tensor>& edge_ids,
                            const c10::optional<torch::Tensor>& edge_weights,
                            const torch::Tensor& seeds, int64_t k,
                            const torch::Tensor& offsets, int64_t total,
                            bool with_edge, bool weighted, bool replace);
torch::Tensor hip_lookup_degree(const torch::Tensor& indptr,
                                const torch::Tensor& nodes);
torch::Tensor hip_sample_negative(const torch::Tensor& indptr,
                                  const torch::Tensor& indices,
                                  int64_t num_cols, int64_t req_num,
                                  int64_t trials, bool padding);
torch::Tensor hip_random_walk(const torch::Tensor& indptr,
                              const torch::Tensor& indices,
                              const torch::Tensor& seeds, int64_t walk_len);
torch::Tensor hip_cal_nbr_prob(const torch::Tensor& indptr,
                               const torch::Tensor& indices,
                               const torch::Tensor& last_prob,
                               const torch::Tensor& nodes, int64_t k);

// --- inducer / subgraph / stitch (hip_inducer.hip) -------------------------
class HIPInducer;
std::shared_ptr<HIPInducer> hip_inducer_create(int64_t reserve);
torch::Tensor hip_inducer_init_node(HIPInducer* ind, const torch::Tensor& s);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor>
hip_inducer_induce_next(HIPInducer* ind, const torch::Tensor& srcs,
                        const torch::Tensor& nbrs,
                        const torch::Tensor& nbrs_num);
torch::Tensor hip_inducer_lookup(HIPInducer* ind, const torch::Tensor& ids);
torch::Tensor hip_inducer_insert(HIPInducer* ind, const torch::Tensor& ids);
// Staged inserts (one totals sync per hop; see HIPInducer comments).
void hip_inducer_reserve(HIPInducer* ind, int64_t total);
std::tuple<torch::Tensor, torch::Tensor> hip_inducer_insert_begin(
    HIPInducer* ind, const torch::Tensor& ids, int64_t idx_base);
torch::Tensor hip_inducer_insert_commit(HIPInducer* ind,
                                        const torch::Tensor& ids,
                                        const torch::Tensor& flags,
                                        const torch::Tensor& ranks,
                                        int64_t n_new);
int64_t hip_inducer_count(HIPInducer* ind);

// --- deferred-sync multi-hop sampler (hip_deferred.hip) ---------------------
// Runs an entire L-hop batch (sample + dedup + relabel) with zero host
// round-trips; all counts stay on device until the caller's single read of
// the returned stats tensor [n_seed, n_new_1..L, total_e_1..L].
class DeferredSampler;
std::shared_ptr<DeferredSampler> deferred_sampler_create(
    std::vector<int64_t> fanout, int64_t batch_cap, int64_t device,
    bool with_eid);
std::tuple<std::vector<torch::Tensor>, std::vector<torch::Tensor>,
           std::vector<torch::Tensor>, std::vector<torch::Tensor>,
           torch::Tensor>
deferred_sampler_run(DeferredSampler* s, const torch::Tensor& indptr,
                     const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const torch::Tensor& seeds);

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           c10::optional<torch::Tensor>>
hip_node_subgraph(const torch::Tensor& indptr, const torch::Tensor& indices,
                  const c10::optional<torch::Tensor>& edge_ids,
                  const torch::Tensor& nodes, bool with_edge);

std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
hip_stitch_sample_results(int64_t ids_count,
                          const std::vector<torch::Tensor>& idx_list,
                          const std::vector<torch::Tensor>& nbrs_list,
                          const std::vector<torch::Tensor>& nbrs_num_list,
                          const std::vector<torch::Tensor>& eids_list);

// --- fused segment aggregation (hip_segment.hip) ----------------------------
torch::Tensor hip_segment_mean_fwd(const torch::Tensor& x,
                                   const torch::Tensor& col,
                                   const torch::Tensor& offsets,
                                   int64_t n_tgt);
torch::Tensor hip_segment_mean_bwd(const torch::Tensor& dy,
                                   const torch::Tensor& col,
                                   const torch::Tensor& offsets,
                                   int64_t n_src);
// Fused [mean-agg | x-prefix] assembly (skips the dim-1 torch.cat).
torch::Tensor hip_segment_mean_cat_fwd(const torch::Tensor& x,
                                       const torch::Tensor& col,
                                       const torch::Tensor& offsets,
                                       int64_t n_tgt);
torch::Tensor hip_segment_mean_cat_bwd(const torch::Tensor& dy,
                                       const torch::Tensor& col,
                                       const torch::Tensor& offsets,
                                       int64_t n_src);

// --- fused GAT edge softmax + aggregation (hip_gat.hip) ---------------------
// Attention logits are computed inside the kernels from (h, att) directly.
// fwd returns (out, m, Z, s_pre); bwd consumes the cached s_pre logits.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_fused_fwd(
    const torch::Tensor& h_tgt, const torch::Tensor& h_src,
    const torch::Tensor& att_src, const torch::Tensor& att_dst,
    const torch::Tensor& src, const torch::Tensor& offsets, double slope);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_fused_bwd(const torch::Tensor& h_tgt, const torch::Tensor& h_src,
                  const torch::Tensor& att_src,
                  const torch::Tensor& att_dst, const torch::Tensor& src,
                  const torch::Tensor& offsets, const torch::Tensor& out,
                  const torch::Tensor& m, const torch::Tensor& z,
                  const torch::Tensor& spre, const torch::Tensor& dout,
                  double slope);

// Multi-relation fused GAT: one launch per hetero layer; h views may be
// row-strided slices of the per-type batched projection.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_multi_fwd(const std::vector<torch::Tensor>& h_tgt,
                  const std::vector<torch::Tensor>& h_src,
                  const std::vector<torch::Tensor>& att_src,
                  const std::vector<torch::Tensor>& att_dst,
                  const std::vector<torch::Tensor>& src,
                  const std::vector<torch::Tensor>& off, double slope,
                  const std::vector<torch::Tensor>& bias);
void hip_gat_multi_bwd(const std::vector<torch::Tensor>& h_tgt,
                       const std::vector<torch::Tensor>& h_src,
                       const std::vector<torch::Tensor>& att_src,
                       const std::vector<torch::Tensor>& att_dst,
                       const std::vector<torch::Tensor>& src,
                       const std::vector<torch::Tensor>& off,
                       const torch::Tensor& out, const torch::Tensor& m,
                       const torch::Tensor& z, const torch::Tensor& spre,
                       const torch::Tensor& dout,
                       const std::vector<torch::Tensor>& dh_tgt,
                       const std::vector<torch::Tensor>& dh_src,
                       const std::vector<torch::Tensor>& datt_src,
                       const std::vector<torch::Tensor>& datt_dst,
                       double slope,
                       const std::vector<torch::Tensor>& dbias);

// --- f32 MFMA projection GEMM (hip_gemm_f32.hip) ----------------------------
torch::Tensor hip_sage_gemm(const torch::Tensor& A, const torch::Tensor& B,
                            const c10::optional<torch::Tensor>& bias,
                            bool relu);

// --- bf16 MFMA projection GEMMs (hip_gemm_bf16.hip) -------------------------
// C = act(A[M,K] @ Bt[N,K]^T + bias): Bt in nn.Linear weight layout.
torch::Tensor hip_gemm_bt_bf16(const torch::Tensor& A,
                               const torch::Tensor& Bt,
                               const c10::optional<torch::Tensor>& bias,
                               bool relu, bool out_fp32);
// (dW, db) = (A[Kb,M]^T @ B[Kb,N], colsum A), fp32 outputs (split-K).
std::tuple<torch::Tensor, c10::optional<torch::Tensor>> hip_gemm_kt_bf16(
    const torch::Tensor& A, const torch::Tensor& B, bool with_db);
torch::Tensor hip_mfma_bf16_selftest(const torch::Tensor& A,
                                     const torch::Tensor& B);

// --- memory plumbing (hip_mem.hip) -----------------------------------------
// Device-dtype alias of (pinned/registered) host memory; keeps `src` alive.
torch::Tensor host_mapped_view(const torch::Tensor& src, int64_t device_index);
// hipHostRegister arbitrary host range (e.g. the shm sample ring).
void pin_host_memory(int64_t addr, int64_t bytes);
void unpin_host_memory(int64_t addr);
void enable_peer_access(int64_t device, int64_t peer);
// IPC sharing of a device tensor: returns handle bytes.
std::string ipc_share(const torch::Tensor& t);
torch::Tensor ipc_open(const std::string& handle, int64_t device,
                       const std::vector<int64_t>& shape,
                       torch::ScalarType dtype);

// --- unified feature store (hip_unified_tensor.hip) ------------------------
class UnifiedFeatureStore;
std::shared_ptr<UnifiedFeatureStore> ufs_create(int64_t device_index);
// Append a row segment; tensor must be HIP-device or a host_mapped_view.
void ufs_append(UnifiedFeatureStore* s, const torch::Tensor& seg);
torch::Tensor ufs_gather(UnifiedFeatureStore* s, const torch::Tensor& rows);
int64_t ufs_rows(UnifiedFeatureStore* s);
int64_t ufs_dim(UnifiedFeatureStore* s);

}  // namespace glt
