/* GPU hash table, subgraph inducer, node-subgraph op and stitch (gfx950).
 *
 * Hash table: open-addressing linear probe on 64-bit keys with atomicCAS
 * insert and an atomicMin first-occurrence index per slot, which makes the
 * compacted local-id assignment deterministic in input order — functional
 * parity with reference hash_table.cuh:35-100 / inducer.cu, fresh
 * implementation re-tiled for wave64 grid-stride loops.
 *
 * Inducer: keeps the table + running unique-node count across hops; each
 * hop inserts the sampled neighbors, assigns local ids (input-order stable
 * via flag + prefix-sum), and relabels rows/cols.  Row expansion uses
 * lane-per-edge binary search over the per-seed offsets (no 32-wide
 * "warp-per-row" loops as in reference inducer.cu:45-64).
 */
#include "hip_common.h"
#include "../include/common.h"
#include "../include/hip_ops.h"

namespace glt {

namespace {

constexpr uint64_t kEmpty = ~0ull;
constexpr uint64_t kIdxInit = 0x7FFFFFFFFFFFFFFFull;

struct TableView {
  uint64_t* keys;
  unsigned long long* first_idx;  // min input position (stable ordering)
  int64_t* local_id;
  uint64_t mask;  // capacity - 1
};

__device__ __forceinline__ uint64_t probe_start(uint64_t key, uint64_t mask) {
  return d_splitmix64(key) & mask;
}

// Insert key with candidate input position; returns slot.
__device__ __forceinline__ uint64_t table_insert(TableView t, uint64_t key,
                                                 uint64_t pos) {
  uint64_t slot = probe_start(key, t.mask);
  for (;;) {
    uint64_t prev = atomicCAS((unsigned long long*)&t.keys[slot],
                              (unsigned long long)kEmpty,
                              (unsigned long long)key);
    if (prev == kEmpty || prev == key) {
      atomicMin(&t.first_idx[slot], (unsigned long long)pos);
      return slot;
    }
    slot = (slot + 1) & t.mask;
  }
}

// Find slot of key; table must contain it.
__device__ __forceinline__ uint64_t table_find(TableView t, uint64_t key) {
  uint64_t slot = probe_start(key, t.mask);
  while (t.keys[slot] != key) slot = (slot + 1) & t.mask;
  return slot;
}

// Find local id, or -1 if absent.
__device__ __forceinline__ int64_t table_lookup(TableView t, uint64_t key) {
  uint64_t slot = probe_start(key, t.mask);
  for (;;) {
    const uint64_t k = t.keys[slot];
    if (k == key) return t.local_id[slot];
    if (k == kEmpty) return -1;
    slot = (slot + 1) & t.mask;
  }
}

// idx_base: staged multi-call insertion (one hop's edge types insert
// back-to-back BEFORE any assign) biases the first-occurrence order key
// so later calls never steal first place from earlier ones — keeps the
// fresh-node order identical to the classic one-sync-per-call flow.
__global__ void insert_kernel(TableView t, const int64_t* __restrict__ ids,
                              int64_t n, int64_t idx_base) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    table_insert(t, (uint64_t)ids[i], (uint64_t)(idx_base + i));
  }
}

// flag[i] = 1 iff position i is the first occurrence of ids[i].
__global__ void flag_first_kernel(TableView t, const int64_t* __restrict__ ids,
                                  int64_t n, int64_t idx_base,
                                  int64_t* __restrict__ flags) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint64_t slot = table_find(t, (uint64_t)ids[i]);
    // Only fresh nodes count: a slot owned by a previous hop has local_id >= 0.
    flags[i] = (t.local_id[slot] < 0 &&
                t.first_idx[slot] == (uint64_t)(idx_base + i))
                   ? 1
                   : 0;
  }
}

// Assign compacted local ids (base + rank) and emit the unique-node list.
__global__ void assign_kernel(TableView t, const int64_t* __restrict__ ids,
                              int64_t n, const int64_t* __restrict__ ranks,
                              const int64_t* __restrict__ flags, int64_t base,
                              int64_t* __restrict__ unique_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (flags[i]) {
      const uint64_t slot = table_find(t, (uint64_t)ids[i]);
      const int64_t local = base + ranks[i] - 1;  // inclusive-scan rank
      t.local_id[slot] = local;
      unique_out[ranks[i] - 1] = ids[i];
    }
  }
}

__global__ void relabel_kernel(TableView t, const int64_t* __restrict__ ids,
                               int64_t n, int64_t* __restrict__ out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = table_lookup(t, (uint64_t)ids[i]);
  }
}

// rows[e] = local id of the seed owning packed output position e.
__global__ void expand_rows_kernel(TableView t,
                                   const int64_t* __restrict__ srcs,
                                   int64_t ns,
                                   const int64_t* __restrict__ offsets,
                                   int64_t total, int64_t* __restrict__ rows) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; e < total;
       e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = row_of(offsets, ns, e);
    rows[e] = table_lookup(t, (uint64_t)srcs[r]);
  }
}

int64_t next_pow2(int64_t x) {
  int64_t p = 1;
  while (p < x) p <<= 1;
  return p;
}

}  // namespace

// ---------------------------------------------------------------------------
// HIPInducer
// ---------------------------------------------------------------------------
class HIPInducer {
 public:
  explicit HIPInducer(int64_t reserve = 4096) : reserve_(reserve) {}

  torch::Tensor init_node(const torch::Tensor& seeds) {
    device_ = seeds.device();
    count_ = 0;
    pending_ = 0;  // clears a staged hop aborted mid-flight (pooled
                   // inducers would otherwise refuse reserve_incoming)
    node_chunks_.clear();
    ensure_capacity(std::max<int64_t>(seeds.size(0), reserve_), true);
    return insert_and_assign(seeds);
  }

  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> induce_next(
      const torch::Tensor& srcs, const torch::Tensor& nbrs,
      const torch::Tensor& nbrs_num) {
    auto stream = current_stream();
    const int64_t ns = srcs.size(0);
    const int64_t ne = nbrs.size(0);
    ensure_capacity(count_ + ne, false);
    auto nodes = insert_and_assign(nbrs);
    // offsets for row expansion
    auto offsets = torch::zeros({ns + 1}, srcs.options());
    { auto v = offsets.narrow(0, 1, ns); torch::cumsum_out(v, nbrs_num, 0); }
    auto rows = torch::empty({ne}, srcs.options());
    auto cols = torch::empty({ne}, srcs.options());
    if (ne > 0) {
      hipLaunchKernelGGL(expand_rows_kernel, dim3(grid_for(ne)), dim3(kBlock),
                         0, stream, view(), srcs.data_ptr<int64_t>(), ns,
                         offsets.data_ptr<int64_t>(), ne,
                         rows.data_ptr<int64_t>());
      hipLaunchKernelGGL(relabel_kernel, dim3(grid_for(ne)), dim3(kBlock), 0,
                         stream, view(), nbrs.data_ptr<int64_t>(), ne,
                         cols.data_ptr<int64_t>());
    }
    return {nodes, rows, cols};
  }

  // Insert ids (dedup against all previous hops); returns the new nodes.
  torch::Tensor insert(const torch::Tensor& ids) {
    ensure_capacity(count_ + ids.size(0), false);
    return insert_and_assign(ids);
  }

  // Staged insert: begin() launches insert+flag+scan with NO host sync
  // (idx_base keeps first-occurrence order across back-to-back begins);
  // the caller batches the n_new reads, then commit() assigns ids.
  // Must be called once per staged hop with the hop's total incoming
  // id count BEFORE any insert_begin: a table growth between begins
  // would rehash away the pending (not-yet-assigned) keys.
  void reserve_incoming(int64_t total) {
    TORCH_CHECK(pending_ == 0, "reserve_incoming during a staged hop");
    ensure_capacity(count_ + total, false);
  }

  std::tuple<torch::Tensor, torch::Tensor> insert_begin(
      const torch::Tensor& ids, int64_t idx_base) {
    const int64_t n = ids.size(0);
    TORCH_CHECK((count_ + pending_ + n) * 2 <= capacity_,
                "insert_begin: call reserve_incoming for the hop first");
    auto flags = torch::empty({n}, ids.options());
    if (n > 0) {
      auto stream = current_stream();
      hipLaunchKernelGGL(insert_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                         stream, view(), ids.data_ptr<int64_t>(), n,
                         idx_base);
      hipLaunchKernelGGL(flag_first_kernel, dim3(grid_for(n)),
                         dim3(kBlock), 0, stream, view(),
                         ids.data_ptr<int64_t>(), n, idx_base,
                         flags.data_ptr<int64_t>());
    }
    auto ranks = torch::cumsum(flags, 0);
    pending_ += n;  // growth headroom until the commits land
    return {flags, ranks};
  }

  torch::Tensor insert_commit(const torch::Tensor& ids,
                              const torch::Tensor& flags,
                              const torch::Tensor& ranks, int64_t n_new) {
    const int64_t n = ids.size(0);
    auto uniq = torch::empty({n_new}, torch::TensorOptions()
                                          .dtype(torch::kInt64)
                                          .device(device_));
    if (n > 0 && n_new > 0) {
      hipLaunchKernelGGL(assign_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                         current_stream(), view(), ids.data_ptr<int64_t>(),
                         n, ranks.data_ptr<int64_t>(),
                         flags.data_ptr<int64_t>(), count_,
                         uniq.data_ptr<int64_t>());
    }
    count_ += n_new;
    pending_ = std::max<int64_t>(0, pending_ - n);
    if (n_new > 0) node_chunks_.push_back(uniq);
    return uniq;
  }

  // Relabel arbitrary global ids through the current table (-1 if absent).
  torch::Tensor lookup(const torch::Tensor& ids) {
    auto out = torch::empty_like(ids);
    const int64_t n = ids.size(0);
    if (n > 0) {
      hipLaunchKernelGGL(relabel_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                         current_stream(), view(), ids.data_ptr<int64_t>(), n,
                         out.data_ptr<int64_t>());
    }
    return out;
  }

  int64_t count() const { return count_; }

  TableView view() {
    TableView t;
    t.keys = reinterpret_cast<uint64_t*>(keys_.data_ptr());
    t.first_idx = reinterpret_cast<unsigned long long*>(first_idx_.data_ptr());
    t.local_id = local_id_.data_ptr<int64_t>();
    t.mask = (uint64_t)capacity_ - 1;
    return t;
  }

 private:

  void reset_table() {
    keys_.fill_(-1);                      // all-ones bytes = kEmpty
    first_idx_.fill_(kIdxInit);
    local_id_.fill_(-1);
  }

  void ensure_capacity(int64_t needed, bool force_reset) {
    const int64_t want = next_pow2(std::max<int64_t>(needed * 2, 64));
    if (want > capacity_ || !keys_.defined()) {
      capacity_ = want;
      auto opts = torch::TensorOptions().dtype(torch::kInt64).device(device_);
      keys_ = torch::empty({capacity_}, opts);
      first_idx_ = torch::empty({capacity_}, opts);
      local_id_ = torch::empty({capacity_}, opts);
      reset_table();
      if (!force_reset && count_ > 0) {
        // Re-insert the already-assigned unique nodes with their ids.
        reinsert(all_nodes());
      }
    } else if (force_reset) {
      reset_table();
    }
  }

  void reinsert(const torch::Tensor& nodes) {
    const int64_t n = nodes.size(0);
    if (n == 0) return;
    auto stream = current_stream();
    hipLaunchKernelGGL(insert_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, view(), nodes.data_ptr<int64_t>(), n,
                       (int64_t)0);
    auto flags = torch::empty({n}, nodes.options());
    hipLaunchKernelGGL(flag_first_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, view(), nodes.data_ptr<int64_t>(), n,
                       (int64_t)0, flags.data_ptr<int64_t>());
    auto ranks = torch::cumsum(flags, 0);
    auto uniq = torch::empty({n}, nodes.options());
    hipLaunchKernelGGL(assign_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, view(), nodes.data_ptr<int64_t>(), n,
                       ranks.data_ptr<int64_t>(), flags.data_ptr<int64_t>(),
                       0, uniq.data_ptr<int64_t>());
  }

  torch::Tensor insert_and_assign(const torch::Tensor& ids) {
    const int64_t n = ids.size(0);
    if (n == 0)
      return torch::empty({0}, torch::TensorOptions()
                                   .dtype(torch::kInt64)
                                   .device(device_));
    auto stream = current_stream();
    hipLaunchKernelGGL(insert_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, view(), ids.data_ptr<int64_t>(), n,
                       (int64_t)0);
    auto flags = torch::empty({n}, ids.options());
    hipLaunchKernelGGL(flag_first_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, view(), ids.data_ptr<int64_t>(), n,
                       (int64_t)0, flags.data_ptr<int64_t>());
    auto ranks = torch::cumsum(flags, 0);
    const int64_t n_new = ranks[n - 1].item<int64_t>();  // sync
    auto uniq = torch::empty({n_new}, ids.options());
    if (n_new > 0 || n > 0) {
      hipLaunchKernelGGL(assign_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                         stream, view(), ids.data_ptr<int64_t>(), n,
                         ranks.data_ptr<int64_t>(), flags.data_ptr<int64_t>(),
                         count_, uniq.data_ptr<int64_t>());
    }
    count_ += n_new;
    if (n_new > 0) node_chunks_.push_back(uniq);
    return uniq;
  }

  // Concatenate lazily: only the (rare) table-growth rebuild needs the full
  // unique-node list, so per-hop O(n) device copies are avoided.
  torch::Tensor all_nodes() {
    if (node_chunks_.empty())
      return torch::empty({0},
                          torch::TensorOptions().dtype(torch::kInt64)
                              .device(device_));
    if (node_chunks_.size() > 1) {
      auto merged = torch::cat(node_chunks_);
      node_chunks_.clear();
      node_chunks_.push_back(merged);
    }
    return node_chunks_[0];
  }

  torch::Device device_{torch::kCPU};
  torch::Tensor keys_, first_idx_, local_id_;
  std::vector<torch::Tensor> node_chunks_;
  int64_t capacity_ = 0;
  int64_t count_ = 0;
  int64_t pending_ = 0;  // begun-but-uncommitted inserts (growth guard)
  int64_t reserve_;
};

// Opaque-handle C API (bindings.cpp is host-compiled).
std::shared_ptr<HIPInducer> hip_inducer_create(int64_t reserve) {
  return std::make_shared<HIPInducer>(reserve);
}
torch::Tensor hip_inducer_init_node(HIPInducer* ind, const torch::Tensor& s) {
  return ind->init_node(s);
}
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor>
hip_inducer_induce_next(HIPInducer* ind, const torch::Tensor& srcs,
                        const torch::Tensor& nbrs,
                        const torch::Tensor& nbrs_num) {
  return ind->induce_next(srcs, nbrs, nbrs_num);
}
torch::Tensor hip_inducer_lookup(HIPInducer* ind, const torch::Tensor& ids) {
  return ind->lookup(ids);
}
torch::Tensor hip_inducer_insert(HIPInducer* ind, const torch::Tensor& ids) {
  return ind->insert(ids);
}
int64_t hip_inducer_count(HIPInducer* ind) { return ind->count(); }
void hip_inducer_reserve(HIPInducer* ind, int64_t total) {
  ind->reserve_incoming(total);
}
std::tuple<torch::Tensor, torch::Tensor> hip_inducer_insert_begin(
    HIPInducer* ind, const torch::Tensor& ids, int64_t idx_base) {
  return ind->insert_begin(ids, idx_base);
}
torch::Tensor hip_inducer_insert_commit(HIPInducer* ind,
                                        const torch::Tensor& ids,
                                        const torch::Tensor& flags,
                                        const torch::Tensor& ranks,
                                        int64_t n_new) {
  return ind->insert_commit(ids, flags, ranks, n_new);
}

// ---------------------------------------------------------------------------
// Node subgraph (full induced edge set among a node set).
// ---------------------------------------------------------------------------
namespace {

__global__ void mark_subgraph_edges_kernel(
    TableView t, const int64_t* __restrict__ indptr,
    const int64_t* __restrict__ indices, const int64_t* __restrict__ uniq,
    int64_t n, const int64_t* __restrict__ edge_offsets, int64_t total,
    int64_t* __restrict__ flags, int64_t* __restrict__ cols_tmp) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; e < total;
       e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t u = row_of(edge_offsets, n, e);
    const int64_t v = uniq[u];
    const int64_t pos = indptr[v] + (e - edge_offsets[u]);
    const int64_t local = table_lookup(t, (uint64_t)indices[pos]);
    flags[e] = local >= 0 ? 1 : 0;
    cols_tmp[e] = local;
  }
}

__global__ void compact_subgraph_kernel(
    const int64_t* __restrict__ flags, const int64_t* __restrict__ ranks,
    const int64_t* __restrict__ cols_tmp,
    const int64_t* __restrict__ edge_offsets,
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ uniq,
    const int64_t* __restrict__ eids_in, int64_t n, int64_t total,
    int64_t* __restrict__ rows, int64_t* __restrict__ cols,
    int64_t* __restrict__ eids_out) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; e < total;
       e += (int64_t)gridDim.x * blockDim.x) {
    if (!flags[e]) continue;
    const int64_t out = ranks[e] - 1;
    const int64_t u = row_of(edge_offsets, n, e);
    rows[out] = u;
    cols[out] = cols_tmp[e];
    if (eids_out) {
      const int64_t v = uniq[u];
      eids_out[out] = eids_in[indptr[v] + (e - edge_offsets[u])];
    }
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           c10::optional<torch::Tensor>>
hip_node_subgraph(const torch::Tensor& indptr, const torch::Tensor& indices,
                  const c10::optional<torch::Tensor>& edge_ids,
                  const torch::Tensor& nodes, bool with_edge) {
  TORCH_CHECK(!with_edge || edge_ids.has_value(), "with_edge requires edge_ids");
  HIPInducer inducer;
  auto uniq = inducer.init_node(nodes);
  const int64_t n = uniq.size(0);
  auto stream = current_stream();
  auto degs = hip_lookup_degree(indptr, uniq);  // declared in hip_ops.h
  auto edge_offsets = torch::zeros({n + 1}, nodes.options());
  if (n > 0) { auto v = edge_offsets.narrow(0, 1, n); torch::cumsum_out(v, degs, 0); }
  const int64_t total = n > 0 ? edge_offsets[n].item<int64_t>() : 0;
  auto flags = torch::zeros({std::max<int64_t>(total, 1)}, nodes.options());
  auto cols_tmp = torch::empty({std::max<int64_t>(total, 1)}, nodes.options());
  if (total > 0) {
    hipLaunchKernelGGL(mark_subgraph_edges_kernel, dim3(grid_for(total)),
                       dim3(kBlock), 0, stream, inducer.view(),
                       indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                       uniq.data_ptr<int64_t>(), n,
                       edge_offsets.data_ptr<int64_t>(), total,
                       flags.data_ptr<int64_t>(),
                       cols_tmp.data_ptr<int64_t>());
  }
  auto ranks = torch::cumsum(flags.narrow(0, 0, std::max<int64_t>(total, 1)), 0);
  const int64_t m = total > 0 ? ranks[total - 1].item<int64_t>() : 0;
  auto rows = torch::empty({m}, nodes.options());
  auto cols = torch::empty({m}, nodes.options());
  auto eids =
      with_edge ? torch::empty({m}, nodes.options()) : torch::Tensor();
  if (total > 0 && m > 0) {
    hipLaunchKernelGGL(
        compact_subgraph_kernel, dim3(grid_for(total)), dim3(kBlock), 0,
        stream, flags.data_ptr<int64_t>(), ranks.data_ptr<int64_t>(),
        cols_tmp.data_ptr<int64_t>(), edge_offsets.data_ptr<int64_t>(),
        indptr.data_ptr<int64_t>(), uniq.data_ptr<int64_t>(),
        with_edge ? edge_ids->data_ptr<int64_t>() : nullptr, n, total,
        rows.data_ptr<int64_t>(), cols.data_ptr<int64_t>(),
        with_edge ? eids.data_ptr<int64_t>() : nullptr);
  }
  return {uniq, rows, cols,
          with_edge ? c10::optional<torch::Tensor>(eids) : c10::nullopt};
}

// ---------------------------------------------------------------------------
// Stitch: scatter per-partition one-hop segments into seed order.
// ---------------------------------------------------------------------------
namespace {
__global__ void stitch_kernel(const int64_t* __restrict__ idx,
                              const int64_t* __restrict__ p_offsets,
                              int64_t m, int64_t p_total,
                              const int64_t* __restrict__ p_nbrs,
                              const int64_t* __restrict__ p_eids,
                              const int64_t* __restrict__ out_offsets,
                              int64_t* __restrict__ out_nbrs,
                              int64_t* __restrict__ out_eids) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       e < p_total; e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = row_of(p_offsets, m, e);
    const int64_t j = e - p_offsets[i];
    const int64_t pos = out_offsets[idx[i]] + j;
    out_nbrs[pos] = p_nbrs[e];
    if (out_eids) out_eids[pos] = p_eids[e];
  }
}
}  // namespace

std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
hip_stitch_sample_results(int64_t ids_count,
                          const std::vector<torch::Tensor>& idx_list,
                          const std::vector<torch::Tensor>& nbrs_list,
                          const std::vector<torch::Tensor>& nbrs_num_list,
                          const std::vector<torch::Tensor>& eids_list) {
  const size_t P = idx_list.size();
  TORCH_CHECK(P > 0, "stitch: no partitions");
  const bool with_edge = !eids_list.empty();
  auto opts = idx_list[0].options();
  auto stream = current_stream();

  auto nbrs_num = torch::zeros({ids_count}, opts);
  for (size_t p = 0; p < P; ++p)
    nbrs_num.index_copy_(0, idx_list[p], nbrs_num_list[p]);
  auto out_offsets = torch::zeros({ids_count + 1}, opts);
  { auto v = out_offsets.narrow(0, 1, ids_count); torch::cumsum_out(v, nbrs_num, 0); }
  const int64_t total = out_offsets[ids_count].item<int64_t>();
  auto nbrs = torch::zeros({total}, opts);
  auto eids = with_edge ? torch::zeros({total}, opts) : torch::Tensor();

  for (size_t p = 0; p < P; ++p) {
    const int64_t m = idx_list[p].size(0);
    if (m == 0) continue;
    auto p_offsets = torch::zeros({m + 1}, opts);
    { auto v = p_offsets.narrow(0, 1, m); torch::cumsum_out(v, nbrs_num_list[p], 0); }
    const int64_t p_total = nbrs_list[p].size(0);
    if (p_total == 0) continue;
    hipLaunchKernelGGL(stitch_kernel, dim3(grid_for(p_total)), dim3(kBlock),
                       0, stream, idx_list[p].data_ptr<int64_t>(),
                       p_offsets.data_ptr<int64_t>(), m, p_total,
                       nbrs_list[p].data_ptr<int64_t>(),
                       with_edge ? eids_list[p].data_ptr<int64_t>() : nullptr,
                       out_offsets.data_ptr<int64_t>(),
                       nbrs.data_ptr<int64_t>(),
                       with_edge ? eids.data_ptr<int64_t>() : nullptr);
  }
  return {nbrs, nbrs_num,
          with_edge ? c10::optional<torch::Tensor>(eids) : c10::nullopt};
}

}  // namespace glt
