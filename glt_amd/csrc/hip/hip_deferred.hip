/* Deferred-sync multi-hop sampling (gfx950).
 *
 * The classic pipeline syncs the host once per hop for the sampled-edge
 * total and once per hop for the new-node count (reference does the same,
 * random_sampler.cu:296-300).  Here every intermediate count lives in
 * device memory and every kernel guards its work on those device scalars,
 * so an L-hop batch runs with ZERO host round trips; the python layer
 * reads one packed [2L] counts tensor at the end (single sync per batch).
 *
 * Buffers are capacity-sized (no-dedup worst case, a pure function of
 * batch x fan-out); tails beyond the device counts are never read.
 */
#include "hip_common.h"
#include "../include/common.h"
#include "../include/hip_ops.h"

namespace glt {

namespace {

constexpr uint64_t kEmpty = ~0ull;

struct DTableView {
  uint64_t* keys;
  unsigned long long* first_idx;
  int64_t* local_id;
  uint64_t mask;
};

__device__ __forceinline__ uint64_t d_probe(uint64_t key, uint64_t mask) {
  return d_splitmix64(key) & mask;
}

__global__ void d_fill_counts_kernel(const int64_t* __restrict__ indptr,
                                     int64_t num_rows,
                                     const int64_t* __restrict__ seeds,
                                     const int64_t* __restrict__ n_dev,
                                     int64_t cap, int64_t k,
                                     int64_t* __restrict__ counts) {
  const int64_t n = *n_dev;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t c = 0;
    if (i < n) {
      const int64_t v = seeds[i];
      const int64_t deg =
          (v >= 0 && v < num_rows) ? indptr[v + 1] - indptr[v] : 0;
      c = deg < k ? deg : k;
    }
    counts[i] = c;
  }
}

template <bool WITH_EID>
__global__ void d_sample_gather_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const int64_t* __restrict__ eids, int64_t num_rows,
    const int64_t* __restrict__ seeds, const int64_t* __restrict__ n_dev,
    int64_t cap, int64_t k, const int64_t* __restrict__ offsets,
    uint64_t call_seed, int64_t* __restrict__ out_nbrs,
    int64_t* __restrict__ out_eids) {
  const int64_t n = *n_dev;
  const int64_t total = offsets[n];
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       e < total; e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = row_of(offsets, n, e);
    const int64_t j = e - offsets[r];
    const int64_t v = seeds[r];
    const int64_t base = indptr[v];
    const int64_t deg = indptr[v + 1] - base;
    int64_t pos;
    if (deg <= k) {
      pos = j;
    } else {
      pos = (int64_t)feistel_perm(d_splitmix64(call_seed + (uint64_t)r),
                                  (uint64_t)j, (uint64_t)deg);
    }
    out_nbrs[e] = indices[base + pos];
    if (WITH_EID) out_eids[e] = eids[base + pos];
  }
}

__global__ void d_insert_kernel(DTableView t,
                                const int64_t* __restrict__ ids,
                                const int64_t* __restrict__ n_dev,
                                int64_t cap) {
  const int64_t n = *n_dev;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (i >= n) continue;
    uint64_t key = (uint64_t)ids[i];
    uint64_t slot = d_probe(key, t.mask);
    for (;;) {
      uint64_t prev = atomicCAS((unsigned long long*)&t.keys[slot],
                                (unsigned long long)kEmpty,
                                (unsigned long long)key);
      if (prev == kEmpty || prev == key) {
        atomicMin(&t.first_idx[slot], (unsigned long long)i);
        break;
      }
      slot = (slot + 1) & t.mask;
    }
  }
}

__global__ void d_flag_kernel(DTableView t, const int64_t* __restrict__ ids,
                              const int64_t* __restrict__ n_dev,
                              int64_t cap, int64_t* __restrict__ flags) {
  const int64_t n = *n_dev;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t f = 0;
    if (i < n) {
      const uint64_t key = (uint64_t)ids[i];
      uint64_t slot = d_probe(key, t.mask);
      while (t.keys[slot] != key) slot = (slot + 1) & t.mask;
      f = (t.local_id[slot] < 0 && t.first_idx[slot] == (uint64_t)i) ? 1
                                                                     : 0;
    }
    flags[i] = f;
  }
}

__global__ void d_assign_kernel(DTableView t,
                                const int64_t* __restrict__ ids,
                                const int64_t* __restrict__ n_dev,
                                int64_t cap,
                                const int64_t* __restrict__ ranks,
                                const int64_t* __restrict__ flags,
                                const int64_t* __restrict__ base_dev,
                                int64_t* __restrict__ unique_out,
                                int64_t* __restrict__ n_new_out) {
  const int64_t n = *n_dev;
  const int64_t base = *base_dev;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (i < n && flags[i]) {
      const uint64_t key = (uint64_t)ids[i];
      uint64_t slot = d_probe(key, t.mask);
      while (t.keys[slot] != key) slot = (slot + 1) & t.mask;
      t.local_id[slot] = base + ranks[i] - 1;
      unique_out[ranks[i] - 1] = ids[i];
    }
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    *n_new_out = n > 0 ? ranks[n - 1] : 0;
  }
}

__global__ void d_bump_count_kernel(int64_t* __restrict__ count_dev,
                                    const int64_t* __restrict__ n_new) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *count_dev += *n_new;
}

__global__ void d_relabel_kernel(DTableView t,
                                 const int64_t* __restrict__ ids,
                                 const int64_t* __restrict__ total_dev,
                                 int64_t cap, int64_t* __restrict__ out) {
  const int64_t n = *total_dev;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (i >= n) continue;
    const uint64_t key = (uint64_t)ids[i];
    uint64_t slot = d_probe(key, t.mask);
    for (;;) {
      const uint64_t kk = t.keys[slot];
      if (kk == key) {
        out[i] = t.local_id[slot];
        break;
      }
      if (kk == kEmpty) {
        out[i] = -1;
        break;
      }
      slot = (slot + 1) & t.mask;
    }
  }
}

__global__ void d_expand_rows_kernel(DTableView t,
                                     const int64_t* __restrict__ srcs,
                                     const int64_t* __restrict__ ns_dev,
                                     int64_t cap_rows,
                                     const int64_t* __restrict__ offsets,
                                     int64_t* __restrict__ rows) {
  const int64_t ns = *ns_dev;
  const int64_t total = offsets[ns];
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       e < total; e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = row_of(offsets, ns, e);
    const uint64_t key = (uint64_t)srcs[r];
    uint64_t slot = d_probe(key, t.mask);
    while (t.keys[slot] != key) slot = (slot + 1) & t.mask;
    rows[e] = t.local_id[slot];
  }
}

int64_t d_next_pow2(int64_t x) {
  int64_t p = 1;
  while (p < x) p <<= 1;
  return p;
}

}  // namespace

// ---------------------------------------------------------------------------
// DeferredSampler: holds all capacity buffers + the hash table and runs an
// L-hop batch without host syncs.  One instance per (device, fanout) — the
// python layer pools them like inducers.
// ---------------------------------------------------------------------------
class DeferredSampler {
 public:
  DeferredSampler(std::vector<int64_t> fanout, int64_t batch_cap,
                  torch::Device device, bool with_eid)
      : fanout_(std::move(fanout)), with_eid_(with_eid), device_(device) {
    TORCH_CHECK(!fanout_.empty(), "deferred sampler needs >=1 hop");
    for (auto k : fanout_)
      TORCH_CHECK(k > 0, "deferred sampler needs positive fan-outs");
    auto opts = torch::TensorOptions().dtype(torch::kInt64).device(device);
    int64_t cap = batch_cap;
    node_cap_ = {cap};
    for (auto k : fanout_) {
      cap *= k;
      node_cap_.push_back(cap);
    }
    total_node_cap_ = 0;
    for (auto c : node_cap_) total_node_cap_ += c;
    // hash table sized for every node of the batch
    table_cap_ = d_next_pow2(total_node_cap_ * 2);
    keys_ = torch::empty({table_cap_}, opts);
    first_idx_ = torch::empty({table_cap_}, opts);
    local_id_ = torch::empty({table_cap_}, opts);
    for (size_t h = 0; h < fanout_.size(); ++h) {
      const int64_t rc = node_cap_[h];
      const int64_t ec = node_cap_[h + 1];
      counts_.push_back(torch::empty({rc}, opts));
      offsets_.push_back(torch::zeros({rc + 1}, opts));
      nbrs_.push_back(torch::empty({ec}, opts));
      eids_.push_back(with_eid ? torch::empty({ec}, opts)
                               : torch::Tensor());
      rows_.push_back(torch::empty({ec}, opts));
      cols_.push_back(torch::empty({ec}, opts));
      uniq_.push_back(torch::empty({ec}, opts));
      flags_.push_back(torch::empty({ec}, opts));
    }
    count_dev_ = torch::zeros({1}, opts);
    seed_uniq_ = torch::empty({batch_cap}, opts);
    seed_flags_ = torch::empty({batch_cap}, opts);
    seed_n_ = torch::zeros({1}, opts);
    stats_ = torch::zeros({(int64_t)(2 * fanout_.size() + 1)}, opts);
  }

  // Runs the whole multi-hop batch; returns
  // (node parts, rows list, cols list, eids list, stats) where stats
  // = [n_seed_uniq, n_new_1.., total_e_1..] lives on DEVICE until the
  // caller's single .cpu() read.
  std::tuple<std::vector<torch::Tensor>, std::vector<torch::Tensor>,
             std::vector<torch::Tensor>, std::vector<torch::Tensor>,
             torch::Tensor>
  run(const torch::Tensor& indptr, const torch::Tensor& indices,
      const c10::optional<torch::Tensor>& edge_ids,
      const torch::Tensor& seeds_in) {
    auto stream = current_stream();
    TORCH_CHECK(seeds_in.scalar_type() == torch::kInt64,
                "deferred: seeds must be int64");
    TORCH_CHECK(seeds_in.device() == device_,
                "deferred: seeds on wrong device");
    TORCH_CHECK(!with_eid_ || edge_ids.has_value(),
                "deferred: with_eid sampler needs edge_ids");
    auto seeds = seeds_in.contiguous();
    const int64_t num_rows = indptr.size(0) - 1;
    const int64_t bs = seeds.size(0);
    TORCH_CHECK(bs <= node_cap_[0], "batch exceeds deferred capacity");
    // reset table + running count
    keys_.fill_(-1);
    first_idx_.fill_(0x7FFFFFFFFFFFFFFFll);
    local_id_.fill_(-1);
    count_dev_.zero_();
    seed_n_.fill_(bs);

    DTableView t = view();
    // dedup seeds
    hipLaunchKernelGGL(d_insert_kernel, dim3(grid_for(bs)), dim3(kBlock), 0,
                       stream, t, seeds.data_ptr<int64_t>(),
                       seed_n_.data_ptr<int64_t>(), bs);
    hipLaunchKernelGGL(d_flag_kernel, dim3(grid_for(bs)), dim3(kBlock), 0,
                       stream, t, seeds.data_ptr<int64_t>(),
                       seed_n_.data_ptr<int64_t>(), bs,
                       seed_flags_.data_ptr<int64_t>());
    auto seed_ranks = torch::cumsum(seed_flags_.narrow(0, 0, bs), 0);
    hipLaunchKernelGGL(d_assign_kernel, dim3(grid_for(bs)), dim3(kBlock), 0,
                       stream, t, seeds.data_ptr<int64_t>(),
                       seed_n_.data_ptr<int64_t>(), bs,
                       seed_ranks.data_ptr<int64_t>(),
                       seed_flags_.data_ptr<int64_t>(),
                       count_dev_.data_ptr<int64_t>(),
                       seed_uniq_.data_ptr<int64_t>(),
                       stats_.data_ptr<int64_t>());  // stats[0] = n_seed
    hipLaunchKernelGGL(d_bump_count_kernel, dim3(1), dim3(64), 0, stream,
                       count_dev_.data_ptr<int64_t>(),
                       stats_.data_ptr<int64_t>());

    const int64_t L = (int64_t)fanout_.size();
    const int64_t* eid_ptr =
        edge_ids.has_value() ? edge_ids->data_ptr<int64_t>() : nullptr;
    torch::Tensor frontier = seed_uniq_;
    torch::Tensor frontier_n = stats_.narrow(0, 0, 1);
    for (int64_t h = 0; h < L; ++h) {
      const int64_t rc = node_cap_[h];
      const int64_t ec = node_cap_[h + 1];
      const int64_t k = fanout_[h];
      const uint64_t cs = SeedManager::instance().next_call_seed();
      hipLaunchKernelGGL(d_fill_counts_kernel, dim3(grid_for(rc)),
                         dim3(kBlock), 0, stream,
                         indptr.data_ptr<int64_t>(), num_rows,
                         frontier.data_ptr<int64_t>(),
                         frontier_n.data_ptr<int64_t>(), rc, k,
                         counts_[h].data_ptr<int64_t>());
      {
        auto v = offsets_[h].narrow(0, 1, rc);
        torch::cumsum_out(v, counts_[h], 0);
      }
      if (with_eid_ && eid_ptr) {
        hipLaunchKernelGGL((d_sample_gather_kernel<true>),
                           dim3(grid_for(ec)), dim3(kBlock), 0, stream,
                           indptr.data_ptr<int64_t>(),
                           indices.data_ptr<int64_t>(), eid_ptr, num_rows,
                           frontier.data_ptr<int64_t>(),
                           frontier_n.data_ptr<int64_t>(), rc, k,
                           offsets_[h].data_ptr<int64_t>(), cs,
                           nbrs_[h].data_ptr<int64_t>(),
                           eids_[h].data_ptr<int64_t>());
      } else {
        hipLaunchKernelGGL((d_sample_gather_kernel<false>),
                           dim3(grid_for(ec)), dim3(kBlock), 0, stream,
                           indptr.data_ptr<int64_t>(),
                           indices.data_ptr<int64_t>(), eid_ptr, num_rows,
                           frontier.data_ptr<int64_t>(),
                           frontier_n.data_ptr<int64_t>(), rc, k,
                           offsets_[h].data_ptr<int64_t>(), cs,
                           nbrs_[h].data_ptr<int64_t>(), nullptr);
      }
      // total edges of this hop -> stats[L + 1 + h] (device-device, async)
      stats_.narrow(0, 1 + L + h, 1).copy_(offsets_[h].narrow(0, rc, 1));
      // rows of hop h (expand frontier local ids over segments)
      hipLaunchKernelGGL(d_expand_rows_kernel, dim3(grid_for(ec)),
                         dim3(kBlock), 0, stream, t,
                         frontier.data_ptr<int64_t>(),
                         frontier_n.data_ptr<int64_t>(), rc,
                         offsets_[h].data_ptr<int64_t>(),
                         rows_[h].data_ptr<int64_t>());
      // insert neighbors, assign fresh local ids
      auto total_dev = offsets_[h].narrow(0, rc, 1);
      hipLaunchKernelGGL(d_insert_kernel, dim3(grid_for(ec)), dim3(kBlock),
                         0, stream, t, nbrs_[h].data_ptr<int64_t>(),
                         total_dev.data_ptr<int64_t>(), ec);
      hipLaunchKernelGGL(d_flag_kernel, dim3(grid_for(ec)), dim3(kBlock), 0,
                         stream, t, nbrs_[h].data_ptr<int64_t>(),
                         total_dev.data_ptr<int64_t>(), ec,
                         flags_[h].data_ptr<int64_t>());
      auto ranks = torch::cumsum(flags_[h], 0);
      hipLaunchKernelGGL(d_assign_kernel, dim3(grid_for(ec)), dim3(kBlock),
                         0, stream, t, nbrs_[h].data_ptr<int64_t>(),
                         total_dev.data_ptr<int64_t>(), ec,
                         ranks.data_ptr<int64_t>(),
                         flags_[h].data_ptr<int64_t>(),
                         count_dev_.data_ptr<int64_t>(),
                         uniq_[h].data_ptr<int64_t>(),
                         stats_.data_ptr<int64_t>() + 1 + h);
      hipLaunchKernelGGL(d_bump_count_kernel, dim3(1), dim3(64), 0, stream,
                         count_dev_.data_ptr<int64_t>(),
                         stats_.data_ptr<int64_t>() + 1 + h);
      // relabel cols
      hipLaunchKernelGGL(d_relabel_kernel, dim3(grid_for(ec)), dim3(kBlock),
                         0, stream, t, nbrs_[h].data_ptr<int64_t>(),
                         total_dev.data_ptr<int64_t>(), ec,
                         cols_[h].data_ptr<int64_t>());
      frontier = uniq_[h];
      frontier_n = stats_.narrow(0, 1 + h, 1);
    }
    std::vector<torch::Tensor> eids_out;
    if (with_eid_)
      for (auto& e : eids_) eids_out.push_back(e);
    // capacity-sized node parts: python slices each with stats after its
    // single end-of-batch sync
    std::vector<torch::Tensor> node_parts = {seed_uniq_};
    for (auto& u : uniq_) node_parts.push_back(u);
    return {node_parts, rows_, cols_, eids_out, stats_};
  }

 private:
  DTableView view() {
    DTableView t;
    t.keys = reinterpret_cast<uint64_t*>(keys_.data_ptr());
    t.first_idx =
        reinterpret_cast<unsigned long long*>(first_idx_.data_ptr());
    t.local_id = local_id_.data_ptr<int64_t>();
    t.mask = (uint64_t)table_cap_ - 1;
    return t;
  }

  std::vector<int64_t> fanout_, node_cap_;
  bool with_eid_;
  torch::Device device_;
  int64_t table_cap_ = 0, total_node_cap_ = 0;
  torch::Tensor keys_, first_idx_, local_id_, count_dev_;
  torch::Tensor seed_uniq_, seed_flags_, seed_n_, stats_;
  std::vector<torch::Tensor> counts_, offsets_, nbrs_, eids_, rows_, cols_,
      uniq_, flags_;
};

std::shared_ptr<DeferredSampler> deferred_sampler_create(
    std::vector<int64_t> fanout, int64_t batch_cap, int64_t device,
    bool with_eid) {
  return std::make_shared<DeferredSampler>(
      std::move(fanout), batch_cap, torch::Device(torch::kCUDA, device),
      with_eid);
}

std::tuple<std::vector<torch::Tensor>, std::vector<torch::Tensor>,
           std::vector<torch::Tensor>, std::vector<torch::Tensor>,
           torch::Tensor>
deferred_sampler_run(DeferredSampler* s, const torch::Tensor& indptr,
                     const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const torch::Tensor& seeds) {
  return s->run(indptr, indices, edge_ids, seeds);
}

}  // namespace glt
