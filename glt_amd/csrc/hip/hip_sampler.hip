/* HIP neighbor / weighted / negative samplers + random walk (gfx950).
 *
 * Design (MI355X-native, not a port of reference random_sampler.cu):
 *  - Uniform without-replacement sampling uses a keyed Feistel permutation
 *    per row (hip_common.h): every output element is computed independently
 *    by one lane (coalesced writes, no atomics, no reservoir, no per-row
 *    thread-block), so the kernel is a pure gather bounded by HBM/L2 random
 *    read bandwidth.  The reference instead runs a 128-thread atomicMax
 *    reservoir per row (reference csrc/cuda/random_sampler.cu:59-109).
 *  - Row lookup for a packed output element is a binary search over the
 *    seed-offset array (L2-resident, <= 17 probes).
 *  - Weighted sampling (with replacement; capability the reference's GPU
 *    path lacks — its weighted_sampler.cuh:28-36 is a stub): wave-per-row,
 *    wave-parallel total-weight reduction, then each lane serves draws by
 *    CDF scan.  Rows with deg <= k are copied wholesale.
 *  - Negative sampling: lane-per-candidate with binary-search membership
 *    test (rows must be column-sorted; glt_amd Topology guarantees it).
 *
 * Scans/compactions use torch ops (rocPRIM-backed) — only the hot inner
 * kernels are hand-written.
 */
#include "hip_common.h"
#include "../include/common.h"

namespace glt {

namespace {

__global__ void fill_counts_kernel(const int64_t* __restrict__ indptr,
                                   int64_t num_rows,
                                   const int64_t* __restrict__ seeds,
                                   int64_t bs, int64_t k,
                                   int64_t* __restrict__ counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < bs;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t v = seeds[i];
    const int64_t deg =
        (v >= 0 && v < num_rows) ? indptr[v + 1] - indptr[v] : 0;
    counts[i] = deg < k ? deg : k;
  }
}

template <bool WITH_EID>
__global__ void sample_gather_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const int64_t* __restrict__ eids, int64_t num_rows,
    const int64_t* __restrict__ seeds, int64_t bs, int64_t k,
    const int64_t* __restrict__ offsets, int64_t total, uint64_t call_seed,
    int64_t* __restrict__ out_nbrs, int64_t* __restrict__ out_eids) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; e < total;
       e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = row_of(offsets, bs, e);
    const int64_t j = e - offsets[r];
    const int64_t v = seeds[r];
    const int64_t base = indptr[v];
    const int64_t deg = indptr[v + 1] - base;
    int64_t pos;
    if (deg <= k) {
      pos = j;
    } else {
      const uint64_t key = d_splitmix64(call_seed + (uint64_t)r);
      pos = (int64_t)feistel_perm(key, (uint64_t)j, (uint64_t)deg);
    }
    out_nbrs[e] = indices[base + pos];
    if (WITH_EID) out_eids[e] = eids[base + pos];
  }
}

// ---------------------------------------------------------------------------
// Weighted neighbor sampling — block-per-row, prefix-CDF in LDS.
//
// Round-1's design scanned the row's CDF linearly PER DRAW (O(k*deg): a
// 10k-degree hub with k=15 did 150k serial weight reads).  Now: the block
// builds a per-row inclusive prefix-CDF once (chunked two-level when
// deg > kCdfCap so hubs of any degree fit LDS), then each draw is one
// binary search over the chunk CDF + a <=chunk-size scan.  REPLACE=false
// adds a without-replacement mode the reference lacks entirely
// (Efraimidis-Spirakis exponential race, k rounds of block-argmin; keys
// are recomputable from the call seed so no per-element state is kept).
// ---------------------------------------------------------------------------
// Small-row fast path (deg <= 64, with-replacement): the row's weights
// live one-per-lane in registers; the CDF is a wave shfl-scan and each
// draw is one ballot + ffs — no LDS, no barriers.  Rows with deg > 64
// are left to the block kernel below (and vice versa).
template <bool WITH_EID>
__global__ void weighted_sample_wave_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const int64_t* __restrict__ eids, const float* __restrict__ weights,
    int64_t num_rows, const int64_t* __restrict__ seeds, int64_t bs,
    int64_t k, const int64_t* __restrict__ offsets, uint64_t call_seed,
    int64_t* __restrict__ out_nbrs, int64_t* __restrict__ out_eids) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t r = wave; r < bs; r += n_waves) {
    const int64_t v = seeds[r];
    if (v < 0 || v >= num_rows) continue;
    const int64_t base = indptr[v];
    const int64_t deg = indptr[v + 1] - base;
    if (deg == 0 || deg > kWave) continue;  // big rows: block kernel
    const int64_t off = offsets[r];
    if (deg <= k) {
      if (lane < deg) {
        out_nbrs[off + lane] = indices[base + lane];
        if (WITH_EID) out_eids[off + lane] = eids[base + lane];
      }
      continue;
    }
    float wv = 0.f;
    if (lane < deg) {
      const float w = weights[base + lane];
      wv = w > 0.f ? w : 0.f;
    }
    float c = wv;  // inclusive wave scan
#pragma unroll
    for (int sft = 1; sft < kWave; sft <<= 1) {
      const float p = __shfl_up(c, sft);
      if (lane >= sft) c += p;
    }
    const float tot = __shfl(c, kWave - 1);
    // all k draws in parallel: lane j serves draw j with a fixed
    // 6-step binary search over the register CDF via lane shuffles
    for (int64_t j = lane; j < k; j += kWave) {
      DRng rng(call_seed ^ (uint64_t)r * 0xD6E8FEB86659FD93ull ^
               (uint64_t)j * 0xA24BAED4963EE407ull);
      int64_t idx;
      if (tot > 0.f) {
        const float u = rng.uniform_float() * tot;
        int lo = 0, hi = (int)deg - 1;
#pragma unroll
        for (int step = 0; step < 6; ++step) {
          const int mid = (lo + hi) >> 1;
          const float cm = __shfl(c, mid);
          if (cm >= u) hi = mid; else lo = mid + 1;
          if (lo > hi) lo = hi;
        }
        idx = lo;
      } else {
        idx = (int64_t)rng.uniform((uint64_t)deg);
      }
      out_nbrs[off + j] = indices[base + idx];
      if (WITH_EID) out_eids[off + j] = eids[base + idx];
    }
  }
}

constexpr int kCdfCap = 4096;
constexpr int kSelCap = 256;

template <bool WITH_EID, bool REPLACE>
__global__ __launch_bounds__(256)
void weighted_sample_block_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const int64_t* __restrict__ eids, const float* __restrict__ weights,
    int64_t num_rows, const int64_t* __restrict__ seeds, int64_t bs,
    int64_t k, const int64_t* __restrict__ offsets, uint64_t call_seed,
    int64_t* __restrict__ out_nbrs, int64_t* __restrict__ out_eids) {
  __shared__ float cdf[kCdfCap];
  __shared__ float scr[256];
  __shared__ int64_t scri[256];
  __shared__ int64_t sel[kSelCap];
  __shared__ float carry_s;
  const int tid = threadIdx.x;
  for (int64_t r = blockIdx.x; r < bs; r += gridDim.x) {
    __syncthreads();  // LDS reuse boundary between row iterations
    const int64_t v = seeds[r];
    if (v < 0 || v >= num_rows) continue;
    const int64_t base = indptr[v];
    const int64_t deg = indptr[v + 1] - base;
    const int64_t off = offsets[r];
    if (deg == 0) continue;
    if (REPLACE && deg <= kWave) continue;  // wave kernel handled it
    if (deg <= k) {
      for (int64_t j = tid; j < deg; j += 256) {
        out_nbrs[off + j] = indices[base + j];
        if (WITH_EID) out_eids[off + j] = eids[base + j];
      }
      continue;
    }
    // ---- build the chunked inclusive prefix-CDF --------------------------
    const int64_t cs = (deg + kCdfCap - 1) / kCdfCap;  // chunk size
    const int nch = (int)((deg + cs - 1) / cs);
    for (int ci = tid; ci < nch; ci += 256) {
      float ssum = 0.f;
      const int64_t lo = (int64_t)ci * cs;
      const int64_t hi = lo + cs < deg ? lo + cs : deg;
      for (int64_t i = lo; i < hi; ++i) {
        const float w = weights[base + i];
        ssum += w > 0.f ? w : 0.f;
      }
      cdf[ci] = ssum;
    }
    if (tid == 0) carry_s = 0.f;
    __syncthreads();
    for (int b0 = 0; b0 < nch; b0 += 256) {
      const float val = (b0 + tid < nch) ? cdf[b0 + tid] : 0.f;
      scr[tid] = val;
      __syncthreads();
#pragma unroll
      for (int sft = 1; sft < 256; sft <<= 1) {
        const float add = tid >= sft ? scr[tid - sft] : 0.f;
        __syncthreads();
        scr[tid] += add;
        __syncthreads();
      }
      const float inc = scr[tid] + carry_s;
      if (b0 + tid < nch) cdf[b0 + tid] = inc;
      __syncthreads();
      if (tid == 255) carry_s = inc;
      __syncthreads();
    }
    const float tot = cdf[nch - 1];
    if (!(tot > 0.f)) {
      // degenerate weights: uniform with-replacement fill so the output
      // is never left uninitialized
      for (int64_t j = tid; j < k; j += 256) {
        DRng rng(call_seed ^ (uint64_t)r * 0xD6E8FEB86659FD93ull ^
                 (uint64_t)j * 0xA24BAED4963EE407ull);
        const int64_t idx = (int64_t)rng.uniform((uint64_t)deg);
        out_nbrs[off + j] = indices[base + idx];
        if (WITH_EID) out_eids[off + j] = eids[base + idx];
      }
      continue;
    }
    if (REPLACE) {
      for (int64_t j = tid; j < k; j += 256) {
        DRng rng(call_seed ^ (uint64_t)r * 0xD6E8FEB86659FD93ull ^
                 (uint64_t)j * 0xA24BAED4963EE407ull);
        const float u = rng.uniform_float() * tot;
        int lo = 0, hi = nch - 1;  // first chunk with cdf[c] >= u
        while (lo < hi) {
          const int mid = (lo + hi) >> 1;
          if (cdf[mid] >= u) hi = mid; else lo = mid + 1;
        }
        const int64_t i0 = (int64_t)lo * cs;
        const int64_t i1 = i0 + cs < deg ? i0 + cs : deg;
        float acc = lo ? cdf[lo - 1] : 0.f;
        int64_t idx = i1 - 1;
        for (int64_t i = i0; i < i1; ++i) {
          const float w = weights[base + i];
          acc += w > 0.f ? w : 0.f;
          if (acc >= u) { idx = i; break; }
        }
        out_nbrs[off + j] = indices[base + idx];
        if (WITH_EID) out_eids[off + j] = eids[base + idx];
      }
    } else {
      // Efraimidis-Spirakis: k smallest keys of -log(u_i)/w_i
      const int kk2 = (int)(k < kSelCap ? k : kSelCap);
      for (int round = 0; round < kk2; ++round) {
        float best = 3.4e38f;
        int64_t bi = -1;
        for (int64_t i = tid; i < deg; i += 256) {
          bool taken = false;
          for (int t = 0; t < round; ++t)
            if (sel[t] == i) { taken = true; break; }
          if (taken) continue;
          const float w = weights[base + i];
          if (!(w > 0.f)) continue;
          DRng rng(call_seed ^ (uint64_t)r * 0xD6E8FEB86659FD93ull ^
                   (uint64_t)i * 0x2545F4914F6CDD1Dull);
          float u = rng.uniform_float();
          u = u < 1e-12f ? 1e-12f : u;
          const float key = -__logf(u) / w;
          if (key < best) { best = key; bi = i; }
        }
        scr[tid] = best;
        scri[tid] = bi;
        __syncthreads();
#pragma unroll
        for (int sft = 128; sft > 0; sft >>= 1) {
          if (tid < sft && scr[tid + sft] < scr[tid]) {
            scr[tid] = scr[tid + sft];
            scri[tid] = scri[tid + sft];
          }
          __syncthreads();
        }
        int64_t chosen = scri[0];
        if (tid == 0) {
          if (chosen < 0) {
            // fewer positive-weight neighbors than k: uniform pad
            DRng rng(call_seed ^ (uint64_t)r * 0xA24BAED4963EE407ull ^
                     (uint64_t)round);
            const int64_t idx = (int64_t)rng.uniform((uint64_t)deg);
            out_nbrs[off + round] = indices[base + idx];
            if (WITH_EID) out_eids[off + round] = eids[base + idx];
            sel[round] = -1 - round;  // never matches an index
          } else {
            out_nbrs[off + round] = indices[base + chosen];
            if (WITH_EID) out_eids[off + round] = eids[base + chosen];
            sel[round] = chosen;
          }
        }
        __syncthreads();
      }
    }
  }
}

__global__ void lookup_degree_kernel(const int64_t* __restrict__ indptr,
                                     int64_t num_rows,
                                     const int64_t* __restrict__ nodes,
                                     int64_t n, int64_t* __restrict__ out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t v = nodes[i];
    out[i] = (v >= 0 && v < num_rows) ? indptr[v + 1] - indptr[v] : 0;
  }
}

__device__ __forceinline__ bool d_edge_in_csr(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    int64_t num_rows, int64_t r, int64_t c) {
  if (r < 0 || r >= num_rows) return false;
  int64_t lo = indptr[r], hi = indptr[r + 1];
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    int64_t x = indices[mid];
    if (x == c) return true;
    if (x < c)
      lo = mid + 1;
    else
      hi = mid;
  }
  return false;
}

__global__ void negative_sample_kernel(const int64_t* __restrict__ indptr,
                                       const int64_t* __restrict__ indices,
                                       int64_t num_rows, int64_t num_cols,
                                       int64_t req_num, int64_t trials,
                                       bool padding, uint64_t call_seed,
                                       int64_t* __restrict__ out_rows,
                                       int64_t* __restrict__ out_cols) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < req_num; i += (int64_t)gridDim.x * blockDim.x) {
    DRng rng(call_seed ^ (uint64_t)i * 0x9FB21C651E98DF25ull);
    int64_t rr = -1, cc = -1;
    for (int64_t t = 0; t < trials; ++t) {
      int64_t r = (int64_t)rng.uniform((uint64_t)num_rows);
      int64_t c = (int64_t)rng.uniform((uint64_t)num_cols);
      if (!d_edge_in_csr(indptr, indices, num_rows, r, c)) {
        rr = r;
        cc = c;
        break;
      }
    }
    if (rr < 0 && padding) {
      rr = (int64_t)rng.uniform((uint64_t)num_rows);
      cc = (int64_t)rng.uniform((uint64_t)num_cols);
    }
    out_rows[i] = rr;
    out_cols[i] = cc;
  }
}

__global__ void random_walk_kernel(const int64_t* __restrict__ indptr,
                                   const int64_t* __restrict__ indices,
                                   int64_t num_rows,
                                   const int64_t* __restrict__ seeds,
                                   int64_t n, int64_t walk_len,
                                   uint64_t call_seed,
                                   int64_t* __restrict__ out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    DRng rng(call_seed ^ (uint64_t)i * 0xD6E8FEB86659FD93ull);
    int64_t cur = seeds[i];
    int64_t* row = out + i * (walk_len + 1);
    row[0] = cur;
    for (int64_t s = 1; s <= walk_len; ++s) {
      if (cur >= 0 && cur < num_rows) {
        const int64_t base = indptr[cur];
        const int64_t deg = indptr[cur + 1] - base;
        if (deg > 0) cur = indices[base + (int64_t)rng.uniform((uint64_t)deg)];
      }
      row[s] = cur;
    }
  }
}

uint64_t fresh_seed() { return SeedManager::instance().next_call_seed(); }

}  // namespace

// Host entry points ---------------------------------------------------------

// --- staged sampling API ----------------------------------------------------
// Stage 1: per-seed counts + exclusive offsets, NO host sync.  The
// hetero multihop runs stage 1 for every edge type of a hop, reads all
// the totals with ONE .cpu() round trip, then runs stage 2 — one
// GIL-held device sync per hop instead of one per (hop, etype), which
// is what throttled the in-process producer thread (ROUND2_NOTES).
std::tuple<torch::Tensor, torch::Tensor> hip_sample_neighbors_offsets(
    const torch::Tensor& indptr, const torch::Tensor& seeds, int64_t k) {
  TORCH_CHECK(indptr.is_cuda() && seeds.is_cuda(),
              "sample_neighbors_offsets: device tensors required");
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t bs = seeds.size(0);
  const int64_t kk = k < 0 ? std::numeric_limits<int64_t>::max() : k;
  auto counts = torch::empty({bs}, seeds.options());
  if (bs > 0) {
    hipLaunchKernelGGL(fill_counts_kernel, dim3(grid_for(bs)), dim3(kBlock),
                       0, current_stream(), indptr.data_ptr<int64_t>(),
                       num_rows, seeds.data_ptr<int64_t>(), bs, kk,
                       counts.data_ptr<int64_t>());
  }
  auto offsets = torch::zeros({bs + 1}, seeds.options());
  if (bs > 0) {
    auto v = offsets.narrow(0, 1, bs);
    torch::cumsum_out(v, counts, 0);
  }
  return {counts, offsets};
}

// Stage 2: gather with the caller-synced total.
std::tuple<torch::Tensor, c10::optional<torch::Tensor>>
hip_sample_neighbors_gather(const torch::Tensor& indptr,
                            const torch::Tensor& indices,
                            const c10::optional<torch::Tensor>& edge_ids,
                            const c10::optional<torch::Tensor>& edge_weights,
                            const torch::Tensor& seeds, int64_t k,
                            const torch::Tensor& offsets, int64_t total,
                            bool with_edge, bool weighted, bool replace) {
  TORCH_CHECK(!with_edge || edge_ids.has_value(),
              "with_edge requires edge_ids");
  TORCH_CHECK(!weighted || edge_weights.has_value(),
              "weighted requires edge_weights");
  const int64_t num_rows = indptr.size(0) - 1;
  const int64_t bs = seeds.size(0);
  const int64_t kk = k < 0 ? std::numeric_limits<int64_t>::max() : k;
  auto stream = current_stream();
  auto nbrs = torch::empty({total}, seeds.options());
  auto out_eids = with_edge ? torch::empty({total}, seeds.options())
                            : torch::Tensor();
  if (total > 0) {
    const uint64_t cs = fresh_seed();
    if (!weighted) {
      auto launch = [&](auto with_eid_tag) {
        constexpr bool WE = decltype(with_eid_tag)::value;
        hipLaunchKernelGGL(
            (sample_gather_kernel<WE>), dim3(grid_for(total)), dim3(kBlock), 0,
            stream, indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
            WE ? edge_ids->data_ptr<int64_t>() : nullptr, num_rows,
            seeds.data_ptr<int64_t>(), bs, kk, offsets.data_ptr<int64_t>(),
            total, cs, nbrs.data_ptr<int64_t>(),
            WE ? out_eids.data_ptr<int64_t>() : nullptr);
      };
      if (with_edge)
        launch(std::true_type{});
      else
        launch(std::false_type{});
    } else {
      TORCH_CHECK(replace || kk <= 256,
                  "weighted without-replacement supports k <= 256");
      const int64_t blocks = std::min<int64_t>(bs, 8192);
      auto launch = [&](auto with_eid_tag, auto replace_tag) {
        constexpr bool WE = decltype(with_eid_tag)::value;
        constexpr bool RP = decltype(replace_tag)::value;
        hipLaunchKernelGGL(
            (weighted_sample_block_kernel<WE, RP>),
            dim3((uint32_t)std::max<int64_t>(blocks, 1)), dim3(256), 0,
            stream, indptr.data_ptr<int64_t>(),
            indices.data_ptr<int64_t>(),
            WE ? edge_ids->data_ptr<int64_t>() : nullptr,
            edge_weights->data_ptr<float>(), num_rows,
            seeds.data_ptr<int64_t>(), bs, kk, offsets.data_ptr<int64_t>(),
            cs, nbrs.data_ptr<int64_t>(),
            WE ? out_eids.data_ptr<int64_t>() : nullptr);
      };
      auto launch_wave = [&](auto with_eid_tag) {
        constexpr bool WE = decltype(with_eid_tag)::value;
        const int64_t wv_blocks =
            std::min<int64_t>((bs * kWave + kBlock - 1) / kBlock,
                              kMaxBlocks);
        hipLaunchKernelGGL(
            (weighted_sample_wave_kernel<WE>),
            dim3((uint32_t)std::max<int64_t>(wv_blocks, 1)), dim3(kBlock),
            0, stream, indptr.data_ptr<int64_t>(),
            indices.data_ptr<int64_t>(),
            WE ? edge_ids->data_ptr<int64_t>() : nullptr,
            edge_weights->data_ptr<float>(), num_rows,
            seeds.data_ptr<int64_t>(), bs, kk, offsets.data_ptr<int64_t>(),
            cs, nbrs.data_ptr<int64_t>(),
            WE ? out_eids.data_ptr<int64_t>() : nullptr);
      };
      if (with_edge) {
        if (replace) {
          launch_wave(std::true_type{});
          launch(std::true_type{}, std::true_type{});
        } else {
          launch(std::true_type{}, std::false_type{});
        }
      } else {
        if (replace) {
          launch_wave(std::false_type{});
          launch(std::false_type{}, std::true_type{});
        } else {
          launch(std::false_type{}, std::false_type{});
        }
      }
    }
  }
  return {nbrs,
          with_edge ? c10::optional<torch::Tensor>(out_eids) : c10::nullopt};
}

std::tuple<torch::Tensor, torch::Tensor, c10::optional<torch::Tensor>>
hip_sample_neighbors(const torch::Tensor& indptr, const torch::Tensor& indices,
                     const c10::optional<torch::Tensor>& edge_ids,
                     const c10::optional<torch::Tensor>& edge_weights,
                     const torch::Tensor& seeds, int64_t k, bool with_edge,
                     bool weighted, bool replace) {
  TORCH_CHECK(indptr.is_cuda() && indices.is_cuda() && seeds.is_cuda(),
              "hip_sample_neighbors: tensors must be on GPU (or mapped views)");
  auto [counts, offsets] = hip_sample_neighbors_offsets(indptr, seeds, k);
  const int64_t bs = seeds.size(0);
  const int64_t total =
      bs > 0 ? offsets[bs].item<int64_t>() : 0;  // hop sync
  auto [nbrs, out_eids] = hip_sample_neighbors_gather(
      indptr, indices, edge_ids, edge_weights, seeds, k, offsets, total,
      with_edge, weighted, replace);
  return {nbrs, counts, out_eids};
}

torch::Tensor hip_lookup_degree(const torch::Tensor& indptr,
                                const torch::Tensor& nodes) {
  const int64_t n = nodes.size(0);
  auto out = torch::empty({n}, nodes.options());
  if (n > 0) {
    hipLaunchKernelGGL(lookup_degree_kernel, dim3(grid_for(n)), dim3(kBlock),
                       0, current_stream(), indptr.data_ptr<int64_t>(),
                       indptr.size(0) - 1, nodes.data_ptr<int64_t>(), n,
                       out.data_ptr<int64_t>());
  }
  return out;
}

torch::Tensor hip_sample_negative(const torch::Tensor& indptr,
                                  const torch::Tensor& indices,
                                  int64_t num_cols, int64_t req_num,
                                  int64_t trials, bool padding) {
  const int64_t num_rows = indptr.size(0) - 1;
  auto opts = indices.options();
  auto rows = torch::empty({req_num}, opts);
  auto cols = torch::empty({req_num}, opts);
  if (req_num > 0) {
    hipLaunchKernelGGL(negative_sample_kernel, dim3(grid_for(req_num)),
                       dim3(kBlock), 0, current_stream(),
                       indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                       num_rows, num_cols, req_num, trials, padding,
                       fresh_seed(), rows.data_ptr<int64_t>(),
                       cols.data_ptr<int64_t>());
  }
  auto ok = rows.ge(0);
  return torch::stack({rows.masked_select(ok), cols.masked_select(ok)});
}

torch::Tensor hip_random_walk(const torch::Tensor& indptr,
                              const torch::Tensor& indices,
                              const torch::Tensor& seeds, int64_t walk_len) {
  const int64_t n = seeds.size(0);
  auto out = torch::empty({n, walk_len + 1}, seeds.options());
  if (n > 0) {
    hipLaunchKernelGGL(random_walk_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                       current_stream(), indptr.data_ptr<int64_t>(),
                       indices.data_ptr<int64_t>(), indptr.size(0) - 1,
                       seeds.data_ptr<int64_t>(), n, walk_len, fresh_seed(),
                       out.data_ptr<int64_t>());
  }
  return out;
}

// Importance-probability propagation, GPU twin of cpu_cal_nbr_prob
// (reference CalNbrProbKernel, random_sampler.cu:167-209).  log-space
// accumulation with atomicAdd per edge.
namespace {
__global__ void cal_nbr_prob_kernel(const int64_t* __restrict__ indptr,
                                    const int64_t* __restrict__ indices,
                                    const float* __restrict__ last_prob,
                                    const int64_t* __restrict__ nodes,
                                    int64_t n, int64_t num_rows, int64_t k,
                                    const int64_t* __restrict__ edge_offsets,
                                    int64_t total_edges,
                                    float* __restrict__ log_acc) {
  for (int64_t e = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       e < total_edges; e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = row_of(edge_offsets, n, e);
    const int64_t v = nodes[i];
    const int64_t base = indptr[v];
    const int64_t deg = indptr[v + 1] - base;
    const int64_t j = e - edge_offsets[i];
    const float p_edge =
        fminf(1.0f, (float)k / (float)deg) * last_prob[v];
    if (p_edge <= 0.f) continue;
    const float log1m = __logf(fmaxf(1e-20f, 1.0f - p_edge));
    atomicAdd(&log_acc[indices[base + j]], log1m);
  }
}

__global__ void combine_prob_kernel(const float* __restrict__ last_prob,
                                    float* __restrict__ log_acc, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float keep = 1.0f - __expf(log_acc[i]);
    log_acc[i] = 1.0f - (1.0f - last_prob[i]) * (1.0f - keep);
  }
}
}  // namespace

torch::Tensor hip_cal_nbr_prob(const torch::Tensor& indptr,
                               const torch::Tensor& indices,
                               const torch::Tensor& last_prob,
                               const torch::Tensor& nodes, int64_t k) {
  const int64_t n = nodes.size(0);
  const int64_t num_rows = indptr.size(0) - 1;
  auto degs = hip_lookup_degree(indptr, nodes);
  auto edge_offsets = torch::zeros({n + 1}, nodes.options());
  { auto v = edge_offsets.narrow(0, 1, n); torch::cumsum_out(v, degs, 0); }
  const int64_t total = n > 0 ? edge_offsets[n].item<int64_t>() : 0;
  auto log_acc = torch::zeros_like(last_prob);
  if (total > 0) {
    hipLaunchKernelGGL(cal_nbr_prob_kernel, dim3(grid_for(total)),
                       dim3(kBlock), 0, current_stream(),
                       indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                       last_prob.data_ptr<float>(), nodes.data_ptr<int64_t>(),
                       n, num_rows, k, edge_offsets.data_ptr<int64_t>(), total,
                       log_acc.data_ptr<float>());
  }
  const int64_t nn = last_prob.size(0);
  hipLaunchKernelGGL(combine_prob_kernel, dim3(grid_for(nn)), dim3(kBlock), 0,
                     current_stream(), last_prob.data_ptr<float>(),
                     log_acc.data_ptr<float>(), nn);
  return log_acc;
}

}  // namespace glt
