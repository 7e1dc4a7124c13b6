/* pybind11 bindings for glt_amd._C.
 *
 * One module covering the CPU C++ core and the gfx950 HIP kernels; ops
 * auto-dispatch on the device of their index tensors.  Queue operations
 * release the GIL (parity: reference py_export_glt.cc:129-146).
 */
#include <torch/extension.h>
#include <pybind11/stl.h>

#include "include/common.h"
#include "include/cpu_ops.h"
#include "include/cpu_inducer.h"
#include "include/hip_ops.h"
#include "include/shm_queue.h"
#include "include/tensor_map.h"

namespace glt {
namespace {

using OptTensor = c10::optional<torch::Tensor>;

std::tuple<torch::Tensor, torch::Tensor, OptTensor> sample_neighbors(
    const torch::Tensor& indptr, const torch::Tensor& indices,
    const OptTensor& edge_ids, const OptTensor& edge_weights,
    const torch::Tensor& seeds, int64_t k, bool with_edge, bool weighted,
    bool replace) {
  if (seeds.is_cuda())
    return hip_sample_neighbors(indptr, indices, edge_ids, edge_weights,
                                seeds, k, with_edge, weighted, replace);
  return cpu_sample_neighbors(indptr, indices, edge_ids, edge_weights, seeds,
                              k, with_edge, weighted, replace);
}

torch::Tensor lookup_degree(const torch::Tensor& indptr,
                            const torch::Tensor& nodes) {
  return nodes.is_cuda() ? hip_lookup_degree(indptr, nodes)
                         : cpu_lookup_degree(indptr, nodes);
}

torch::Tensor sample_negative(const torch::Tensor& indptr,
                              const torch::Tensor& indices, int64_t num_cols,
                              int64_t req_num, int64_t trials, bool padding) {
  return indices.is_cuda()
             ? hip_sample_negative(indptr, indices, num_cols, req_num, trials,
                                   padding)
             : cpu_sample_negative(indptr, indices, num_cols, req_num, trials,
                                   padding);
}

torch::Tensor random_walk(const torch::Tensor& indptr,
                          const torch::Tensor& indices,
                          const torch::Tensor& seeds, int64_t walk_len) {
  return seeds.is_cuda() ? hip_random_walk(indptr, indices, seeds, walk_len)
                         : cpu_random_walk(indptr, indices, seeds, walk_len);
}

torch::Tensor cal_nbr_prob(const torch::Tensor& indptr,
                           const torch::Tensor& indices,
                           const torch::Tensor& last_prob,
                           const torch::Tensor& nodes, int64_t k) {
  return nodes.is_cuda() ? hip_cal_nbr_prob(indptr, indices, last_prob, nodes, k)
                         : cpu_cal_nbr_prob(indptr, indices, last_prob, nodes, k);
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, OptTensor>
node_subgraph(const torch::Tensor& indptr, const torch::Tensor& indices,
              const OptTensor& edge_ids, const torch::Tensor& nodes,
              bool with_edge) {
  return nodes.is_cuda()
             ? hip_node_subgraph(indptr, indices, edge_ids, nodes, with_edge)
             : cpu_node_subgraph(indptr, indices, edge_ids, nodes, with_edge);
}

std::tuple<torch::Tensor, torch::Tensor, OptTensor> stitch_sample_results(
    int64_t ids_count, const std::vector<torch::Tensor>& idx_list,
    const std::vector<torch::Tensor>& nbrs_list,
    const std::vector<torch::Tensor>& nbrs_num_list,
    const std::vector<torch::Tensor>& eids_list) {
  const bool cuda = !idx_list.empty() && idx_list[0].is_cuda();
  return cuda ? hip_stitch_sample_results(ids_count, idx_list, nbrs_list,
                                          nbrs_num_list, eids_list)
              : cpu_stitch_sample_results(ids_count, idx_list, nbrs_list,
                                          nbrs_num_list, eids_list);
}

// Device inducer wrapper (HIPInducer itself is opaque to the host compiler).
struct DeviceInducer {
  std::shared_ptr<HIPInducer> p;
  explicit DeviceInducer(int64_t reserve) : p(hip_inducer_create(reserve)) {}
  torch::Tensor init_node(const torch::Tensor& s) {
    return hip_inducer_init_node(p.get(), s);
  }
  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> induce_next(
      const torch::Tensor& srcs, const torch::Tensor& nbrs,
      const torch::Tensor& cnt) {
    return hip_inducer_induce_next(p.get(), srcs, nbrs, cnt);
  }
  torch::Tensor lookup(const torch::Tensor& ids) {
    return hip_inducer_lookup(p.get(), ids);
  }
  torch::Tensor insert(const torch::Tensor& ids) {
    return hip_inducer_insert(p.get(), ids);
  }
  void reserve_incoming(int64_t total) {
    hip_inducer_reserve(p.get(), total);
  }
  std::tuple<torch::Tensor, torch::Tensor> insert_begin(
      const torch::Tensor& ids, int64_t idx_base) {
    return hip_inducer_insert_begin(p.get(), ids, idx_base);
  }
  torch::Tensor insert_commit(const torch::Tensor& ids,
                              const torch::Tensor& flags,
                              const torch::Tensor& ranks, int64_t n_new) {
    return hip_inducer_insert_commit(p.get(), ids, flags, ranks, n_new);
  }
  int64_t count() { return hip_inducer_count(p.get()); }
};

// Deferred-sync multi-hop sampler (opaque HIP-side implementation).
struct DeferredSamplerPy {
  std::shared_ptr<DeferredSampler> p;
  DeferredSamplerPy(std::vector<int64_t> fanout, int64_t batch_cap,
                    int64_t device, bool with_eid)
      : p(deferred_sampler_create(std::move(fanout), batch_cap, device,
                                  with_eid)) {}
  auto run(const torch::Tensor& indptr, const torch::Tensor& indices,
           const c10::optional<torch::Tensor>& edge_ids,
           const torch::Tensor& seeds) {
    return deferred_sampler_run(p.get(), indptr, indices, edge_ids, seeds);
  }
};

struct FeatureStorePy {
  std::shared_ptr<UnifiedFeatureStore> p;
  explicit FeatureStorePy(int64_t device) : p(ufs_create(device)) {}
  void append(const torch::Tensor& seg) { ufs_append(p.get(), seg); }
  torch::Tensor gather(const torch::Tensor& rows) {
    return ufs_gather(p.get(), rows);
  }
  int64_t rows() { return ufs_rows(p.get()); }
  int64_t dim() { return ufs_dim(p.get()); }
};

torch::Tensor ipc_open_py(const std::string& handle, int64_t device,
                          const std::vector<int64_t>& shape,
                          int64_t dtype_code) {
  return ipc_open(handle, device, shape,
                  static_cast<torch::ScalarType>(dtype_code));
}

}  // namespace
}  // namespace glt

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using namespace glt;
  m.doc() = "glt_amd native core (CPU C++ + gfx950 HIP)";

  m.def("manual_seed",
        [](uint64_t s) { SeedManager::instance().set_seed(s); });

  m.def(
      "sample_neighbors",
      [](const torch::Tensor& indptr, const torch::Tensor& indices,
         const torch::Tensor& seeds, int64_t k, const OptTensor& edge_ids,
         const OptTensor& edge_weights, bool with_edge, bool weighted,
         bool replace) {
        return sample_neighbors(indptr, indices, edge_ids, edge_weights,
                                seeds, k, with_edge, weighted, replace);
      },
      py::arg("indptr"), py::arg("indices"), py::arg("seeds"), py::arg("k"),
      py::arg("edge_ids") = py::none(), py::arg("edge_weights") = py::none(),
      py::arg("with_edge") = false, py::arg("weighted") = false,
      py::arg("replace") = true);
  m.def("sample_neighbors_offsets", &hip_sample_neighbors_offsets,
        py::arg("indptr"), py::arg("seeds"), py::arg("k"));
  m.def(
      "sample_neighbors_gather",
      [](const torch::Tensor& indptr, const torch::Tensor& indices,
         const torch::Tensor& seeds, int64_t k,
         const torch::Tensor& offsets, int64_t total,
         const OptTensor& edge_ids, const OptTensor& edge_weights,
         bool with_edge, bool weighted, bool replace) {
        return hip_sample_neighbors_gather(indptr, indices, edge_ids,
                                           edge_weights, seeds, k, offsets,
                                           total, with_edge, weighted,
                                           replace);
      },
      py::arg("indptr"), py::arg("indices"), py::arg("seeds"), py::arg("k"),
      py::arg("offsets"), py::arg("total"),
      py::arg("edge_ids") = py::none(), py::arg("edge_weights") = py::none(),
      py::arg("with_edge") = false, py::arg("weighted") = false,
      py::arg("replace") = true);
  m.def("lookup_degree", &lookup_degree);
  m.def("sample_negative", &sample_negative, py::arg("indptr"),
        py::arg("indices"), py::arg("num_cols"), py::arg("req_num"),
        py::arg("trials") = 5, py::arg("padding") = false);
  m.def("random_walk", &random_walk);
  m.def("cal_nbr_prob", &cal_nbr_prob);
  m.def(
      "node_subgraph",
      [](const torch::Tensor& indptr, const torch::Tensor& indices,
         const torch::Tensor& nodes, const OptTensor& edge_ids,
         bool with_edge) {
        return node_subgraph(indptr, indices, edge_ids, nodes, with_edge);
      },
      py::arg("indptr"), py::arg("indices"), py::arg("nodes"),
      py::arg("edge_ids") = py::none(), py::arg("with_edge") = false);
  m.def("stitch_sample_results", &stitch_sample_results, py::arg("ids_count"),
        py::arg("idx_list"), py::arg("nbrs_list"), py::arg("nbrs_num_list"),
        py::arg("eids_list") = std::vector<torch::Tensor>{});

  py::class_<CPUInducer>(m, "CPUInducer")
      .def(py::init<int64_t>(), py::arg("reserve") = 1024)
      .def("init_node", &CPUInducer::init_node)
      .def("induce_next", &CPUInducer::induce_next)
      .def("reset", &CPUInducer::reset);

  py::class_<CPUHeteroInducer>(m, "CPUHeteroInducer")
      .def(py::init<int64_t>(), py::arg("reserve") = 1024)
      .def("init_node", &CPUHeteroInducer::init_node)
      .def("induce_next", &CPUHeteroInducer::induce_next)
      .def("reset", &CPUHeteroInducer::reset);

  py::class_<DeviceInducer>(m, "DeviceInducer")
      .def(py::init<int64_t>(), py::arg("reserve") = 4096)
      .def("init_node", &DeviceInducer::init_node)
      .def("induce_next", &DeviceInducer::induce_next)
      .def("lookup", &DeviceInducer::lookup)
      .def("insert", &DeviceInducer::insert)
      .def("reserve_incoming", &DeviceInducer::reserve_incoming)
      .def("insert_begin", &DeviceInducer::insert_begin)
      .def("insert_commit", &DeviceInducer::insert_commit)
      .def("count", &DeviceInducer::count);

  py::class_<DeferredSamplerPy>(m, "DeferredSampler")
      .def(py::init<std::vector<int64_t>, int64_t, int64_t, bool>(),
           py::arg("fanout"), py::arg("batch_cap"), py::arg("device") = 0,
           py::arg("with_eid") = false)
      .def("run", &DeferredSamplerPy::run, py::arg("indptr"),
           py::arg("indices"), py::arg("edge_ids"), py::arg("seeds"));

  py::class_<FeatureStorePy>(m, "UnifiedFeatureStore")
      .def(py::init<int64_t>(), py::arg("device"))
      .def("append", &FeatureStorePy::append)
      .def("gather", &FeatureStorePy::gather)
      .def("rows", &FeatureStorePy::rows)
      .def("dim", &FeatureStorePy::dim);

  m.def("sage_gemm", &hip_sage_gemm, py::arg("A"), py::arg("B"),
        py::arg("bias") = py::none(), py::arg("relu") = false);
  m.def("gemm_bt_bf16", &hip_gemm_bt_bf16, py::arg("A"), py::arg("Bt"),
        py::arg("bias") = py::none(), py::arg("relu") = false,
        py::arg("out_fp32") = false);
  m.def("gemm_kt_bf16", &hip_gemm_kt_bf16, py::arg("A"), py::arg("B"),
        py::arg("with_db") = false);
  m.def("mfma_bf16_selftest", &hip_mfma_bf16_selftest);
  m.def("gat_fused_fwd", &hip_gat_fused_fwd);
  m.def("gat_fused_bwd", &hip_gat_fused_bwd);
  m.def("gat_multi_fwd", &hip_gat_multi_fwd);
  m.def("gat_multi_bwd", &hip_gat_multi_bwd);
  m.def("segment_mean_fwd", &hip_segment_mean_fwd);
  m.def("segment_mean_bwd", &hip_segment_mean_bwd);
  m.def("segment_mean_cat_fwd", &hip_segment_mean_cat_fwd);
  m.def("segment_mean_cat_bwd", &hip_segment_mean_cat_bwd);

  // Memory plumbing
  m.def("host_mapped_view", &host_mapped_view, py::arg("src"),
        py::arg("device") = 0);
  m.def("pin_host_memory", &pin_host_memory);
  m.def("unpin_host_memory", &unpin_host_memory);
  m.def("enable_peer_access", &enable_peer_access);
  m.def("ipc_share", [](const torch::Tensor& t) {
    return py::bytes(ipc_share(t));
  });
  m.def("ipc_open", &ipc_open_py);

  // Sample channel
  py::register_exception<QueueTimeoutError>(m, "QueueTimeoutError");
  py::class_<SampleQueue>(m, "SampleQueue")
      .def(py::init<uint32_t, uint64_t>(), py::arg("capacity"),
           py::arg("ring_bytes"))
      .def(py::init<int>(), py::arg("shmid"))
      .def_property_readonly("shmid", &SampleQueue::shmid)
      .def("send", &SampleQueue::send,
           py::call_guard<py::gil_scoped_release>())
      .def("receive", &SampleQueue::receive, py::arg("timeout_ms") = -1,
           py::call_guard<py::gil_scoped_release>())
      .def("empty", &SampleQueue::empty)
      .def("pending", &SampleQueue::pending)
      .def("pin_memory", [](SampleQueue& q) {
        pin_host_memory((int64_t)(uintptr_t)q.ring_data(),
                        (int64_t)q.ring_bytes());
      });
}
