/* UnifiedFeatureStore — the multi-segment feature gather engine (gfx950).
 *
 * A logical 2-D row store whose row space is the concatenation of physical
 * segments living on (a) this GPU's HBM3E, (b) peer GPUs reachable over
 * xGMI (hip IPC + peer access), (c) pinned host memory mapped into the
 * device address space (UVA).  The gather kernel resolves each row's owning
 * segment from a register-resident offsets array (<= 16 segments) and
 * copies the row with 16-byte vector loads.
 *
 * Capability parity: reference csrc/cuda/unified_tensor.cu (GatherTensorKernel
 * :48-81, segment offsets scan :35-45, IPC :135-152).  Fresh design:
 * byte-typed rows (one kernel for every dtype instead of an 18-way dtype
 * factory, unified_tensor.cu:114-131), wave-per-row with vector width chosen
 * per call, XCD-friendly grid-stride.
 */
#include "hip_common.h"
#include "../include/common.h"
#include "../include/hip_ops.h"

namespace glt {

namespace {

constexpr int kMaxSegs = 16;

struct SegTable {
  const char* base[kMaxSegs];
  int64_t row_start[kMaxSegs + 1];  // ascending; row r in seg s iff
                                    // row_start[s] <= r < row_start[s+1]
  int n;
};

// Element-level gather: thread t copies VEC bytes of slot (t % slots) of
// output row (t / slots).  Full lane utilization for short feature rows
// (a 400-byte fp32 dim-100 row is only 25 16-byte slots — a wave-per-row
// tiling would idle 39 of 64 lanes); writes stay perfectly coalesced and
// row reads are contiguous VEC-chunks resolved per segment.
template <int VEC>
__global__ void gather_rows_kernel(SegTable segs, int64_t row_bytes,
                                   const int64_t* __restrict__ rows,
                                   int64_t n, char* __restrict__ out) {
  const int64_t slots = row_bytes / VEC;
  const int64_t total = n * slots;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = idx / slots;
    const int64_t j = idx - i * slots;
    const int64_t r = rows[i];
    int s = 0;
    while (s + 1 < segs.n && r >= segs.row_start[s + 1]) ++s;
    const char* src = segs.base[s] + (r - segs.row_start[s]) * row_bytes;
    char* dst = out + i * row_bytes;
    if (VEC == 16) {
      reinterpret_cast<uint4*>(dst)[j] =
          reinterpret_cast<const uint4*>(src)[j];
    } else if (VEC == 8) {
      reinterpret_cast<uint2*>(dst)[j] =
          reinterpret_cast<const uint2*>(src)[j];
    } else if (VEC == 4) {
      reinterpret_cast<uint32_t*>(dst)[j] =
          reinterpret_cast<const uint32_t*>(src)[j];
    } else {
      dst[j] = src[j];
    }
  }
}

}  // namespace

class UnifiedFeatureStore {
 public:
  explicit UnifiedFeatureStore(int64_t device_index)
      : device_(torch::kCUDA, (int8_t)device_index) {}

  void append(const torch::Tensor& seg) {
    TORCH_CHECK(seg.is_cuda(),
                "segments must be device tensors or host-mapped views");
    TORCH_CHECK(seg.dim() == 2, "segment must be 2-D [rows, dim]");
    TORCH_CHECK(seg.is_contiguous(), "segment must be contiguous");
    TORCH_CHECK((int)segs_.size() < kMaxSegs, "too many segments");
    if (!segs_.empty()) {
      TORCH_CHECK(seg.size(1) == dim_ && seg.scalar_type() == dtype_,
                  "segment shape/dtype mismatch");
    } else {
      dim_ = seg.size(1);
      dtype_ = seg.scalar_type();
    }
    segs_.push_back(seg);
    rows_ += seg.size(0);
  }

  torch::Tensor gather(const torch::Tensor& rows) {
    TORCH_CHECK(!segs_.empty(), "empty feature store");
    TORCH_CHECK(rows.is_cuda(), "row indices must be on device");
    const int64_t n = rows.size(0);
    auto out = torch::empty(
        {n, dim_},
        torch::TensorOptions().dtype(dtype_).device(device_));
    if (n == 0) return out;

    SegTable t;
    t.n = (int)segs_.size();
    int64_t acc = 0;
    for (int i = 0; i < t.n; ++i) {
      t.base[i] = reinterpret_cast<const char*>(segs_[i].data_ptr());
      t.row_start[i] = acc;
      acc += segs_[i].size(0);
    }
    t.row_start[t.n] = acc;

    const int64_t row_bytes = dim_ * (int64_t)elementSize(dtype_);
    const int64_t vec = row_bytes % 16 == 0 ? 16
                        : row_bytes % 8 == 0 ? 8
                        : row_bytes % 4 == 0 ? 4 : 1;
    const int blocks = grid_for(n * (row_bytes / vec));
    auto stream = current_stream();
    char* out_p = reinterpret_cast<char*>(out.data_ptr());
    const int64_t* rows_p = rows.data_ptr<int64_t>();
#define GLT_LAUNCH_GATHER(V)                                             \
  hipLaunchKernelGGL((gather_rows_kernel<V>), dim3(blocks), dim3(kBlock), \
                     0, stream, t, row_bytes, rows_p, n, out_p)
    if (row_bytes % 16 == 0)
      GLT_LAUNCH_GATHER(16);
    else if (row_bytes % 8 == 0)
      GLT_LAUNCH_GATHER(8);
    else if (row_bytes % 4 == 0)
      GLT_LAUNCH_GATHER(4);
    else
      GLT_LAUNCH_GATHER(1);
#undef GLT_LAUNCH_GATHER
    return out;
  }

  int64_t rows() const { return rows_; }
  int64_t dim() const { return dim_; }

 private:
  torch::Device device_;
  std::vector<torch::Tensor> segs_;
  int64_t rows_ = 0;
  int64_t dim_ = 0;
  torch::ScalarType dtype_ = torch::kFloat32;
};

std::shared_ptr<UnifiedFeatureStore> ufs_create(int64_t device_index) {
  return std::make_shared<UnifiedFeatureStore>(device_index);
}
void ufs_append(UnifiedFeatureStore* s, const torch::Tensor& seg) {
  s->append(seg);
}
torch::Tensor ufs_gather(UnifiedFeatureStore* s, const torch::Tensor& rows) {
  return s->gather(rows);
}
int64_t ufs_rows(UnifiedFeatureStore* s) { return s->rows(); }
int64_t ufs_dim(UnifiedFeatureStore* s) { return s->dim(); }

}  // namespace glt
