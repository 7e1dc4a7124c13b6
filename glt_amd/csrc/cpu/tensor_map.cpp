/* TensorMap serializer + SampleQueue.
 *
 * Serializes a Dict[str, Tensor] sample message into a ShmQueue block and
 * loads it back zero-copy: returned tensors alias the shared-memory ring via
 * torch::from_blob with a releasing deleter, so the ring block is recycled
 * only once every tensor of the message has been freed (out-of-order release).
 * Capability parity: reference csrc/tensor_map.cc + csrc/sample_queue.cc
 * (layout |key|dtype|shape|data| per entry, include/tensor_map.h:24-28).
 *
 * Message layout (all little-endian, data 16-byte aligned):
 *   u32 magic 'GAMQ' | u32 n_entries
 *   per entry: u32 key_len | key | i32 dtype | u32 ndim | i64 shape[ndim]
 *              | u64 nbytes | pad16 | raw data
 */
#include "../include/common.h"
#include "../include/tensor_map.h"

#include <memory>
#include <string>
#include <vector>

namespace glt {

namespace {
constexpr uint32_t kMsgMagic = 0x47414D51u;

inline uint64_t align16(uint64_t x) { return (x + 15) & ~uint64_t(15); }

struct Writer {
  char* p;
  template <typename T>
  void put(const T& v) {
    std::memcpy(p, &v, sizeof(T));
    p += sizeof(T);
  }
  void put_bytes(const void* src, uint64_t n) {
    std::memcpy(p, src, n);
    p += n;
  }
};

struct Reader {
  const char* p;
  template <typename T>
  T get() {
    T v;
    std::memcpy(&v, p, sizeof(T));
    p += sizeof(T);
    return v;
  }
};
}  // namespace

uint64_t tensor_map_serialized_size(
    const std::vector<std::pair<std::string, torch::Tensor>>& entries) {
  uint64_t sz = 8;  // magic + count
  for (const auto& kv : entries) {
    sz += 4 + kv.first.size();
    sz += 4 + 4 + 8 * (uint64_t)kv.second.dim() + 8;
    sz = align16(sz);
    sz += (uint64_t)kv.second.nbytes();
  }
  return sz;
}

void tensor_map_serialize(
    const std::vector<std::pair<std::string, torch::Tensor>>& entries,
    char* base, uint64_t cap) {
  Writer w{base};
  w.put(kMsgMagic);
  w.put((uint32_t)entries.size());
  for (const auto& kv : entries) {
    torch::Tensor t = kv.second.contiguous();
    if (t.is_cuda()) t = t.cpu();  // D2H; callers may pre-stage on pinned mem
    w.put((uint32_t)kv.first.size());
    w.put_bytes(kv.first.data(), kv.first.size());
    w.put((int32_t)t.scalar_type());
    w.put((uint32_t)t.dim());
    for (int64_t d = 0; d < t.dim(); ++d) w.put((int64_t)t.size(d));
    w.put((uint64_t)t.nbytes());
    uint64_t off = (uint64_t)(w.p - base);
    w.p = base + align16(off);
    w.put_bytes(t.data_ptr(), t.nbytes());
  }
  TORCH_CHECK((uint64_t)(w.p - base) <= cap, "tensor_map serialize overflow");
}

// Parse a message at `base`; each tensor aliases the block and calls
// `on_release()` once destroyed.
std::vector<std::pair<std::string, torch::Tensor>> tensor_map_load(
    const char* base, uint64_t size, std::function<void()> on_release) {
  Reader r{base};
  TORCH_CHECK(r.get<uint32_t>() == kMsgMagic, "bad sample message magic");
  const uint32_t n = r.get<uint32_t>();
  std::vector<std::pair<std::string, torch::Tensor>> out;
  out.reserve(n);
  for (uint32_t i = 0; i < n; ++i) {
    const uint32_t klen = r.get<uint32_t>();
    std::string key(r.p, klen);
    r.p += klen;
    const auto dtype = (torch::ScalarType)r.get<int32_t>();
    const uint32_t ndim = r.get<uint32_t>();
    std::vector<int64_t> shape(ndim);
    for (uint32_t d = 0; d < ndim; ++d) shape[d] = r.get<int64_t>();
    const uint64_t nbytes = r.get<uint64_t>();
    uint64_t off = (uint64_t)(r.p - base);
    r.p = base + align16(off);
    auto holder = on_release;  // copied per tensor
    auto t = torch::from_blob(
        const_cast<char*>(r.p), shape,
        [holder](void*) {
          if (holder) holder();
        },
        torch::TensorOptions().dtype(dtype));
    out.emplace_back(std::move(key), std::move(t));
    r.p += nbytes;
    TORCH_CHECK((uint64_t)(r.p - base) <= size, "sample message truncated");
  }
  return out;
}

}  // namespace glt
