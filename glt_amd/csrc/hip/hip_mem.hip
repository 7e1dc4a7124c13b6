/* Memory plumbing: host-mapped device views, host pinning, xGMI peer
 * access and hip IPC tensor sharing (gfx950).
 *
 * host_mapped_view() is the MI355X-native replacement for the reference's
 * ZERO_COPY graph/feature modes (reference graph.cu:109-128,
 * unified_tensor.cu:202-231): instead of a C++ Graph class carrying raw
 * registered pointers, any pinned/registered host tensor gets a CUDA-device
 * alias tensor (torch::from_blob over hipHostGetDevicePointer) so every
 * kernel and torch op sees an ordinary device tensor whose reads ride UVA
 * over PCIe.
 */
#include "hip_common.h"
#include "../include/common.h"
#include "../include/hip_ops.h"

#include <mutex>
#include <unordered_map>

namespace glt {

namespace {
// Keep the source tensor (and our registration) alive for as long as any
// mapped view exists.
struct MapHolder {
  torch::Tensor src;
  bool registered;
  void* host_ptr;
  ~MapHolder() {
    if (registered) (void)hipHostUnregister(host_ptr);
  }
};
}  // namespace

torch::Tensor host_mapped_view(const torch::Tensor& src,
                               int64_t device_index) {
  TORCH_CHECK(!src.is_cuda(), "host_mapped_view takes a CPU tensor");
  TORCH_CHECK(src.is_contiguous(), "host_mapped_view requires contiguous");
  auto holder = std::make_shared<MapHolder>();
  holder->src = src;
  holder->host_ptr = src.data_ptr();
  holder->registered = false;
  GLT_HIP_CHECK(hipSetDevice((int)device_index));
  if (!src.is_pinned()) {
    hipError_t e = hipHostRegister(holder->host_ptr, src.nbytes(),
                                   hipHostRegisterMapped);
    if (e == hipErrorHostMemoryAlreadyRegistered) {
      (void)hipGetLastError();
    } else {
      TORCH_CHECK(e == hipSuccess,
                  "hipHostRegister failed: ", hipGetErrorString(e));
      holder->registered = true;
    }
  }
  void* dev_ptr = nullptr;
  GLT_HIP_CHECK(hipHostGetDevicePointer(&dev_ptr, holder->host_ptr, 0));
  auto opts = torch::TensorOptions()
                  .dtype(src.scalar_type())
                  .device(torch::kCUDA, (int8_t)device_index);
  return torch::from_blob(
      dev_ptr, src.sizes(), [holder](void*) mutable { holder.reset(); },
      opts);
}

namespace {
std::mutex g_pin_mtx;
std::unordered_map<int64_t, int64_t> g_pinned;  // addr -> bytes
}  // namespace

void pin_host_memory(int64_t addr, int64_t bytes) {
  std::lock_guard<std::mutex> g(g_pin_mtx);
  if (g_pinned.count(addr)) return;
  hipError_t e =
      hipHostRegister(reinterpret_cast<void*>(addr), bytes,
                      hipHostRegisterMapped);
  if (e == hipErrorHostMemoryAlreadyRegistered) {
    (void)hipGetLastError();
    return;
  }
  TORCH_CHECK(e == hipSuccess, "hipHostRegister: ", hipGetErrorString(e));
  g_pinned[addr] = bytes;
}

void unpin_host_memory(int64_t addr) {
  std::lock_guard<std::mutex> g(g_pin_mtx);
  auto it = g_pinned.find(addr);
  if (it == g_pinned.end()) return;
  (void)hipHostUnregister(reinterpret_cast<void*>(addr));
  g_pinned.erase(it);
}

void enable_peer_access(int64_t device, int64_t peer) {
  if (device == peer) return;
  int can = 0;
  GLT_HIP_CHECK(hipDeviceCanAccessPeer(&can, (int)device, (int)peer));
  TORCH_CHECK(can, "xGMI peer access not available between ", device, " and ",
              peer);
  GLT_HIP_CHECK(hipSetDevice((int)device));
  hipError_t e = hipDeviceEnablePeerAccess((int)peer, 0);
  if (e == hipErrorPeerAccessAlreadyEnabled) {
    (void)hipGetLastError();
    return;
  }
  TORCH_CHECK(e == hipSuccess, "enable_peer_access: ", hipGetErrorString(e));
}

std::string ipc_share(const torch::Tensor& t) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous(), "ipc_share needs contiguous device tensor");
  hipIpcMemHandle_t h;
  GLT_HIP_CHECK(hipIpcGetMemHandle(&h, t.data_ptr()));
  return std::string(reinterpret_cast<char*>(&h), sizeof(h));
}

torch::Tensor ipc_open(const std::string& handle, int64_t device,
                       const std::vector<int64_t>& shape,
                       torch::ScalarType dtype) {
  TORCH_CHECK(handle.size() == sizeof(hipIpcMemHandle_t), "bad ipc handle");
  hipIpcMemHandle_t h;
  std::memcpy(&h, handle.data(), sizeof(h));
  GLT_HIP_CHECK(hipSetDevice((int)device));
  void* ptr = nullptr;
  GLT_HIP_CHECK(
      hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess));
  auto opts = torch::TensorOptions().dtype(dtype).device(torch::kCUDA,
                                                         (int8_t)device);
  return torch::from_blob(
      ptr, shape, [](void* p) { (void)hipIpcCloseMemHandle(p); }, opts);
}

}  // namespace glt
