#pragma once

#include <torch/extension.h>

#include <functional>
#include <memory>
#include <string>
#include <utility>
#include <vector>

#include "shm_queue.h"

namespace glt {

using TensorEntries = std::vector<std::pair<std::string, torch::Tensor>>;

uint64_t tensor_map_serialized_size(const TensorEntries& entries);
void tensor_map_serialize(const TensorEntries& entries, char* base,
                          uint64_t cap);
std::vector<std::pair<std::string, torch::Tensor>> tensor_map_load(
    const char* base, uint64_t size, std::function<void()> on_release);

// SampleQueue: Dict[str, Tensor] messages over a ShmQueue.  receive() is
// zero-copy: tensors alias the shm ring until freed.
class SampleQueue {
 public:
  SampleQueue(uint32_t capacity, uint64_t ring_bytes)
      : q_(std::make_shared<ShmQueue>(capacity, ring_bytes)) {}
  explicit SampleQueue(int shmid) : q_(std::make_shared<ShmQueue>(shmid)) {}

  int shmid() const { return q_->shmid(); }
  uint64_t pending() const { return q_->pending(); }
  void* ring_data() { return q_->ring_data(); }
  uint64_t ring_bytes() const { return q_->ring_bytes(); }

  void send(const TensorEntries& entries) {
    const uint64_t sz = tensor_map_serialized_size(entries);
    auto t = q_->reserve(sz);
    tensor_map_serialize(entries, t.data, t.size);
    q_->commit(t);
  }

  TensorEntries receive(int64_t timeout_ms) {
    auto t = q_->dequeue(timeout_ms);
    auto q = q_;
    const uint32_t slot = t.slot;
    auto msg =
        tensor_map_load(t.data, t.size, [q, slot]() { q->release(slot); });
    // One ring ref per tensor; then drop the dequeue ref.
    if (!msg.empty()) q_->add_refs(slot, (uint32_t)msg.size());
    q_->release(slot);
    return msg;
  }

  bool empty() const { return q_->pending() == 0; }

  std::shared_ptr<ShmQueue> queue() { return q_; }

 private:
  std::shared_ptr<ShmQueue> q_;
};

}  // namespace glt
