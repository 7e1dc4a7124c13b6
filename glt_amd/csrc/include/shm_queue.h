/* Cross-process shared-memory ring queue for sample messages.
 *
 * Capability parity: reference csrc/shm_queue.cc / include/shm_queue.h
 * (SysV shm ring buffer with per-block semaphores, tail-fragment handling,
 * out-of-order release).  This is a fresh design:
 *   - one process-shared mutex + two condvars (space / ready) in the segment,
 *   - monotonic virtual byte offsets (physical = v % ring_bytes) with
 *     tail-skip when a message would straddle the wrap point,
 *   - descriptor ring handed to the consumer in allocation (FIFO) order,
 *   - blocks released out of order by zero-copy readers (refcounted) and
 *     retired in order to advance the free boundary.
 * Picklable by shmid (SysV), like the reference (py_export_glt.cc:138-146).
 */
#pragma once

#include <pthread.h>
#include <stdexcept>
#include <cstddef>
#include <cstdint>

namespace glt {

class QueueTimeoutError : public std::runtime_error {
 public:
  QueueTimeoutError() : std::runtime_error("glt_amd: shm queue timeout") {}
};

class ShmQueue {
 public:
  // Create a new segment holding up to `capacity` in-flight messages and
  // `ring_bytes` of payload.
  ShmQueue(uint32_t capacity, uint64_t ring_bytes);
  // Attach to an existing segment by SysV shm id.
  explicit ShmQueue(int shmid);
  ~ShmQueue();
  ShmQueue(const ShmQueue&) = delete;
  ShmQueue& operator=(const ShmQueue&) = delete;

  int shmid() const { return shmid_; }

  struct Ticket {
    char* data = nullptr;
    uint64_t size = 0;
    uint32_t slot = 0;
  };

  // Two-phase producer API (zero-copy serialize directly into the ring).
  Ticket reserve(uint64_t size);
  void commit(const Ticket& t);
  // Convenience copy-in enqueue.
  void enqueue(const void* data, uint64_t size);

  // FIFO consumer: blocks up to timeout_ms (<0 = forever); throws
  // QueueTimeoutError on timeout.  The returned block stays valid until
  // `release(slot)` has been called `refcnt` times (set by add_refs).
  Ticket dequeue(int64_t timeout_ms);
  void add_refs(uint32_t slot, uint32_t extra);
  void release(uint32_t slot);

  // Whole payload region (for hipHostRegister pinning).
  void* ring_data();
  uint64_t ring_bytes() const;
  uint64_t capacity() const;
  // Number of committed-but-unread messages.
  uint64_t pending() const;

  struct Desc;
  struct Meta;

 private:
  Meta* meta_ = nullptr;
  char* data_ = nullptr;
  Desc* descs_ = nullptr;
  int shmid_ = -1;
  bool owner_ = false;

  void attach(int shmid);
};

}  // namespace glt
