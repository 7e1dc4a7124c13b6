#include "../include/shm_queue.h"

#include <sys/ipc.h>
#include <sys/shm.h>
#include <time.h>
#include <errno.h>

#include <atomic>
#include <cstring>
#include <stdexcept>

namespace glt {

namespace {
enum BlockState : uint32_t {
  kFree = 0,
  kWriting = 1,
  kReady = 2,
  kDone = 3,
};
}  // namespace

struct ShmQueue::Desc {
  uint64_t v_off;   // virtual payload offset (monotonic)
  uint64_t size;
  std::atomic<uint32_t> state;
  std::atomic<uint32_t> refcnt;
};

struct ShmQueue::Meta {
  uint64_t magic;
  uint32_t capacity;
  uint64_t ring_bytes;
  pthread_mutex_t mtx;
  pthread_cond_t cv_space;
  pthread_cond_t cv_ready;
  // All offsets are virtual (monotonic); physical = v % ring_bytes.
  uint64_t alloc_off;   // next payload byte to allocate
  uint64_t free_off;    // oldest live payload byte
  uint64_t head_desc;   // next descriptor slot to allocate
  uint64_t tail_desc;   // oldest non-retired descriptor
  uint64_t read_desc;   // next descriptor to hand to a consumer
};

static constexpr uint64_t kMagic = 0x474C545F414D4451ull;  // "GLT_AMDQ"

static uint64_t segment_bytes(uint32_t capacity, uint64_t ring_bytes) {
  return sizeof(ShmQueue::Meta) + sizeof(ShmQueue::Desc) * (uint64_t)capacity +
         64 + ring_bytes;  // +64: payload ring is 64-byte aligned
}

static char* ring_base(ShmQueue::Desc* descs, uint32_t capacity) {
  uintptr_t p = reinterpret_cast<uintptr_t>(descs) +
                sizeof(ShmQueue::Desc) * (uint64_t)capacity;
  return reinterpret_cast<char*>((p + 63) & ~uintptr_t(63));
}

ShmQueue::ShmQueue(uint32_t capacity, uint64_t ring_bytes) {
  if (capacity == 0 || ring_bytes == 0)
    throw std::invalid_argument("ShmQueue: capacity and bytes must be > 0");
  const uint64_t total = segment_bytes(capacity, ring_bytes);
  shmid_ = shmget(IPC_PRIVATE, total, IPC_CREAT | 0600);
  if (shmid_ < 0)
    throw std::runtime_error(std::string("shmget failed: ") + strerror(errno));
  owner_ = true;
  void* base = shmat(shmid_, nullptr, 0);
  if (base == (void*)-1)
    throw std::runtime_error(std::string("shmat failed: ") + strerror(errno));
  meta_ = reinterpret_cast<Meta*>(base);
  descs_ = reinterpret_cast<Desc*>(reinterpret_cast<char*>(base) + sizeof(Meta));
  data_ = ring_base(descs_, capacity);

  std::memset(meta_, 0, sizeof(Meta));
  meta_->capacity = capacity;
  meta_->ring_bytes = ring_bytes;
  for (uint32_t i = 0; i < capacity; ++i) {
    descs_[i].state.store(kFree, std::memory_order_relaxed);
    descs_[i].refcnt.store(0, std::memory_order_relaxed);
  }
  pthread_mutexattr_t ma;
  pthread_mutexattr_init(&ma);
  pthread_mutexattr_setpshared(&ma, PTHREAD_PROCESS_SHARED);
  pthread_mutexattr_setrobust(&ma, PTHREAD_MUTEX_ROBUST);
  pthread_mutex_init(&meta_->mtx, &ma);
  pthread_mutexattr_destroy(&ma);
  pthread_condattr_t ca;
  pthread_condattr_init(&ca);
  pthread_condattr_setpshared(&ca, PTHREAD_PROCESS_SHARED);
  pthread_condattr_setclock(&ca, CLOCK_MONOTONIC);
  pthread_cond_init(&meta_->cv_space, &ca);
  pthread_cond_init(&meta_->cv_ready, &ca);
  pthread_condattr_destroy(&ca);
  meta_->magic = kMagic;
}

ShmQueue::ShmQueue(int shmid) { attach(shmid); }

void ShmQueue::attach(int shmid) {
  shmid_ = shmid;
  owner_ = false;
  void* base = shmat(shmid, nullptr, 0);
  if (base == (void*)-1)
    throw std::runtime_error(std::string("shmat failed: ") + strerror(errno));
  meta_ = reinterpret_cast<Meta*>(base);
  if (meta_->magic != kMagic)
    throw std::runtime_error("ShmQueue: bad segment magic");
  descs_ = reinterpret_cast<Desc*>(reinterpret_cast<char*>(base) + sizeof(Meta));
  data_ = ring_base(descs_, meta_->capacity);
}

ShmQueue::~ShmQueue() {
  if (meta_) shmdt(meta_);
  if (owner_ && shmid_ >= 0) shmctl(shmid_, IPC_RMID, nullptr);
}

void* ShmQueue::ring_data() { return data_; }
uint64_t ShmQueue::ring_bytes() const { return meta_->ring_bytes; }
uint64_t ShmQueue::capacity() const { return meta_->capacity; }

uint64_t ShmQueue::pending() const {
  // racy read is fine for introspection
  return meta_->head_desc - meta_->read_desc;
}

namespace {
struct LockGuard {
  pthread_mutex_t* m;
  explicit LockGuard(pthread_mutex_t* mm) : m(mm) {
    int rc = pthread_mutex_lock(m);
    if (rc == EOWNERDEAD) pthread_mutex_consistent(m);
  }
  ~LockGuard() { pthread_mutex_unlock(m); }
};

inline timespec deadline_after_ms(int64_t ms) {
  timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  ts.tv_sec += ms / 1000;
  ts.tv_nsec += (ms % 1000) * 1000000L;
  if (ts.tv_nsec >= 1000000000L) {
    ts.tv_sec += 1;
    ts.tv_nsec -= 1000000000L;
  }
  return ts;
}
}  // namespace

ShmQueue::Ticket ShmQueue::reserve(uint64_t size) {
  if (size > meta_->ring_bytes)
    throw std::invalid_argument("ShmQueue: message larger than ring");
  LockGuard g(&meta_->mtx);
  for (;;) {
    // Retire finished descriptors in order to advance the free boundary.
    while (meta_->tail_desc < meta_->head_desc) {
      Desc& d = descs_[meta_->tail_desc % meta_->capacity];
      if (d.state.load(std::memory_order_acquire) != kDone) break;
      meta_->free_off = d.v_off + d.size;
      d.state.store(kFree, std::memory_order_release);
      ++meta_->tail_desc;
    }
    if (meta_->head_desc - meta_->tail_desc < meta_->capacity) {
      uint64_t v = meta_->alloc_off;
      const uint64_t phys = v % meta_->ring_bytes;
      if (phys + size > meta_->ring_bytes) v += meta_->ring_bytes - phys;  // skip tail fragment
      if (meta_->head_desc == meta_->tail_desc) {
        // queue empty: the skipped tail fragment holds no live data, so the
        // free boundary may jump with the allocation pointer (otherwise a
        // message with size > phys of an empty ring deadlocks).
        meta_->free_off = v;
      }
      if (v + size - meta_->free_off <= meta_->ring_bytes) {
        const uint32_t slot = (uint32_t)(meta_->head_desc % meta_->capacity);
        Desc& d = descs_[slot];
        d.v_off = v;
        d.size = size;
        d.refcnt.store(0, std::memory_order_relaxed);
        d.state.store(kWriting, std::memory_order_release);
        meta_->alloc_off = v + size;
        ++meta_->head_desc;
        Ticket t;
        t.data = data_ + (v % meta_->ring_bytes);
        t.size = size;
        t.slot = slot;
        return t;
      }
    }
    pthread_cond_wait(&meta_->cv_space, &meta_->mtx);
  }
}

void ShmQueue::commit(const Ticket& t) {
  LockGuard g(&meta_->mtx);
  descs_[t.slot].state.store(kReady, std::memory_order_release);
  pthread_cond_broadcast(&meta_->cv_ready);
}

void ShmQueue::enqueue(const void* src, uint64_t size) {
  Ticket t = reserve(size);
  std::memcpy(t.data, src, size);
  commit(t);
}

ShmQueue::Ticket ShmQueue::dequeue(int64_t timeout_ms) {
  timespec dl;
  if (timeout_ms >= 0) dl = deadline_after_ms(timeout_ms);
  LockGuard g(&meta_->mtx);
  for (;;) {
    if (meta_->read_desc < meta_->head_desc) {
      const uint32_t slot = (uint32_t)(meta_->read_desc % meta_->capacity);
      Desc& d = descs_[slot];
      if (d.state.load(std::memory_order_acquire) == kReady) {
        d.refcnt.store(1, std::memory_order_relaxed);
        ++meta_->read_desc;
        Ticket t;
        t.data = data_ + (d.v_off % meta_->ring_bytes);
        t.size = d.size;
        t.slot = slot;
        return t;
      }
    }
    int rc;
    if (timeout_ms >= 0) {
      rc = pthread_cond_timedwait(&meta_->cv_ready, &meta_->mtx, &dl);
      if (rc == ETIMEDOUT) throw QueueTimeoutError();
    } else {
      pthread_cond_wait(&meta_->cv_ready, &meta_->mtx);
    }
  }
}

void ShmQueue::add_refs(uint32_t slot, uint32_t extra) {
  descs_[slot].refcnt.fetch_add(extra, std::memory_order_acq_rel);
}

void ShmQueue::release(uint32_t slot) {
  Desc& d = descs_[slot];
  if (d.refcnt.fetch_sub(1, std::memory_order_acq_rel) == 1) {
    LockGuard g(&meta_->mtx);
    d.state.store(kDone, std::memory_order_release);
    pthread_cond_broadcast(&meta_->cv_space);
  }
}

}  // namespace glt
