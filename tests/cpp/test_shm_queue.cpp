// C++-level unit test for the ShmQueue ring (torch-free subset), compiled
// and run by tests/test_cpp_units.py with plain g++ (capability parity:
// reference test/cpp/test_shm_queue.cu fork-based producer/consumer).
#include "../../glt_amd/csrc/include/shm_queue.h"

#include <sys/wait.h>
#include <unistd.h>

#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

using glt::ShmQueue;

static void fill(char* p, uint64_t n, unsigned seed) {
  for (uint64_t i = 0; i < n; ++i) p[i] = (char)((seed * 131 + i * 7) & 0xFF);
}

static bool check(const char* p, uint64_t n, unsigned seed) {
  for (uint64_t i = 0; i < n; ++i)
    if (p[i] != (char)((seed * 131 + i * 7) & 0xFF)) return false;
  return true;
}

int main() {
  // 1) single-process wrap-around with varying sizes
  {
    ShmQueue q(4, 1 << 12);
    for (int round = 0; round < 300; ++round) {
      uint64_t sz = 64 + (round * 97) % 2800;
      std::vector<char> buf(sz);
      fill(buf.data(), sz, round);
      q.enqueue(buf.data(), sz);
      auto t = q.dequeue(2000);
      assert(t.size == sz);
      assert(check(t.data, sz, round));
      q.release(t.slot);
    }
  }
  // 2) out-of-order release: hold 3 blocks, release middle-first
  {
    ShmQueue q(8, 1 << 14);
    std::vector<ShmQueue::Ticket> held;
    for (int i = 0; i < 3; ++i) {
      std::vector<char> buf(512);
      fill(buf.data(), 512, 100 + i);
      q.enqueue(buf.data(), 512);
      held.push_back(q.dequeue(2000));
    }
    q.release(held[1].slot);
    q.release(held[2].slot);
    q.release(held[0].slot);
    // ring must be fully reclaimable afterwards
    std::vector<char> big(1 << 13);
    fill(big.data(), big.size(), 7);
    q.enqueue(big.data(), big.size());
    auto t = q.dequeue(2000);
    assert(check(t.data, t.size, 7));
    q.release(t.slot);
  }
  // 3) cross-process producer (fork), FIFO preserved
  {
    ShmQueue q(4, 1 << 12);
    const int N = 200;
    pid_t pid = fork();
    if (pid == 0) {
      ShmQueue child(q.shmid());
      for (int i = 0; i < N; ++i) {
        std::vector<char> buf(128 + i % 700);
        fill(buf.data(), buf.size(), i);
        child.enqueue(buf.data(), buf.size());
      }
      _exit(0);
    }
    for (int i = 0; i < N; ++i) {
      auto t = q.dequeue(10000);
      assert(t.size == (uint64_t)(128 + i % 700));
      assert(check(t.data, t.size, i));
      q.release(t.slot);
    }
    int st = 0;
    waitpid(pid, &st, 0);
    assert(WIFEXITED(st) && WEXITSTATUS(st) == 0);
  }
  // 4) timeout
  {
    ShmQueue q(2, 1 << 10);
    bool threw = false;
    try {
      q.dequeue(50);
    } catch (const glt::QueueTimeoutError&) {
      threw = true;
    }
    assert(threw);
  }
  printf("cpp shm_queue tests OK\n");
  return 0;
}
