/* glt_amd common definitions.
 *
 * MI355X-native GNN sampling engine — shared CPU/HIP declarations.
 * Re-designed from scratch; behavioral parity targets are cited against
 * the reference (alibaba/graphlearn-for-pytorch) as file:line where relevant.
 */
#pragma once

#include <torch/extension.h>

#include <cstdint>
#include <atomic>
#include <random>

namespace glt {

// ---------------------------------------------------------------------------
// Deterministic seeding.
//
// A single process-wide base seed (set from python via manual_seed) combined
// with a monotonically increasing call counter, split per row/lane with
// splitmix64.  This gives reproducible sampling runs when the user seeds,
// while every call still draws fresh randomness.
// Parity target: RandomSeedManager (reference include/common.h, bound at
// python/py_export_glt.cc:84-87).
// ---------------------------------------------------------------------------
class SeedManager {
 public:
  static SeedManager& instance() {
    static SeedManager inst;
    return inst;
  }
  void set_seed(uint64_t s) {
    base_.store(s, std::memory_order_relaxed);
    counter_.store(0, std::memory_order_relaxed);
    seeded_.store(true, std::memory_order_relaxed);
  }
  bool seeded() const { return seeded_.load(std::memory_order_relaxed); }
  // One fresh 64-bit stream id per sampling call.
  uint64_t next_call_seed() {
    uint64_t c = counter_.fetch_add(1, std::memory_order_relaxed);
    uint64_t b = seeded_.load(std::memory_order_relaxed)
                     ? base_.load(std::memory_order_relaxed)
                     : std::random_device{}();
    return b * 0x9E3779B97F4A7C15ull + c + 1;
  }

 private:
  std::atomic<uint64_t> base_{0x243F6A8885A308D3ull};
  std::atomic<uint64_t> counter_{0};
  std::atomic<bool> seeded_{false};
};

// splitmix64: cheap, high-quality 64-bit mixer; used to derive per-row RNG
// streams on both CPU and GPU so results are device-independent in structure
// (not bitwise — different generators).
inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// Fast 64-bit PCG-ish generator for CPU sampling paths (cheaper than mt19937
// to construct per row).
struct Rng64 {
  uint64_t state;
  explicit Rng64(uint64_t seed) : state(splitmix64(seed)) {}
  inline uint64_t next() {
    // xorshift64*
    uint64_t x = state;
    x ^= x >> 12;
    x ^= x << 25;
    x ^= x >> 27;
    state = x;
    return x * 0x2545F4914F6CDD1Dull;
  }
  // uniform integer in [0, n) without modulo bias (Lemire).
  inline uint64_t uniform(uint64_t n) {
    __uint128_t m = (__uint128_t)next() * (__uint128_t)n;
    return (uint64_t)(m >> 64);
  }
  inline float uniform_float() {  // [0, 1)
    return (next() >> 40) * (1.0f / 16777216.0f);
  }
};

inline void check_int64_1d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.dim() == 1, name, " must be 1-D");
  TORCH_CHECK(t.scalar_type() == torch::kInt64, name, " must be int64");
}

}  // namespace glt
