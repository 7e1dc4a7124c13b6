/* HIP/CDNA4 (gfx950) common device helpers for glt_amd.
 *
 * Wave64-first: every kernel here is written for 64-lane wavefronts and the
 * MI355X memory system (per-XCD L2, 256 MiB LLC, HBM3E).  Grid sizing follows
 * the memory-bound rule: cap at ~2048 blocks and grid-stride the rest.
 */
#pragma once

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPCachingAllocator.h>

#include <cstdint>

namespace glt {

#define GLT_HIP_CHECK(expr)                                              \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

constexpr int kWave = 64;
constexpr int kBlock = 256;
// Memory-bound grid cap: 256 CUs x 8 blocks (guide §6 G11).
constexpr int kMaxBlocks = 2048;

inline int grid_for(int64_t work, int block = kBlock) {
  int64_t b = (work + block - 1) / block;
  return (int)std::min<int64_t>(b < 1 ? 1 : b, kMaxBlocks);
}

inline hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// Scratch allocations through PyTorch's caching allocator so sampler scratch
// shares the pool with training tensors (parity: reference common.cuh:40-47).
inline at::Tensor scratch_bytes(int64_t bytes, const at::Device& dev) {
  return at::empty({bytes},
                   at::TensorOptions().dtype(at::kByte).device(dev));
}

// ---------------------------------------------------------------------------
// Device-side RNG & hashing (mirrors csrc/include/common.h host versions).
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint64_t d_splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

struct DRng {
  uint64_t state;
  __device__ __forceinline__ explicit DRng(uint64_t seed)
      : state(d_splitmix64(seed)) {}
  __device__ __forceinline__ uint64_t next() {
    uint64_t x = state;
    x ^= x >> 12;
    x ^= x << 25;
    x ^= x >> 27;
    state = x;
    return x * 0x2545F4914F6CDD1Dull;
  }
  __device__ __forceinline__ uint64_t uniform(uint64_t n) {
    return (uint64_t)(((__uint128_t)next() * (__uint128_t)n) >> 64);
  }
  __device__ __forceinline__ float uniform_float() {
    return (next() >> 40) * (1.0f / 16777216.0f);
  }
};

// ---------------------------------------------------------------------------
// Keyed pseudo-random permutation over [0, n) — the without-replacement
// sampling primitive.  A 6-round Feistel network on the next power-of-two
// domain with cycle-walking back into [0, n).  Each output position is
// computed independently (lane-per-output-element), so uniform neighbor
// sampling needs NO per-row state, NO atomics and NO reservoir pass —
// unlike the reference's 128-thread atomicMax reservoir
// (reference random_sampler.cu:59-109).  O(1) registers, ~30 VALU ops.
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint32_t feistel_round(uint32_t x, uint32_t key) {
  x = (x ^ key) * 0x9E3779B9u;
  x ^= x >> 16;
  x *= 0x85EBCA6Bu;
  x ^= x >> 13;
  return x;
}

// perm over [0, n); key64 must be row-unique.  j must be < n.
__device__ __forceinline__ uint64_t feistel_perm(uint64_t key64, uint64_t j,
                                                 uint64_t n) {
  // bits per half: domain 2^(2*hb) >= n, hb >= 1
  uint32_t total_bits = 64 - __clzll((unsigned long long)(n - 1) | 1ull);
  uint32_t hb = (total_bits + 1) >> 1;
  if (hb < 1) hb = 1;
  const uint32_t hmask = (1u << hb) - 1u;
  const uint32_t k0 = (uint32_t)key64, k1 = (uint32_t)(key64 >> 32);
  uint64_t x = j;
  do {
    uint32_t l = (uint32_t)(x >> hb) & hmask;
    uint32_t r = (uint32_t)x & hmask;
#pragma unroll
    for (int round = 0; round < 6; ++round) {
      uint32_t nl = r;
      r = l ^ (feistel_round(r, k0 + round * 0x7F4A7C15u + (round & 1 ? k1 : 0)) &
               hmask);
      l = nl;
    }
    x = ((uint64_t)l << hb) | r;
  } while (x >= n);  // cycle-walk; expected < 2 iterations
  return x;
}

// Binary search: greatest i such that offsets[i] <= e, offsets ascending of
// length m+1 (returns row index in [0, m)).
__device__ __forceinline__ int64_t row_of(const int64_t* __restrict__ offsets,
                                          int64_t m, int64_t e) {
  int64_t lo = 0, hi = m;  // invariant: offsets[lo] <= e < offsets[hi]
  while (hi - lo > 1) {
    int64_t mid = (lo + hi) >> 1;
    if (offsets[mid] <= e)
      lo = mid;
    else
      hi = mid;
  }
  return lo;
}

}  // namespace glt
