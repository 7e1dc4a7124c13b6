/* Hand-written bf16 MFMA GEMM family for GNN projection shapes (gfx950).
 *
 * Why: hipBLASLt's heuristic picks ~50 TF/s tiles for the tall-skinny
 * batches this framework produces (M~1e5, N=256, K=200/512 — measured
 * 322 us for the 17 GFLOP flagship projection, profiles/r02_*), a tiny
 * fraction of the 16x16x32_bf16 MFMA rate.  Three kernels:
 *
 *  1. gemm_bt_kernel:  C[M,N] = relu(A[M,K] @ Bt[N,K]^T + bias)
 *     - Bt is the nn.Linear weight layout [out,in] as-is: NO transpose
 *       materialization anywhere on the forward path.
 *     - bf16 in, bf16 out, fp32 accumulate, bias+ReLU fused in the
 *       epilogue (kills the separate activation round-trip).
 *  2. gemm_kt_kernel:  dW[M,N] = A[Kb,M]^T @ B[Kb,N], db[M] = colsum(A)
 *     - the weight-gradient shape (K huge = batch rows): split-K over
 *       grid.z into per-chunk partial planes (a torch sum folds them);
 *       the bias gradient rides the same kernel (the separate
 *       [rows,256] bf16 column-reduce measured 78 us/call).
 *     - fp32 out: master-grad dtype, so the .to(fp32) casts disappear.
 *  3. mfma_bf16_selftest: one 16x16x32 / 32x32x16 tile from explicit
 *     matrices (fragment-layout ground truth for the GPU tests).
 *
 * Fragment layouts (cdna4 §10): 16x16x32_bf16: A lane l holds
 * A[i=l&15][k=(l>>4)*8+e], B lane l holds B[k=(l>>4)*8+e][j=l&15],
 * C/D col=l&15, row=(l>>4)*4+reg.  Both kernels read fragments as
 * contiguous 16-byte LDS vectors: A and Bt are staged row-major (their
 * K dim is innermost in global memory already), the kt kernel stages
 * transposed during the write pass instead.
 */
#include <cstdlib>

#include "hip_common.h"
#include "../include/common.h"

namespace glt {

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4b;

constexpr int GBM = 64, GBN = 64, GBK = 64;  // block tile
constexpr int LDP = 8;                       // LDS row pad (bf16 elems)

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }
__device__ __forceinline__ __bf16 f2bf(float v) { return (__bf16)v; }

// C[M,N] = act(A[M,K] @ Bt[N,K]^T + bias).  4 waves as 2x2; each wave a
// 32x32 tile of 2x2 16x16 fragments; K staged in GBK=64 steps.
template <bool RELU, bool BF16_OUT>
__global__ __launch_bounds__(256)
void gemm_bt_kernel(const __bf16* __restrict__ A,
                    const __bf16* __restrict__ Bt,
                    const float* __restrict__ bias,
                    void* __restrict__ Cv,
                    int64_t M, int64_t K, int64_t N) {
  __shared__ __bf16 As[GBM][GBK + LDP];
  __shared__ __bf16 Bs[GBN][GBK + LDP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;            // 32-row wave tile
  const int wc = wave & 1;             // 32-col wave tile

  const int64_t block_row = (int64_t)blockIdx.x * GBM;
  const int64_t block_col = (int64_t)blockIdx.y * GBN;

  f32x4b acc[2][2] = {};

  // staging map: 256 threads, each loads 16 bf16 (32 B) per row set;
  // a 64x64 bf16 tile is 256 x 16 elements.
  const int s_row = tid >> 2;          // 0..63
  const int s_col = (tid & 3) * 16;    // 0,16,32,48

  const int fi = lane & 15;            // fragment row/col
  const int fk8 = (lane >> 4) * 8;     // fragment k base

  for (int64_t k0 = 0; k0 < K; k0 += GBK) {
    const int64_t kmax = K - k0;
    // stage A row-major
    {
      const int64_t g_row = block_row + s_row;
      bf16x8 v0 = {}, v1 = {};
      if (g_row < M) {
        const int64_t base = g_row * K + k0 + s_col;
        if (s_col + 15 < kmax) {
          v0 = *reinterpret_cast<const bf16x8*>(&A[base]);
          v1 = *reinterpret_cast<const bf16x8*>(&A[base + 8]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (s_col + e < kmax) v0[e] = A[base + e];
            if (s_col + 8 + e < kmax) v1[e] = A[base + 8 + e];
          }
        }
      }
      *reinterpret_cast<bf16x8*>(&As[s_row][s_col]) = v0;
      *reinterpret_cast<bf16x8*>(&As[s_row][s_col + 8]) = v1;
    }
    // stage Bt row-major (same geometry: rows are output cols)
    {
      const int64_t g_row = block_col + s_row;
      bf16x8 v0 = {}, v1 = {};
      if (g_row < N) {
        const int64_t base = g_row * K + k0 + s_col;
        if (s_col + 15 < kmax) {
          v0 = *reinterpret_cast<const bf16x8*>(&Bt[base]);
          v1 = *reinterpret_cast<const bf16x8*>(&Bt[base + 8]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (s_col + e < kmax) v0[e] = Bt[base + e];
            if (s_col + 8 + e < kmax) v1[e] = Bt[base + 8 + e];
          }
        }
      }
      *reinterpret_cast<bf16x8*>(&Bs[s_row][s_col]) = v0;
      *reinterpret_cast<bf16x8*>(&Bs[s_row][s_col + 8]) = v1;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < GBK; kk += 32) {
      // NOTE: mfma B fragment is B[k][j]; our Bs rows are j with k
      // contiguous — exactly the A-layout, and A@Bt^T is symmetric in
      // (A,Bt), so both fragments load the same way.
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + fi][kk + fk8]);
      bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + 16 + fi][kk + fk8]);
      bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + fi][kk + fk8]);
      bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + 16 + fi][kk + fk8]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0,
                                                          acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1,
                                                          acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0,
                                                          acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1,
                                                          acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D: col = lane&15, row = (lane>>4)*4 + reg.
  // mfma computed acc = As_frag @ Bs_frag^T with both fragments in
  // "A layout": result row i = A row, result col j = Bs row = output col.
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
#pragma unroll
  for (int fi2 = 0; fi2 < 2; ++fi2) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t row = block_row + wr * 32 + fi2 * 16 + c_row0 + r;
        const int64_t col = block_col + wc * 32 + fj * 16 + c_col;
        if (row < M && col < N) {
          float v = acc[fi2][fj][r];
          if (bias != nullptr) v += bias[col];
          if (RELU) v = v > 0.f ? v : 0.f;
          if (BF16_OUT)
            reinterpret_cast<__bf16*>(Cv)[row * N + col] = f2bf(v);
          else
            reinterpret_cast<float*>(Cv)[row * N + col] = v;
        }
      }
    }
  }
}


// Persistent-B forward variant for K <= 256: the whole Bt n-tile
// ([64][K] bf16 <= 32 KB) is staged ONCE per workgroup, then A m-tiles
// stream through with a single sync pair each and one full-K MFMA pass
// — the generic kernel re-stages B and syncs per 64-K step.
constexpr int PKP = 256;  // padded K capacity

template <bool RELU, bool BF16_OUT>
__global__ __launch_bounds__(256)
void gemm_bt_persist_kernel(const __bf16* __restrict__ A,
                            const __bf16* __restrict__ Bt,
                            const float* __restrict__ bias,
                            void* __restrict__ Cv,
                            int64_t M, int64_t K, int64_t N) {
  __shared__ __bf16 As[GBM][PKP + LDP];
  __shared__ __bf16 Bs[GBN][PKP + LDP];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const int64_t block_col = (int64_t)blockIdx.y * GBN;
  const int s_row = tid >> 2;          // 0..63
  const int s_col0 = (tid & 3) * 64;   // 4 threads cover 256 cols
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  const int kp = (int)((K + 31) / 32) * 32;

  // stage Bt rows once (zero-padded K tail)
  {
    const int64_t g_row = block_col + s_row;
#pragma unroll
    for (int c8 = 0; c8 < 64; c8 += 8) {
      const int col = s_col0 + c8;
      bf16x8 v = {};
      if (col < kp && g_row < N) {
        const int64_t base = g_row * K + col;
        if (col + 7 < K) {
          v = *reinterpret_cast<const bf16x8*>(&Bt[base]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (col + e < K) v[e] = Bt[base + e];
        }
      }
      if (col < kp)
        *reinterpret_cast<bf16x8*>(&Bs[s_row][col]) = v;
    }
  }
  __syncthreads();

  for (int64_t m0 = (int64_t)blockIdx.x * GBM; m0 < M;
       m0 += (int64_t)gridDim.x * GBM) {
    {
      const int64_t g_row = m0 + s_row;
#pragma unroll
      for (int c8 = 0; c8 < 64; c8 += 8) {
        const int col = s_col0 + c8;
        bf16x8 v = {};
        if (col < kp && g_row < M) {
          const int64_t base = g_row * K + col;
          if (col + 7 < K) {
            v = *reinterpret_cast<const bf16x8*>(&A[base]);
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e)
              if (col + e < K) v[e] = A[base + e];
          }
        }
        if (col < kp)
          *reinterpret_cast<bf16x8*>(&As[s_row][col]) = v;
      }
    }
    __syncthreads();
    f32x4b acc[2][2] = {};
    for (int kk = 0; kk < kp; kk += 32) {
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + fi][kk + fk8]);
      bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + 16 + fi][kk + fk8]);
      bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + fi][kk + fk8]);
      bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + 16 + fi][kk + fk8]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0,
                                                          acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1,
                                                          acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0,
                                                          acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1,
                                                          acc[1][1], 0, 0, 0);
    }
    __syncthreads();
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;
#pragma unroll
    for (int fi2 = 0; fi2 < 2; ++fi2) {
#pragma unroll
      for (int fj = 0; fj < 2; ++fj) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t row = m0 + wr * 32 + fi2 * 16 + c_row0 + r;
          const int64_t col = block_col + wc * 32 + fj * 16 + c_col;
          if (row < M && col < N) {
            float v = acc[fi2][fj][r];
            if (bias != nullptr) v += bias[col];
            if (RELU) v = v > 0.f ? v : 0.f;
            if (BF16_OUT)
              reinterpret_cast<__bf16*>(Cv)[row * N + col] = f2bf(v);
            else
              reinterpret_cast<float*>(Cv)[row * N + col] = v;
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dW[M,N] = A[Kb,M]^T @ B[Kb,N] (+ db[M] = colsum A), split-K over
// grid.z with per-chunk partial planes.  A/B tiles are transposed
// during the LDS write pass (paired k-rows -> b32 writes) so fragments
// read as contiguous 16-byte vectors for the bf16 16x16x32 MFMA.
// (An f32-16x16x4 variant with transpose-free fragments measured
// SLOWER — 40 vs 56 TF/s — the f32 MFMA issue rate is the wall.)
// ---------------------------------------------------------------------------
template <bool WITH_DB>
__global__ __launch_bounds__(256)
void gemm_kt_kernel(const __bf16* __restrict__ A,
                    const __bf16* __restrict__ B,
                    float* __restrict__ C,
                    float* __restrict__ db,
                    int64_t Kb, int64_t M, int64_t N, int64_t k_per_z) {
  __shared__ __bf16 As[GBM][GBK + LDP];   // As[m][k]
  __shared__ __bf16 Bs[GBN][GBK + LDP];   // Bs[n][k]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int64_t block_row = (int64_t)blockIdx.x * GBM;   // m tile
  const int64_t block_col = (int64_t)blockIdx.y * GBN;   // n tile
  const int64_t kz0 = (int64_t)blockIdx.z * k_per_z;
  const int64_t kz1 = std::min(kz0 + k_per_z, Kb);

  f32x4b acc[2][2] = {};

  // transposing stage: thread owns TWO adjacent k rows x 8 cols, so
  // each LDS write is a b32 pair (halves the transpose instruction
  // count vs scalar b16 writes).  32 k-pairs x 8 col-groups = 256.
  const int t_k = (tid >> 3) * 2;      // 0..62 step 2
  const int t_c = (tid & 7) * 8;       // 0..56 step 8

  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;

  for (int64_t k0 = kz0; k0 < kz1; k0 += GBK) {
    const int64_t g_k = k0 + t_k;
    // stage A^T: As[m][k] = A[k][m]
    {
      bf16x8 v0 = {}, v1 = {};
      if (g_k < kz1 && block_row + t_c < M) {
        const int64_t base = g_k * M + block_row + t_c;
        if (block_row + t_c + 7 < M) {
          v0 = *reinterpret_cast<const bf16x8*>(&A[base]);
          if (g_k + 1 < kz1)
            v1 = *reinterpret_cast<const bf16x8*>(&A[base + M]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (block_row + t_c + e < M) {
              v0[e] = A[base + e];
              if (g_k + 1 < kz1) v1[e] = A[base + M + e];
            }
          }
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        __bf16 pair[2] = {v0[e], v1[e]};
        *reinterpret_cast<uint32_t*>(&As[t_c + e][t_k]) =
            *reinterpret_cast<const uint32_t*>(pair);
      }
    }
    // stage B^T: Bs[n][k] = B[k][n]
    {
      bf16x8 v0 = {}, v1 = {};
      if (g_k < kz1 && block_col + t_c < N) {
        const int64_t base = g_k * N + block_col + t_c;
        if (block_col + t_c + 7 < N) {
          v0 = *reinterpret_cast<const bf16x8*>(&B[base]);
          if (g_k + 1 < kz1)
            v1 = *reinterpret_cast<const bf16x8*>(&B[base + N]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (block_col + t_c + e < N) {
              v0[e] = B[base + e];
              if (g_k + 1 < kz1) v1[e] = B[base + N + e];
            }
          }
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        __bf16 pair[2] = {v0[e], v1[e]};
        *reinterpret_cast<uint32_t*>(&Bs[t_c + e][t_k]) =
            *reinterpret_cast<const uint32_t*>(pair);
      }
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < GBK; kk += 32) {
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + fi][kk + fk8]);
      bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          &As[wr * 32 + 16 + fi][kk + fk8]);
      bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + fi][kk + fk8]);
      bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          &Bs[wc * 32 + 16 + fi][kk + fk8]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0,
                                                          acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1,
                                                          acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0,
                                                          acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1,
                                                          acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // z-partials land in per-chunk planes C[z][M][N] (plain stores; a
  // torch sum(0) folds them — measured far cheaper than 128-way fp32
  // atomic contention on every output cell).
  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
  float* Cz = C + (int64_t)blockIdx.z * M * N;
#pragma unroll
  for (int fi2 = 0; fi2 < 2; ++fi2) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t row = block_row + wr * 32 + fi2 * 16 + c_row0 + r;
        const int64_t col = block_col + wc * 32 + fj * 16 + c_col;
        if (row < M && col < N) Cz[row * N + col] = acc[fi2][fj][r];
      }
    }
  }
  if (WITH_DB && blockIdx.y == 0) {
    // db[m] = sum_k A[k][m] for this block's m tile and k chunk: re-read
    // A from global (L2-hot from the staging pass just above), each
    // thread owning one (m, k-phase) stripe, then one LDS fold.
    __shared__ float dbs[GBM][4];
    const int m_l = tid & 63, phase = tid >> 6;
    float s = 0.f;
    if (block_row + m_l < M) {
      for (int64_t k = kz0 + phase; k < kz1 && k < Kb; k += 4)
        s += bf2f(A[k * M + block_row + m_l]);
    }
    dbs[m_l][phase] = s;
    __syncthreads();
    if (tid < GBM && block_row + tid < M)
      atomicAdd(&db[block_row + tid], dbs[tid][0] + dbs[tid][1] +
                                          dbs[tid][2] + dbs[tid][3]);
  }
}

// ---------------------------------------------------------------------------
// 128x128 / 8-wave variant of gemm_kt: halves the A/B re-read
// amplification (traffic ∝ tiles-in-other-dim) for the big dW shapes.
// Waves as 2(m)x4(n), each wave 64x32 = 4x2 16x16 fragments.
// ---------------------------------------------------------------------------
constexpr int KBM = 128, KBN = 128;

template <bool WITH_DB>
__global__ __launch_bounds__(512)
void gemm_kt128_kernel(const __bf16* __restrict__ A,
                       const __bf16* __restrict__ B,
                       float* __restrict__ C,
                       float* __restrict__ db,
                       int64_t Kb, int64_t M, int64_t N,
                       int64_t k_per_z) {
  __shared__ __bf16 As[KBM][GBK + LDP];   // As[m][k]
  __shared__ __bf16 Bs[KBN][GBK + LDP];   // Bs[n][k]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;               // 0..7
  const int wm = wave >> 2;                // 0..1 (64 rows each)
  const int wn = wave & 3;                 // 0..3 (32 cols each)

  const int64_t block_row = (int64_t)blockIdx.x * KBM;
  const int64_t block_col = (int64_t)blockIdx.y * KBN;
  const int64_t kz0 = (int64_t)blockIdx.z * k_per_z;
  const int64_t kz1 = std::min(kz0 + k_per_z, Kb);

  f32x4b acc[4][2] = {};

  // transposing stage: 512 threads = 16 col-groups x 32 k-pairs
  const int t_k = (tid >> 4) * 2;          // 0..62 step 2
  const int t_c = (tid & 15) * 8;          // 0..120 step 8

  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;

  for (int64_t k0 = kz0; k0 < kz1; k0 += GBK) {
    const int64_t g_k = k0 + t_k;
    {
      bf16x8 v0 = {}, v1 = {};
      if (g_k < kz1 && block_row + t_c < M) {
        const int64_t base = g_k * M + block_row + t_c;
        if (block_row + t_c + 7 < M) {
          v0 = *reinterpret_cast<const bf16x8*>(&A[base]);
          if (g_k + 1 < kz1)
            v1 = *reinterpret_cast<const bf16x8*>(&A[base + M]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (block_row + t_c + e < M) {
              v0[e] = A[base + e];
              if (g_k + 1 < kz1) v1[e] = A[base + M + e];
            }
          }
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        __bf16 pair[2] = {v0[e], v1[e]};
        *reinterpret_cast<uint32_t*>(&As[t_c + e][t_k]) =
            *reinterpret_cast<const uint32_t*>(pair);
      }
    }
    {
      bf16x8 v0 = {}, v1 = {};
      if (g_k < kz1 && block_col + t_c < N) {
        const int64_t base = g_k * N + block_col + t_c;
        if (block_col + t_c + 7 < N) {
          v0 = *reinterpret_cast<const bf16x8*>(&B[base]);
          if (g_k + 1 < kz1)
            v1 = *reinterpret_cast<const bf16x8*>(&B[base + N]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            if (block_col + t_c + e < N) {
              v0[e] = B[base + e];
              if (g_k + 1 < kz1) v1[e] = B[base + N + e];
            }
          }
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        __bf16 pair[2] = {v0[e], v1[e]};
        *reinterpret_cast<uint32_t*>(&Bs[t_c + e][t_k]) =
            *reinterpret_cast<const uint32_t*>(pair);
      }
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < GBK; kk += 32) {
      bf16x8 a[4], b[2];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *reinterpret_cast<const bf16x8*>(
            &As[wm * 64 + i * 16 + fi][kk + fk8]);
#pragma unroll
      for (int j = 0; j < 2; ++j)
        b[j] = *reinterpret_cast<const bf16x8*>(
            &Bs[wn * 32 + j * 16 + fi][kk + fk8]);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  const int c_col = lane & 15;
  const int c_row0 = (lane >> 4) * 4;
  float* Cz = C + (int64_t)blockIdx.z * M * N;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t row = block_row + wm * 64 + i * 16 + c_row0 + r;
        const int64_t col = block_col + wn * 32 + j * 16 + c_col;
        if (row < M && col < N) Cz[row * N + col] = acc[i][j][r];
      }
    }
  }
  if (WITH_DB && blockIdx.y == 0) {
    __shared__ float dbs[KBM][4];
    const int m_l = tid & 127, phase = tid >> 7;
    float s = 0.f;
    if (block_row + m_l < M) {
      for (int64_t k = kz0 + phase; k < kz1; k += 4)
        s += bf2f(A[k * M + block_row + m_l]);
    }
    dbs[m_l][phase] = s;
    __syncthreads();
    if (tid < KBM && block_row + tid < M)
      atomicAdd(&db[block_row + tid], dbs[tid][0] + dbs[tid][1] +
                                          dbs[tid][2] + dbs[tid][3]);
  }
}

// ---------------------------------------------------------------------------
// Single-tile fragment-layout selftests.
// ---------------------------------------------------------------------------
__global__ void selftest_16x16x32(const __bf16* A, const __bf16* B,
                                  float* C) {
  const int lane = threadIdx.x;
  bf16x8 af, bf_;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    af[e] = A[(lane & 15) * 32 + (lane >> 4) * 8 + e];       // A[i][k]
    bf_[e] = B[((lane >> 4) * 8 + e) * 16 + (lane & 15)];    // B[k][j]
  }
  f32x4b acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf_, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

}  // namespace

// C = act(A @ Bt^T + bias): A [M,K] bf16, Bt [N,K] bf16 (nn.Linear
// weight layout), bias fp32 or none.
torch::Tensor hip_gemm_bt_bf16(const torch::Tensor& A,
                               const torch::Tensor& Bt,
                               const c10::optional<torch::Tensor>& bias,
                               bool relu, bool out_fp32) {
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda(), "device tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                  Bt.scalar_type() == torch::kBFloat16,
              "bf16 inputs required");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2 && A.size(1) == Bt.size(1),
              "shape mismatch");
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  auto Ac = A.contiguous();
  auto Bc = Bt.contiguous();
  auto C = torch::empty({M, N}, A.options().dtype(
                                    out_fp32 ? torch::kFloat32
                                             : torch::kBFloat16));
  if (M == 0) return C;
  const float* bias_p = nullptr;
  torch::Tensor bias_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    bias_p = bias_f.data_ptr<float>();
  }
  // NOTE: a persistent-B variant (whole <=256-K weight tile staged once,
  // A streaming with one sync pair per m-tile) measured SLOWER: its
  // 68 KB LDS footprint halves occupancy (2 WGs/CU vs 4+) and the lost
  // latency hiding on the A stream outweighs the saved syncs/B-restage
  // (L0 fwd 167 us vs 112; flagship 661 vs 720 b/s).  Enable with
  // GLT_GEMM_PERSIST=1 for further experiments.
  if (K <= PKP && std::getenv("GLT_GEMM_PERSIST") != nullptr) {
    const int64_t m_t = (M + GBM - 1) / GBM;
    dim3 grid((uint32_t)std::min<int64_t>(m_t, 1024),
              (uint32_t)((N + GBN - 1) / GBN));
    auto* kfn = out_fp32
                    ? (relu ? gemm_bt_persist_kernel<true, false>
                            : gemm_bt_persist_kernel<false, false>)
                    : (relu ? gemm_bt_persist_kernel<true, true>
                            : gemm_bt_persist_kernel<false, true>);
    hipLaunchKernelGGL(kfn, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                       reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                       bias_p, C.data_ptr(), M, K, N);
    return C;
  }
  dim3 grid((uint32_t)((M + GBM - 1) / GBM),
            (uint32_t)((N + GBN - 1) / GBN));
  auto* kfn = out_fp32
                  ? (relu ? gemm_bt_kernel<true, false>
                          : gemm_bt_kernel<false, false>)
                  : (relu ? gemm_bt_kernel<true, true>
                          : gemm_bt_kernel<false, true>);
  hipLaunchKernelGGL(kfn, grid, dim3(256), 0, current_stream(),
                     reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                     reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                     bias_p, C.data_ptr(), M, K, N);
  return C;
}

// (dW, db) = (A^T @ B, colsum(A)): A [Kb,M] bf16, B [Kb,N] bf16;
// outputs fp32 (master-grad dtype).  with_db=false skips db.
std::tuple<torch::Tensor, c10::optional<torch::Tensor>> hip_gemm_kt_bf16(
    const torch::Tensor& A, const torch::Tensor& B, bool with_db) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda(), "device tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
                  B.scalar_type() == torch::kBFloat16,
              "bf16 inputs required");
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(0) == B.size(0),
              "shape mismatch");
  const int64_t Kb = A.size(0), M = A.size(1), N = B.size(1);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  // 128x128/8-wave tiles for big dW shapes (halves the A/B re-read
  // amplification); 64x64 for small/tail-heavy outputs
  const bool big = (M >= KBM && N >= 96 && Kb >= 4096);
  const int64_t bm = big ? KBM : GBM, bn = big ? KBN : GBN;
  const int64_t m_t = (M + bm - 1) / bm, n_t = (N + bn - 1) / bn;
  // split-K sized so the grid fills the 256 CUs a few times over
  int64_t z = 1;
  if (Kb > GBK) {
    const int64_t want_wg = big ? 512 : 1024;
    const int64_t want = (want_wg + m_t * n_t - 1) / (m_t * n_t);
    const int64_t max_z = (Kb + GBK - 1) / GBK;
    z = std::max<int64_t>(1, std::min(want, max_z));
  }
  const int64_t k_per_z = ((Kb + z - 1) / z + GBK - 1) / GBK * GBK;
  z = std::max<int64_t>(1, (Kb + k_per_z - 1) / k_per_z);
  c10::optional<torch::Tensor> db;
  float* db_p = nullptr;
  if (with_db) {
    db = torch::zeros({M}, A.options().dtype(torch::kFloat32));
    db_p = db->data_ptr<float>();
  }
  if (Kb == 0) {
    return {torch::zeros({M, N}, A.options().dtype(torch::kFloat32)), db};
  }
  // z>1: per-chunk partial planes + one sum(0) (no output atomics)
  auto P = torch::empty({z, M, N}, A.options().dtype(torch::kFloat32));
  dim3 grid((uint32_t)m_t, (uint32_t)n_t, (uint32_t)z);
  if (big) {
    auto* kfn = with_db ? gemm_kt128_kernel<true>
                        : gemm_kt128_kernel<false>;
    hipLaunchKernelGGL(kfn, grid, dim3(512), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                       reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                       P.data_ptr<float>(), db_p, Kb, M, N, k_per_z);
  } else {
    auto* kfn = with_db ? gemm_kt_kernel<true> : gemm_kt_kernel<false>;
    hipLaunchKernelGGL(kfn, grid, dim3(256), 0, current_stream(),
                       reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                       reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                       P.data_ptr<float>(), db_p, Kb, M, N, k_per_z);
  }
  auto C = z == 1 ? P.squeeze(0) : P.sum(0);
  return {C, db};
}

torch::Tensor hip_mfma_bf16_selftest(const torch::Tensor& A,
                                     const torch::Tensor& B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) &&
              B.sizes() == torch::IntArrayRef({32, 16}),
              "selftest is 16x32 @ 32x16");
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(selftest_16x16x32, dim3(1), dim3(64), 0,
                     current_stream(),
                     reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                     reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                     C.data_ptr<float>());
  return C;
}

}  // namespace glt
