/* Fused segment-mean aggregation for GNN convs (gfx950).
 *
 * The glt_amd sampler emits batch edges sorted by target local id, so the
 * per-target segments are contiguous: forward is one wave-per-target kernel
 * (no atomics, fused mean), backward scatters dY/deg into dX with fp32
 * atomics (same contention class as torch index_add_, but fused with the
 * degree division and vectorized).
 *
 * Replaces the 4-kernel torch sequence (bincount + index_select +
 * index_add_ + div) in SAGEConv (glt_amd/models/layers.py).
 *
 * dtype: templated over fp32 and bf16 inputs with fp32 accumulation —
 * the kernels are HBM-bound (see profiles/), so bf16 halves the roofline
 * traffic.  Backward always accumulates into an fp32 arena (bf16 atomics
 * lose too much precision for degree-weighted sums); the python wrapper
 * casts once at the end.
 */
#include <hip/hip_bf16.h>

#include "hip_common.h"
#include "../include/common.h"

namespace glt {

namespace {

// out[t, :] = mean_{e in [off[t], off[t+1])} x[col[e], :]
// One wave per target row; lanes cover the feature dim.
template <typename scalar_t>
__global__ void seg_mean_fwd_kernel(const scalar_t* __restrict__ x,
                                    const int64_t* __restrict__ col,
                                    const int64_t* __restrict__ off,
                                    int64_t n_tgt, int64_t feat,
                                    scalar_t* __restrict__ out) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    const float inv = e > s ? 1.0f / (float)(e - s) : 0.0f;
    if (feat <= 2 * kWave) {
      // lane owns channels f and f+64: register accumulators + row
      // addresses batched 8 deep (independent gather misses)
      const int64_t f0 = lane, f1 = lane + kWave;
      float a0 = 0.f, a1 = 0.f;
      for (int64_t k0 = s; k0 < e; k0 += 8) {
        const int nb = (int)((e - k0) < 8 ? (e - k0) : 8);
        int64_t cc[8];
#pragma unroll
        for (int q = 0; q < 8; ++q)
          cc[q] = (q < nb ? col[k0 + q] : col[k0]) * feat;
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          if (q < nb) {
            if (f0 < feat) a0 += static_cast<float>(x[cc[q] + f0]);
            if (f1 < feat) a1 += static_cast<float>(x[cc[q] + f1]);
          }
        }
      }
      if (f0 < feat) out[t * feat + f0] = static_cast<scalar_t>(a0 * inv);
      if (f1 < feat) out[t * feat + f1] = static_cast<scalar_t>(a1 * inv);
      continue;
    }
    for (int64_t f = lane; f < feat; f += kWave) {
      float acc = 0.f;
      for (int64_t k = s; k < e; ++k)
        acc += static_cast<float>(x[col[k] * feat + f]);
      out[t * feat + f] = static_cast<scalar_t>(acc * inv);
    }
  }
}

// bf16 fast path: each lane owns TWO adjacent channels (one 4-byte
// ushort2 load per row visit) — the scalar template's 2-byte gathers
// were measured at only 1.36x fp32 on the HBM-bound forward.
__global__ void seg_mean_fwd_bf162_kernel(const __hip_bfloat162* __restrict__ x,
                                          const int64_t* __restrict__ col,
                                          const int64_t* __restrict__ off,
                                          int64_t n_tgt, int64_t feat2,
                                          __hip_bfloat162* __restrict__ out) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    const float inv = e > s ? 1.0f / (float)(e - s) : 0.0f;
    if (feat2 <= kWave) {
      // one channel-pair per lane: accumulate in registers with the
      // row addresses batched 8 deep, so the gather issues 8
      // independent misses instead of a serial col->row->col chain
      const int64_t f = lane;
      const bool act = f < feat2;
      float2 acc = {0.f, 0.f};
      for (int64_t k0 = s; k0 < e; k0 += 8) {
        const int nb = (int)((e - k0) < 8 ? (e - k0) : 8);
        int64_t cc[8];
#pragma unroll
        for (int q = 0; q < 8; ++q)
          cc[q] = (q < nb ? col[k0 + q] : col[k0]) * feat2;
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          if (q < nb && act) {
            const float2 v = __bfloat1622float2(x[cc[q] + f]);
            acc.x += v.x;
            acc.y += v.y;
          }
        }
      }
      if (act)
        out[t * feat2 + f] =
            __float22bfloat162_rn({acc.x * inv, acc.y * inv});
      continue;
    }
    for (int64_t f = lane; f < feat2; f += kWave) {
      float2 acc = {0.f, 0.f};
      for (int64_t k = s; k < e; ++k) {
        const float2 v = __bfloat1622float2(x[col[k] * feat2 + f]);
        acc.x += v.x;
        acc.y += v.y;
      }
      out[t * feat2 + f] =
          __float22bfloat162_rn({acc.x * inv, acc.y * inv});
    }
  }
}

__global__ void seg_mean_cat_fwd_bf162_kernel(
    const __hip_bfloat162* __restrict__ x,
    const int64_t* __restrict__ col, const int64_t* __restrict__ off,
    int64_t n_tgt, int64_t feat2, __hip_bfloat162* __restrict__ out) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t ostride = 2 * feat2;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    const float inv = e > s ? 1.0f / (float)(e - s) : 0.0f;
    if (feat2 <= kWave) {
      const int64_t f = lane;
      const bool act = f < feat2;
      float2 acc = {0.f, 0.f};
      for (int64_t k0 = s; k0 < e; k0 += 8) {
        const int nb = (int)((e - k0) < 8 ? (e - k0) : 8);
        int64_t cc[8];
#pragma unroll
        for (int q = 0; q < 8; ++q)
          cc[q] = (q < nb ? col[k0 + q] : col[k0]) * feat2;
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          if (q < nb && act) {
            const float2 v = __bfloat1622float2(x[cc[q] + f]);
            acc.x += v.x;
            acc.y += v.y;
          }
        }
      }
      if (act) {
        out[t * ostride + f] =
            __float22bfloat162_rn({acc.x * inv, acc.y * inv});
        out[t * ostride + feat2 + f] = x[t * feat2 + f];
      }
      continue;
    }
    for (int64_t f = lane; f < feat2; f += kWave) {
      float2 acc = {0.f, 0.f};
      for (int64_t k = s; k < e; ++k) {
        const float2 v = __bfloat1622float2(x[col[k] * feat2 + f]);
        acc.x += v.x;
        acc.y += v.y;
      }
      out[t * ostride + f] =
          __float22bfloat162_rn({acc.x * inv, acc.y * inv});
      out[t * ostride + feat2 + f] = x[t * feat2 + f];
    }
  }
}

// dx[col[e], :] += dy[t, :] / deg(t)  (dx is always fp32)
template <typename scalar_t>
__global__ void seg_mean_bwd_kernel(const scalar_t* __restrict__ dy,
                                    const int64_t* __restrict__ col,
                                    const int64_t* __restrict__ off,
                                    int64_t n_tgt, int64_t feat,
                                    float* __restrict__ dx) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    if (e <= s) continue;
    const float inv = 1.0f / (float)(e - s);
    for (int64_t k = s; k < e; ++k) {
      const int64_t c = col[k];
      for (int64_t f = lane; f < feat; f += kWave) {
        atomicAdd(&dx[c * feat + f],
                  static_cast<float>(dy[t * feat + f]) * inv);
      }
    }
  }
}

// Fused [mean-agg | x_root] assembly: out[t, :F] = segment mean,
// out[t, F:] = x[t, :].  Saves the dim-1 torch.cat (two copyBuffer passes
// over [n, F] per layer, ~260 us/step in the flagship profile) and its
// launches; valid for the homo SAGE path where targets are the row prefix
// of x (the sampler's local-id invariant).
template <typename scalar_t>
__global__ void seg_mean_cat_fwd_kernel(const scalar_t* __restrict__ x,
                                        const int64_t* __restrict__ col,
                                        const int64_t* __restrict__ off,
                                        int64_t n_tgt, int64_t feat,
                                        scalar_t* __restrict__ out) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t ostride = 2 * feat;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    const float inv = e > s ? 1.0f / (float)(e - s) : 0.0f;
    if (feat <= 2 * kWave) {
      const int64_t f0 = lane, f1 = lane + kWave;
      float a0 = 0.f, a1 = 0.f;
      for (int64_t k0 = s; k0 < e; k0 += 8) {
        const int nb = (int)((e - k0) < 8 ? (e - k0) : 8);
        int64_t cc[8];
#pragma unroll
        for (int q = 0; q < 8; ++q)
          cc[q] = (q < nb ? col[k0 + q] : col[k0]) * feat;
#pragma unroll
        for (int q = 0; q < 8; ++q) {
          if (q < nb) {
            if (f0 < feat) a0 += static_cast<float>(x[cc[q] + f0]);
            if (f1 < feat) a1 += static_cast<float>(x[cc[q] + f1]);
          }
        }
      }
      if (f0 < feat) {
        out[t * ostride + f0] = static_cast<scalar_t>(a0 * inv);
        out[t * ostride + feat + f0] = x[t * feat + f0];
      }
      if (f1 < feat) {
        out[t * ostride + f1] = static_cast<scalar_t>(a1 * inv);
        out[t * ostride + feat + f1] = x[t * feat + f1];
      }
      continue;
    }
    for (int64_t f = lane; f < feat; f += kWave) {
      float acc = 0.f;
      for (int64_t k = s; k < e; ++k)
        acc += static_cast<float>(x[col[k] * feat + f]);
      out[t * ostride + f] = static_cast<scalar_t>(acc * inv);
      out[t * ostride + feat + f] = x[t * feat + f];
    }
  }
}

// dx[t, :] += dy[t, F:];  dx[col[e], :] += dy[t, :F] / deg(t)
template <typename scalar_t>
__global__ void seg_mean_cat_bwd_kernel(const scalar_t* __restrict__ dy,
                                        const int64_t* __restrict__ col,
                                        const int64_t* __restrict__ off,
                                        int64_t n_tgt, int64_t feat,
                                        float* __restrict__ dx) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t ostride = 2 * feat;
  for (int64_t t = wave; t < n_tgt; t += n_waves) {
    const int64_t s = off[t], e = off[t + 1];
    for (int64_t f = lane; f < feat; f += kWave)
      atomicAdd(&dx[t * feat + f],
                static_cast<float>(dy[t * ostride + feat + f]));
    if (e <= s) continue;
    const float inv = 1.0f / (float)(e - s);
    for (int64_t k = s; k < e; ++k) {
      const int64_t c = col[k];
      for (int64_t f = lane; f < feat; f += kWave)
        atomicAdd(&dx[c * feat + f],
                  static_cast<float>(dy[t * ostride + f]) * inv);
    }
  }
}

int wave_grid(int64_t rows) {
  const int64_t waves = std::min<int64_t>(rows, (int64_t)kMaxBlocks * 4);
  return (int)std::min<int64_t>((waves * kWave + kBlock - 1) / kBlock,
                                kMaxBlocks);
}

#define GLT_DISPATCH_SEG(TYPE, NAME, ...)                          \
  [&] {                                                            \
    switch (TYPE) {                                                \
      case torch::kFloat32: {                                      \
        using scalar_t = float;                                    \
        return __VA_ARGS__();                                      \
      }                                                            \
      case torch::kBFloat16: {                                     \
        using scalar_t = c10::BFloat16;                            \
        return __VA_ARGS__();                                      \
      }                                                            \
      default:                                                     \
        TORCH_CHECK(false, NAME ": unsupported dtype (fp32/bf16)");\
    }                                                              \
  }()

}  // namespace

torch::Tensor hip_segment_mean_fwd(const torch::Tensor& x,
                                   const torch::Tensor& col,
                                   const torch::Tensor& offsets,
                                   int64_t n_tgt) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(),
              "segment_mean: x must be contiguous on device");
  const int64_t feat = x.size(1);
  auto out = torch::empty({n_tgt, feat}, x.options());
  if (n_tgt > 0) {
    if (x.scalar_type() == torch::kBFloat16 && feat % 2 == 0) {
      hipLaunchKernelGGL(seg_mean_fwd_bf162_kernel,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(),
                         reinterpret_cast<const __hip_bfloat162*>(
                             x.data_ptr()),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat / 2,
                         reinterpret_cast<__hip_bfloat162*>(
                             out.data_ptr()));
      return out;
    }
    GLT_DISPATCH_SEG(x.scalar_type(), "segment_mean_fwd", [&] {
      hipLaunchKernelGGL(seg_mean_fwd_kernel<scalar_t>,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(), x.data_ptr<scalar_t>(),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat,
                         out.data_ptr<scalar_t>());
    });
  }
  return out;
}

torch::Tensor hip_segment_mean_cat_fwd(const torch::Tensor& x,
                                       const torch::Tensor& col,
                                       const torch::Tensor& offsets,
                                       int64_t n_tgt) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(),
              "segment_mean_cat: x must be contiguous on device");
  TORCH_CHECK(x.size(0) >= n_tgt,
              "segment_mean_cat: targets must be a row prefix of x");
  const int64_t feat = x.size(1);
  auto out = torch::empty({n_tgt, 2 * feat}, x.options());
  if (n_tgt > 0) {
    if (x.scalar_type() == torch::kBFloat16 && feat % 2 == 0) {
      hipLaunchKernelGGL(seg_mean_cat_fwd_bf162_kernel,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(),
                         reinterpret_cast<const __hip_bfloat162*>(
                             x.data_ptr()),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat / 2,
                         reinterpret_cast<__hip_bfloat162*>(
                             out.data_ptr()));
      return out;
    }
    GLT_DISPATCH_SEG(x.scalar_type(), "segment_mean_cat_fwd", [&] {
      hipLaunchKernelGGL(seg_mean_cat_fwd_kernel<scalar_t>,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(), x.data_ptr<scalar_t>(),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat,
                         out.data_ptr<scalar_t>());
    });
  }
  return out;
}

// Returns fp32 regardless of dy dtype (fp32 atomic accumulation arena);
// the python wrapper casts to dy.dtype once.
torch::Tensor hip_segment_mean_cat_bwd(const torch::Tensor& dy,
                                       const torch::Tensor& col,
                                       const torch::Tensor& offsets,
                                       int64_t n_src) {
  const int64_t n_tgt = dy.size(0);
  const int64_t feat = dy.size(1) / 2;
  auto dx = torch::zeros({n_src, feat}, dy.options().dtype(torch::kFloat32));
  if (n_tgt > 0) {
    auto dyc = dy.contiguous();
    GLT_DISPATCH_SEG(dy.scalar_type(), "segment_mean_cat_bwd", [&] {
      hipLaunchKernelGGL(seg_mean_cat_bwd_kernel<scalar_t>,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(), dyc.data_ptr<scalar_t>(),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat,
                         dx.data_ptr<float>());
    });
  }
  return dx;
}

torch::Tensor hip_segment_mean_bwd(const torch::Tensor& dy,
                                   const torch::Tensor& col,
                                   const torch::Tensor& offsets,
                                   int64_t n_src) {
  const int64_t n_tgt = dy.size(0);
  const int64_t feat = dy.size(1);
  auto dx = torch::zeros({n_src, feat}, dy.options().dtype(torch::kFloat32));
  if (n_tgt > 0) {
    auto dyc = dy.contiguous();
    GLT_DISPATCH_SEG(dy.scalar_type(), "segment_mean_bwd", [&] {
      hipLaunchKernelGGL(seg_mean_bwd_kernel<scalar_t>,
                         dim3(wave_grid(n_tgt)), dim3(kBlock), 0,
                         current_stream(), dyc.data_ptr<scalar_t>(),
                         col.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, feat,
                         dx.data_ptr<float>());
    });
  }
  return dx;
}

}  // namespace glt
