/* Fused GAT edge-softmax + weighted aggregation (gfx950).
 *
 * Replaces the whole torch chain per GATConv — including the per-node
 * attention-logit precompute (h*att).sum(-1) — with one forward and one
 * backward kernel.  The logits are recomputed per edge from rows the
 * aggregation loads anyway (a 6-step wave reduction on data already in
 * registers), which deletes ~8 small launches per relation per layer in
 * RGAT's launch-bound regime.
 *
 * Layout: edges sorted by target (glt_amd batch invariant); one wave per
 * (target, head), lanes over the channel dim; the softmax runs online
 * (running max / rescaled sum) along the target's edge segment, so scores
 * are never materialized.
 *
 *   s_e   = leaky_relu(<h_tgt[t,h,:], att_dst[h,:]> +
 *                      <h_src[src_e,h,:], att_src[h,:]>)
 *   p_e   = exp(s_e - m_t) / Z_t
 *   out_t = sum_e p_e * h_src[src_e,h,:]
 *
 * Backward recomputes s_e from the saved (m, Z) statistics:
 *   dh_src[src_e]  += p_e * dout_t + ds_e * att_src[h]   (fp32 atomics)
 *   ds_e            = p_e * (<dout_t, h_src_e> - <dout_t, out_t>)
 *                     * leaky'(s_pre_e)
 *   datt_src[h]    += ds_e * h_src[src_e];  dh_tgt[t] += (sum ds) * att_dst
 *   datt_dst[h]    += (sum_e ds_e) * h_tgt[t]
 */
#include <hip/hip_bf16.h>

#include "hip_common.h"
#include "../include/common.h"

namespace glt {

namespace {

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int sft = kWave / 2; sft > 0; sft >>= 1) v += __shfl_down(v, sft);
  return __shfl(v, 0);
}

template <typename T>  // h/out dtype: float or bf16 (math stays fp32)
__global__ void gat_fused_fwd_kernel(
    const T* __restrict__ h_tgt,          // [Nt.., H, C]
    const T* __restrict__ h_src,          // [Ns, H, C]
    const float* __restrict__ att_src,    // [H, C]
    const float* __restrict__ att_dst,    // [H, C]
    const int64_t* __restrict__ src,      // [E]
    const int64_t* __restrict__ offsets,  // [Nt+1]
    int64_t n_tgt, int64_t H, int64_t C, float slope,
    T* __restrict__ out,                  // [Nt, H, C]
    float* __restrict__ m_out,            // [Nt, H]
    float* __restrict__ z_out,            // [Nt, H]
    float* __restrict__ spre_out) {       // [E, H] pre-activation logits
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t total = n_tgt * H;
  for (int64_t w = wave; w < total; w += n_waves) {
    const int64_t t = w / H;
    const int64_t h = w - t * H;
    const int64_t s0 = offsets[t], s1 = offsets[t + 1];
    const float* ats = att_src + h * C;
    const float* atd = att_dst + h * C;
    const float as0 = lane < C ? ats[lane] : 0.f;
    const float as1 = kWave + lane < C ? ats[kWave + lane] : 0.f;
    const T* tv = h_tgt + (t * H + h) * C;
    const float ad = wave_sum(
        (lane < C ? (float)tv[lane] * atd[lane] : 0.f) +
        (kWave + lane < C ? (float)tv[kWave + lane] * atd[kWave + lane]
                          : 0.f));
    float m = -1e30f, Z = 0.f;
    float acc0 = 0.f, acc1 = 0.f;  // lanes cover C (up to 2 passes)
    // 4-edge load batching (see backward): the online-softmax chain
    // stays serial but its global loads overlap.
    for (int64_t e = s0; e < s1; e += 4) {
      const int nb = (int)((s1 - e) < 4 ? (s1 - e) : 4);
      int64_t sn[4];
      float hh0[4], hh1[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) sn[q] = q < nb ? src[e + q] : sn[0];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const T* hv = h_src + (sn[q] * H + h) * C;
        hh0[q] = lane < C ? (float)hv[lane] : 0.f;
        hh1[q] = kWave + lane < C ? (float)hv[kWave + lane] : 0.f;
      }
      float sc4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) sc4[q] = hh0[q] * as0 + hh1[q] * as1;
#pragma unroll
      for (int sft = kWave / 2; sft > 0; sft >>= 1) {
#pragma unroll
        for (int q = 0; q < 4; ++q)
          sc4[q] += __shfl_down(sc4[q], sft);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q >= nb) continue;
        float sv = ad + __shfl(sc4[q], 0);
        if (lane == 0) spre_out[(e + q) * H + h] = sv;
        sv = sv > 0.f ? sv : sv * slope;
        float scale = 1.f;
        float p;
        if (sv > m) {
          scale = __expf(m - sv);
          p = 1.f;
          m = sv;
        } else {
          p = __expf(sv - m);
        }
        Z = Z * scale + p;
        acc0 = acc0 * scale + p * hh0[q];
        acc1 = acc1 * scale + p * hh1[q];
      }
    }
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    T* ov = out + (t * H + h) * C;
    if (lane < C) ov[lane] = (T)(acc0 * inv);
    if (kWave + lane < C) ov[kWave + lane] = (T)(acc1 * inv);
    if (lane == 0) {
      m_out[t * H + h] = m;
      z_out[t * H + h] = Z;
    }
  }
}

template <typename T>
__global__ void gat_fused_bwd_kernel(
    const T* __restrict__ h_tgt, const T* __restrict__ h_src,
    const float* __restrict__ att_src, const float* __restrict__ att_dst,
    const int64_t* __restrict__ src, const int64_t* __restrict__ offsets,
    const T* __restrict__ out, const float* __restrict__ m_in,
    const float* __restrict__ z_in, const float* __restrict__ spre_in,
    const T* __restrict__ dout,
    int64_t n_tgt, int64_t H, int64_t C, float slope, int64_t S,
    float* __restrict__ dh_tgt, float* __restrict__ dh_src,
    float* __restrict__ datt_src, float* __restrict__ datt_dst) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  // S-way segment split: small-target layers (hop 0: 512 targets x 4
  // heads = 2048 waves) cannot fill 256 CUs with (t, h) waves alone —
  // every per-edge term in the backward is independent given the saved
  // (m, Z, out) statistics, so S sub-waves each take a slice of the
  // segment and flush their partial reductions atomically.
  const int64_t total = n_tgt * H * S;
  for (int64_t w = wave; w < total; w += n_waves) {
    const int64_t t = w / (H * S);
    const int64_t rem = w - t * H * S;
    const int64_t h = rem / S;
    const int64_t q = rem - h * S;
    const int64_t f0 = offsets[t], f1 = offsets[t + 1];
    if (f1 <= f0) continue;
    const int64_t per = (f1 - f0 + S - 1) / S;
    const int64_t s0 = f0 + q * per;
    const int64_t s1 = s0 + per < f1 ? s0 + per : f1;
    if (s1 <= s0) continue;
    const float* ats = att_src + h * C;
    const float* atd = att_dst + h * C;
    const float as0 = lane < C ? ats[lane] : 0.f;
    const float as1 = kWave + lane < C ? ats[kWave + lane] : 0.f;
    const T* tv = h_tgt + (t * H + h) * C;
    const float t0 = lane < C ? (float)tv[lane] : 0.f;
    const float t1 = kWave + lane < C ? (float)tv[kWave + lane] : 0.f;
    const float ad0 = lane < C ? atd[lane] : 0.f;
    const float ad1 = kWave + lane < C ? atd[kWave + lane] : 0.f;
    const float m = m_in[t * H + h];
    const float Z = z_in[t * H + h];
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    const T* dv = dout + (t * H + h) * C;
    const T* ov = out + (t * H + h) * C;
    const float d0 = lane < C ? (float)dv[lane] : 0.f;
    const float d1 = kWave + lane < C ? (float)dv[kWave + lane] : 0.f;
    const float dot_o = wave_sum(
        (lane < C ? d0 * (float)ov[lane] : 0.f) +
        (kWave + lane < C ? d1 * (float)ov[kWave + lane] : 0.f));
    float dad_acc = 0.f;        // sum of ds over the segment
    float das0 = 0.f, das1 = 0.f;  // datt_src accumulator (per lane)
    // 4-edge software pipeline: the segment loop was a serial chain of
    // dependent global loads (src[e] -> h_src row -> math), ~10 L2-miss
    // latencies back to back per wave; batching the loads of 4 edges
    // gives the memory system 4 independent misses in flight and lets
    // the 4 wave-reductions interleave.
    for (int64_t e = s0; e < s1; e += 4) {
      const int nb = (int)((s1 - e) < 4 ? (s1 - e) : 4);
      int64_t sn[4];
      float hh0[4], hh1[4], spre4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        sn[q] = q < nb ? src[e + q] : sn[0];
        if (q < nb) spre4[q] = spre_in[(e + q) * H + h];
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const T* hv = h_src + (sn[q] * H + h) * C;
        hh0[q] = lane < C ? (float)hv[lane] : 0.f;
        hh1[q] = kWave + lane < C ? (float)hv[kWave + lane] : 0.f;
      }
      float dot_h4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q)
        dot_h4[q] = hh0[q] * d0 + hh1[q] * d1;
      // interleaved wave reductions (independent shfl chains)
#pragma unroll
      for (int sft = kWave / 2; sft > 0; sft >>= 1) {
#pragma unroll
        for (int q = 0; q < 4; ++q)
          dot_h4[q] += __shfl_down(dot_h4[q], sft);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q >= nb) continue;
        const float s_pre = spre4[q];
        const float sa = s_pre > 0.f ? s_pre : s_pre * slope;
        const float p = __expf(sa - m) * inv;
        float ds = p * (__shfl(dot_h4[q], 0) - dot_o);
        ds *= (s_pre > 0.f ? 1.f : slope);
        float* dhv = dh_src + (sn[q] * H + h) * C;
        if (lane < C) atomicAdd(&dhv[lane], p * d0 + ds * as0);
        if (kWave + lane < C)
          atomicAdd(&dhv[kWave + lane], p * d1 + ds * as1);
        das0 += ds * hh0[q];
        das1 += ds * hh1[q];
        dad_acc += ds;
      }
    }
    // one atomic flush per wave (not per edge)
    float* dtv = dh_tgt + (t * H + h) * C;
    if (lane < C) {
      atomicAdd(&dtv[lane], dad_acc * ad0);
      atomicAdd(&datt_src[h * C + lane], das0);
      atomicAdd(&datt_dst[h * C + lane], dad_acc * t0);
    }
    if (kWave + lane < C) {
      atomicAdd(&dtv[kWave + lane], dad_acc * ad1);
      atomicAdd(&datt_src[h * C + kWave + lane], das1);
      atomicAdd(&datt_dst[h * C + kWave + lane], dad_acc * t1);
    }
  }
}

int gat_grid(int64_t waves_needed) {
  const int64_t w = std::min<int64_t>(waves_needed, (int64_t)kMaxBlocks * 4);
  return (int)std::min<int64_t>((w * kWave + kBlock - 1) / kBlock,
                                kMaxBlocks);
}

}  // namespace

namespace {

// ---------------------------------------------------------------------------
// Multi-relation fused GAT: ONE launch per hetero layer.  RGAT's step is
// launch-bound (~285 kernels/step measured); per-relation attention
// kernels, their offsets glue, the .contiguous() slice copies and the
// slice-grad zeros+adds all collapse into one fwd + one bwd call over a
// packed relation table.  h rows are accessed STRIDED (base + n*stride
// + h*C + c), so the per-type batched-GEMM output [N_t, R_t*H*C] is
// consumed in place and dh accumulates atomically straight into one
// per-type arena.
// ---------------------------------------------------------------------------
constexpr int kMaxRel = 8;

template <typename T>
struct GatPack {
  const T* h_tgt[kMaxRel];
  const T* h_src[kMaxRel];
  int64_t tgt_stride[kMaxRel];
  int64_t src_stride[kMaxRel];
  const float* att_src[kMaxRel];
  const float* att_dst[kMaxRel];
  const int64_t* src[kMaxRel];
  const int64_t* off[kMaxRel];
  T* out[kMaxRel];              // contiguous [n_tgt_r, H, C] slices
  float* m[kMaxRel];
  float* z[kMaxRel];
  float* spre[kMaxRel];
  float* dh_tgt[kMaxRel];       // bwd only (fp32 arenas, strided)
  float* dh_src[kMaxRel];
  int64_t dht_stride[kMaxRel];
  int64_t dhs_stride[kMaxRel];
  const T* dout[kMaxRel];       // bwd only (contiguous slices)
  float* datt_src[kMaxRel];
  float* datt_dst[kMaxRel];
  const float* bias[kMaxRel];   // fwd: optional [H*C] bias per relation
  float* dbias[kMaxRel];        // bwd: its grad (atomic colsum of dout)
  int64_t n_tgt[kMaxRel];
  int64_t cum[kMaxRel + 1];     // cumulative n_tgt (work-item decode)
  int n_rel;
};

template <typename T>
__global__ void gat_multi_fwd_kernel(GatPack<T> P, int64_t H, int64_t C,
                                     float slope, int64_t total) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t w = wave; w < total * H; w += n_waves) {
    const int64_t tg = w / H;
    const int64_t h = w - tg * H;
    int r = 0;
    while (r + 1 < P.n_rel && tg >= P.cum[r + 1]) ++r;
    // r is wave-uniform by construction (w is the wave index); telling
    // the compiler so keeps every P.*[r] select scalar instead of
    // per-lane kernarg fetches
    r = __builtin_amdgcn_readfirstlane(r);
    const int64_t t = tg - P.cum[r];
    const int64_t s0 = P.off[r][t], s1 = P.off[r][t + 1];
    const float* ats = P.att_src[r] + h * C;
    const float* atd = P.att_dst[r] + h * C;
    const float as0 = lane < C ? ats[lane] : 0.f;
    const float as1 = kWave + lane < C ? ats[kWave + lane] : 0.f;
    const T* tv = P.h_tgt[r] + t * P.tgt_stride[r] + h * C;
    const float ad = wave_sum(
        (lane < C ? (float)tv[lane] * atd[lane] : 0.f) +
        (kWave + lane < C ? (float)tv[kWave + lane] * atd[kWave + lane]
                          : 0.f));
    float m = -1e30f, Z = 0.f;
    float acc0 = 0.f, acc1 = 0.f;
    const int64_t hstride = P.src_stride[r];
    const T* hbase = P.h_src[r] + h * C;
    for (int64_t e = s0; e < s1; e += 4) {
      const int nb = (int)((s1 - e) < 4 ? (s1 - e) : 4);
      int64_t sn[4];
      float hh0[4], hh1[4];
#pragma unroll
      for (int q = 0; q < 4; ++q)
        sn[q] = q < nb ? P.src[r][e + q] : sn[0];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const T* hv = hbase + sn[q] * hstride;
        hh0[q] = lane < C ? (float)hv[lane] : 0.f;
        hh1[q] = kWave + lane < C ? (float)hv[kWave + lane] : 0.f;
      }
      float sc4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) sc4[q] = hh0[q] * as0 + hh1[q] * as1;
#pragma unroll
      for (int sft = kWave / 2; sft > 0; sft >>= 1) {
#pragma unroll
        for (int q = 0; q < 4; ++q)
          sc4[q] += __shfl_down(sc4[q], sft);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q >= nb) continue;
        float sv = ad + __shfl(sc4[q], 0);
        if (lane == 0) P.spre[r][(e + q) * H + h] = sv;
        sv = sv > 0.f ? sv : sv * slope;
        float scale = 1.f, pp;
        if (sv > m) {
          scale = __expf(m - sv);
          pp = 1.f;
          m = sv;
        } else {
          pp = __expf(sv - m);
        }
        Z = Z * scale + pp;
        acc0 = acc0 * scale + pp * hh0[q];
        acc1 = acc1 * scale + pp * hh1[q];
      }
    }
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    T* ov = P.out[r] + (t * H + h) * C;
    const float* bv = P.bias[r];
    if (lane < C)
      ov[lane] = (T)(acc0 * inv + (bv ? bv[h * C + lane] : 0.f));
    if (kWave + lane < C)
      ov[kWave + lane] =
          (T)(acc1 * inv + (bv ? bv[h * C + kWave + lane] : 0.f));
    if (lane == 0) {
      P.m[r][t * H + h] = m;
      P.z[r][t * H + h] = Z;
    }
  }
}

template <typename T>
__global__ void gat_multi_bwd_kernel(GatPack<T> P, int64_t H, int64_t C,
                                     float slope, int64_t S,
                                     int64_t total) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  for (int64_t w = wave; w < total * H * S; w += n_waves) {
    const int64_t tg = w / (H * S);
    const int64_t rem = w - tg * H * S;
    const int64_t h = rem / S;
    const int64_t q0 = rem - h * S;
    int r = 0;
    while (r + 1 < P.n_rel && tg >= P.cum[r + 1]) ++r;
    r = __builtin_amdgcn_readfirstlane(r);
    const int64_t t = tg - P.cum[r];
    const int64_t f0 = P.off[r][t], f1 = P.off[r][t + 1];
    if (f1 <= f0) continue;
    const int64_t per = (f1 - f0 + S - 1) / S;
    const int64_t s0 = f0 + q0 * per;
    const int64_t s1 = s0 + per < f1 ? s0 + per : f1;
    if (s1 <= s0) continue;
    const float* ats = P.att_src[r] + h * C;
    const float* atd = P.att_dst[r] + h * C;
    const float as0 = lane < C ? ats[lane] : 0.f;
    const float as1 = kWave + lane < C ? ats[kWave + lane] : 0.f;
    const T* tv = P.h_tgt[r] + t * P.tgt_stride[r] + h * C;
    const float t0 = lane < C ? (float)tv[lane] : 0.f;
    const float t1 = kWave + lane < C ? (float)tv[kWave + lane] : 0.f;
    const float ad0 = lane < C ? atd[lane] : 0.f;
    const float ad1 = kWave + lane < C ? atd[kWave + lane] : 0.f;
    const float m = P.m[r][t * H + h];
    const float Z = P.z[r][t * H + h];
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    const T* dv = P.dout[r] + (t * H + h) * C;
    const T* ov = P.out[r] + (t * H + h) * C;
    const float d0 = lane < C ? (float)dv[lane] : 0.f;
    const float d1 = kWave + lane < C ? (float)dv[kWave + lane] : 0.f;
    const float dot_o = wave_sum(
        (lane < C ? d0 * (float)ov[lane] : 0.f) +
        (kWave + lane < C ? d1 * (float)ov[kWave + lane] : 0.f));
    float dad_acc = 0.f;
    float das0 = 0.f, das1 = 0.f;
    const int64_t hstride = P.src_stride[r];
    const int64_t dstride = P.dhs_stride[r];
    const T* hbase = P.h_src[r] + h * C;
    float* dhbase = P.dh_src[r] + h * C;
    for (int64_t e = s0; e < s1; e += 4) {
      const int nb = (int)((s1 - e) < 4 ? (s1 - e) : 4);
      int64_t sn[4];
      float hh0[4], hh1[4], spre4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        sn[q] = q < nb ? P.src[r][e + q] : sn[0];
        if (q < nb) spre4[q] = P.spre[r][(e + q) * H + h];
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const T* hv = hbase + sn[q] * hstride;
        hh0[q] = lane < C ? (float)hv[lane] : 0.f;
        hh1[q] = kWave + lane < C ? (float)hv[kWave + lane] : 0.f;
      }
      float dot_h4[4];
#pragma unroll
      for (int q = 0; q < 4; ++q)
        dot_h4[q] = hh0[q] * d0 + hh1[q] * d1;
#pragma unroll
      for (int sft = kWave / 2; sft > 0; sft >>= 1) {
#pragma unroll
        for (int q = 0; q < 4; ++q)
          dot_h4[q] += __shfl_down(dot_h4[q], sft);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (q >= nb) continue;
        const float s_pre = spre4[q];
        const float sa = s_pre > 0.f ? s_pre : s_pre * slope;
        const float pp = __expf(sa - m) * inv;
        float ds = pp * (__shfl(dot_h4[q], 0) - dot_o);
        ds *= (s_pre > 0.f ? 1.f : slope);
        float* dhv = dhbase + sn[q] * dstride;
        if (lane < C) atomicAdd(&dhv[lane], pp * d0 + ds * as0);
        if (kWave + lane < C)
          atomicAdd(&dhv[kWave + lane], pp * d1 + ds * as1);
        das0 += ds * hh0[q];
        das1 += ds * hh1[q];
        dad_acc += ds;
      }
    }
    float* dtv = P.dh_tgt[r] + t * P.dht_stride[r] + h * C;
    if (lane < C) {
      atomicAdd(&dtv[lane], dad_acc * ad0);
      atomicAdd(&P.datt_src[r][h * C + lane], das0);
      atomicAdd(&P.datt_dst[r][h * C + lane], dad_acc * t0);
    }
    if (kWave + lane < C) {
      atomicAdd(&dtv[kWave + lane], dad_acc * ad1);
      atomicAdd(&P.datt_src[r][h * C + kWave + lane], das1);
      atomicAdd(&P.datt_dst[r][h * C + kWave + lane], dad_acc * t1);
    }
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_fused_fwd(
    const torch::Tensor& h_tgt, const torch::Tensor& h_src,
    const torch::Tensor& att_src, const torch::Tensor& att_dst,
    const torch::Tensor& src, const torch::Tensor& offsets, double slope) {
  const int64_t n_tgt = offsets.numel() - 1;
  const int64_t H = h_src.size(1), C = h_src.size(2);
  const int64_t E = src.numel();
  TORCH_CHECK(C <= 2 * kWave, "GAT fused kernel supports C <= 128");
  TORCH_CHECK(h_tgt.size(0) >= n_tgt, "h_tgt must cover all targets");
  const bool bf16 = h_src.scalar_type() == torch::kBFloat16;
  auto fopt = h_src.options().dtype(torch::kFloat32);
  auto out = torch::empty({n_tgt, H, C}, h_src.options());
  auto m = torch::empty({n_tgt, H}, fopt);
  auto z = torch::empty({n_tgt, H}, fopt);
  auto spre = torch::empty({E, H}, fopt);
  auto as = att_src.to(torch::kFloat32).contiguous();
  auto ad = att_dst.to(torch::kFloat32).contiguous();
  if (n_tgt > 0) {
    if (bf16) {
      hipLaunchKernelGGL(gat_fused_fwd_kernel<__bf16>,
                         dim3(gat_grid(n_tgt * H)),
                         dim3(kBlock), 0, current_stream(),
                         (const __bf16*)h_tgt.data_ptr(),
                         (const __bf16*)h_src.data_ptr(),
                         as.data_ptr<float>(), ad.data_ptr<float>(),
                         src.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, H, C,
                         (float)slope, (__bf16*)out.data_ptr(),
                         m.data_ptr<float>(), z.data_ptr<float>(),
                         spre.data_ptr<float>());
    } else {
      hipLaunchKernelGGL(gat_fused_fwd_kernel<float>,
                         dim3(gat_grid(n_tgt * H)),
                         dim3(kBlock), 0, current_stream(),
                         h_tgt.data_ptr<float>(), h_src.data_ptr<float>(),
                         as.data_ptr<float>(), ad.data_ptr<float>(),
                         src.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(), n_tgt, H, C,
                         (float)slope, out.data_ptr<float>(),
                         m.data_ptr<float>(), z.data_ptr<float>(),
                         spre.data_ptr<float>());
    }
  }
  return {out, m, z, spre};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_fused_bwd(const torch::Tensor& h_tgt, const torch::Tensor& h_src,
                  const torch::Tensor& att_src,
                  const torch::Tensor& att_dst, const torch::Tensor& src,
                  const torch::Tensor& offsets, const torch::Tensor& out,
                  const torch::Tensor& m, const torch::Tensor& z,
                  const torch::Tensor& spre, const torch::Tensor& dout,
                  double slope) {
  const int64_t n_tgt = offsets.numel() - 1;
  const int64_t H = h_src.size(1), C = h_src.size(2);
  const bool bf16 = h_src.scalar_type() == torch::kBFloat16;
  auto fopt = h_src.options().dtype(torch::kFloat32);
  // grad arenas are always fp32 (atomic accumulation); the python
  // wrapper casts dh back to the h dtype once
  auto dh_tgt = torch::zeros(h_tgt.sizes(), fopt);
  auto dh_src = torch::zeros(h_src.sizes(), fopt);
  auto das = torch::zeros(att_src.sizes(), fopt);
  auto dad = torch::zeros(att_dst.sizes(), fopt);
  auto as = att_src.to(torch::kFloat32).contiguous();
  auto ad = att_dst.to(torch::kFloat32).contiguous();
  if (n_tgt > 0) {
    // fill the chip: split segments when (t, h) waves alone are few
    int64_t S = 32768 / std::max<int64_t>(n_tgt * H, 1);
    S = std::max<int64_t>(1, std::min<int64_t>(S, 8));
    auto dc = dout.contiguous();
    if (bf16) {
      hipLaunchKernelGGL(gat_fused_bwd_kernel<__bf16>,
                         dim3(gat_grid(n_tgt * H * S)),
                         dim3(kBlock), 0, current_stream(),
                         (const __bf16*)h_tgt.data_ptr(),
                         (const __bf16*)h_src.data_ptr(),
                         as.data_ptr<float>(), ad.data_ptr<float>(),
                         src.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(),
                         (const __bf16*)out.data_ptr(),
                         m.data_ptr<float>(),
                         z.data_ptr<float>(), spre.data_ptr<float>(),
                         (const __bf16*)dc.data_ptr(), n_tgt, H, C,
                         (float)slope, S, dh_tgt.data_ptr<float>(),
                         dh_src.data_ptr<float>(), das.data_ptr<float>(),
                         dad.data_ptr<float>());
    } else {
      hipLaunchKernelGGL(gat_fused_bwd_kernel<float>,
                         dim3(gat_grid(n_tgt * H * S)),
                         dim3(kBlock), 0, current_stream(),
                         h_tgt.data_ptr<float>(), h_src.data_ptr<float>(),
                         as.data_ptr<float>(), ad.data_ptr<float>(),
                         src.data_ptr<int64_t>(),
                         offsets.data_ptr<int64_t>(),
                         out.data_ptr<float>(), m.data_ptr<float>(),
                         z.data_ptr<float>(), spre.data_ptr<float>(),
                         dc.data_ptr<float>(), n_tgt, H, C,
                         (float)slope, S, dh_tgt.data_ptr<float>(),
                         dh_src.data_ptr<float>(), das.data_ptr<float>(),
                         dad.data_ptr<float>());
    }
  }
  return {dh_tgt, dh_src, das, dad};
}


// ---------------------------------------------------------------------------
// Multi-relation hosts.  h views may be STRIDED row-wise (slices of the
// per-type batched projection [N_t, R_t*H*C]); dims 1,2 must be dense.
// ---------------------------------------------------------------------------
namespace {

template <typename T>
void fill_pack_common(GatPack<T>& P,
                      const std::vector<torch::Tensor>& h_tgt,
                      const std::vector<torch::Tensor>& h_src,
                      const std::vector<torch::Tensor>& att_src,
                      const std::vector<torch::Tensor>& att_dst,
                      const std::vector<torch::Tensor>& src,
                      const std::vector<torch::Tensor>& off,
                      int64_t H, int64_t C) {
  const int R = (int)h_tgt.size();
  P.n_rel = R;
  P.cum[0] = 0;
  for (int r = 0; r < R; ++r) {
    TORCH_CHECK(h_tgt[r].stride(1) == C && h_tgt[r].stride(2) == 1 &&
                    h_src[r].stride(1) == C && h_src[r].stride(2) == 1,
                "gat_multi: inner dims must be dense");
    P.h_tgt[r] = reinterpret_cast<const T*>(h_tgt[r].data_ptr());
    P.h_src[r] = reinterpret_cast<const T*>(h_src[r].data_ptr());
    P.tgt_stride[r] = h_tgt[r].stride(0);
    P.src_stride[r] = h_src[r].stride(0);
    P.att_src[r] = att_src[r].data_ptr<float>();
    P.att_dst[r] = att_dst[r].data_ptr<float>();
    P.src[r] = src[r].data_ptr<int64_t>();
    P.off[r] = off[r].data_ptr<int64_t>();
    P.n_tgt[r] = off[r].numel() - 1;
    P.cum[r + 1] = P.cum[r] + P.n_tgt[r];
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
hip_gat_multi_fwd(const std::vector<torch::Tensor>& h_tgt,
                  const std::vector<torch::Tensor>& h_src,
                  const std::vector<torch::Tensor>& att_src,
                  const std::vector<torch::Tensor>& att_dst,
                  const std::vector<torch::Tensor>& src,
                  const std::vector<torch::Tensor>& off, double slope,
                  const std::vector<torch::Tensor>& bias) {
  const int R = (int)h_tgt.size();
  TORCH_CHECK(R >= 1 && R <= kMaxRel, "1..8 relations supported");
  const int64_t H = h_src[0].size(1), C = h_src[0].size(2);
  TORCH_CHECK(C <= 2 * kWave, "GAT fused kernel supports C <= 128");
  int64_t nt_tot = 0, e_tot = 0;
  for (int r = 0; r < R; ++r) {
    nt_tot += off[r].numel() - 1;
    e_tot += src[r].numel();
  }
  const bool bf16 = h_src[0].scalar_type() == torch::kBFloat16;
  auto fopt = h_src[0].options().dtype(torch::kFloat32);
  auto out = torch::empty({nt_tot, H, C}, h_src[0].options());
  auto m = torch::empty({nt_tot, H}, fopt);
  auto z = torch::empty({nt_tot, H}, fopt);
  auto spre = torch::empty({e_tot, H}, fopt);
  auto run = [&](auto type_tag) {
    using T = decltype(type_tag);
    GatPack<T> P;
    fill_pack_common<T>(P, h_tgt, h_src, att_src, att_dst, src, off, H, C);
    int64_t nt_off = 0, e_off = 0;
    for (int r = 0; r < R; ++r) {
      P.out[r] = reinterpret_cast<T*>(out.data_ptr()) + nt_off * H * C;
      P.m[r] = m.data_ptr<float>() + nt_off * H;
      P.z[r] = z.data_ptr<float>() + nt_off * H;
      P.spre[r] = spre.data_ptr<float>() + e_off * H;
      P.bias[r] = (r < (int)bias.size() && bias[r].defined() &&
                   bias[r].numel() > 0)
                      ? bias[r].data_ptr<float>()
                      : nullptr;
      P.dbias[r] = nullptr;
      nt_off += P.n_tgt[r];
      e_off += src[r].numel();
    }
    if (nt_tot > 0) {
      hipLaunchKernelGGL(gat_multi_fwd_kernel<T>,
                         dim3(gat_grid(nt_tot * H)), dim3(kBlock), 0,
                         current_stream(), P, H, C, (float)slope, nt_tot);
    }
  };
  if (bf16) run(__bf16{}); else run(float{});
  return {out, m, z, spre};
}

void hip_gat_multi_bwd(const std::vector<torch::Tensor>& h_tgt,
                       const std::vector<torch::Tensor>& h_src,
                       const std::vector<torch::Tensor>& att_src,
                       const std::vector<torch::Tensor>& att_dst,
                       const std::vector<torch::Tensor>& src,
                       const std::vector<torch::Tensor>& off,
                       const torch::Tensor& out, const torch::Tensor& m,
                       const torch::Tensor& z, const torch::Tensor& spre,
                       const torch::Tensor& dout,
                       const std::vector<torch::Tensor>& dh_tgt,
                       const std::vector<torch::Tensor>& dh_src,
                       const std::vector<torch::Tensor>& datt_src,
                       const std::vector<torch::Tensor>& datt_dst,
                       double slope,
                       const std::vector<torch::Tensor>& dbias) {
  const int R = (int)h_tgt.size();
  const int64_t H = h_src[0].size(1), C = h_src[0].size(2);
  int64_t nt_tot = 0;
  for (int r = 0; r < R; ++r) nt_tot += off[r].numel() - 1;
  const bool bf16 = h_src[0].scalar_type() == torch::kBFloat16;
  auto dc = dout.contiguous();
  auto run = [&](auto type_tag) {
    using T = decltype(type_tag);
    GatPack<T> P;
    fill_pack_common<T>(P, h_tgt, h_src, att_src, att_dst, src, off, H, C);
    int64_t nt_off = 0, e_off = 0;
    for (int r = 0; r < R; ++r) {
      P.out[r] = reinterpret_cast<T*>(out.data_ptr()) + nt_off * H * C;
      P.dout[r] = reinterpret_cast<const T*>(dc.data_ptr()) +
                  nt_off * H * C;
      P.m[r] = const_cast<float*>(m.data_ptr<float>()) + nt_off * H;
      P.z[r] = const_cast<float*>(z.data_ptr<float>()) + nt_off * H;
      P.spre[r] = const_cast<float*>(spre.data_ptr<float>()) + e_off * H;
      TORCH_CHECK(dh_tgt[r].stride(1) == C && dh_src[r].stride(1) == C,
                  "gat_multi_bwd: dh inner dims must be dense");
      P.dh_tgt[r] = dh_tgt[r].data_ptr<float>();
      P.dh_src[r] = dh_src[r].data_ptr<float>();
      P.dht_stride[r] = dh_tgt[r].stride(0);
      P.dhs_stride[r] = dh_src[r].stride(0);
      P.datt_src[r] = datt_src[r].data_ptr<float>();
      P.datt_dst[r] = datt_dst[r].data_ptr<float>();
      P.bias[r] = nullptr;
      P.dbias[r] = (r < (int)dbias.size() && dbias[r].defined() &&
                    dbias[r].numel() > 0)
                       ? dbias[r].data_ptr<float>()
                       : nullptr;
      nt_off += P.n_tgt[r];
      e_off += src[r].numel();
    }
    if (nt_tot > 0) {
      int64_t S = 32768 / std::max<int64_t>(nt_tot * H, 1);
      S = std::max<int64_t>(1, std::min<int64_t>(S, 8));
      hipLaunchKernelGGL(gat_multi_bwd_kernel<T>,
                         dim3(gat_grid(nt_tot * H * S)), dim3(kBlock), 0,
                         current_stream(), P, H, C, (float)slope, S,
                         nt_tot);
    }
  };
  if (bf16) run(__bf16{}); else run(float{});
}

}  // namespace glt
