/* Fused GAT edge-softmax + weighted aggregation (gfx950).
 *
 * Replaces the 6-kernel torch chain per GATConv (gather-add, leaky_relu,
 * scatter-amax, exp, scatter-sum, weighted index_add) and its sort-heavy
 * backward with one forward and one backward kernel.
 *
 * Layout: edges sorted by target (glt_amd batch invariant); one wave per
 * (target, head), lanes over the channel dim; the softmax runs online
 * (running max / rescaled sum) along the target's edge segment, so scores
 * are never materialized.
 *
 *   s_e   = leaky_relu(alpha_dst[t,h] + alpha_src[src_e,h])
 *   p_e   = exp(s_e - m_t) / Z_t
 *   out_t = sum_e p_e * h_src[src_e,h,:]
 *
 * Backward recomputes s_e from the saved (m, Z) statistics:
 *   dh_src[src_e]   += p_e * dout_t                  (fp32 atomics)
 *   ds_e             = p_e * (<dout_t, h_src_e> - <dout_t, out_t>)
 *   dalpha_src/dst  += ds_e * leaky'(s_pre_e)
 */
#include "hip_common.h"
#include "../include/common.h"

namespace glt {

namespace {

__global__ void gat_fused_fwd_kernel(
    const float* __restrict__ h_src,      // [Ns, H, C]
    const float* __restrict__ a_src,      // [Ns, H]
    const float* __restrict__ a_dst,      // [Nt, H]
    const int64_t* __restrict__ src,      // [E]
    const int64_t* __restrict__ offsets,  // [Nt+1]
    int64_t n_tgt, int64_t H, int64_t C, float slope,
    float* __restrict__ out,              // [Nt, H, C]
    float* __restrict__ m_out,            // [Nt, H]
    float* __restrict__ z_out) {          // [Nt, H]
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t total = n_tgt * H;
  for (int64_t w = wave; w < total; w += n_waves) {
    const int64_t t = w / H;
    const int64_t h = w - t * H;
    const int64_t s0 = offsets[t], s1 = offsets[t + 1];
    const float ad = a_dst[t * H + h];
    float m = -1e30f, Z = 0.f;
    float acc0 = 0.f, acc1 = 0.f;  // lanes cover C (up to 2 passes)
    for (int64_t e = s0; e < s1; ++e) {
      const int64_t sn = src[e];
      float s = ad + a_src[sn * H + h];
      s = s > 0.f ? s : s * slope;
      float scale = 1.f;
      float p;
      if (s > m) {
        scale = __expf(m - s);
        p = 1.f;
        m = s;
      } else {
        p = __expf(s - m);
      }
      Z = Z * scale + p;
      const float* hv = h_src + (sn * H + h) * C;
      if (lane < C) acc0 = acc0 * scale + p * hv[lane];
      if (kWave + lane < C) acc1 = acc1 * scale + p * hv[kWave + lane];
    }
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    float* ov = out + (t * H + h) * C;
    if (lane < C) ov[lane] = acc0 * inv;
    if (kWave + lane < C) ov[kWave + lane] = acc1 * inv;
    if (lane == 0) {
      m_out[t * H + h] = m;
      z_out[t * H + h] = Z;
    }
  }
}

__global__ void gat_fused_bwd_kernel(
    const float* __restrict__ h_src, const float* __restrict__ a_src,
    const float* __restrict__ a_dst, const int64_t* __restrict__ src,
    const int64_t* __restrict__ offsets, const float* __restrict__ out,
    const float* __restrict__ m_in, const float* __restrict__ z_in,
    const float* __restrict__ dout, int64_t n_tgt, int64_t H, int64_t C,
    float slope, float* __restrict__ dh_src, float* __restrict__ da_src,
    float* __restrict__ da_dst) {
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wave =
      (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) / kWave;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) / kWave;
  const int64_t total = n_tgt * H;
  for (int64_t w = wave; w < total; w += n_waves) {
    const int64_t t = w / H;
    const int64_t h = w - t * H;
    const int64_t s0 = offsets[t], s1 = offsets[t + 1];
    if (s1 <= s0) continue;
    const float ad = a_dst[t * H + h];
    const float m = m_in[t * H + h];
    const float Z = z_in[t * H + h];
    const float inv = Z > 0.f ? 1.f / Z : 0.f;
    const float* dv = dout + (t * H + h) * C;
    const float* ov = out + (t * H + h) * C;
    float d0 = lane < C ? dv[lane] : 0.f;
    float d1 = kWave + lane < C ? dv[kWave + lane] : 0.f;
    // <dout, out>
    float dot_o = (lane < C ? d0 * ov[lane] : 0.f) +
                  (kWave + lane < C ? d1 * ov[kWave + lane] : 0.f);
#pragma unroll
    for (int sft = kWave / 2; sft > 0; sft >>= 1)
      dot_o += __shfl_down(dot_o, sft);
    dot_o = __shfl(dot_o, 0);
    float dad_acc = 0.f;
    for (int64_t e = s0; e < s1; ++e) {
      const int64_t sn = src[e];
      float s_pre = ad + a_src[sn * H + h];
      float s = s_pre > 0.f ? s_pre : s_pre * slope;
      const float p = __expf(s - m) * inv;
      const float* hv = h_src + (sn * H + h) * C;
      float* dhv = dh_src + (sn * H + h) * C;
      float dot_h = (lane < C ? d0 * hv[lane] : 0.f) +
                    (kWave + lane < C ? d1 * hv[kWave + lane] : 0.f);
#pragma unroll
      for (int sft = kWave / 2; sft > 0; sft >>= 1)
        dot_h += __shfl_down(dot_h, sft);
      dot_h = __shfl(dot_h, 0);
      if (lane < C) atomicAdd(&dhv[lane], p * d0);
      if (kWave + lane < C) atomicAdd(&dhv[kWave + lane], p * d1);
      float ds = p * (dot_h - dot_o);
      ds *= (s_pre > 0.f ? 1.f : slope);
      if (lane == 0) atomicAdd(&da_src[sn * H + h], ds);
      dad_acc += ds;
    }
    if (lane == 0) atomicAdd(&da_dst[t * H + h], dad_acc);
  }
}

int gat_grid(int64_t waves_needed) {
  const int64_t w = std::min<int64_t>(waves_needed, (int64_t)kMaxBlocks * 4);
  return (int)std::min<int64_t>((w * kWave + kBlock - 1) / kBlock,
                                kMaxBlocks);
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> hip_gat_fused_fwd(
    const torch::Tensor& h_src, const torch::Tensor& a_src,
    const torch::Tensor& a_dst, const torch::Tensor& src,
    const torch::Tensor& offsets, double slope) {
  const int64_t n_tgt = offsets.numel() - 1;
  const int64_t H = h_src.size(1), C = h_src.size(2);
  TORCH_CHECK(C <= 2 * kWave, "GAT fused kernel supports C <= 128");
  auto out = torch::empty({n_tgt, H, C}, h_src.options());
  auto m = torch::empty({n_tgt, H}, h_src.options());
  auto z = torch::empty({n_tgt, H}, h_src.options());
  if (n_tgt > 0) {
    hipLaunchKernelGGL(gat_fused_fwd_kernel, dim3(gat_grid(n_tgt * H)),
                       dim3(kBlock), 0, current_stream(),
                       h_src.data_ptr<float>(), a_src.data_ptr<float>(),
                       a_dst.data_ptr<float>(), src.data_ptr<int64_t>(),
                       offsets.data_ptr<int64_t>(), n_tgt, H, C,
                       (float)slope, out.data_ptr<float>(),
                       m.data_ptr<float>(), z.data_ptr<float>());
  }
  return {out, m, z};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> hip_gat_fused_bwd(
    const torch::Tensor& h_src, const torch::Tensor& a_src,
    const torch::Tensor& a_dst, const torch::Tensor& src,
    const torch::Tensor& offsets, const torch::Tensor& out,
    const torch::Tensor& m, const torch::Tensor& z,
    const torch::Tensor& dout, double slope) {
  const int64_t n_tgt = offsets.numel() - 1;
  const int64_t H = h_src.size(1), C = h_src.size(2);
  auto dh = torch::zeros_like(h_src);
  auto das = torch::zeros_like(a_src);
  auto dad = torch::zeros_like(a_dst);
  if (n_tgt > 0) {
    hipLaunchKernelGGL(gat_fused_bwd_kernel, dim3(gat_grid(n_tgt * H)),
                       dim3(kBlock), 0, current_stream(),
                       h_src.data_ptr<float>(), a_src.data_ptr<float>(),
                       a_dst.data_ptr<float>(), src.data_ptr<int64_t>(),
                       offsets.data_ptr<int64_t>(),
                       out.data_ptr<float>(), m.data_ptr<float>(),
                       z.data_ptr<float>(),
                       dout.contiguous().data_ptr<float>(), n_tgt, H, C,
                       (float)slope, dh.data_ptr<float>(),
                       das.data_ptr<float>(), dad.data_ptr<float>());
  }
  return {dh, das, dad};
}

}  // namespace glt
