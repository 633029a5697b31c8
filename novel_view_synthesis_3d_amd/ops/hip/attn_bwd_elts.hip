// Fused elementwise pieces of the attention backward — CDNA4 gfx950.
//
// The attention backward (ops/hip_ops.py _Attention.backward) recomputes
// P from Q,K and the saved forward logsumexp, then runs 4 rocBLAS GEMMs.
// Eager torch would materialize S and P in fp32 (~6 passes over B*H*L*L);
// these two kernels do it in one pass each, bf16 in/out, fp32 math:
//   attn_p_from_lse: P = exp(S*scale - lse)            (S = QK^T, bf16)
//   attn_ds:         dS = scale * P * (dP - delta)     (delta = rowsum(dO*O))

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;

// p[b,h,i,j] = exp(s[b,h,i,j]*scale - lse[b,i,h])
__global__ void attn_p_kernel(const bf16* __restrict__ s,
                              const float* __restrict__ lse,  // (B,L,H)
                              bf16* __restrict__ p,
                              float scale, int B, int H, int L, int Lk) {
  const long total = (long)B * H * L * Lk;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i0 < total; i0 += stride) {
    Pack<bf16, 8> vs = pload<bf16, 8>(s + i0);
    const long row = i0 / Lk;        // (b*H + h)*L + i
    const int qi = (int)(row % L);
    const int h = (int)((row / L) % H);
    const int b = (int)(row / ((long)L * H));
    const float l = lse[((long)b * L + qi) * H + h];
    Pack<bf16, 8> vp;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      from_f32(__expf(to_f32(vs.v[j]) * scale - l), vp.v[j]);
    }
    pstore<bf16, 8>(p + i0, vp);
  }
}

// ds[b,h,i,j] = scale * p[b,h,i,j] * (dp[b,h,i,j] - delta[b,h,i])
__global__ void attn_ds_kernel(const bf16* __restrict__ p,
                               const bf16* __restrict__ dp,
                               const float* __restrict__ delta,  // (B,H,L)
                               bf16* __restrict__ ds,
                               float scale, int B, int H, int L, int Lk) {
  const long total = (long)B * H * L * Lk;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i0 < total; i0 += stride) {
    Pack<bf16, 8> vp = pload<bf16, 8>(p + i0);
    Pack<bf16, 8> vdp = pload<bf16, 8>(dp + i0);
    const float d = delta[i0 / Lk];
    Pack<bf16, 8> vo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      from_f32(scale * to_f32(vp.v[j]) * (to_f32(vdp.v[j]) - d), vo.v[j]);
    }
    pstore<bf16, 8>(ds + i0, vo);
  }
}

}  // namespace

torch::Tensor attn_p_from_lse(torch::Tensor s, torch::Tensor lse,
                              double scale) {
  TORCH_CHECK(s.is_cuda() && s.is_contiguous()
              && s.scalar_type() == torch::kBFloat16);
  const int B = s.size(0), H = s.size(1), L = s.size(2), Lk = s.size(3);
  TORCH_CHECK(Lk % 8 == 0);
  auto p = torch::empty_like(s);
  const long total = s.numel() / 8;
  const int grid = (int)std::min<long>((total + 255) / 256, 8192);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_p_kernel, dim3(grid), dim3(256), 0, stream,
      reinterpret_cast<const bf16*>(s.data_ptr()), lse.data_ptr<float>(),
      reinterpret_cast<bf16*>(p.data_ptr()), (float)scale, B, H, L, Lk);
  return p;
}

torch::Tensor attn_ds(torch::Tensor p, torch::Tensor dp, torch::Tensor delta,
                      double scale) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && dp.is_contiguous());
  const int B = p.size(0), H = p.size(1), L = p.size(2), Lk = p.size(3);
  auto ds = torch::empty_like(p);
  const long total = p.numel() / 8;
  const int grid = (int)std::min<long>((total + 255) / 256, 8192);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_ds_kernel, dim3(grid), dim3(256), 0, stream,
      reinterpret_cast<const bf16*>(p.data_ptr()),
      reinterpret_cast<const bf16*>(dp.data_ptr()),
      delta.contiguous().data_ptr<float>(),
      reinterpret_cast<bf16*>(ds.data_ptr()), (float)scale, B, H, L, Lk);
  return ds;
}
