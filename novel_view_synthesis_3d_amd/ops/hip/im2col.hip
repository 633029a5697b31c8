// im2col for the 3x3 SAME stride-1 conv — CDNA4 gfx950.
//
// Materializes A[m][k] (M = IMG*H*W pixels, k = (plane, ci), K = 9*Cin,
// zero-padded borders) so the conv WGRAD becomes one hipBLASLt GEMM
// A^T (9Cin x M) @ dy (M x Cout) — the long-K reduction that MIOpen's wrw
// kernels handle poorly on gfx950 (measured 4x slower than the GEMM peak
// path; see profiles/). Thread mapping: one 16-byte pack per thread-step,
// coalesced on both sides.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;

// m indexes the OUTPUT grid (IMG, Ho, Wo); input pixel = out*stride + d - pad
// nplanes == 9: full (dy,dx) window; nplanes == 3: row-shift only
// (dy in {-1,0,1}, dx = 0 — the wgrad 3-row-shift decomposition)
__global__ void im2col3x3_kernel(const bf16* __restrict__ x,  // (IMG,H,W,C)
                                 bf16* __restrict__ out, // (m1-m0, nplanes*C)
                                 int IMG, int H, int W, int C,
                                 int Ho, int Wo, int stride_s,
                                 int pad_h, int pad_w, int nplanes,
                                 long m0, long m1) {
  const int packs_per_plane = C / 8;
  const long total = (m1 - m0) * nplanes * packs_per_plane;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cp = (int)(i % packs_per_plane);
    long rest = i / packs_per_plane;
    const int plane = (int)(rest % nplanes);
    const long m = m0 + rest / nplanes;
    const int wpix = (int)(m % Wo);
    const int hpix = (int)((m / Wo) % Ho);
    const int img = (int)(m / ((long)Wo * Ho));
    const int pdy = (nplanes == 3) ? plane - 1 : plane / 3 - 1;
    const int pdx = (nplanes == 3) ? 0 : plane % 3 - 1;
    const int hy = hpix * stride_s + pdy + 1 - pad_h;
    const int wx = wpix * stride_s + pdx + 1 - pad_w;
    Pack<bf16, 8> v;
    if (hy >= 0 && hy < H && wx >= 0 && wx < W) {
      v = pload<bf16, 8>(x + (((long)img * H + hy) * W + wx) * C + cp * 8);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = __float2bfloat16(0.f);
    }
    pstore<bf16, 8>(out + (m - m0) * ((long)nplanes * C)
                    + (long)plane * C + cp * 8, v);
  }
}

}  // namespace

torch::Tensor im2col3x3(torch::Tensor x, int64_t stride_s,
                        int64_t nplanes, int64_t m0, int64_t m1,
                        c10::optional<torch::Tensor> out_buf) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous()
              && x.scalar_type() == torch::kBFloat16);
  const int nd = x.dim();
  TORCH_CHECK(nd == 4 || nd == 5);
  const int IMG = nd == 5 ? x.size(0) * x.size(1) : x.size(0);
  const int H = x.size(nd - 3), W = x.size(nd - 2), C = x.size(nd - 1);
  TORCH_CHECK(C % 8 == 0, "Cin must be a multiple of 8");
  const int s = (int)stride_s;
  TORCH_CHECK(nplanes == 9 || (nplanes == 3 && s == 1));
  // FLAX SAME: out = ceil(H/s), pad_lo = total//2 (asymmetric)
  const int Ho = (H + s - 1) / s, Wo = (W + s - 1) / s;
  const int pad_h = std::max((Ho - 1) * s + 3 - H, 0) / 2;
  const int pad_w = std::max((Wo - 1) * s + 3 - W, 0) / 2;
  const long M = (long)IMG * Ho * Wo;
  if (m1 < 0) m1 = M;
  TORCH_CHECK(0 <= m0 && m0 < m1 && m1 <= M);
  torch::Tensor out;
  if (out_buf.has_value()) {
    out = out_buf->narrow(0, 0, m1 - m0);
  } else {
    out = torch::empty({m1 - m0, nplanes * C}, x.options());
  }
  const int block = 256;
  const long total = (m1 - m0) * nplanes * (C / 8);
  const int grid = (int)std::min<long>((total + block - 1) / block, 16384);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(im2col3x3_kernel, dim3(grid), dim3(block), 0, stream,
      reinterpret_cast<const bf16*>(x.data_ptr()),
      reinterpret_cast<bf16*>(out.data_ptr()), IMG, H, W, C,
      Ho, Wo, s, pad_h, pad_w, (int)nplanes, m0, m1);
  return out;
}
