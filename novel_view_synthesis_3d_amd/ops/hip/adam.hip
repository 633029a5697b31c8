// Fused multi-tensor Adam — CDNA4 gfx950. SURVEY.md §2.4 K18.
//
// The reference uses optax.adam (/root/reference/train.py:45,74-76); eager
// torch launches per-tensor foreach kernels. Here ONE launch updates every
// parameter: the host packs (p, g, m, v) pointers + a chunk map into device
// buffers once (pointers are stable across steps), and each block processes
// one 64K-element chunk with 16-byte vectorized fp32 loads/stores.
//
//   m = b1*m + (1-b1)*g;  v = b2*v + (1-b2)*g^2
//   p -= lr * (m/bc1) / (sqrt(v/bc2) + eps),  bc_i = 1 - beta_i^step

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int CHUNK = 1 << 16;  // elements per block
constexpr int T = 256;

__global__ void adam_kernel(const unsigned long long* __restrict__ ptrs, // (n,4)
                            const int* __restrict__ chunk_tensor,
                            const long* __restrict__ chunk_off,
                            const long* __restrict__ numels,
                            float lr, float b1, float b2, float eps,
                            float bc1, float bc2) {
  const int ci = blockIdx.x;
  const int tidx = chunk_tensor[ci];
  const long off = chunk_off[ci];
  const long n = min((long)CHUNK, numels[tidx] - off);
  float* p = reinterpret_cast<float*>(ptrs[tidx * 4 + 0]) + off;
  const float* g = reinterpret_cast<const float*>(ptrs[tidx * 4 + 1]) + off;
  float* m = reinterpret_cast<float*>(ptrs[tidx * 4 + 2]) + off;
  float* v = reinterpret_cast<float*>(ptrs[tidx * 4 + 3]) + off;

  const float inv_bc1 = 1.f / bc1;
  const float inv_bc2 = 1.f / bc2;

  long i = (long)threadIdx.x * 4;
  const long stride = (long)T * 4;
  // vectorized body (n multiple of 4 within the bulk)
  for (; i + 3 < n; i += stride) {
    Pack<float, 4> pg = pload<float, 4>(g + i);
    Pack<float, 4> pm = pload<float, 4>(m + i);
    Pack<float, 4> pv = pload<float, 4>(v + i);
    Pack<float, 4> pp = pload<float, 4>(p + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float gj = pg.v[j];
      const float mj = b1 * pm.v[j] + (1.f - b1) * gj;
      const float vj = b2 * pv.v[j] + (1.f - b2) * gj * gj;
      pm.v[j] = mj;
      pv.v[j] = vj;
      pp.v[j] -= lr * (mj * inv_bc1) / (sqrtf(vj * inv_bc2) + eps);
    }
    pstore<float, 4>(m + i, pm);
    pstore<float, 4>(v + i, pv);
    pstore<float, 4>(p + i, pp);
  }
  // scalar tail
  for (i = (n & ~3L) + threadIdx.x; i < n; i += T) {
    const float gj = g[i];
    const float mj = b1 * m[i] + (1.f - b1) * gj;
    const float vj = b2 * v[i] + (1.f - b2) * gj * gj;
    m[i] = mj;
    v[i] = vj;
    p[i] -= lr * (mj * inv_bc1) / (sqrtf(vj * inv_bc2) + eps);
  }
}

}  // namespace

void fused_adam(torch::Tensor ptrs, torch::Tensor chunk_tensor,
                torch::Tensor chunk_off, torch::Tensor numels,
                double lr, double b1, double b2, double eps, int64_t step) {
  TORCH_CHECK(ptrs.is_cuda() && ptrs.scalar_type() == torch::kUInt64);
  const int nchunks = chunk_tensor.size(0);
  const float bc1 = 1.f - powf((float)b1, (float)step);
  const float bc2 = 1.f - powf((float)b2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(adam_kernel, dim3(nchunks), dim3(T), 0, stream,
      reinterpret_cast<const unsigned long long*>(ptrs.data_ptr()),
      chunk_tensor.data_ptr<int>(), chunk_off.data_ptr<long>(),
      numels.data_ptr<long>(),
      (float)lr, (float)b1, (float)b2, (float)eps, bc1, bc2);
}
