// Resampling + residual elementwise kernels — CDNA4 gfx950.
// SURVEY.md §2.4 K8 (nearest 2x upsample), K9 (2x2 avg-pool downsample),
// K10 (residual add * 1/sqrt(2)), forward and backward
// (/root/reference/model/xunet.py:14-21,92,127). All shapes (B,F,H,W,C)
// bf16/fp32, C % 8 == 0 for the bf16 16-byte path (true for every model
// channel count; generic fallback stays in eager torch).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;

// out = nearest-2x of in; pack index over OUTPUT (coalesced stores)
template <typename T>
__global__ void up2x_kernel(const T* __restrict__ in, T* __restrict__ out,
                            long total, int H, int W, int Cp) {
  const long stride = (long)gridDim.x * blockDim.x;
  const int H2 = H * 2, W2 = W * 2;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cp = (int)(i % Cp);
    long rest = i / Cp;
    const int w2 = (int)(rest % W2); rest /= W2;
    const int h2 = (int)(rest % H2); rest /= H2;
    const long bf = rest;
    const long src = ((bf * H + (h2 >> 1)) * W + (w2 >> 1)) * Cp + cp;
    Pack<T, 8> v = pload<T, 8>(in + src * 8);
    pstore<T, 8>(out + i * 8, v);
  }
}

// dL/din[h,w] = sum of dout over the 2x2 block
template <typename T>
__global__ void up2x_bwd_kernel(const T* __restrict__ dout,
                                T* __restrict__ din,
                                long total, int H, int W, int Cp) {
  const long stride = (long)gridDim.x * blockDim.x;
  const int W2 = W * 2, H2 = H * 2;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cp = (int)(i % Cp);
    long rest = i / Cp;
    const int w = (int)(rest % W); rest /= W;
    const int h = (int)(rest % H); rest /= H;
    const long bf = rest;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int dy = 0; dy < 2; ++dy)
#pragma unroll
      for (int dx = 0; dx < 2; ++dx) {
        const long src = ((bf * H2 + 2 * h + dy) * W2 + 2 * w + dx) * Cp + cp;
        Pack<T, 8> v = pload<T, 8>(dout + src * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += to_f32(v.v[j]);
      }
    Pack<T, 8> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) from_f32(acc[j], o.v[j]);
    pstore<T, 8>(din + i * 8, o);
  }
}

// avgpool fwd: out[h,w] = mean of in 2x2; bwd: din[h2,w2] = dout[h,w]/4
template <typename T, bool AVG>
__global__ void pool2x_kernel(const T* __restrict__ in, T* __restrict__ out,
                              long total, int Ho, int Wo, int Cp) {
  const long stride = (long)gridDim.x * blockDim.x;
  const int H2 = Ho * 2, W2 = Wo * 2;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cp = (int)(i % Cp);
    long rest = i / Cp;
    const int w = (int)(rest % Wo); rest /= Wo;
    const int h = (int)(rest % Ho); rest /= Ho;
    const long bf = rest;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int dy = 0; dy < 2; ++dy)
#pragma unroll
      for (int dx = 0; dx < 2; ++dx) {
        const long src = ((bf * H2 + 2 * h + dy) * W2 + 2 * w + dx) * Cp + cp;
        Pack<T, 8> v = pload<T, 8>(in + src * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += to_f32(v.v[j]);
      }
    Pack<T, 8> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) from_f32(acc[j] * 0.25f, o.v[j]);
    pstore<T, 8>(out + i * 8, o);
  }
}

template <typename T>
__global__ void pool2x_bwd_kernel(const T* __restrict__ dout,
                                  T* __restrict__ din,
                                  long total, int H2, int W2, int Cp) {
  const long stride = (long)gridDim.x * blockDim.x;
  const int Ho = H2 / 2, Wo = W2 / 2;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int cp = (int)(i % Cp);
    long rest = i / Cp;
    const int w2 = (int)(rest % W2); rest /= W2;
    const int h2 = (int)(rest % H2); rest /= H2;
    const long bf = rest;
    const long src = ((bf * Ho + (h2 >> 1)) * Wo + (w2 >> 1)) * Cp + cp;
    Pack<T, 8> v = pload<T, 8>(dout + src * 8);
    Pack<T, 8> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) from_f32(to_f32(v.v[j]) * 0.25f, o.v[j]);
    pstore<T, 8>(din + i * 8, o);
  }
}

// y = (a + b) * scale
template <typename T>
__global__ void add_scale_kernel(const T* __restrict__ a,
                                 const T* __restrict__ b,
                                 T* __restrict__ y, float scale, long total) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    Pack<T, 8> va = pload<T, 8>(a + i * 8);
    Pack<T, 8> vb = pload<T, 8>(b + i * 8);
    Pack<T, 8> o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      from_f32((to_f32(va.v[j]) + to_f32(vb.v[j])) * scale, o.v[j]);
    }
    pstore<T, 8>(y + i * 8, o);
  }
}

template <typename F>
void dispatch_dtype(const torch::Tensor& x, F&& f) {
  if (x.scalar_type() == torch::kBFloat16) {
    f((bf16*)nullptr);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat);
    f((float*)nullptr);
  }
}

int grid_for(long packs) {
  return (int)std::min<long>((packs + 255) / 256, 8192);
}

}  // namespace

torch::Tensor up2x_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 5
              && x.size(4) % 8 == 0);
  const long BF = x.size(0) * x.size(1);
  const int H = x.size(2), W = x.size(3), Cp = x.size(4) / 8;
  auto out = torch::empty({x.size(0), x.size(1), 2 * H, 2 * W, x.size(4)},
                          x.options());
  const long total = BF * 4 * H * W * Cp;
  auto stream = at::hip::getCurrentHIPStream();
  dispatch_dtype(x, [&](auto* tag) {
    using T = std::remove_pointer_t<decltype(tag)>;
    hipLaunchKernelGGL(up2x_kernel<T>, dim3(grid_for(total)), dim3(256), 0,
        stream, reinterpret_cast<const T*>(x.data_ptr()),
        reinterpret_cast<T*>(out.data_ptr()), total, H, W, Cp);
  });
  return out;
}

torch::Tensor up2x_bwd(torch::Tensor dout) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && dout.dim() == 5
              && dout.size(4) % 8 == 0);
  const long BF = dout.size(0) * dout.size(1);
  const int H = dout.size(2) / 2, W = dout.size(3) / 2;
  const int Cp = dout.size(4) / 8;
  auto din = torch::empty({dout.size(0), dout.size(1), H, W, dout.size(4)},
                          dout.options());
  const long total = BF * H * W * Cp;
  auto stream = at::hip::getCurrentHIPStream();
  dispatch_dtype(dout, [&](auto* tag) {
    using T = std::remove_pointer_t<decltype(tag)>;
    hipLaunchKernelGGL(up2x_bwd_kernel<T>, dim3(grid_for(total)), dim3(256),
        0, stream, reinterpret_cast<const T*>(dout.data_ptr()),
        reinterpret_cast<T*>(din.data_ptr()), total, H, W, Cp);
  });
  return din;
}

torch::Tensor pool2x_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 5
              && x.size(4) % 8 == 0 && x.size(2) % 2 == 0);
  const long BF = x.size(0) * x.size(1);
  const int Ho = x.size(2) / 2, Wo = x.size(3) / 2, Cp = x.size(4) / 8;
  auto out = torch::empty({x.size(0), x.size(1), Ho, Wo, x.size(4)},
                          x.options());
  const long total = BF * Ho * Wo * Cp;
  auto stream = at::hip::getCurrentHIPStream();
  dispatch_dtype(x, [&](auto* tag) {
    using T = std::remove_pointer_t<decltype(tag)>;
    hipLaunchKernelGGL((pool2x_kernel<T, true>), dim3(grid_for(total)),
        dim3(256), 0, stream, reinterpret_cast<const T*>(x.data_ptr()),
        reinterpret_cast<T*>(out.data_ptr()), total, Ho, Wo, Cp);
  });
  return out;
}

torch::Tensor pool2x_bwd(torch::Tensor dout) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && dout.dim() == 5
              && dout.size(4) % 8 == 0);
  const long BF = dout.size(0) * dout.size(1);
  const int H2 = dout.size(2) * 2, W2 = dout.size(3) * 2;
  const int Cp = dout.size(4) / 8;
  auto din = torch::empty({dout.size(0), dout.size(1), H2, W2, dout.size(4)},
                          dout.options());
  const long total = BF * H2 * W2 * Cp;
  auto stream = at::hip::getCurrentHIPStream();
  dispatch_dtype(dout, [&](auto* tag) {
    using T = std::remove_pointer_t<decltype(tag)>;
    hipLaunchKernelGGL(pool2x_bwd_kernel<T>, dim3(grid_for(total)), dim3(256),
        0, stream, reinterpret_cast<const T*>(dout.data_ptr()),
        reinterpret_cast<T*>(din.data_ptr()), total, H2, W2, Cp);
  });
  return din;
}

torch::Tensor add_scale(torch::Tensor a, torch::Tensor b, double scale) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous()
              && a.numel() == b.numel() && a.numel() % 8 == 0);
  auto y = torch::empty_like(a);
  const long total = a.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream();
  dispatch_dtype(a, [&](auto* tag) {
    using T = std::remove_pointer_t<decltype(tag)>;
    hipLaunchKernelGGL(add_scale_kernel<T>, dim3(grid_for(total)), dim3(256),
        0, stream, reinterpret_cast<const T*>(a.data_ptr()),
        reinterpret_cast<const T*>(b.data_ptr()),
        reinterpret_cast<T*>(y.data_ptr()), (float)scale, total);
  });
  return y;
}
