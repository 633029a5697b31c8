// Common helpers for the nvs3d CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define NVS_WAVE 64

// ---------------------------------------------------------------------------
// Packed vector load/store: Pack<T,V> is an aligned V-element chunk. hipcc
// emits global_load_dwordx4 / dwordx2 for the 16B/8B cases.
// ---------------------------------------------------------------------------
template <typename T, int V>
struct alignas(sizeof(T) * V) Pack {
  T v[V];
};

template <typename T, int V>
__device__ __forceinline__ Pack<T, V> pload(const T* p) {
  return *reinterpret_cast<const Pack<T, V>*>(p);
}

template <typename T, int V>
__device__ __forceinline__ void pstore(T* p, const Pack<T, V>& x) {
  *reinterpret_cast<Pack<T, V>*>(p) = x;
}

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ void from_f32(float x, float& o) { o = x; }
__device__ __forceinline__ void from_f32(float x, __hip_bfloat16& o) {
  o = __float2bfloat16(x);
}

// ---------------------------------------------------------------------------
// Wave + block reductions
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = NVS_WAVE / 2; off > 0; off >>= 1) {
    x += __shfl_down(x, off, NVS_WAVE);
  }
  return x;
}

__device__ __forceinline__ float sigmoidf_fast(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

static inline int ceil_div_host(long a, long b) {
  return static_cast<int>((a + b - 1) / b);
}

#define NVS_CHECK_HIP(expr)                                        \
  do {                                                             \
    hipError_t _e = (expr);                                        \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ",                   \
                hipGetErrorString(_e));                            \
  } while (0)
