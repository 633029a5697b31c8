// Fused joint-frame GroupNorm (+FiLM +SiLU) forward/backward — CDNA4 gfx950.
//
// Implements SURVEY.md §2.4 K3+K5+K4: GroupNorm with statistics computed
// jointly over both frames and all spatial positions per (batch, group)
// (reference /root/reference/model/xunet.py:46-61), fused with the FiLM
// modulation h*(1+scale)+shift and SiLU, forward and backward. The eager
// oracle is ops/reference.py joint_groupnorm.
//
// Layout: x is (B, F*H*W = R rows, C) contiguous NHWC-with-frames. Thread
// mapping: each thread owns a FIXED channel span [c0, c0+V) (V | Cg so the
// span stays inside one group) and strides over rows — fully coalesced
// 16-byte loads, per-thread register gamma/beta, one LDS atomic per thread
// per reduction. Grid = B * P row-chunks so the launch fills 256 CUs
// (one block per (b, chunk); stats finalized per-block by re-reducing the
// tiny (B, G, P) partial buffer).
//
// Math (all accumulation fp32):
//   xhat = (x - mu) * r;  u = xhat*gamma + beta;  v = u*(1+s) + t;
//   y = silu(v) = v * sigmoid(v)
// backward:
//   dv = dy * sig(v)*(1 + v*(1-sig(v)));  ds = dv*u; dt = dv
//   du = dv*(1+s);  dgamma_c = sum du*xhat;  dbeta_c = sum du
//   dxhat = du*gamma; per (b,g): S1 = sum dxhat, S2 = sum dxhat*xhat
//   dx = r * (dxhat - (S1 + xhat*S2)/N)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int MAX_GROUPS = 32;

// Counter-based dropout mask: PCG hash of (element index, seed). The
// backward REGENERATES the mask from the same seed — no mask tensor.
// Both 32-bit halves of the element offset feed the hash, so tensors with
// >= 2^32 elements don't silently reuse mask values across the wrap.
__device__ __forceinline__ bool keep_elem(unsigned long long idx,
                                          unsigned seed, unsigned thresh) {
  unsigned v = (unsigned)idx * 0x9E3779B9u
             + (unsigned)(idx >> 32) * 0x85EBCA6Bu + seed;
  v = v * 747796405u + 2891336453u;
  unsigned w = ((v >> ((v >> 28) + 4u)) ^ v) * 277803737u;
  return ((w >> 22) ^ w) >= thresh;
}

struct GnShape {
  int B, R, C, G, Cg;   // rows R = F*H*W; Cg = C/G
  int P;                // row chunks per batch
  int rowThreads;       // C / V
  int T;                // block threads = rowThreads * rowsPerIter
  float drop_scale;     // 1/(1-p); 0 = no dropout
  // device-side seed (1-elem int32 tensor): read at kernel execution time,
  // not baked in at launch, so dropout masks keep advancing under hipGraph
  // replay (the host-drawn-seed D2-style freeze is gone)
  const unsigned* __restrict__ seed_ptr;
  unsigned drop_thresh; // p * 2^32
};

// ---------------------------------------------------------------------------
// Pass 1 (forward): per-(b,chunk) partial sum/sumsq per group.
// partials layout: (B, G, P, 2) fp32
// ---------------------------------------------------------------------------
template <typename T, int V>
__global__ void gn_fwd_partials(const T* __restrict__ x,
                                float* __restrict__ partials,
                                GnShape s) {
  __shared__ float lsum[MAX_GROUPS], lsumsq[MAX_GROUPS];
  const int b = blockIdx.x / s.P;
  const int chunk = blockIdx.x % s.P;
  const int tid = threadIdx.x;
  if (tid < s.G) { lsum[tid] = 0.f; lsumsq[tid] = 0.f; }
  __syncthreads();

  const int rowsPerIter = s.T / s.rowThreads;
  const int rt = tid % s.rowThreads;
  const int ri = tid / s.rowThreads;
  const int c0 = rt * V;
  const int g = c0 / s.Cg;

  const int rc = (s.R + s.P - 1) / s.P;
  const int r0 = chunk * rc;
  const int r1 = min(s.R, r0 + rc);

  float s1 = 0.f, s2 = 0.f;
  const T* xb = x + (size_t)b * s.R * s.C;
  for (int r = r0 + ri; r < r1; r += rowsPerIter) {
    Pack<T, V> p = pload<T, V>(xb + (size_t)r * s.C + c0);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float xv = to_f32(p.v[j]);
      s1 += xv;
      s2 += xv * xv;
    }
  }
  atomicAdd(&lsum[g], s1);
  atomicAdd(&lsumsq[g], s2);
  __syncthreads();
  if (tid < s.G) {
    float* dst = partials + (((size_t)b * s.G + tid) * s.P + chunk) * 2;
    dst[0] = lsum[tid];
    dst[1] = lsumsq[tid];
  }
}

// ---------------------------------------------------------------------------
// Pass 2 (forward): finalize stats (block-local re-reduce of partials) and
// apply normalize + affine (+FiLM)(+SiLU).
// ---------------------------------------------------------------------------
template <typename T, int V, bool FILM, bool SILU>
__global__ void gn_fwd_apply(const T* __restrict__ x,
                             const float* __restrict__ partials,
                             const float* __restrict__ gamma,
                             const float* __restrict__ beta,
                             const T* __restrict__ film,  // (..,2C): scale|shift
                             T* __restrict__ y,
                             float* __restrict__ mean_out,   // (B,G)
                             float* __restrict__ rstd_out,   // (B,G)
                             float eps, GnShape s) {
  __shared__ float lmean[MAX_GROUPS], lrstd[MAX_GROUPS];
  const int b = blockIdx.x / s.P;
  const int chunk = blockIdx.x % s.P;
  const int tid = threadIdx.x;
  if (tid < s.G) {
    float s1 = 0.f, s2 = 0.f;
    const float* src = partials + ((size_t)b * s.G + tid) * s.P * 2;
    for (int p = 0; p < s.P; ++p) { s1 += src[2 * p]; s2 += src[2 * p + 1]; }
    const float n = (float)s.R * s.Cg;
    const float mu = s1 / n;
    const float var = fmaxf(s2 / n - mu * mu, 0.f);
    const float r = rsqrtf(var + eps);
    lmean[tid] = mu;
    lrstd[tid] = r;
    if (chunk == 0) {
      mean_out[(size_t)b * s.G + tid] = mu;
      rstd_out[(size_t)b * s.G + tid] = r;
    }
  }
  __syncthreads();

  const int rowsPerIter = s.T / s.rowThreads;
  const int rt = tid % s.rowThreads;
  const int ri = tid / s.rowThreads;
  const int c0 = rt * V;
  const int g = c0 / s.Cg;
  const float mu = lmean[g];
  const float r = lrstd[g];
  const unsigned seed = s.drop_scale != 0.f ? *s.seed_ptr : 0u;

  float gm[V], bt[V];
#pragma unroll
  for (int j = 0; j < V; ++j) { gm[j] = gamma[c0 + j]; bt[j] = beta[c0 + j]; }

  const int rc = (s.R + s.P - 1) / s.P;
  const int r0 = chunk * rc;
  const int r1 = min(s.R, r0 + rc);
  const size_t base = (size_t)b * s.R * s.C;

  const size_t fbase = (size_t)b * s.R * 2 * s.C;
  for (int row = r0 + ri; row < r1; row += rowsPerIter) {
    const size_t off = base + (size_t)row * s.C + c0;
    const size_t foff = fbase + (size_t)row * 2 * s.C + c0;
    Pack<T, V> px = pload<T, V>(x + off);
    Pack<T, V> ps, pt;
    if (FILM) { ps = pload<T, V>(film + foff);
                pt = pload<T, V>(film + foff + s.C); }
    Pack<T, V> po;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float u = (to_f32(px.v[j]) - mu) * r * gm[j] + bt[j];
      if (FILM) u = u * (1.f + to_f32(ps.v[j])) + to_f32(pt.v[j]);
      if (SILU) u = u * sigmoidf_fast(u);
      if (s.drop_scale != 0.f) {
        u = keep_elem(off + j, seed, s.drop_thresh)
                ? u * s.drop_scale : 0.f;
      }
      from_f32(u, po.v[j]);
    }
    pstore<T, V>(y + off, po);
  }
}

// ---------------------------------------------------------------------------
// Backward pass 1: per-(b,chunk) S1/S2 partials per group + dgamma/dbeta
// (global fp32 atomics after LDS pre-reduction) + elementwise ds/dt.
// ---------------------------------------------------------------------------
template <typename T, int V, bool FILM, bool SILU>
__global__ void gn_bwd_partials(const T* __restrict__ dy,
                                const T* __restrict__ x,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                const T* __restrict__ film,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                float* __restrict__ partials,   // (B,G,P,2)
                                float* __restrict__ dgamma,     // (C,) zeroed
                                float* __restrict__ dbeta,      // (C,) zeroed
                                T* __restrict__ dfilm,
                                GnShape s) {
  __shared__ float ls1[MAX_GROUPS], ls2[MAX_GROUPS];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* ldg = reinterpret_cast<float*>(smem);          // C floats
  float* ldb = ldg + s.C;                               // C floats
  const int b = blockIdx.x / s.P;
  const int chunk = blockIdx.x % s.P;
  const int tid = threadIdx.x;
  if (tid < s.G) { ls1[tid] = 0.f; ls2[tid] = 0.f; }
  for (int c = tid; c < s.C; c += s.T) { ldg[c] = 0.f; ldb[c] = 0.f; }
  __syncthreads();

  const int rowsPerIter = s.T / s.rowThreads;
  const int rt = tid % s.rowThreads;
  const int ri = tid / s.rowThreads;
  const int c0 = rt * V;
  const int g = c0 / s.Cg;
  const float mu = mean[(size_t)b * s.G + g];
  const float r = rstd[(size_t)b * s.G + g];
  const unsigned seed = s.drop_scale != 0.f ? *s.seed_ptr : 0u;

  float gm[V], bt[V], dgm[V], dbt[V];
#pragma unroll
  for (int j = 0; j < V; ++j) {
    gm[j] = gamma[c0 + j]; bt[j] = beta[c0 + j];
    dgm[j] = 0.f; dbt[j] = 0.f;
  }

  const int rc = (s.R + s.P - 1) / s.P;
  const int r0 = chunk * rc;
  const int r1 = min(s.R, r0 + rc);
  const size_t base = (size_t)b * s.R * s.C;
  float s1 = 0.f, s2 = 0.f;

  const size_t fbase = (size_t)b * s.R * 2 * s.C;
  for (int row = r0 + ri; row < r1; row += rowsPerIter) {
    const size_t off = base + (size_t)row * s.C + c0;
    const size_t foff = fbase + (size_t)row * 2 * s.C + c0;
    Pack<T, V> px = pload<T, V>(x + off);
    Pack<T, V> pdy = pload<T, V>(dy + off);
    Pack<T, V> ps, pt;
    if (FILM) { ps = pload<T, V>(film + foff);
                pt = pload<T, V>(film + foff + s.C); }
    Pack<T, V> pds, pdt;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float xhat = (to_f32(px.v[j]) - mu) * r;
      const float u = xhat * gm[j] + bt[j];
      float v = u;
      if (FILM) v = u * (1.f + to_f32(ps.v[j])) + to_f32(pt.v[j]);
      float dv = to_f32(pdy.v[j]);
      if (s.drop_scale != 0.f) {
        dv = keep_elem(off + j, seed, s.drop_thresh)
                 ? dv * s.drop_scale : 0.f;
      }
      if (SILU) {
        const float sg = sigmoidf_fast(v);
        dv *= sg * (1.f + v * (1.f - sg));
      }
      if (FILM) {
        from_f32(dv * u, pds.v[j]);
        from_f32(dv, pdt.v[j]);
      }
      const float du = FILM ? dv * (1.f + to_f32(ps.v[j])) : dv;
      dgm[j] += du * xhat;
      dbt[j] += du;
      const float dxh = du * gm[j];
      s1 += dxh;
      s2 += dxh * xhat;
    }
    if (FILM) {
      pstore<T, V>(dfilm + foff, pds);
      pstore<T, V>(dfilm + foff + s.C, pdt);
    }
  }
  atomicAdd(&ls1[g], s1);
  atomicAdd(&ls2[g], s2);
#pragma unroll
  for (int j = 0; j < V; ++j) {
    atomicAdd(&ldg[c0 + j], dgm[j]);
    atomicAdd(&ldb[c0 + j], dbt[j]);
  }
  __syncthreads();
  if (tid < s.G) {
    float* dst = partials + (((size_t)b * s.G + tid) * s.P + chunk) * 2;
    dst[0] = ls1[tid];
    dst[1] = ls2[tid];
  }
  for (int c = tid; c < s.C; c += s.T) {
    atomicAdd(&dgamma[c], ldg[c]);
    atomicAdd(&dbeta[c], ldb[c]);
  }
}

// ---------------------------------------------------------------------------
// Backward pass 2: dx.
// ---------------------------------------------------------------------------
template <typename T, int V, bool FILM, bool SILU>
__global__ void gn_bwd_apply(const T* __restrict__ dy,
                             const T* __restrict__ x,
                             const float* __restrict__ gamma,
                             const float* __restrict__ beta,
                             const T* __restrict__ film,
                             const float* __restrict__ mean,
                             const float* __restrict__ rstd,
                             const float* __restrict__ partials,
                             T* __restrict__ dx,
                             GnShape s) {
  __shared__ float lS1[MAX_GROUPS], lS2[MAX_GROUPS];
  const int b = blockIdx.x / s.P;
  const int chunk = blockIdx.x % s.P;
  const int tid = threadIdx.x;
  if (tid < s.G) {
    float a = 0.f, c = 0.f;
    const float* src = partials + ((size_t)b * s.G + tid) * s.P * 2;
    for (int p = 0; p < s.P; ++p) { a += src[2 * p]; c += src[2 * p + 1]; }
    const float n = (float)s.R * s.Cg;
    lS1[tid] = a / n;
    lS2[tid] = c / n;
  }
  __syncthreads();

  const int rowsPerIter = s.T / s.rowThreads;
  const int rt = tid % s.rowThreads;
  const int ri = tid / s.rowThreads;
  const int c0 = rt * V;
  const int g = c0 / s.Cg;
  const float mu = mean[(size_t)b * s.G + g];
  const float r = rstd[(size_t)b * s.G + g];
  const float m1 = lS1[g];
  const float m2 = lS2[g];
  const unsigned seed = s.drop_scale != 0.f ? *s.seed_ptr : 0u;

  float gm[V], bt[V];
#pragma unroll
  for (int j = 0; j < V; ++j) { gm[j] = gamma[c0 + j]; bt[j] = beta[c0 + j]; }

  const int rc = (s.R + s.P - 1) / s.P;
  const int r0 = chunk * rc;
  const int r1 = min(s.R, r0 + rc);
  const size_t base = (size_t)b * s.R * s.C;

  const size_t fbase = (size_t)b * s.R * 2 * s.C;
  for (int row = r0 + ri; row < r1; row += rowsPerIter) {
    const size_t off = base + (size_t)row * s.C + c0;
    const size_t foff = fbase + (size_t)row * 2 * s.C + c0;
    Pack<T, V> px = pload<T, V>(x + off);
    Pack<T, V> pdy = pload<T, V>(dy + off);
    Pack<T, V> ps, pt;
    if (FILM) { ps = pload<T, V>(film + foff);
                pt = pload<T, V>(film + foff + s.C); }
    Pack<T, V> pdx;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float xhat = (to_f32(px.v[j]) - mu) * r;
      const float u = xhat * gm[j] + bt[j];
      float v = u;
      if (FILM) v = u * (1.f + to_f32(ps.v[j])) + to_f32(pt.v[j]);
      float dv = to_f32(pdy.v[j]);
      if (s.drop_scale != 0.f) {
        dv = keep_elem(off + j, seed, s.drop_thresh)
                 ? dv * s.drop_scale : 0.f;
      }
      if (SILU) {
        const float sg = sigmoidf_fast(v);
        dv *= sg * (1.f + v * (1.f - sg));
      }
      const float du = FILM ? dv * (1.f + to_f32(ps.v[j])) : dv;
      const float dxh = du * gm[j];
      from_f32(r * (dxh - m1 - xhat * m2), pdx.v[j]);
    }
    pstore<T, V>(dx + off, pdx);
  }
}

// ---------------------------------------------------------------------------
// Host-side shape/launch selection
// ---------------------------------------------------------------------------
GnShape make_shape(long B, long R, long C, long G) {
  GnShape s;
  s.B = (int)B; s.R = (int)R; s.C = (int)C; s.G = (int)G;
  s.Cg = (int)(C / G);
  return s;
}

int pick_vec(int Cg, int elem_bytes) {
  const int maxv = 16 / elem_bytes;  // 16B packs
  for (int v = maxv; v >= 1; v >>= 1) {
    if (Cg % v == 0) return v;
  }
  return 1;
}

bool pick_block(GnShape& s, int V) {
  if (s.C % V != 0) return false;
  s.rowThreads = s.C / V;
  if (s.rowThreads > 1024) return false;
  int bestT = 0;
  for (int k = 1; k * s.rowThreads <= 1024; ++k) {
    int T = k * s.rowThreads;
    if (T % 64 == 0) bestT = T;
  }
  if (bestT == 0) return false;
  s.T = bestT;
  // target ~2048 blocks in flight
  long rowsPerIter = s.T / s.rowThreads;
  long maxP = (s.R + rowsPerIter - 1) / rowsPerIter;
  long wantP = std::max(1L, 2048L / std::max(1, s.B));
  s.P = (int)std::min(maxP, wantP);
  return true;
}

#define DISPATCH_V(V_, ...)                                   \
  switch (V_) {                                               \
    case 8: { constexpr int V = 8; __VA_ARGS__; break; }      \
    case 4: { constexpr int V = 4; __VA_ARGS__; break; }      \
    case 2: { constexpr int V = 2; __VA_ARGS__; break; }      \
    default: { constexpr int V = 1; __VA_ARGS__; break; }     \
  }

#define DISPATCH_BOOL(B_, NAME, ...)                          \
  if (B_) { constexpr bool NAME = true; __VA_ARGS__; }        \
  else    { constexpr bool NAME = false; __VA_ARGS__; }

}  // namespace

// ---------------------------------------------------------------------------
// ATen entry points
// ---------------------------------------------------------------------------
static const unsigned* seed_ptr_of(const c10::optional<torch::Tensor>& seed,
                                   double p_drop) {
  if (!(p_drop > 0)) return nullptr;
  TORCH_CHECK(seed.has_value(), "p_drop > 0 needs a device seed tensor");
  TORCH_CHECK(seed->is_cuda() && seed->scalar_type() == torch::kInt &&
              seed->numel() == 1, "seed must be a 1-elem int32 CUDA tensor");
  return reinterpret_cast<const unsigned*>(seed->data_ptr<int32_t>());
}

std::vector<torch::Tensor> gn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  c10::optional<torch::Tensor> film_in,
                                  int64_t groups, double eps, bool silu,
                                  double p_drop,
                                  c10::optional<torch::Tensor> drop_seed) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous CUDA");
  TORCH_CHECK(x.dim() == 5, "x must be (B,F,H,W,C)");
  const long B = x.size(0);
  const long R = x.size(1) * x.size(2) * x.size(3);
  const long C = x.size(4);
  TORCH_CHECK(C % groups == 0 && groups <= MAX_GROUPS);
  const bool film = film_in.has_value();
  if (film) {
    TORCH_CHECK(film_in->is_contiguous());
    TORCH_CHECK(film_in->size(4) == 2 * C, "film must be (..,2C) scale|shift");
    TORCH_CHECK(film_in->scalar_type() == x.scalar_type());
  }

  GnShape s = make_shape(B, R, C, groups);
  s.drop_scale = p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 0.f;
  s.seed_ptr = seed_ptr_of(drop_seed, p_drop);
  s.drop_thresh = (unsigned)(p_drop * 4294967296.0);
  const int elem = x.scalar_type() == torch::kFloat ? 4 : 2;
  int V = pick_vec(s.Cg, elem);
  TORCH_CHECK(pick_block(s, V), "unsupported GN shape C=", C);

  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(torch::kFloat);
  auto partials = torch::empty({B, groups, s.P, 2}, opts);
  auto mean = torch::empty({B, groups}, opts);
  auto rstd = torch::empty({B, groups}, opts);

  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid(s.B * s.P), block(s.T);
  auto gammaf = gamma.to(torch::kFloat).contiguous();
  auto betaf = beta.to(torch::kFloat).contiguous();

  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
      "gn_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, float>;
    DISPATCH_V(V, ({
      hipLaunchKernelGGL((gn_fwd_partials<T, V>), grid, block, 0, stream,
          reinterpret_cast<const T*>(x.data_ptr()),
          partials.data_ptr<float>(), s);
      DISPATCH_BOOL(film, FILM, ({
        DISPATCH_BOOL(silu, SILU, ({
          hipLaunchKernelGGL((gn_fwd_apply<T, V, FILM, SILU>), grid, block, 0,
              stream,
              reinterpret_cast<const T*>(x.data_ptr()),
              partials.data_ptr<float>(),
              gammaf.data_ptr<float>(), betaf.data_ptr<float>(),
              film ? reinterpret_cast<const T*>(film_in->data_ptr()) : nullptr,
              reinterpret_cast<T*>(y.data_ptr()),
              mean.data_ptr<float>(), rstd.data_ptr<float>(),
              (float)eps, s);
        }));
      }));
    }));
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> gn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  c10::optional<torch::Tensor> film_in,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  int64_t groups, bool silu,
                                  double p_drop,
                                  c10::optional<torch::Tensor> drop_seed) {
  TORCH_CHECK(dy.is_cuda() && x.is_contiguous());
  auto dyc = dy.contiguous();
  const long B = x.size(0);
  const long R = x.size(1) * x.size(2) * x.size(3);
  const long C = x.size(4);
  const bool film = film_in.has_value();

  GnShape s = make_shape(B, R, C, groups);
  s.drop_scale = p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 0.f;
  s.seed_ptr = seed_ptr_of(drop_seed, p_drop);
  s.drop_thresh = (unsigned)(p_drop * 4294967296.0);
  const int elem = x.scalar_type() == torch::kFloat ? 4 : 2;
  int V = pick_vec(s.Cg, elem);
  TORCH_CHECK(pick_block(s, V), "unsupported GN shape C=", C);

  auto opts = x.options().dtype(torch::kFloat);
  auto partials = torch::empty({B, groups, s.P, 2}, opts);
  auto dgamma = torch::zeros({C}, opts);
  auto dbeta = torch::zeros({C}, opts);
  auto dx = torch::empty_like(x);
  torch::Tensor dfilm;
  if (film) {
    dfilm = torch::empty_like(*film_in);
  }

  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid(s.B * s.P), block(s.T);
  auto gammaf = gamma.to(torch::kFloat).contiguous();
  auto betaf = beta.to(torch::kFloat).contiguous();
  const size_t lds = 2 * C * sizeof(float);

  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
      "gn_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, float>;
    DISPATCH_V(V, ({
      DISPATCH_BOOL(film, FILM, ({
        DISPATCH_BOOL(silu, SILU, ({
          hipLaunchKernelGGL((gn_bwd_partials<T, V, FILM, SILU>), grid, block,
              lds, stream,
              reinterpret_cast<const T*>(dyc.data_ptr()),
              reinterpret_cast<const T*>(x.data_ptr()),
              gammaf.data_ptr<float>(), betaf.data_ptr<float>(),
              film ? reinterpret_cast<const T*>(film_in->data_ptr()) : nullptr,
              mean.data_ptr<float>(), rstd.data_ptr<float>(),
              partials.data_ptr<float>(),
              dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
              film ? reinterpret_cast<T*>(dfilm.data_ptr()) : nullptr,
              s);
          hipLaunchKernelGGL((gn_bwd_apply<T, V, FILM, SILU>), grid, block, 0,
              stream,
              reinterpret_cast<const T*>(dyc.data_ptr()),
              reinterpret_cast<const T*>(x.data_ptr()),
              gammaf.data_ptr<float>(), betaf.data_ptr<float>(),
              film ? reinterpret_cast<const T*>(film_in->data_ptr()) : nullptr,
              mean.data_ptr<float>(), rstd.data_ptr<float>(),
              partials.data_ptr<float>(),
              reinterpret_cast<T*>(dx.data_ptr()), s);
        }));
      }));
    }));
  });
  if (film) return {dx, dgamma, dbeta, dfilm};
  return {dx, dgamma, dbeta};
}
