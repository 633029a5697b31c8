// Linear-layer weight gradient (tall-skinny reduction GEMM) — CDNA4 gfx950.
//
// SURVEY.md §2.4 K6 backward: for y = x W^T + b (nn.Linear / flax Dense,
// reference model/xunet.py:54-61,91,100-102,156-157),
//   dW[n][k] = sum_m dy[m][n] * x[m][k],     db[n] = sum_m dy[m][n].
//
// hipBLASLt collapses on these shapes (N*K tiny, M ~ 10^5-10^6: measured
// 65-115 TF, >1 ms for the FiLM/skip-Dense wgrads at full config). Here the
// whole dW tile lives in one block's accumulators and M is split across
// blocks (fp32 atomic reduction), so dy and x stream from HBM once per
// (n,k)-tile.
//
// Both operands are m-major, so both MFMA fragments need m-contiguous
// lanes: chunks are staged ROW-MAJOR ([m][C_tile], row stride padded +64 B)
// and consumed with ds_read_b64_tr_b16 — the transpose read takes per-lane
// granule addresses, so no subtile reshuffle is needed. Row-major staging
// keeps glds fully line-coalesced (the first version's [csub][m][16]
// subtile images scattered global reads into 32 B pieces — 1.5 TB/s
// effective); the +64 B row pad puts the rows of each 32-lane transpose
// service group in distinct bank classes (conflict-free without an XOR swizzle).
//
// Geometry: block tile 256(n) x 256(k), 8 waves as 4(n) x 2(k), wave tile
// 64 x 128 = 8 accumulator planes of mfma_f32_32x32x16_bf16 (8 waves =
// 2/SIMD so the accumulator fits in VGPRs; 16 waves would spill). Grid =
// NT x KT x SK; chunks of 32 m-rows double-buffered via glds.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using as3_bf16x4p = __attribute__((address_space(3))) bf16x4*;

using as1_cvp = const __attribute__((address_space(1))) void*;
using as3_vp = __attribute__((address_space(3))) void*;
__device__ __forceinline__ as1_cvp as_global(const void* p) {
  return (as1_cvp)(unsigned long long)(uintptr_t)p;
}
__device__ __forceinline__ as3_vp as_shared(void* p) {
  return (as3_vp)(unsigned int)(uintptr_t)p;
}

__device__ __forceinline__ bf16x8 tr16x8w(const char* p0, const char* p1) {
  bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (as3_bf16x4p)(const_cast<char*>(p0)));
  bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (as3_bf16x4p)(const_cast<char*>(p1)));
  return __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
}

constexpr int LBN = 256;   // n per block tile
constexpr int LBK = 256;   // k per block tile
constexpr int LWT = 512;   // 8 waves
// row stride: C_tile*2 bytes + 64 pad (row step = 16 banks: the 32-lane tr service group lands on 32 distinct banks)
constexpr int LROW = LBN * 2 + 64;

struct LwShape {
  long M;
  int N, K;
  int nt, kt, sk;
  int with_bias;
};

// LMT (m rows per staged chunk) is a template knob: 32 keeps 2 blocks/CU
// (74 KB LDS), 64 halves the per-chunk glds-latency stalls at 1 block/CU
// (147 KB) — the chunk loop is latency-bound, measured A/B picks it.
template <int LMT>
__global__ __launch_bounds__(LWT)
void linear_wgrad_kernel(const bf16* __restrict__ dy,   // (M, N)
                         const bf16* __restrict__ x,    // (M, K)
                         const bf16* __restrict__ zbuf,
                         float* __restrict__ dwacc,     // (N, K) zeroed
                         float* __restrict__ dbacc,     // (N,) zeroed | null
                         LwShape s) {
  int bid = blockIdx.x;
  const int ntk = s.nt * s.kt;
  const int tile = bid % ntk;
  const int sk = bid / ntk;
  const int n0 = (tile / s.kt) * LBN;
  const int k0 = (tile % s.kt) * LBK;

  // m range for this split-K block (chunk-aligned)
  const long chunks = (s.M + LMT - 1) / LMT;
  const long per = (chunks + s.sk - 1) / s.sk;
  const long c0 = sk * per;
  const long c1 = min(chunks, c0 + per);
  if (c0 >= c1) return;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int img_bytes = LMT * LROW;       // one operand image
  constexpr int buf_bytes = 2 * img_bytes;    // [dy][x]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wn = wave >> 1;   // 0..3: n strip (64 rows)
  const int wk = wave & 1;    // 0..1: k strip (128 cols)
  const bool live_n = n0 + wn * 64 < s.N;
  const bool live_k = k0 + wk * 128 < s.K;

  // per-lane tr-read base: granule of lane l covers channels
  // cbase + (g&1)*16 + 4*(m16&3) at row (g>>1)*8 + (m16>>2) (+ rr*4 + p0)
  const int g = lane >> 4;
  const int m16 = lane & 15;
  const int tr_row = (g >> 1) * 8 + (m16 >> 2);
  const int tr_chb = ((g & 1) * 16 + 4 * (m16 & 3)) * 2;
  const int tr_off = tr_row * LROW + tr_chb;

  auto stage = [&](long chunk, int buf) {
    char* dst = smem + buf * buf_bytes;
    const long m0 = chunk * LMT;
    // dy chunk -> row-major [m][LBN] (+pad); lane-linear glds, sources
    // walk global lines contiguously within each row
    for (int o = wave * 1024 + lane * 16; o < img_bytes; o += 8 * 1024) {
      const int m = o / LROW;
      const int w = o % LROW;
      const bf16* src = zbuf;
      if (w < LBN * 2) {
        const long mm = m0 + m;
        const int n = n0 + (w >> 1);
        if (mm < s.M && n < s.N) src = dy + mm * s.N + n;
      }
      __builtin_amdgcn_global_load_lds(as_global(src),
          as_shared(dst + (o - lane * 16)), 16, 0, 0);
    }
    char* dstx = dst + img_bytes;
    for (int o = wave * 1024 + lane * 16; o < img_bytes; o += 8 * 1024) {
      const int m = o / LROW;
      const int w = o % LROW;
      const bf16* src = zbuf;
      if (w < LBK * 2) {
        const long mm = m0 + m;
        const int k = k0 + (w >> 1);
        if (mm < s.M && k < s.K) src = x + mm * s.K + k;
      }
      __builtin_amdgcn_global_load_lds(as_global(src),
          as_shared(dstx + (o - lane * 16)), 16, 0, 0);
    }
  };

  float acc[8][16];
#pragma unroll
  for (int t = 0; t < 8; ++t)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[t][e] = 0.f;
  float dbsum[2] = {0.f, 0.f};
  const bool do_bias = s.with_bias && k0 == 0 && wk == 0 && live_n;

  // Depth-2 glds prefetch over a 3-buffer ring with COUNTED vmcnt across
  // RAW barriers (the chunk loop is glds-latency-bound; a vmcnt(0) drain
  // per chunk left it at half roofline). Per-wave glds per chunk is fixed
  // by the chunk map (the low-id waves cover one extra 1 KB piece per
  // operand), so the wait that leaves exactly chunk c+1 in flight is a
  // per-wave constant: iteration c = {wait until only c+1 outstanding;
  // raw barrier; issue c+2; compute c}. The barrier guarantees every wave
  // finished computing c-1, so buffer (c+2)%3 == (c-1)%3 is free.
  const int gper = 2 * ((img_bytes - wave * 1024 + 8191) / 8192);
  stage(c0, 0);
  if (c0 + 1 < c1) stage(c0 + 1, 1);

  for (long c = c0; c < c1; ++c) {
    const int buf = (int)((c - c0) % 3);
    if (c + 1 < c1) {
      if (gper == 6) {
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      }
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    if (c + 2 < c1) stage(c + 2, (int)((c + 2 - c0) % 3));
    const char* dyb = smem + buf * buf_bytes + tr_off;
    const char* xb = dyb + img_bytes;
    if (live_n) {
#pragma unroll
      for (int p0 = 0; p0 < LMT; p0 += 16) {
        bf16x8 af[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          const char* pa = dyb + p0 * LROW + (wn * 64 + i * 32) * 2;
          af[i] = tr16x8w(pa, pa + 4 * LROW);
        }
        if (do_bias) {
#pragma unroll
          for (int i = 0; i < 2; ++i)
#pragma unroll
            for (int e = 0; e < 8; ++e) dbsum[i] += (float)af[i][e];
        }
        if (live_k) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const char* pb = xb + p0 * LROW + (wk * 128 + j * 32) * 2;
            bf16x8 bf = tr16x8w(pb, pb + 4 * LROW);
#pragma unroll
            for (int i = 0; i < 2; ++i) {
              *reinterpret_cast<f32x16*>(acc[i * 4 + j]) =
                  __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                      af[i], bf, *reinterpret_cast<f32x16*>(acc[i * 4 + j]),
                      0, 0, 0);
            }
          }
        }
      }
    }
  }

  // epilogue: fp32 atomics. D layout: col = lane&31, row = (e&3) + 8*(e>>2)
  // + 4*(lane>>5).
  if (live_n && live_k) {
    const int kcol0 = k0 + wk * 128 + (lane & 31);
    const int nrow0 = n0 + wn * 64 + 4 * (lane >> 5);
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int kk = kcol0 + j * 32;
        if (kk >= s.K) continue;
#pragma unroll
        for (int e = 0; e < 16; ++e) {
          const int n = nrow0 + i * 32 + (e & 3) + 8 * (e >> 2);
          if (n < s.N) {
            atomicAdd(dwacc + (long)n * s.K + kk, acc[i * 4 + j][e]);
          }
        }
      }
  }
  if (do_bias) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int n = n0 + wn * 64 + i * 32 + (lane & 31);
      if (n < s.N) atomicAdd(dbacc + n, dbsum[i]);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> linear_wgrad(torch::Tensor dy, torch::Tensor x,
                                        bool with_bias) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 &&
              x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(dy.dim() == 2 && x.dim() == 2 && dy.size(0) == x.size(0));
  LwShape s;
  s.M = dy.size(0);
  s.N = (int)dy.size(1);
  s.K = (int)x.size(1);
  TORCH_CHECK(s.N % 8 == 0 && s.K % 8 == 0, "N/K must be 16B-packable");
  s.nt = (s.N + LBN - 1) / LBN;
  s.kt = (s.K + LBK - 1) / LBK;
  s.with_bias = with_bias ? 1 : 0;
  const int LMT = 32;  // 3-buffer prefetch ring (110 KB LDS)
  const long chunks = (s.M + LMT - 1) / LMT;
  // 2 blocks/CU (LDS 74KB) so one block's staging stalls hide under the
  // other's compute — the chunk loop is glds-latency-bound
  static const char* lwgs = getenv("NVS3D_LWG_SK");
  const int target = lwgs ? atoi(lwgs) : 512;
  s.sk = (int)std::max(1L, std::min(chunks,
                                    (long)(target / (s.nt * s.kt) + 1)));

  auto opts = x.options().dtype(torch::kFloat);
  auto dw = torch::zeros({s.N, s.K}, opts);
  torch::Tensor db;
  if (with_bias) db = torch::zeros({s.N}, opts);

  static torch::Tensor zbuf;
  if (!zbuf.defined() || zbuf.device() != x.device()) {
    zbuf = torch::zeros({64}, x.options());
  }
  const size_t lds = 3 * 2 * (size_t)(LMT * LROW);
  auto stream = at::hip::getCurrentHIPStream();
  auto kfn = linear_wgrad_kernel<32>;
  hipLaunchKernelGGL(kfn,
      dim3(s.sk * s.nt * s.kt), dim3(LWT), lds, stream,
      reinterpret_cast<const bf16*>(dy.data_ptr()),
      reinterpret_cast<const bf16*>(x.data_ptr()),
      reinterpret_cast<const bf16*>(zbuf.data_ptr()),
      dw.data_ptr<float>(),
      with_bias ? db.data_ptr<float>() : nullptr, s);
  if (with_bias) return {dw, db};
  return {dw};
}
