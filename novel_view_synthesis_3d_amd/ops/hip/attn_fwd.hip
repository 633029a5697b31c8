// Flash-style MFMA attention forward — CDNA4 gfx950. SURVEY.md §2.4 K7.
//
// The reference's attention is flax nn.dot_product_attention over H*W
// tokens (/root/reference/model/xunet.py:94-127): softmax(QK^T/sqrt(d))V,
// no mask, no output projection; 4 calls per attn block (self x2 frames,
// cross x2 directions — cross = same kernel, kv from the other frame).
// Shapes here: L in {256, 1024}, d in {16, 128, 256}, heads 4, bf16.
//
// Structure: one workgroup = 4 waves = 64 q rows of one (batch, head);
// each wave owns 16 q rows. Per KV-tile (64 keys):
//   K staged [64][d] in LDS (XOR-swizzled rows), V staged TRANSPOSED
//   [d][64] (so the PV B-fragment reads are contiguous 16B), then
//   S = QK^T via mfma_f32_16x16x32_bf16 (Q pre-scaled by 1/sqrt(d), held
//   in registers in A-fragment layout), online softmax with per-lane
//   4-row stats + 16-lane shfl_xor reduction, P staged through a per-wave
//   LDS tile in A-fragment layout, O += P.V accumulated in f32 fragments.
// Output + logsumexp written at the end (lse feeds the GEMM-recompute
// backward in ops/hip_ops.py).
//
// Numerics: softmax and accumulation in fp32; exp via __expf (args <= 0).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int QB = 16;     // q rows per wave
constexpr int KB = 64;     // keys per tile
constexpr int WAVES = 4;   // q rows per block = 64
constexpr float NEG_INF = -1e30f;

struct AttnShape {
  int B, L, H, D;
  int Lk;               // key sequence length (same L here)
  float scale;
  // strides in ELEMENTS (d contiguous, head stride = D): lets q/k/v be
  // views into one fused QKV projection (B, L, 3C) without copies
  long sqb, sqt;        // q batch, token
  long skb, skt;        // k
  long svb, svt;        // v
  int kv_swap;          // batch b attends to k/v of batch b^1 (the model's
                        // cross-FRAME attention with both frames batched)
};

__device__ __forceinline__ int swz_row(int row, int byte_in_row,
                                       int row_bytes) {
  return (row * row_bytes + byte_in_row) ^ ((row & 7) << 4);
}

template <int D>
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const bf16* __restrict__ q,
                     const bf16* __restrict__ k,
                     const bf16* __restrict__ v,
                     bf16* __restrict__ out,
                     float* __restrict__ lse,   // (B, L, H)
                     AttnShape s) {
  constexpr int DC = (D + 31) / 32;       // 32-wide d chunks for QK^T
  constexpr int DF = (D + 15) / 16;       // 16-wide d frags for PV/O
  // LDS: K [KB][D] swz | V_T [D][KB] swz | P (4 waves)[16][KB] swz
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* ldsK = smem;                       // KB*D*2
  char* ldsV = ldsK + KB * D * 2;          // D*KB*2
  char* ldsP = ldsV + D * KB * 2;          // WAVES*16*KB*2

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // block -> (b, h, qtile)
  const int qtiles = s.L / (QB * WAVES);
  int bid = blockIdx.x;
  const int qt = bid % qtiles; bid /= qtiles;
  const int head = bid % s.H;
  const int b = bid / s.H;

  const int HD = s.H * s.D;  // output layout stays contiguous (B,L,H,D)
  const int bkv = s.kv_swap ? (b ^ 1) : b;
  const bf16* qbase = q + (long)b * s.sqb + head * s.D;
  const bf16* kbase = k + (long)bkv * s.skb + head * s.D;
  const bf16* vbase = v + (long)bkv * s.svb + head * s.D;

  // ---- load Q into A-fragment registers, pre-scaled ----
  // chunk dc: lane l holds Q[q0 + (l&15)][dc*32 + (l>>4)*8 .. +8]
  const int q0 = qt * QB * WAVES + wave * QB;
  const int qrow = q0 + (lane & 15);
  bf16x8 qf[DC];
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
    const int off = dc * 32 + (lane >> 4) * 8;
    if (off < s.D) {
      Pack<bf16, 8> p = pload<bf16, 8>(qbase + (long)qrow * s.sqt + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qf[dc][j] = (__bf16)__float2bfloat16(
            __bfloat162float(p.v[j]) * s.scale);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[dc][j] = (__bf16)0.0f;
    }
  }

  float m[4], lsum[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = NEG_INF; lsum[r] = 0.f; }
  float o[DF][4];
#pragma unroll
  for (int f = 0; f < DF; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) o[f][r] = 0.f;

  const int ntiles = s.Lk / KB;
  for (int t = 0; t < ntiles; ++t) {
    // ---- stage K [KB][D] and V^T [D][KB] ----
    __syncthreads();  // previous tile's reads done
    {
      // K: KB*D/8 packs over 256 threads
      constexpr int PACKS = KB * D / 8;
#pragma unroll
      for (int it = 0; it < (PACKS + 255) / 256; ++it) {
        const int p = tid + it * 256;
        if (p < PACKS) {
          const int key = p / (D / 8);
          const int d0 = (p % (D / 8)) * 8;
          Pack<bf16, 8> kv8 =
              pload<bf16, 8>(kbase + (long)(t * KB + key) * s.skt + d0);
          *reinterpret_cast<Pack<bf16, 8>*>(
              ldsK + swz_row(key, d0 * 2, D * 2)) = kv8;
          // V: same source geometry, transposed scatter into [D][KB]
          Pack<bf16, 8> vv8 =
              pload<bf16, 8>(vbase + (long)(t * KB + key) * s.svt + d0);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            *reinterpret_cast<bf16*>(
                ldsV + swz_row(d0 + j, key * 2, KB * 2)) = vv8.v[j];
          }
        }
      }
    }
    __syncthreads();

    // ---- S = Q K^T over this tile: 4 col-fragments of 16 keys ----
    f32x4 sf[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      sf[c] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
        const int key = c * 16 + (lane & 15);
        const int off = dc * 32 + (lane >> 4) * 8;
        bf16x8 kf;
        if (off < s.D) {
          kf = *reinterpret_cast<const bf16x8*>(
              ldsK + swz_row(key, off * 2, D * 2));
        } else {
          kf = bf16x8{};
        }
        // A = Q fragment (rows = q), B = K^T fragment (cols = keys):
        // S[q][key] lands in C layout with rows = q, as the softmax needs.
        sf[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[dc], kf, sf[c],
                                                        0, 0, 0);
      }
    }

    // ---- online softmax on sf (rows = q) ----
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = fmaxf(fmaxf(sf[0][r], sf[1][r]),
                       fmaxf(sf[2][r], sf[3][r]));
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1) {
        mx = fmaxf(mx, __shfl_xor(mx, w, 64));
      }
      const float mn = fmaxf(m[r], mx);
      alpha[r] = __expf(m[r] - mn);
      m[r] = mn;
      float ps = 0.f;
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const float e = __expf(sf[c][r] - mn);
        sf[c][r] = e;
        ps += e;
      }
#pragma unroll
      for (int w = 8; w >= 1; w >>= 1) {
        ps += __shfl_xor(ps, w, 64);
      }
      lsum[r] = lsum[r] * alpha[r] + ps;
#pragma unroll
      for (int f = 0; f < DF; ++f) o[f][r] *= alpha[r];
    }

    // ---- P -> per-wave LDS tile [16][KB] (A-frag layout for PV) ----
    char* pw = ldsP + wave * (QB * KB * 2);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
        const int col = c * 16 + (lane & 15);
        *reinterpret_cast<bf16*>(pw + swz_row(row, col * 2, KB * 2)) =
            __float2bfloat16(sf[c][r]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);

    // ---- O += P V ----
#pragma unroll
    for (int kc = 0; kc < KB / 32; ++kc) {
      const int koff = kc * 32 + (lane >> 4) * 8;
      bf16x8 pf = *reinterpret_cast<const bf16x8*>(
          pw + swz_row(lane & 15, koff * 2, KB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            ldsV + swz_row(f * 16 + (lane & 15), koff * 2, KB * 2));
        *reinterpret_cast<f32x4*>(o[f]) =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pf, vf, *reinterpret_cast<f32x4*>(o[f]), 0, 0, 0);
      }
    }
  }

  // ---- normalize + write out (staged through LDS for coalescing) ----
  __syncthreads();
  char* ow = ldsK + wave * (QB * D * 2);  // reuse K buffer per wave
#pragma unroll
  for (int f = 0; f < DF; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      const int col = f * 16 + (lane & 15);
      const float val = o[f][r] / lsum[r];
      *reinterpret_cast<bf16*>(ow + (row * D + col) * 2) =
          __float2bfloat16(val);
    }
  }
  // lse: one lane per row-group writes 4 rows
  if ((lane & 15) == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = q0 + (lane >> 4) * 4 + r;
      lse[((long)b * s.L + row) * s.H + head] = m[r] + __logf(lsum[r]);
    }
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  // coalesced store: 16 rows x D/8 packs per wave
  {
    constexpr int PACKS = QB * D / 8;
#pragma unroll
    for (int it = 0; it < (PACKS + 63) / 64; ++it) {
      const int p = lane + it * 64;
      if (p < PACKS) {
        const int row = p / (D / 8);
        const int d0 = (p % (D / 8)) * 8;
        Pack<bf16, 8> pv = *reinterpret_cast<Pack<bf16, 8>*>(
            ow + (row * D + d0) * 2);
        pstore<bf16, 8>(out + ((long)(q0 + row)) * HD
                        + (long)b * s.L * HD + head * s.D + d0, pv);
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool kv_swap) {
  // q/k/v: (B, L, H, D) bf16 — may be strided VIEWS (e.g. into a fused
  // (B, L, 3C) QKV projection) as long as d is contiguous and the head
  // stride is D.
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  auto check_strides = [](const torch::Tensor& t) {
    TORCH_CHECK(t.stride(3) == 1 && t.stride(2) == t.size(3),
                "attn operand needs contiguous (head, d) tail");
  };
  check_strides(q); check_strides(k); check_strides(v);
  const int B = q.size(0), L = q.size(1), H = q.size(2), D = q.size(3);
  TORCH_CHECK(k.size(1) == L, "self/cross attention has equal q/kv length");
  TORCH_CHECK(L % 64 == 0, "L must be a multiple of 64");
  TORCH_CHECK(D == 16 || D == 32 || D == 64 || D == 128 || D == 256,
              "unsupported head dim ", D);

  AttnShape s;
  s.B = B; s.L = L; s.H = H; s.D = D; s.Lk = k.size(1);
  s.scale = 1.0f / std::sqrt((float)D);
  s.sqb = q.stride(0); s.sqt = q.stride(1);
  s.skb = k.stride(0); s.skt = k.stride(1);
  s.svb = v.stride(0); s.svt = v.stride(1);
  s.kv_swap = kv_swap ? 1 : 0;
  TORCH_CHECK(!kv_swap || B % 2 == 0, "kv_swap pairs batches (b, b^1)");

  auto out = torch::empty_like(q);
  auto lse = torch::empty({B, L, H}, q.options().dtype(torch::kFloat));
  const int grid = B * H * (L / 64);
  auto stream = at::hip::getCurrentHIPStream();

#define LAUNCH_D(DV)                                                     \
  {                                                                      \
    const size_t lds = (size_t)64 * DV * 2 * 2 + 4 * 16 * 64 * 2;        \
    hipLaunchKernelGGL((attn_fwd_kernel<DV>), dim3(grid), dim3(256),     \
        lds, stream,                                                     \
        reinterpret_cast<const bf16*>(q.data_ptr()),                     \
        reinterpret_cast<const bf16*>(k.data_ptr()),                     \
        reinterpret_cast<const bf16*>(v.data_ptr()),                     \
        reinterpret_cast<bf16*>(out.data_ptr()),                         \
        lse.data_ptr<float>(), s);                                       \
  }
  switch (D) {
    case 16: LAUNCH_D(16); break;
    case 32: LAUNCH_D(32); break;
    case 64: LAUNCH_D(64); break;
    case 128: LAUNCH_D(128); break;
    case 256: LAUNCH_D(256); break;
  }
#undef LAUNCH_D
  return {out, lse};
}
