// Flash-style MFMA attention backward — CDNA4 gfx950. SURVEY.md §2.4 K7 bwd
// (reference /root/reference/model/xunet.py:103 under train.py:70).
//
// Replaces the round-1 GEMM-recompute path (rocBLAS batched matmuls that
// materialized (B,H,L,L) S/P/dP through HBM). Tile-wise recompute from
// q/k/lse instead — nothing L x L ever touches HBM:
//
//   delta_i = sum_d do[i,d] * o[i,d]                      (attn_delta)
//   S       = scale * q k^T;  P = exp(S - lse_i)
//   dv_j    = sum_i P[i,j] do_i                           (attn_bwd_dkv)
//   ds      = P * (do v^T - delta_i) * scale
//   dk_j    = sum_i ds[i,j] q_i                           (attn_bwd_dkv)
//   dq_i    = sum_j ds[i,j] k_j                           (attn_bwd_dq)
//
// Two kernels so neither needs cross-block atomics: dkv blocks own a
// 128-key tile (dk/dv in registers, written once), dq blocks own a 64-row
// q tile. Both recompute S tile-wise; the extra MFMA work is far cheaper
// than the old path's HBM round trips. Staging mirrors attn_fwd.hip:
// row-major XOR-swizzled LDS tiles for row-fragment reads, transposed
// scatter tiles where a fragment needs the contraction dim contiguous,
// per-wave LDS tiles to convert C-layout P/ds into A-fragments.
//
// Supported: d in {16, 32, 64, 128}, L % 128 == 0 (q and kv); the d=256 /
// short-L shapes keep the GEMM-recompute fallback (ops/hip_ops.py).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

struct ABwdShape {
  int B, L, H, D;
  int Lk;
  float scale;
  long sqb, sqt;   // q strides (elements); head stride = D, d contiguous
  long skb, skt;   // k
  long svb, svt;   // v
  int kv_swap;     // batch b's queries attend batch b^1's k/v (batched
                   // cross-frame attention; see attn_fwd.hip)
};

__device__ __forceinline__ int swz_row(int row, int byte_in_row,
                                       int row_bytes) {
  return (row * row_bytes + byte_in_row) ^ ((row & 7) << 4);
}

// ---------------------------------------------------------------------------
// delta[b,l,h] = sum_d do * o   (both (B,L,H,D) bf16 contiguous)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void attn_delta_kernel(const bf16* __restrict__ dout,
                       const bf16* __restrict__ o,
                       float* __restrict__ delta, int rows, int D) {
  // one 64-lane wave handles (64*8)/D rows... keep it simple: each wave
  // processes rows in D/8-lane groups: lane li covers 8 d-elements.
  const int lanes_per_row = D / 8;
  const int rows_per_block = 256 / lanes_per_row;
  const long r0 = (long)blockIdx.x * rows_per_block;
  const int li = threadIdx.x % lanes_per_row;
  const long row = r0 + threadIdx.x / lanes_per_row;
  if (row >= rows) return;
  const long off = row * D + li * 8;
  Pack<bf16, 8> pd = pload<bf16, 8>(dout + off);
  Pack<bf16, 8> po = pload<bf16, 8>(o + off);
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) s += to_f32(pd.v[j]) * to_f32(po.v[j]);
  // reduce across the lanes_per_row group (power of two, <= 32)
  for (int w = lanes_per_row >> 1; w >= 1; w >>= 1) {
    s += __shfl_xor(s, w, 64);
  }
  if (li == 0) delta[row] = s;
}

// ---------------------------------------------------------------------------
// dkv kernel: one block = (b, h, 128-key tile); 8 waves x 16 keys.
// Loops over 64-row q tiles staging Q/DO row-major (+ transposed copies for
// the contraction-over-i fragments).
// ---------------------------------------------------------------------------
constexpr int KVB = 128;   // keys per block
constexpr int QTB = 64;    // q rows per tile

template <int D>
__global__ __launch_bounds__(512)
void attn_bwd_dkv_kernel(const bf16* __restrict__ q,
                         const bf16* __restrict__ k,
                         const bf16* __restrict__ v,
                         const bf16* __restrict__ dout,  // (B,L,H,D) contig
                         const float* __restrict__ lse,  // (B,L,H)
                         const float* __restrict__ delta,
                         bf16* __restrict__ dk,          // (B,Lk,H,D) contig
                         bf16* __restrict__ dv,
                         ABwdShape s) {
  constexpr int DC = (D + 31) / 32;
  constexpr int DF = (D + 15) / 16;
  // LDS: 2 x { Q [64][D] | DO [64][D] | Q^T [D][64] | DO^T [D][64] |
  // lse/delta (64 f32 each) } double-buffered (the next q-tile stages
  // under this one's MFMAs) | per-wave P/ds tile (8)[16][64]
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int QSET = 4 * QTB * D * 2 + 2 * QTB * 4;  // one buffer set
  char* ldsP = smem + 2 * QSET;                 // 8 waves x 16 x 64 bf16

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  int bid = blockIdx.x;
  const int kvtiles = s.Lk / KVB;
  const int kvt = bid % kvtiles; bid /= kvtiles;
  const int head = bid % s.H;
  const int b = bid / s.H;

  // this block owns keys of batch b; under kv_swap those keys are attended
  // by batch b^1's queries, so q/do/lse/delta come from b^1
  const int bq = s.kv_swap ? (b ^ 1) : b;
  const bf16* qbase = q + (long)bq * s.sqb + head * s.D;
  const bf16* kbase = k + (long)b * s.skb + head * s.D;
  const bf16* vbase = v + (long)b * s.svb + head * s.D;
  const int HD = s.H * s.D;
  const bf16* dobase = dout + (long)bq * s.L * HD + head * s.D;
  const float* lsebase = lse + (long)bq * s.L * s.H + head;
  const float* delbase = delta + (long)bq * s.L * s.H + head;

  // ---- preload this wave's K and V rows as A-fragments (rows = keys) ----
  // lane: row = k0 + (l&15), k = dc*32 + (l>>4)*8 .. +8.  K prescaled.
  const int k0 = kvt * KVB + wave * 16;
  const int krow = k0 + (lane & 15);
  bf16x8 ka[DC], va[DC];
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
    const int off = dc * 32 + (lane >> 4) * 8;
    if (off < s.D) {
      Pack<bf16, 8> pk = pload<bf16, 8>(kbase + (long)krow * s.skt + off);
      Pack<bf16, 8> pv = pload<bf16, 8>(vbase + (long)krow * s.svt + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ka[dc][j] = (__bf16)__float2bfloat16(to_f32(pk.v[j]) * s.scale);
        va[dc][j] = (__bf16)pv.v[j];
      }
    } else {
      ka[dc] = bf16x8{};
      va[dc] = bf16x8{};
    }
  }

  float dvacc[DF][4], dkacc[DF][4];
#pragma unroll
  for (int f = 0; f < DF; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) { dvacc[f][r] = 0.f; dkacc[f][r] = 0.f; }

  char* pw = ldsP + wave * (16 * QTB * 2);

  // stage q-tile t into buffer set `buf` (row-major swz + transposed
  // scatter + lse/delta)
  auto stage_qtile = [&](int t, int buf) {
    char* ldsQ = smem + buf * QSET;
    char* ldsDO = ldsQ + QTB * D * 2;
    char* ldsQT = ldsDO + QTB * D * 2;
    char* ldsDOT = ldsQT + D * QTB * 2;
    float* ldsLse = reinterpret_cast<float*>(ldsDOT + D * QTB * 2);
    float* ldsDel = ldsLse + QTB;
    constexpr int PACKS = QTB * D / 8;
#pragma unroll
    for (int it = 0; it < (PACKS + 511) / 512; ++it) {
      const int p = tid + it * 512;
      if (p < PACKS) {
        const int row = p / (D / 8);
        const int d0 = (p % (D / 8)) * 8;
        Pack<bf16, 8> pq =
            pload<bf16, 8>(qbase + (long)(t * QTB + row) * s.sqt + d0);
        *reinterpret_cast<Pack<bf16, 8>*>(
            ldsQ + swz_row(row, d0 * 2, D * 2)) = pq;
        Pack<bf16, 8> pdo =
            pload<bf16, 8>(dobase + (long)(t * QTB + row) * HD + d0);
        *reinterpret_cast<Pack<bf16, 8>*>(
            ldsDO + swz_row(row, d0 * 2, D * 2)) = pdo;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          *reinterpret_cast<bf16*>(
              ldsQT + swz_row(d0 + j, row * 2, QTB * 2)) = pq.v[j];
          *reinterpret_cast<bf16*>(
              ldsDOT + swz_row(d0 + j, row * 2, QTB * 2)) = pdo.v[j];
        }
      }
    }
    if (tid < QTB) {
      ldsLse[tid] = lsebase[(long)(t * QTB + tid) * s.H];
      ldsDel[tid] = delbase[(long)(t * QTB + tid) * s.H];
    }
  };

  const int qtiles = s.L / QTB;
  stage_qtile(0, 0);
  __syncthreads();
  for (int t = 0; t < qtiles; ++t) {
    const int buf = t & 1;
    if (t + 1 < qtiles) stage_qtile(t + 1, buf ^ 1);
    const char* ldsQ = smem + buf * QSET;
    const char* ldsDO = ldsQ + QTB * D * 2;
    const char* ldsQT = ldsDO + QTB * D * 2;
    const char* ldsDOT = ldsQT + D * QTB * 2;
    const float* ldsLse = reinterpret_cast<const float*>(
        ldsDOT + D * QTB * 2);
    const float* ldsDel = ldsLse + QTB;

    // ---- S^T[j][i] (rows = keys, cols = q rows), P^T = exp(S^T - lse_i)
    f32x4 st[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      st[c] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
        const int qi = c * 16 + (lane & 15);
        const int off = dc * 32 + (lane >> 4) * 8;
        bf16x8 qf = off < s.D
            ? *reinterpret_cast<const bf16x8*>(
                  ldsQ + swz_row(qi, off * 2, D * 2))
            : bf16x8{};
        st[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka[dc], qf, st[c],
                                                        0, 0, 0);
      }
    }
    // dp^T[j][i] = sum_d V[j][d] do[i][d]
    f32x4 dpt[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      dpt[c] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
        const int qi = c * 16 + (lane & 15);
        const int off = dc * 32 + (lane >> 4) * 8;
        bf16x8 df = off < s.D
            ? *reinterpret_cast<const bf16x8*>(
                  ldsDO + swz_row(qi, off * 2, D * 2))
            : bf16x8{};
        dpt[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va[dc], df, dpt[c],
                                                         0, 0, 0);
      }
    }

    // ---- P^T -> per-wave LDS tile [16 keys][64 i] (A-frag layout) ----
    // C layout: col i = c*16 + (lane&15), row key = (lane>>4)*4 + r.
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qi = c * 16 + (lane & 15);
      const float l_i = ldsLse[qi];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(st[c][r] - l_i);
        st[c][r] = p;  // keep P^T for ds^T below
        *reinterpret_cast<bf16*>(
            pw + swz_row((lane >> 4) * 4 + r, qi * 2, QTB * 2)) =
            __float2bfloat16(p);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);

    // ---- dv += P^T . DO  (contraction over i via DO^T tile) ----
#pragma unroll
    for (int kc = 0; kc < QTB / 32; ++kc) {
      const int koff = kc * 32 + (lane >> 4) * 8;
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          pw + swz_row(lane & 15, koff * 2, QTB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 dof = *reinterpret_cast<const bf16x8*>(
            ldsDOT + swz_row(f * 16 + (lane & 15), koff * 2, QTB * 2));
        *reinterpret_cast<f32x4*>(dvacc[f]) =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pa, dof, *reinterpret_cast<f32x4*>(dvacc[f]), 0, 0, 0);
      }
    }

    // ---- ds^T = P^T * (dp^T - delta_i) * scale -> same wave tile ----
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qi = c * 16 + (lane & 15);
      const float d_i = ldsDel[qi];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float ds = st[c][r] * (dpt[c][r] - d_i) * s.scale;
        *reinterpret_cast<bf16*>(
            pw + swz_row((lane >> 4) * 4 + r, qi * 2, QTB * 2)) =
            __float2bfloat16(ds);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);

    // ---- dk += ds^T . Q  (contraction over i via Q^T tile) ----
#pragma unroll
    for (int kc = 0; kc < QTB / 32; ++kc) {
      const int koff = kc * 32 + (lane >> 4) * 8;
      bf16x8 dsa = *reinterpret_cast<const bf16x8*>(
          pw + swz_row(lane & 15, koff * 2, QTB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
            ldsQT + swz_row(f * 16 + (lane & 15), koff * 2, QTB * 2));
        *reinterpret_cast<f32x4*>(dkacc[f]) =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                dsa, qtf, *reinterpret_cast<f32x4*>(dkacc[f]), 0, 0, 0);
      }
    }
    // next tile's staging writes (buf^1) visible + this buf's reads done
    // before t+2 overwrites it
    __syncthreads();
  }

  // ---- write dk/dv (C layout row = key, col = d) through LDS ----
  // reuse the smem front: per wave [16][D]
  char* ow = smem + wave * (16 * D * 2);
#pragma unroll
  for (int f = 0; f < DF; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      const int col = f * 16 + (lane & 15);
      *reinterpret_cast<bf16*>(ow + (row * D + col) * 2) =
          __float2bfloat16(dvacc[f][r]);
    }
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  {
    constexpr int PACKS = 16 * D / 8;
#pragma unroll
    for (int it = 0; it < (PACKS + 63) / 64; ++it) {
      const int p = lane + it * 64;
      if (p < PACKS) {
        const int row = p / (D / 8);
        const int d0 = (p % (D / 8)) * 8;
        Pack<bf16, 8> pv = *reinterpret_cast<Pack<bf16, 8>*>(
            ow + (row * D + d0) * 2);
        pstore<bf16, 8>(dv + ((long)b * s.Lk + k0 + row) * HD
                        + head * s.D + d0, pv);
      }
    }
  }
  __syncthreads();
#pragma unroll
  for (int f = 0; f < DF; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      const int col = f * 16 + (lane & 15);
      *reinterpret_cast<bf16*>(ow + (row * D + col) * 2) =
          __float2bfloat16(dkacc[f][r]);
    }
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  {
    constexpr int PACKS = 16 * D / 8;
#pragma unroll
    for (int it = 0; it < (PACKS + 63) / 64; ++it) {
      const int p = lane + it * 64;
      if (p < PACKS) {
        const int row = p / (D / 8);
        const int d0 = (p % (D / 8)) * 8;
        Pack<bf16, 8> pv = *reinterpret_cast<Pack<bf16, 8>*>(
            ow + (row * D + d0) * 2);
        pstore<bf16, 8>(dk + ((long)b * s.Lk + k0 + row) * HD
                        + head * s.D + d0, pv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dq kernel: one block = (b, h, 64-row q tile); 4 waves x 16 rows.
// Loops over 64-key kv tiles staging K row-major + K^T + V row-major.
// ---------------------------------------------------------------------------
constexpr int KB2 = 64;

template <int D>
__global__ __launch_bounds__(256)
void attn_bwd_dq_kernel(const bf16* __restrict__ q,
                        const bf16* __restrict__ k,
                        const bf16* __restrict__ v,
                        const bf16* __restrict__ dout,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,
                        bf16* __restrict__ dq,   // (B,L,H,D) contig
                        ABwdShape s) {
  constexpr int DC = (D + 31) / 32;
  constexpr int DF = (D + 15) / 16;
  // LDS: K [KB2][D] | K^T [D][KB2] | V [KB2][D] | per-wave ds (4)[16][KB2]
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* ldsK = smem;
  char* ldsKT = ldsK + KB2 * D * 2;
  char* ldsV = ldsKT + D * KB2 * 2;
  char* ldsS = ldsV + KB2 * D * 2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  int bid = blockIdx.x;
  const int qtiles = s.L / QTB;
  const int qt = bid % qtiles; bid /= qtiles;
  const int head = bid % s.H;
  const int b = bid / s.H;

  // this block owns queries of batch b; under kv_swap they attend b^1's k/v
  const int bkv = s.kv_swap ? (b ^ 1) : b;
  const bf16* qbase = q + (long)b * s.sqb + head * s.D;
  const bf16* kbase = k + (long)bkv * s.skb + head * s.D;
  const bf16* vbase = v + (long)bkv * s.svb + head * s.D;
  const int HD = s.H * s.D;
  const bf16* dobase = dout + (long)b * s.L * HD + head * s.D;

  // ---- per-wave Q (prescaled) and DO fragments in registers ----
  const int q0 = qt * QTB + wave * 16;
  const int qrow = q0 + (lane & 15);
  bf16x8 qf[DC], dof[DC];
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
    const int off = dc * 32 + (lane >> 4) * 8;
    if (off < s.D) {
      Pack<bf16, 8> pq = pload<bf16, 8>(qbase + (long)qrow * s.sqt + off);
      Pack<bf16, 8> pd = pload<bf16, 8>(dobase + (long)qrow * HD + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qf[dc][j] = (__bf16)__float2bfloat16(to_f32(pq.v[j]) * s.scale);
        dof[dc][j] = (__bf16)pd.v[j];
      }
    } else {
      qf[dc] = bf16x8{};
      dof[dc] = bf16x8{};
    }
  }
  // lse/delta for this lane's 4 rows (C-layout rows)
  float lse_r[4], del_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + (lane >> 4) * 4 + r;
    lse_r[r] = lse[((long)b * s.L + row) * s.H + head];
    del_r[r] = delta[((long)b * s.L + row) * s.H + head];
  }

  float dqacc[DF][4];
#pragma unroll
  for (int f = 0; f < DF; ++f)
#pragma unroll
    for (int r = 0; r < 4; ++r) dqacc[f][r] = 0.f;

  char* sw = ldsS + wave * (16 * KB2 * 2);

  const int ntiles = s.Lk / KB2;
  for (int t = 0; t < ntiles; ++t) {
    __syncthreads();
    {
      constexpr int PACKS = KB2 * D / 8;
#pragma unroll
      for (int it = 0; it < (PACKS + 255) / 256; ++it) {
        const int p = tid + it * 256;
        if (p < PACKS) {
          const int key = p / (D / 8);
          const int d0 = (p % (D / 8)) * 8;
          Pack<bf16, 8> pk =
              pload<bf16, 8>(kbase + (long)(t * KB2 + key) * s.skt + d0);
          *reinterpret_cast<Pack<bf16, 8>*>(
              ldsK + swz_row(key, d0 * 2, D * 2)) = pk;
          Pack<bf16, 8> pv =
              pload<bf16, 8>(vbase + (long)(t * KB2 + key) * s.svt + d0);
          *reinterpret_cast<Pack<bf16, 8>*>(
              ldsV + swz_row(key, d0 * 2, D * 2)) = pv;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            *reinterpret_cast<bf16*>(
                ldsKT + swz_row(d0 + j, key * 2, KB2 * 2)) = pk.v[j];
          }
        }
      }
    }
    __syncthreads();

    // ---- S (rows = q), P = exp(S - lse); dp (rows = q); ds ----
    f32x4 sf[4], dp[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      sf[c] = f32x4{0.f, 0.f, 0.f, 0.f};
      dp[c] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
        const int key = c * 16 + (lane & 15);
        const int off = dc * 32 + (lane >> 4) * 8;
        bf16x8 kf, vf;
        if (off < s.D) {
          kf = *reinterpret_cast<const bf16x8*>(
              ldsK + swz_row(key, off * 2, D * 2));
          vf = *reinterpret_cast<const bf16x8*>(
              ldsV + swz_row(key, off * 2, D * 2));
        } else {
          kf = bf16x8{};
          vf = bf16x8{};
        }
        sf[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[dc], kf, sf[c],
                                                        0, 0, 0);
        dp[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[dc], vf, dp[c],
                                                        0, 0, 0);
      }
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(sf[c][r] - lse_r[r]);
        const float ds = p * (dp[c][r] - del_r[r]) * s.scale;
        *reinterpret_cast<bf16*>(
            sw + swz_row((lane >> 4) * 4 + r, (c * 16 + (lane & 15)) * 2,
                         KB2 * 2)) = __float2bfloat16(ds);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);

    // ---- dq += ds . K  (contraction over keys via K^T tile) ----
#pragma unroll
    for (int kc = 0; kc < KB2 / 32; ++kc) {
      const int koff = kc * 32 + (lane >> 4) * 8;
      bf16x8 dsa = *reinterpret_cast<const bf16x8*>(
          sw + swz_row(lane & 15, koff * 2, KB2 * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 ktf = *reinterpret_cast<const bf16x8*>(
            ldsKT + swz_row(f * 16 + (lane & 15), koff * 2, KB2 * 2));
        *reinterpret_cast<f32x4*>(dqacc[f]) =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                dsa, ktf, *reinterpret_cast<f32x4*>(dqacc[f]), 0, 0, 0);
      }
    }
  }

  // ---- write dq through LDS ----
  __syncthreads();
  char* ow = ldsK + wave * (16 * D * 2);
#pragma unroll
  for (int f = 0; f < DF; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = (lane >> 4) * 4 + r;
      const int col = f * 16 + (lane & 15);
      *reinterpret_cast<bf16*>(ow + (row * D + col) * 2) =
          __float2bfloat16(dqacc[f][r]);
    }
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  {
    constexpr int PACKS = 16 * D / 8;
#pragma unroll
    for (int it = 0; it < (PACKS + 63) / 64; ++it) {
      const int p = lane + it * 64;
      if (p < PACKS) {
        const int row = p / (D / 8);
        const int d0 = (p % (D / 8)) * 8;
        Pack<bf16, 8> pv = *reinterpret_cast<Pack<bf16, 8>*>(
            ow + (row * D + d0) * 2);
        pstore<bf16, 8>(dq + ((long)b * s.L + q0 + row) * HD
                        + head * s.D + d0, pv);
      }
    }
  }
}

ABwdShape make_shape(const torch::Tensor& q, const torch::Tensor& k,
                     const torch::Tensor& v) {
  ABwdShape s;
  s.B = q.size(0); s.L = q.size(1); s.H = q.size(2); s.D = q.size(3);
  s.Lk = k.size(1);
  s.scale = 1.0f / std::sqrt((float)s.D);
  s.sqb = q.stride(0); s.sqt = q.stride(1);
  s.skb = k.stride(0); s.skt = k.stride(1);
  s.svb = v.stride(0); s.svt = v.stride(1);
  return s;
}

}  // namespace

torch::Tensor attn_delta(torch::Tensor dout, torch::Tensor o) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && o.is_contiguous());
  const int B = dout.size(0), L = dout.size(1), H = dout.size(2),
            D = dout.size(3);
  TORCH_CHECK(D % 8 == 0 && D <= 512);
  auto delta = torch::empty({B, L, H}, dout.options().dtype(torch::kFloat));
  const long rows = (long)B * L * H;
  const int rows_per_block = 256 / (D / 8);
  const int grid = (int)((rows + rows_per_block - 1) / rows_per_block);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_delta_kernel, dim3(grid), dim3(256), 0, stream,
      reinterpret_cast<const bf16*>(dout.data_ptr()),
      reinterpret_cast<const bf16*>(o.data_ptr()),
      delta.data_ptr<float>(), (int)rows, D);
  return delta;
}

std::vector<torch::Tensor> attn_bwd_fused(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor dout,
                                          torch::Tensor lse,
                                          torch::Tensor delta, bool kv_swap) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  auto check_strides = [](const torch::Tensor& t) {
    TORCH_CHECK(t.stride(3) == 1 && t.stride(2) == t.size(3),
                "attn operand needs contiguous (head, d) tail");
  };
  check_strides(q); check_strides(k); check_strides(v);
  TORCH_CHECK(dout.is_contiguous() && lse.is_contiguous()
              && delta.is_contiguous());
  ABwdShape s = make_shape(q, k, v);
  s.kv_swap = kv_swap ? 1 : 0;
  TORCH_CHECK(!kv_swap || (s.B % 2 == 0 && s.L == s.Lk),
              "kv_swap pairs batches (b, b^1)");
  TORCH_CHECK(s.L % 128 == 0 && s.Lk % 128 == 0,
              "L must be a multiple of 128");
  TORCH_CHECK(s.D == 16 || s.D == 32 || s.D == 64 || s.D == 128,
              "unsupported head dim for fused bwd: ", s.D);

  auto dq = torch::empty({s.B, s.L, s.H, s.D},
                         q.options().layout(torch::kStrided));
  auto dk = torch::empty({s.B, s.Lk, s.H, s.D}, dq.options());
  auto dv = torch::empty({s.B, s.Lk, s.H, s.D}, dq.options());
  auto stream = at::hip::getCurrentHIPStream();

#define LAUNCH_BWD(DV)                                                     \
  {                                                                        \
    const size_t lds_kv = 2 * ((size_t)4 * QTB * DV * 2 + 2 * QTB * 4)    \
                          + 8 * 16 * QTB * 2;                              \
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<DV>),                          \
        dim3(s.B * s.H * (s.Lk / KVB)), dim3(512), lds_kv, stream,         \
        reinterpret_cast<const bf16*>(q.data_ptr()),                       \
        reinterpret_cast<const bf16*>(k.data_ptr()),                       \
        reinterpret_cast<const bf16*>(v.data_ptr()),                       \
        reinterpret_cast<const bf16*>(dout.data_ptr()),                    \
        lse.data_ptr<float>(), delta.data_ptr<float>(),                    \
        reinterpret_cast<bf16*>(dk.data_ptr()),                            \
        reinterpret_cast<bf16*>(dv.data_ptr()), s);                        \
    const size_t lds_q = (size_t)3 * KB2 * DV * 2 + 4 * 16 * KB2 * 2;      \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<DV>),                           \
        dim3(s.B * s.H * (s.L / QTB)), dim3(256), lds_q, stream,           \
        reinterpret_cast<const bf16*>(q.data_ptr()),                       \
        reinterpret_cast<const bf16*>(k.data_ptr()),                       \
        reinterpret_cast<const bf16*>(v.data_ptr()),                       \
        reinterpret_cast<const bf16*>(dout.data_ptr()),                    \
        lse.data_ptr<float>(), delta.data_ptr<float>(),                    \
        reinterpret_cast<bf16*>(dq.data_ptr()), s);                        \
  }
  switch (s.D) {
    case 16: LAUNCH_BWD(16); break;
    case 32: LAUNCH_BWD(32); break;
    case 64: LAUNCH_BWD(64); break;
    case 128: LAUNCH_BWD(128); break;
  }
#undef LAUNCH_BWD
  return {dq, dk, dv};
}
