// Implicit-GEMM 3x3 SAME stride-1 conv, NHWC bf16, MFMA — CDNA4 gfx950.
//
// the reference's only conv type (kernel (1,3,3),
// /root/reference/model/xunet.py:81,85,229,276). GEMM view:
//   out[M, N] = sum_k A[M, k] * B[k, N]
//   M = B*F*H*W pixels, N = Cout, k = (plane dy*3+dx, ci), K = 9*Cin.
// A is the im2col image — materialized on the fly into LDS with boundary
// masking (zero padding); B is the weight, whose OHWI storage
// (Cout, 3, 3, Cin) is ALREADY (n, k) row-major, so both stage coalesced.
//
// Structure (cdna_hip_programming.md §5 ladder step 2-3): 128x128 tile,
// BK=64, 4 waves (2x2 of 64x64), mfma_f32_16x16x32_bf16 with 4x4
// accumulator fragments per wave, double-buffered LDS with the
// ((row&7)<<4) byte-XOR swizzle (§6 G4) on both tiles, register-staged
// A/B (boundary masks need per-lane predication, so no glds), epilogue
// through LDS for coalesced bf16 stores, bias fused.
//
// The same kernel computes dgrad: conv3x3(dy, w~) with
// w~[ci, ey, ex, co] = w[co, 2-ey, 2-ex, ci] (host-side transform).
// Constraints: Cin % 64 == 0, Cout % 128 == 0 (dispatch falls back to the
// MIOpen path otherwise — stem 3ch / posemb 144ch / head 3ch convs).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// address-space casts for global_load_lds (via integer, the sanctioned way)
using as1_cvp = const __attribute__((address_space(1))) void*;
using as3_vp = __attribute__((address_space(3))) void*;
__device__ __forceinline__ as1_cvp as_global(const void* p) {
  return (as1_cvp)(unsigned long long)(uintptr_t)p;
}
__device__ __forceinline__ as3_vp as_shared(void* p) {
  return (as3_vp)(unsigned int)(uintptr_t)p;
}

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 64;
constexpr int THREADS = 256;
// LDS tile: [rows][BK] bf16, byte-swizzled. row stride = BK*2 = 128B.
__device__ __forceinline__ int swz(int row, int k_elem) {
  int byte = row * (BK * 2) + k_elem * 2;
  return byte ^ ((row & 7) << 4);
}

struct ConvShape {
  int IMG;       // B*F images
  int H, W;      // spatial
  int Cin, Cout;
  int M;         // IMG*H*W
  int ksteps;    // 9*Cin / BK
  int steps_per_plane;  // Cin / BK
};

__global__ __launch_bounds__(THREADS)
void conv3x3_igemm(const bf16* __restrict__ x,   // (IMG,H,W,Cin)
                   const bf16* __restrict__ w,   // (Cout, 9*Cin) rows=co
                   const float* __restrict__ bias,  // (Cout,) or null
                   const bf16* __restrict__ zbuf,   // 128B of zeros (padding)
                   const bf16* __restrict__ residual,  // out-shaped or null
                   float out_scale,              // y=(conv+bias+res)*scale
                   bf16* __restrict__ out,       // (IMG,H,W,Cout)
                   ConvShape s, int nblocks_m) {
  // XCD-aware block swizzle (bijective form, §5.5 T1): consecutive
  // swizzled ids walk M-blocks within one N-column so each XCD's L2 keeps
  // the weight panel.
  int bid = blockIdx.x;
  const int nwg = nblocks_m * (s.Cout / BN);
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid % nblocks_m;
  const int bn = bid / nblocks_m;
  const int m0 = bm * BM;
  const int n0 = bn * BN;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* ldsA = reinterpret_cast<bf16*>(smem);               // 2 x 16KB
  bf16* ldsB = reinterpret_cast<bf16*>(smem + 2 * BM * BK * 2);  // 2 x 16KB

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;   // 0..1: M half
  const int wc = wave & 1;    // 0..1: N half

  // --- glds staging ---------------------------------------------------
  // A and B tiles stream global->LDS via global_load_lds (16B/lane): the
  // LDS image is lane-linear, so the XOR swizzle moves to the per-lane
  // SOURCE address (rule 21), and boundary zero-fill is done by pointing
  // out-of-bounds lanes at a 128B zero buffer. One glds instruction fills
  // 1 KiB: 16 per 16KB tile = 4 per wave; slot s of this wave covers LDS
  // bytes [(wave*4+s)*1024, +1024).
  // Lane's element under the swizzle: row = o/128 (XOR keeps the row),
  // k-byte = (o ^ ((row&7)<<4)) % 128, with o = slot*1024 + lane*16.
  int a_h[4], a_w[4], a_ok[4], a_kp[4], b_kp[4], b_co[4];
  long a_base[4];
#pragma unroll
  for (int slot = 0; slot < 4; ++slot) {
    const int o = (wave * 4 + slot) * 1024 + lane * 16;
    const int row = o >> 7;
    const int kb = (o ^ ((row & 7) << 4)) & 127;
    a_kp[slot] = kb >> 4;
    b_kp[slot] = kb >> 4;
    b_co[slot] = n0 + row;
    const int m = m0 + row;
    a_w[slot] = m % s.W;
    a_h[slot] = (m / s.W) % s.H;
    a_ok[slot] = m < s.M;
    a_base[slot] = ((long)(m / (s.W * s.H)) * s.H) * s.W * s.Cin;
  }

  float acc[4][4][4];  // [mi][nj][reg] f32x4 fragments
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  auto stage_glds = [&](int kstep, int buf) {
    const int plane = kstep / s.steps_per_plane;
    const int ci0 = (kstep % s.steps_per_plane) * BK;
    const int dy = plane / 3 - 1;
    const int dx = plane % 3 - 1;
    char* baseA = reinterpret_cast<char*>(ldsA) + buf * BM * BK * 2
                  + wave * 4 * 1024;
    char* baseB = reinterpret_cast<char*>(ldsB) + buf * BN * BK * 2
                  + wave * 4 * 1024;
#pragma unroll
    for (int slot = 0; slot < 4; ++slot) {
      const int hh = a_h[slot] + dy;
      const int ww = a_w[slot] + dx;
      const bool valid = a_ok[slot] & (hh >= 0) & (hh < s.H) & (ww >= 0)
                         & (ww < s.W);
      const bf16* srcA = valid
          ? x + a_base[slot] + ((long)hh * s.W + ww) * s.Cin + ci0
              + a_kp[slot] * 8
          : zbuf;
      __builtin_amdgcn_global_load_lds(as_global(srcA),
          as_shared(baseA + slot * 1024), 16, 0, 0);
      const bf16* srcB = w + (long)b_co[slot] * (9 * s.Cin) + kstep * BK
                         + b_kp[slot] * 8;
      __builtin_amdgcn_global_load_lds(as_global(srcB),
          as_shared(baseB + slot * 1024), 16, 0, 0);
    }
  };

  auto compute = [&](int buf) {
    const char* baseA = reinterpret_cast<const char*>(ldsA) + buf * BM * BK * 2;
    const char* baseB = reinterpret_cast<const char*>(ldsB) + buf * BN * BK * 2;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bf[4];
      const int kbase = kk * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wr * 64 + i * 16 + (lane & 15);
        af[i] = *reinterpret_cast<const bf16x8*>(baseA + swz(row, kbase));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = wc * 64 + j * 16 + (lane & 15);
        bf[j] = *reinterpret_cast<const bf16x8*>(baseB + swz(row, kbase));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          *reinterpret_cast<f32x4*>(acc[i][j]) =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[i], bf[j], *reinterpret_cast<f32x4*>(acc[i][j]),
                  0, 0, 0);
        }
    }
  };

  // --- main loop: glds double-buffer, one barrier per K-step ----------
  stage_glds(0, 0);
  __syncthreads();  // drains the glds (vmcnt(0) inside the barrier)
  for (int t = 0; t < s.ksteps; ++t) {
    const int cur = t & 1;
    if (t + 1 < s.ksteps) stage_glds(t + 1, cur ^ 1);
    compute(cur);
    __syncthreads();  // glds for t+1 landed; LDS reads of buf cur done
  }

  // --- epilogue: acc -> (bias add) -> bf16 via LDS -> coalesced stores --
  bf16* ldsC = reinterpret_cast<bf16*>(smem);  // [BM][BN] bf16 = 32 KB
  __syncthreads();  // done with A/B buffers
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = wc * 64 + j * 16 + (lane & 15);
    const float bj = bias != nullptr ? bias[n0 + col] : 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int rbase = wr * 64 + i * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        ldsC[(rbase + r) * BN + col] = __float2bfloat16(acc[i][j][r] + bj);
      }
    }
  }
  __syncthreads();
  // store: 128 rows x 16 packs(16B) = 2048 packs / 256 threads = 8 each
  // optional fused residual: y = (conv + bias + residual) * out_scale
  // (reference ResnetBlock tail, xunet.py:92)
#pragma unroll
  for (int it = 0; it < 8; ++it) {
    const int p = tid + it * THREADS;
    const int row = p >> 4;
    const int cp = p & 15;
    const int m = m0 + row;
    if (m < s.M) {
      Pack<bf16, 8> v = *reinterpret_cast<Pack<bf16, 8>*>(
          ldsC + row * BN + cp * 8);
      const long goff = (long)m * s.Cout + n0 + cp * 8;
      if (residual != nullptr) {
        Pack<bf16, 8> rv = pload<bf16, 8>(residual + goff);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          from_f32((to_f32(v.v[j]) + to_f32(rv.v[j])) * out_scale, v.v[j]);
        }
      } else if (out_scale != 1.0f) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          from_f32(to_f32(v.v[j]) * out_scale, v.v[j]);
        }
      }
      pstore<bf16, 8>(out + goff, v);
    }
  }
}

}  // namespace


// ---------------------------------------------------------------------------
// 256x256-tile variant (8 waves as 2Mx4N, BK=64) — the higher-throughput
// structure of cdna_hip_programming.md §5: halves the staging traffic per
// FLOP and runs 64 MFMA per wave between barriers (vs 32 in the 128² kernel).
// Per K-tile: 4 quadrant phases with no internal barriers (each wave reads
// its own A/B halves); ONE vmcnt(0)+barrier per K-tile. A/B staged by glds
// in 4 half-tiles (128 rows x 64 k each), double-buffered (4 x 16KB x 2 =
// 128 KiB LDS), same source-side XOR swizzle and zero-buffer padding.
// Selected when M % 256 == 0 and Cout % 256 == 0 (L0-L2 of the full model).
// ---------------------------------------------------------------------------

namespace {

constexpr int BM2 = 256;
constexpr int BN2 = 256;
constexpr int T2 = 512;

__global__ __launch_bounds__(T2)
void conv3x3_igemm_256(const bf16* __restrict__ x,
                       const bf16* __restrict__ w,   // (Cout, 9*Cin)
                       const float* __restrict__ bias,
                       const bf16* __restrict__ zbuf,
                       const bf16* __restrict__ residual,
                       float out_scale,
                       bf16* __restrict__ out,
                       ConvShape s, int nblocks_m) {
  int bid = blockIdx.x;
  const int nwg = nblocks_m * (s.Cout / BN2);
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid % nblocks_m;
  const int bn = bid / nblocks_m;
  const int m0 = bm * BM2;
  const int n0 = bn * BN2;

  // LDS: [buf(2)][operand A=0/B=1][half(2)][16KB]
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto half_base = [&](int buf, int op, int half) -> char* {
    return smem + (((buf * 2 + op) * 2 + half) << 14);
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;   // 0..7
  const int wm = wave >> 2;    // A half this wave consumes
  const int wn = wave & 3;     // B cols: half wn>>1, sub (wn&1)*64

  // glds slot decode: wave issues 2 glds per half-tile; slot s covers LDS
  // bytes [(wave*2+s)*1024, +1024) of that half (lane-linear). Element under
  // the ((row&7)<<4) XOR swizzle: row = o>>7, kp = ((o ^ ((row&7)<<4)) & 127) >> 4.
  int arow_h[2][2], arow_w[2][2], arow_ok[2][2], kp_[2];  // [half][slot]
  long arow_base[2][2];
  int bco[2][2];
#pragma unroll
  for (int half = 0; half < 2; ++half) {
#pragma unroll
    for (int slot = 0; slot < 2; ++slot) {
      const int o = (wave * 2 + slot) * 1024 + lane * 16;
      const int row = o >> 7;             // 0..127 within the half
      kp_[slot] = ((o ^ ((row & 7) << 4)) & 127) >> 4;
      const int m = m0 + half * 128 + row;
      arow_w[half][slot] = m % s.W;
      arow_h[half][slot] = (m / s.W) % s.H;
      arow_ok[half][slot] = m < s.M;
      arow_base[half][slot] = ((long)(m / (s.W * s.H)) * s.H) * s.W * s.Cin;
      bco[half][slot] = n0 + half * 128 + row;
    }
  }

  auto stage_tile = [&](int kstep, int buf) {
    const int plane = kstep / s.steps_per_plane;
    const int ci0 = (kstep % s.steps_per_plane) * BK;
    const int dy = plane / 3 - 1;
    const int dx = plane % 3 - 1;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int slot = 0; slot < 2; ++slot) {
        const int hh = arow_h[half][slot] + dy;
        const int ww = arow_w[half][slot] + dx;
        const bool valid = arow_ok[half][slot] & (hh >= 0) & (hh < s.H)
                           & (ww >= 0) & (ww < s.W);
        const bf16* srcA = valid
            ? x + arow_base[half][slot] + ((long)hh * s.W + ww) * s.Cin
                + ci0 + kp_[slot] * 8
            : zbuf;
        __builtin_amdgcn_global_load_lds(as_global(srcA),
            as_shared(half_base(buf, 0, half) + (wave * 2 + slot) * 1024),
            16, 0, 0);
        const bf16* srcB = w + (long)bco[half][slot] * (9 * s.Cin)
                           + kstep * BK + kp_[slot] * 8;
        __builtin_amdgcn_global_load_lds(as_global(srcB),
            as_shared(half_base(buf, 1, half) + (wave * 2 + slot) * 1024),
            16, 0, 0);
      }
    }
  };

  float acc[8][4][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  stage_tile(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < s.ksteps; ++t) {
    const int cur = t & 1;
    const char* baseA = half_base(cur, 0, wm);
    const char* baseB = half_base(cur, 1, wn >> 1);
    // B fragments for the whole K-tile (shared by all 4 quadrants)
    bf16x8 bfr[4][2];
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int row = (wn & 1) * 64 + j * 16 + (lane & 15);
        bfr[j][kk] = *reinterpret_cast<const bf16x8*>(
            baseB + swz(row, kk * 32 + (lane >> 4) * 8));
      }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (q == 0 && t + 1 < s.ksteps) {
        stage_tile(t + 1, cur ^ 1);  // overlaps the whole tile's compute
      }
      bf16x8 afr[2][2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int row = q * 32 + i * 16 + (lane & 15);
          afr[i][kk] = *reinterpret_cast<const bf16x8*>(
              baseA + swz(row, kk * 32 + (lane >> 4) * 8));
        }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            *reinterpret_cast<f32x4*>(acc[q * 2 + i][j]) =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr[i][kk], bfr[j][kk],
                    *reinterpret_cast<f32x4*>(acc[q * 2 + i][j]), 0, 0, 0);
          }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // epilogue: stage the 256x256 bf16 tile through LDS (128 KiB) and store
  // coalesced, with fused bias (+ residual)*scale
  bf16* ldsC = reinterpret_cast<bf16*>(smem);
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = wn * 64 + j * 16 + (lane & 15);
    const float bj = bias != nullptr ? bias[n0 + col] : 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int rbase = wm * 128 + i * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        ldsC[(rbase + r) * BN2 + col] = __float2bfloat16(acc[i][j][r] + bj);
      }
    }
  }
  __syncthreads();
  {
    // 256 rows x 32 packs = 8192 packs / 512 threads = 16 each
#pragma unroll
    for (int it = 0; it < 16; ++it) {
      const int p = tid + it * T2;
      const int row = p >> 5;
      const int cp = p & 31;
      const int m = m0 + row;
      if (m < s.M) {
        Pack<bf16, 8> v = *reinterpret_cast<Pack<bf16, 8>*>(
            ldsC + row * BN2 + cp * 8);
        const long goff = (long)m * s.Cout + n0 + cp * 8;
        if (residual != nullptr) {
          Pack<bf16, 8> rv = pload<bf16, 8>(residual + goff);
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            from_f32((to_f32(v.v[jj]) + to_f32(rv.v[jj])) * out_scale,
                     v.v[jj]);
          }
        } else if (out_scale != 1.0f) {
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            from_f32(to_f32(v.v[jj]) * out_scale, v.v[jj]);
          }
        }
        pstore<bf16, 8>(out + goff, v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Pipelined 256x256 variant (v2): counted-vmcnt glds across RAW barriers
// (cdna_hip_programming.md §5.5 T3+T4). The K-tile is consumed in TWO
// phases (A-quadrants 0-1, then 2-3); staging is piece-granular — a piece
// is 64 rows x BK of one operand half (8 KB), one glds per wave per piece
// via the slot->piece chunk map (slot s = piece s, byte s*8192 + wave*1024).
// Next tile's B pieces issue during phase 0, A pieces (even-row pieces
// first) during phase 1, so each wait keeps 2-4 glds in flight instead of
// draining vmcnt(0) at every barrier:
//   mid-tile:  vmcnt(4)  [this tile's A-odd pieces landed; next B in flight]
//   boundary:  vmcnt(2)  [next tile's B + A-even landed; A-odd in flight]
// ---------------------------------------------------------------------------


__global__ __launch_bounds__(T2)
void conv3x3_igemm_256_v2(const bf16* __restrict__ x,
                          const bf16* __restrict__ w,   // (Cout, 9*Cin)
                          const float* __restrict__ bias,
                          const bf16* __restrict__ zbuf,
                          const bf16* __restrict__ residual,
                          float out_scale,
                          bf16* __restrict__ out,
                          ConvShape s, int nblocks_m) {
  int bid = blockIdx.x;
  const int nwg = nblocks_m * (s.Cout / BN2);
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int bm = bid % nblocks_m;
  const int bn = bid / nblocks_m;
  const int m0 = bm * BM2;
  const int n0 = bn * BN2;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto half_base = [&](int buf, int op, int half) -> char* {
    return smem + (((buf * 2 + op) * 2 + half) << 14);
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  // chunk map: slot s (= piece), half h -> byte o = s*8192 + wave*1024
  int arow_h[2][2], arow_w[2][2], arow_ok[2][2], kp_[2];  // [half][slot]
  long arow_base[2][2];
  int bco[2][2];
#pragma unroll
  for (int half = 0; half < 2; ++half) {
#pragma unroll
    for (int slot = 0; slot < 2; ++slot) {
      const int o = slot * 8192 + wave * 1024 + lane * 16;
      const int row = o >> 7;             // 0..127 within the half
      kp_[slot] = ((o ^ ((row & 7) << 4)) & 127) >> 4;
      const int m = m0 + half * 128 + row;
      arow_w[half][slot] = m % s.W;
      arow_h[half][slot] = (m / s.W) % s.H;
      arow_ok[half][slot] = m < s.M;
      arow_base[half][slot] = ((long)(m / (s.W * s.H)) * s.H) * s.W * s.Cin;
      bco[half][slot] = n0 + half * 128 + row;
    }
  }

  // one wave-glds for piece (op, half, slot) of kstep into buf
  auto stage_piece = [&](int kstep, int buf, int op, int half, int slot) {
    const int plane = kstep / s.steps_per_plane;
    const int ci0 = (kstep % s.steps_per_plane) * BK;
    const int dyp = plane / 3 - 1;
    const int dxp = plane % 3 - 1;
    char* dst = half_base(buf, op, half) + slot * 8192 + wave * 1024;
    if (op == 0) {
      const int hh = arow_h[half][slot] + dyp;
      const int ww = arow_w[half][slot] + dxp;
      const bool valid = arow_ok[half][slot] & (hh >= 0) & (hh < s.H)
                         & (ww >= 0) & (ww < s.W);
      const bf16* srcA = valid
          ? x + arow_base[half][slot] + ((long)hh * s.W + ww) * s.Cin
              + ci0 + kp_[slot] * 8
          : zbuf;
      __builtin_amdgcn_global_load_lds(as_global(srcA), as_shared(dst),
                                       16, 0, 0);
    } else {
      const bf16* srcB = w + (long)bco[half][slot] * (9 * s.Cin)
                         + kstep * BK + kp_[slot] * 8;
      __builtin_amdgcn_global_load_lds(as_global(srcB), as_shared(dst),
                                       16, 0, 0);
    }
  };
  auto stage_op = [&](int kstep, int buf, int op) {
    // pieces: slot 0 (even rows) of both halves first, then slot 1
#pragma unroll
    for (int slot = 0; slot < 2; ++slot)
#pragma unroll
      for (int half = 0; half < 2; ++half)
        stage_piece(kstep, buf, op, half, slot);
  };
  // one 2-glds step of the per-phase issue schedule: B piece-0s, B
  // piece-1s, A piece-0s, A piece-1s (phase = which pair)
  auto stage_phase = [&](int kstep, int buf, int phase) {
    const int op = phase < 2 ? 1 : 0;
    const int slot = phase & 1;
#pragma unroll
    for (int half = 0; half < 2; ++half)
      stage_piece(kstep, buf, op, half, slot);
  };

  float acc[8][4][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[i][j][r] = 0.f;

  // prologue: tile 0 fully staged
  stage_op(0, 0, 1);
  stage_op(0, 0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < s.ksteps; ++t) {
    const int cur = t & 1;
    const bool more = t + 1 < s.ksteps;
    const char* baseA = half_base(cur, 0, wm);
    const char* baseB = half_base(cur, 1, wn >> 1);
    // B fragments for the whole K-tile (all 4 quadrant phases)
    bf16x8 bfr[4][2];
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int row = (wn & 1) * 64 + j * 16 + (lane & 15);
        bfr[j][kk] = *reinterpret_cast<const bf16x8*>(
            baseB + swz(row, kk * 32 + (lane >> 4) * 8));
      }
    // 4 quadrant phases, template-style: {A-quadrant ds_reads | 2-glds
    // issue | barrier | MFMA under setprio | barrier}; counted vmcnt only
    // where a consumed piece could still be in flight:
    //   before phase 2 of this tile: its A piece-1 (issued at phase 3 of
    //   the previous tile) -> vmcnt(4) [next tile's B pieces in flight]
    //   at the boundary: next tile's B + A piece-0 -> vmcnt(2)
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      bf16x8 afr[2][2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int row = q * 32 + i * 16 + (lane & 15);
          afr[i][kk] = *reinterpret_cast<const bf16x8*>(
              baseA + swz(row, kk * 32 + (lane >> 4) * 8));
        }
      if (more) stage_phase(t + 1, cur ^ 1, q);
      if (q == 2) {
        if (more) {
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            *reinterpret_cast<f32x4*>(acc[q * 2 + i][j]) =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr[i][kk], bfr[j][kk],
                    *reinterpret_cast<f32x4*>(acc[q * 2 + i][j]), 0, 0, 0);
          }
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    if (more) {
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  // epilogue identical to v1: stage C through LDS, coalesced bf16 stores
  bf16* ldsC = reinterpret_cast<bf16*>(smem);
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int col = wn * 64 + j * 16 + (lane & 15);
    const float bj = bias != nullptr ? bias[n0 + col] : 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int rbase = wm * 128 + i * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        ldsC[(rbase + r) * BN2 + col] = __float2bfloat16(acc[i][j][r] + bj);
      }
    }
  }
  __syncthreads();
  {
#pragma unroll
    for (int it = 0; it < 16; ++it) {
      const int p = tid + it * T2;
      const int row = p >> 5;
      const int cp = p & 31;
      const int m = m0 + row;
      if (m < s.M) {
        Pack<bf16, 8> v = *reinterpret_cast<Pack<bf16, 8>*>(
            ldsC + row * BN2 + cp * 8);
        const long goff = (long)m * s.Cout + n0 + cp * 8;
        if (residual != nullptr) {
          Pack<bf16, 8> rv = pload<bf16, 8>(residual + goff);
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            from_f32((to_f32(v.v[jj]) + to_f32(rv.v[jj])) * out_scale,
                     v.v[jj]);
          }
        } else if (out_scale != 1.0f) {
#pragma unroll
          for (int jj = 0; jj < 8; ++jj) {
            from_f32(to_f32(v.v[jj]) * out_scale, v.v[jj]);
          }
        }
        pstore<bf16, 8>(out + goff, v);
      }
    }
  }
}

}  // namespace

torch::Tensor conv3x3_fwd(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> residual,
                          double out_scale) {
  // x: (B,F,H,W,Cin) or (IMG,H,W,Cin) bf16 contiguous; w: (Cout,3,3,Cin)
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16);
  auto xs = x.sizes();
  const int nd = x.dim();
  TORCH_CHECK(nd == 4 || nd == 5);
  ConvShape s;
  s.IMG = nd == 5 ? xs[0] * xs[1] : xs[0];
  s.H = xs[nd - 3]; s.W = xs[nd - 2]; s.Cin = xs[nd - 1];
  s.Cout = w.size(0);
  TORCH_CHECK(w.size(1) == 3 && w.size(2) == 3 && w.size(3) == s.Cin);
  TORCH_CHECK(s.Cin % BK == 0, "Cin must be multiple of 64");
  TORCH_CHECK(s.Cout % BN == 0, "Cout must be multiple of 128");
  s.M = s.IMG * s.H * s.W;
  s.steps_per_plane = s.Cin / BK;
  s.ksteps = 9 * s.steps_per_plane;

  std::vector<int64_t> oshape(xs.begin(), xs.end());
  oshape[nd - 1] = s.Cout;
  auto out = torch::empty(oshape, x.options());

  torch::Tensor biasf;
  if (bias.has_value()) biasf = bias->to(torch::kFloat).contiguous();
  static torch::Tensor zbuf;  // 128B zero pad source for OOB glds lanes
  if (!zbuf.defined() || zbuf.device() != x.device()) {
    zbuf = torch::zeros({64}, x.options());
  }

  auto stream = at::hip::getCurrentHIPStream();
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous()
                && residual->sizes() == out.sizes()
                && residual->scalar_type() == torch::kBFloat16);
  }
  // 256x256-tile variant where it fills the chip (L0-L2 full-config convs)
  const bool use256 = (s.M % 256 == 0) && (s.Cout % 256 == 0)
      && ((long)(s.M / 256) * (s.Cout / 256) >= 128);
  if (use256) {
    const int nb_m = s.M / BM2;
    // v1 (one vmcnt(0)+barrier per K-tile) is the default: the piece-wise
    // counted-vmcnt v2 MEASURED 5-13% SLOWER on every model shape
    // (gpurun_out/kb_v1.json vs kb_v2.json — the guide's "coarse
    // phase-split without the fine per-phase interleave hurts" case);
    // NVS3D_CONV256=v2 keeps the negative result reproducible
    static const char* env = getenv("NVS3D_CONV256");
    const bool v2 = env && env[0] == 'v' && env[1] == '2';
    auto kfn = v2 ? conv3x3_igemm_256_v2 : conv3x3_igemm_256;
    hipLaunchKernelGGL(kfn, dim3(nb_m * (s.Cout / BN2)),
        dim3(T2), 128 * 1024, stream,
        reinterpret_cast<const bf16*>(x.data_ptr()),
        reinterpret_cast<const bf16*>(w.data_ptr()),
        bias.has_value() ? biasf.data_ptr<float>() : nullptr,
        reinterpret_cast<const bf16*>(zbuf.data_ptr()),
        residual.has_value()
            ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr,
        (float)out_scale,
        reinterpret_cast<bf16*>(out.data_ptr()), s, nb_m);
    return out;
  }
  const int nblocks_m = (s.M + BM - 1) / BM;
  const int grid = nblocks_m * (s.Cout / BN);
  const size_t lds = 2 * (BM + BN) * BK * 2;  // 64 KB
  hipLaunchKernelGGL(conv3x3_igemm, dim3(grid), dim3(THREADS), lds, stream,
      reinterpret_cast<const bf16*>(x.data_ptr()),
      reinterpret_cast<const bf16*>(w.data_ptr()),
      bias.has_value() ? biasf.data_ptr<float>() : nullptr,
      reinterpret_cast<const bf16*>(zbuf.data_ptr()),
      residual.has_value()
          ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr,
      (float)out_scale,
      reinterpret_cast<bf16*>(out.data_ptr()), s, nblocks_m);
  return out;
}
