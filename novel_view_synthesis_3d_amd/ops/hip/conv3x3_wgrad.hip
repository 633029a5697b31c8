// Conv3x3 weight gradient — direct MFMA kernel, CDNA4 gfx950.
//
// SURVEY.md §2.4 K1 backward (reference /root/reference/model/xunet.py:81,85
// under jax.value_and_grad, train.py:70): for the SAME, stride-1, 3x3 frame
// conv,  dW[co,ky,kx,ci] = sum_{img,y,x} dy[img,y,x,co] * x[img,y+ky-1,x+kx-1,ci].
//
// GEMM view per (ky,kx) tap: D[M=co][N=ci] = sum_K A[co][K] B[K][ci] with
// K = pixels. Both operands are stored pixel-major ([K][C]), so both MFMA
// fragments need K(pixel)-contiguous elements per lane: row-tiles are
// staged ROW-MAJOR per pixel ([pixel][C_tile], padded row stride) and
// consumed with ds_read_b64_tr_b16 hardware transpose reads — the read
// takes per-lane granule addresses, so the row-major image serves the
// transposed fragments directly while glds stays line-coalesced, and the
// row pad spreads each transpose group's 4 pixel-rows over distinct bank
// classes.
//
// Structure:
//  * One block = (co-tile 128) x (ci-tile 64) x (a contiguous range of
//    row-tiles). 8 waves as 4(M) x 2(N), each wave one 32x32 output tile
//    per tap via mfma_f32_32x32x16_bf16 (K-step = 16 pixels).
//  * ALL NINE (ky,kx) taps accumulate concurrently (9 f32x16 accumulators).
//    The ky dimension pairs the dy row r with x rows r-1,r,r+1 held in an
//    LDS ring; the kx dimension is a +-1 pixel shift = a +-row-stride LDS
//    offset against the x row image, staged with a 1-pixel halo per row
//    (halo = real neighbor pixels of the column tile, or zero at image
//    edges). Each staged byte of x feeds 9 taps: x and dy stream from HBM
//    exactly once per (co,ci)-tile.
//  * Rows are processed in groups of gr (4/2/1 for W<=16/32/64+) per
//    barrier round, prefetching the next group's dy + x rows under the
//    current group's MFMAs.
//  * Split-K: each block atomically adds its fp32 9-tap tile into dw_acc
//    (Cout,3,3,Cin); bias grad rides along on the A fragments of the
//    ci-tile-0 blocks.
//
// Constraints: Cin % 64 == 0, Cout % 128 == 0, W % 16 == 0 and
// (W <= 128 or W % 128 == 0). Dispatch falls back to the im2col+GEMM path
// otherwise (ops/hip_ops.py).

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

__device__ __forceinline__ bf16x8 tr16x8(const char* p0, const char* p1) {
  bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (as3_bf16x4p)(const_cast<char*>(p0)));
  bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (as3_bf16x4p)(const_cast<char*>(p1)));
  return __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
}

constexpr int WBM = 128;   // co per block
constexpr int WBN = 64;    // ci per block
constexpr int WT = 512;    // 8 waves
// per-PIXEL row strides (bytes): C_tile*2 + pad. dy: 256+32 (rows step 8
// banks; group-1 overlap is a minor 2-way on the single A read); x:
// 128+32 (rows step 40 banks, distinct mod 4).
constexpr int SROW_DY = WBM * 2 + 32;   // 288
constexpr int SROW_X = WBN * 2 + 32;    // 160

struct WgShape {
  int IMG, H, W, Cin, Cout;
  int PW;          // column-tile width (min(W,128)), multiple of 16
  int pwlog;       // log2(PW)
  int nct;         // column tiles per row = W / PW
  int gr;          // rows staged per barrier round (4/2/1 for PW 16/32/64+)
  int rs;          // x ring slots (power of 2)
  int units;       // IMG*H*nct row-tiles
  int nb_m, nb_n;  // Cout/WBM, Cin/WBN
  int sk;          // split-K factor
  int with_bias;
};

__global__ __launch_bounds__(WT)
void conv3x3_wgrad_kernel(const bf16* __restrict__ x,
                          const bf16* __restrict__ dy,
                          const bf16* __restrict__ zbuf,
                          float* __restrict__ dwacc,  // (Cout,3,3,Cin) zeroed
                          float* __restrict__ dbacc,  // (Cout,) zeroed | null
                          WgShape s) {
  // block decode: bid = sk * (nb_m*nb_n) + (bm*nb_n + bn), with the
  // bijective XCD remap so blocks sharing a split-K data range (identical
  // dy/x reads, different (co,ci) tiles) sit on one XCD's L2.
  int bid = blockIdx.x;
  const int nwg = s.sk * s.nb_m * s.nb_n;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int mn = bid % (s.nb_m * s.nb_n);
  const int sk = bid / (s.nb_m * s.nb_n);
  const int bm = mn / s.nb_n;
  const int bn = mn % s.nb_n;
  const int co0 = bm * WBM;
  const int ci0 = bn * WBN;

  // this block's contiguous row-tile range [u0, u1), ct-major so that
  // consecutive units usually share the x-row ring
  const int per = (s.units + s.sk - 1) / s.sk;
  const int u0 = sk * per;
  const int u1 = min(s.units, u0 + per);
  if (u0 >= u1) return;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: [dy buf 0][dy buf 1][x slot 0..rs-1]
  const int dy_bytes = (s.gr << s.pwlog) * SROW_DY;
  const int x_bytes = (s.PW + 2) * SROW_X;
  char* xbase = smem + 2 * dy_bytes;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 1;   // 0..3: co quarter (32 rows)
  const int wn = wave & 1;    // 0..1: ci half (32 cols)

  // per-lane tr-read base: granule covers channels cbase + (g&1)*16 +
  // 4*(m16&3) at pixel-row (g>>1)*8 + (m16>>2)
  const int g = lane >> 4;
  const int m16 = lane & 15;
  const int tr_prow = (g >> 1) * 8 + (m16 >> 2);
  const int tr_chb = ((g & 1) * 16 + 4 * (m16 & 3)) * 2;
  const int a_off = tr_prow * SROW_DY + wm * 64 + tr_chb;
  const int b_off = tr_prow * SROW_X + wn * 64 + tr_chb;

  const long rowstride_x = (long)s.W * s.Cin;
  const long rowstride_dy = (long)s.W * s.Cout;

  // stage `nrows` dy rows [rbase, rbase+nrows) of one (img, ct) into buf:
  // row-major [group pixel][128 co] (+pad)
  auto stage_dy = [&](int img, int rbase, int nrows, int ct, int buf) {
    const long base = ((long)img * s.H + rbase) * rowstride_dy
                      + (long)ct * s.PW * s.Cout + co0;
    char* dst = smem + buf * dy_bytes;
    for (int o = wave * 1024 + lane * 16; o < dy_bytes; o += 8 * 1024) {
      const int pxf = o / SROW_DY;
      const int w = o % SROW_DY;
      const bf16* src = zbuf;
      if (w < WBM * 2) {
        const int rig = pxf >> s.pwlog;
        const int px = pxf & (s.PW - 1);
        if (rig < nrows) {
          src = dy + base + (long)rig * rowstride_dy
                + (long)px * s.Cout + (w >> 1);
        }
      }
      __builtin_amdgcn_global_load_lds(as_global(src),
          as_shared(dst + (o - lane * 16)), 16, 0, 0);
    }
  };

  auto stage_x = [&](int img, int r, int ct, int slot) {
    const long base = ((long)img * s.H + r) * rowstride_x + ci0;
    const int px0 = ct * s.PW - 1;  // image pixel of halo index 0
    char* dst = xbase + slot * x_bytes;
    for (int o = wave * 1024 + lane * 16; o < x_bytes; o += 8 * 1024) {
      const int pxi = o / SROW_X;
      const int w = o % SROW_X;
      const bf16* src = zbuf;
      if (w < WBN * 2) {
        const int px = px0 + pxi;
        if (px >= 0 && px < s.W) {
          src = x + base + (long)px * s.Cin + (w >> 1);
        }
      }
      __builtin_amdgcn_global_load_lds(as_global(src),
          as_shared(dst + (o - lane * 16)), 16, 0, 0);
    }
  };

  // ---- accumulators ---------------------------------------------------
  float acc[9][16];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[t][e] = 0.f;
  float dbsum = 0.f;
  const bool do_bias = s.with_bias && bn == 0 && wn == 0;

  // ---- compute one staged row ----------------------------------------
  // dy row r = group row `ri` of dybuf[buf]; x rows r-1..r+1 in ring slots
  // (rr+1)&(rs-1); vmask enables the ky taps whose x row exists.
  auto compute_row = [&](int buf, int ri, int r, int vmask) {
    const char* abase = smem + buf * dy_bytes + a_off
                        + (ri << s.pwlog) * SROW_DY;
#pragma unroll 1
    for (int p0 = 0; p0 < s.PW; p0 += 16) {
      const char* pa = abase + p0 * SROW_DY;
      bf16x8 af = tr16x8(pa, pa + 4 * SROW_DY);
      if (do_bias) {
#pragma unroll
        for (int e = 0; e < 8; ++e) dbsum += (float)af[e];
      }
#pragma unroll
      for (int ky = 0; ky < 3; ++ky) {
        if (!(vmask & (1 << ky))) continue;
        const int slot = (r + ky) & (s.rs - 1);  // x row r+ky-1
        // x image pixel index = dy pixel + kx (halo offset folded in)
        const char* bb = xbase + slot * x_bytes + b_off + p0 * SROW_X;
#pragma unroll
        for (int kx = 0; kx < 3; ++kx) {
          const char* pb = bb + kx * SROW_X;
          bf16x8 bf = tr16x8(pb, pb + 4 * SROW_X);
          *reinterpret_cast<f32x16*>(acc[ky * 3 + kx]) =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  af, bf, *reinterpret_cast<f32x16*>(acc[ky * 3 + kx]),
                  0, 0, 0);
        }
      }
    }
  };

  // ---- main loop over row groups --------------------------------------
  const int rsm = s.rs - 1;
  int u = u0;
  while (u < u1) {
    const int ct = u / (s.IMG * s.H);
    const int vr = u % (s.IMG * s.H);
    const int img = vr / s.H;
    const int r0 = vr % s.H;
    // segment = run of rows of this (img, ct); img/ct boundaries end it
    const int rend = min(s.H, r0 + (u1 - u));

    // prologue: first group's dy rows + x rows [r0-1, r0+cnt]
    const int cnt0 = min(s.gr, rend - r0);
    for (int rr = r0 - 1; rr <= r0 + cnt0; ++rr) {
      if (rr >= 0 && rr < s.H) stage_x(img, rr, ct, (rr + 1) & rsm);
    }
    stage_dy(img, r0, cnt0, ct, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    int gbuf = 0;
    for (int gs = r0; gs < rend; gs += s.gr) {
      const int cnt = min(s.gr, rend - gs);
      const int nxt = gs + cnt;
      if (nxt < rend) {
        const int ncnt = min(s.gr, rend - nxt);
        stage_dy(img, nxt, ncnt, ct, gbuf ^ 1);
        for (int rr = nxt + 1; rr <= nxt + ncnt; ++rr) {
          if (rr < s.H) stage_x(img, rr, ct, (rr + 1) & rsm);
        }
      }
      for (int ri = 0; ri < cnt; ++ri) {
        const int r = gs + ri;
        const int vmask = (r > 0 ? 1 : 0) | 2 | (r + 1 < s.H ? 4 : 0);
        compute_row(gbuf, ri, r, vmask);
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      gbuf ^= 1;
    }
    u += rend - r0;
  }

  // ---- epilogue: fp32 atomic reduction into dwacc ---------------------
  // D layout (mfma_f32_32x32x16): col = lane&31 (ci), row = (e&3) + 8*(e>>2)
  // + 4*(lane>>5) (co), e in [0,16).
  const int ci = ci0 + wn * 32 + (lane & 31);
  const int co_base = co0 + wm * 32 + 4 * (lane >> 5);
#pragma unroll
  for (int ky = 0; ky < 3; ++ky)
#pragma unroll
    for (int kx = 0; kx < 3; ++kx)
#pragma unroll
      for (int e = 0; e < 16; ++e) {
        const int co = co_base + (e & 3) + 8 * (e >> 2);
        atomicAdd(dwacc + (((long)co * 3 + ky) * 3 + kx) * s.Cin + ci,
                  acc[ky * 3 + kx][e]);
      }
  if (do_bias) {
    // lanes l and l+32 hold the same co (different pixels)
    atomicAdd(dbacc + co0 + wm * 32 + (lane & 31), dbsum);
  }
}

}  // namespace

std::vector<torch::Tensor> conv3x3_wgrad(torch::Tensor x, torch::Tensor dy,
                                         bool with_bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              dy.scalar_type() == torch::kBFloat16);
  const int nd = x.dim();
  TORCH_CHECK(nd == 4 || nd == 5);
  WgShape s;
  auto xs = x.sizes();
  s.IMG = nd == 5 ? xs[0] * xs[1] : xs[0];
  s.H = xs[nd - 3]; s.W = xs[nd - 2]; s.Cin = xs[nd - 1];
  s.Cout = dy.size(nd - 1);
  TORCH_CHECK(dy.numel() == (long)s.IMG * s.H * s.W * s.Cout);
  TORCH_CHECK(s.Cin % WBN == 0 && s.Cout % WBM == 0);
  TORCH_CHECK(s.W % 16 == 0 && (s.W <= 128 || s.W % 128 == 0));

  s.PW = std::min(s.W, 128);
  s.pwlog = (int)std::round(std::log2((double)s.PW));
  TORCH_CHECK((1 << s.pwlog) == s.PW, "PW must be a power of two");
  s.nct = s.W / s.PW;
  s.gr = s.PW <= 16 ? 4 : (s.PW <= 32 ? 2 : 1);
  s.rs = s.gr == 1 ? 4 : (s.gr == 2 ? 8 : 16);
  s.units = s.IMG * s.H * s.nct;
  s.nb_m = s.Cout / WBM;
  s.nb_n = s.Cin / WBN;
  s.with_bias = with_bias ? 1 : 0;
  // split-K sized for ~2 block-waves over 256 CUs (1 block/CU at this LDS)
  const int target = 512;
  s.sk = std::max(1, std::min(s.units,
                              target / std::max(1, s.nb_m * s.nb_n)));

  auto opts = x.options().dtype(torch::kFloat);
  auto dw = torch::zeros({s.Cout, 3, 3, s.Cin}, opts);
  torch::Tensor db;
  if (with_bias) db = torch::zeros({s.Cout}, opts);

  static torch::Tensor zbuf;
  if (!zbuf.defined() || zbuf.device() != x.device()) {
    zbuf = torch::zeros({64}, x.options());
  }

  const size_t lds = 2 * (size_t)((s.gr * s.PW) * SROW_DY)
                     + (size_t)s.rs * ((s.PW + 2) * SROW_X);
  TORCH_CHECK(lds <= 160 * 1024, "wgrad LDS overflow: ", lds);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = s.sk * s.nb_m * s.nb_n;
  hipLaunchKernelGGL(conv3x3_wgrad_kernel, dim3(grid), dim3(WT), lds, stream,
      reinterpret_cast<const bf16*>(x.data_ptr()),
      reinterpret_cast<const bf16*>(dy.data_ptr()),
      reinterpret_cast<const bf16*>(zbuf.data_ptr()),
      dw.data_ptr<float>(),
      with_bias ? db.data_ptr<float>() : nullptr, s);
  if (with_bias) return {dw, db};
  return {dw};
}
