// NHWC implicit-GEMM convolution on MFMA matrix cores (gfx950, bf16).
//
// The conv/GEMM core the reference gets from cuDNN (SURVEY.md §2.11
// "MI355X plan implication"), hand-written for CDNA4:
//   y[n,oh,ow,k] = sum_{r,s,c} x[n, oh*st+r-p, ow*st+s-p, c] * w[k,r,s,c]
// viewed as GEMM  C[M=N*OH*OW][Nd=K] = A[M][Kd=R*S*C] * B[Nd][Kd]^T
// with A materialized implicitly (im2col addressing in the staging
// stage) and B = the conv weight read directly in its channels-last
// (K,R,S,C) memory image — no host-side im2col, no weight transform for
// forward.
//
// One kernel template serves forward (FPROP) and data-grad (DGRAD —
// A = scatter-gathered dy with transposed-conv addressing, B = the
// weight permuted to (C,R,S,K) memory once per backward).
//
// Structure = the documented CDNA4 GEMM recipe (cdna_hip_programming.md
// §5): 128x128 block tile, BK=64, 4 waves each owning a 64x64 sub-tile
// of 16x16x32 bf16 MFMA fragments, double-buffered LDS filled by
// 16-byte global_load_lds (lane-linear dest; XOR bank swizzle applied on
// the SOURCE chunk index and re-applied on the ds_read side — rule 21),
// out-of-bounds/padding chunks redirected to a zero page so every lane
// always issues its DMA.
//
// Constraints (dispatcher falls back for the rest): bf16, groups=1,
// dilation=1, C %8 == 0 (FPROP) / K %8 == 0 (DGRAD) so no 16-B chunk
// crosses an (r,s) boundary.

#include "conv_igemm.h"
#include <stdint.h>
#include <stdexcept>
#include <stdlib.h>
#include <string>

using bf16 = conv_bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 64
#define THREADS 256

using RowCoords = ConvRowCoords;

// ---- chunk -> global address generators -------------------------------
// A-chunk: logical (row=m in [0,BM), cg in [0,8)) of the current K-step.
// Returns the 16-byte-aligned source for elements kk0+cg*8 .. +7.

// MODE: 0 = FPROP, 1 = DGRAD (any stride, gather with validity tests),
//       2 = DGRAD stride-2 parity class (all taps valid by construction)
//
// The A operand's row -> pixel decomposition is loop-invariant (the
// GEMM row never changes across K-steps), so it is computed once per
// staging slot (conv_a_row_coords) and only the (r, s, c/k) part runs
// per step (a_step_addr).
template <int MODE>
DEV RowCoords a_row_coords(const ConvParams& p, long m) {
  return conv_a_row_coords<MODE>(p, m);
}

template <int MODE>
DEV const bf16* a_step_addr(const ConvParams& p, const RowCoords& rc,
                            long kkg) {
  if (!rc.valid) return p.zero;
  if (MODE == 0) {
    const int c0 = (int)(kkg % p.Cin);
    const int rs = (int)(kkg / p.Cin);
    const int r = rs / p.S, s = rs - (rs / p.S) * p.S;
    const int ih = rc.y * p.stride + r - p.pad;
    const int iw = rc.x * p.stride + s - p.pad;
    if (ih < 0 || ih >= p.H || iw < 0 || iw >= p.W) return p.zero;
    return p.a + (((long)rc.n * p.H + ih) * p.W + iw) * p.Cin + c0;
  } else if (MODE == 1) {
    const int k0 = (int)(kkg % p.K);
    const int rs = (int)(kkg / p.K);
    const int r = rs / p.S, s = rs - (rs / p.S) * p.S;
    const int tih = rc.y + p.pad - r;
    const int tiw = rc.x + p.pad - s;
    if (tih < 0 || tiw < 0 || (tih % p.stride) || (tiw % p.stride))
      return p.zero;
    const int oh = tih / p.stride, ow = tiw / p.stride;
    if (oh >= p.OH || ow >= p.OW) return p.zero;
    return p.a + (((long)rc.n * p.OH + oh) * p.OW + ow) * p.K + k0;
  } else {
    const int k0 = (int)(kkg % p.K);
    const int rs = (int)(kkg / p.K);
    const int ri = rs / p.ns, si = rs - (rs / p.ns) * p.ns;
    const int r = p.r0 + 2 * ri, s = p.s0 + 2 * si;
    const int oh = (rc.y + p.pad - r) >> 1;
    const int ow = (rc.x + p.pad - s) >> 1;
    if (oh < 0 || ow < 0 || oh >= p.OH || ow >= p.OW) return p.zero;
    return p.a + (((long)rc.n * p.OH + oh) * p.OW + ow) * p.K + k0;
  }
}

template <int MODE>
DEV const bf16* b_chunk_addr(const ConvParams& p, long row, long kkg) {
  if (row >= p.Nd || kkg >= p.Kd) return p.zero;
  if (MODE != 2) return p.b + row * p.Kd + kkg;
  // class subset of the (C,R,S,K) weight image
  const int k0 = (int)(kkg % p.K);
  const int rs = (int)(kkg / p.K);
  const int ri = rs / p.ns, si = rs - (rs / p.ns) * p.ns;
  const int r = p.r0 + 2 * ri, s = p.s0 + 2 * si;
  return p.b + (row * p.R + r) * (long)p.S * p.K + (long)s * p.K + k0;
}

// ---- the kernel -------------------------------------------------------
// LDS: A[2][128][64] + B[2][128][64] bf16 = 64 KiB. Lane-linear glds
// image: slot ca in [0,1024) holds logical (row = ca>>3, cg' = ca&7)
// where the DATA stored is logical cg = cg' ^ (row & 7). ds_read applies
// the same XOR.

// Tile geometry is templated so small output-channel counts get a
// narrow N tile instead of half-empty MFMA work: 4 waves arranged
// (WR x WC) each owning a (BM_/WR x BN_/WC) sub-tile.
template <int MODE, int BM_, int BN_, int WR, int WC>
__global__ __launch_bounds__(THREADS, 2)
void conv_igemm_kernel(ConvParams p) {
  constexpr int WM = BM_ / WR;        // wave tile M
  constexpr int WN = BN_ / WC;        // wave tile N
  constexpr int MF = WM / 16;         // M fragments per wave
  constexpr int NF = WN / 16;         // N fragments per wave
  constexpr int CA = BM_ * BK / 8 / THREADS;  // A chunks per thread
  constexpr int CB = BN_ * BK / 8 / THREADS;  // B chunks per thread
  static_assert(WR * WC == 4, "4 waves");
  static_assert(CA >= 1 && CB >= 1, "tile vs thread count");

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* lds = reinterpret_cast<bf16*>(smem);
  auto lds_tile = [&](int buf, int ab) {
    return lds + buf * (BM_ + BN_) * BK + (ab ? BM_ * BK : 0);
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  const int nbn = (int)((p.Nd + BN_ - 1) / BN_);
  int block = blockIdx.x;
  // XCD-aware swizzle: contiguous chunks per XCD (bijective form)
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rmd = nwg % 8;
    const int xcd = block % 8, idx = block / 8;
    block = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
            + idx;
  }
  const long bm = (long)(block / nbn) * BM_;
  const long bn = (long)(block % nbn) * BN_;

  const int nsteps = (int)((p.Kd + BK - 1) / BK);

  // ---- staging slots: slot ca -> (row = ca>>3, stored cg = ca&7),
  // holding source logical cg = (ca&7) ^ (row&7)
  int a_cg[CA], b_row[CB], b_cg[CB];
  RowCoords a_rc[CA];
#pragma unroll
  for (int l = 0; l < CA; ++l) {
    const int ca = l * THREADS + tid;
    const int arow = ca >> 3;
    a_cg[l] = (ca & 7) ^ (arow & 7);
    a_rc[l] = a_row_coords<MODE>(p, bm + arow);
  }
#pragma unroll
  for (int l = 0; l < CB; ++l) {
    const int ca = l * THREADS + tid;
    b_row[l] = ca >> 3;
    b_cg[l] = (ca & 7) ^ (b_row[l] & 7);
  }

  auto stage = [&](int buf, int step) {
    const long kk0 = (long)step * BK;
    bf16* la = lds_tile(buf, 0);
    bf16* lb = lds_tile(buf, 1);
#pragma unroll
    for (int l = 0; l < CA; ++l) {
      const long kkga = kk0 + a_cg[l] * 8;
      const bf16* src = a_step_addr<MODE>(p, a_rc[l],
                                          kkga < p.Kd ? kkga : 0);
      if (kkga >= p.Kd) src = p.zero;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(la +
              (l * THREADS + tid) * 8), 16, 0, 0);
    }
#pragma unroll
    for (int l = 0; l < CB; ++l) {
      const bf16* src = b_chunk_addr<MODE>(p, bn + b_row[l],
                                           kk0 + b_cg[l] * 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lb +
              (l * THREADS + tid) * 8), 16, 0, 0);
    }
  };

  // ---- fragment read offsets (XOR re-applied) -----------------------
  const int fr = lane & 15;         // row-in-frag
  const int fk = lane >> 4;         // k-subgroup
  const int wm = (wid / WC) * WM;   // wave M offset
  const int wn = (wid % WC) * WN;   // wave N offset

  auto frag_ptr = [&](bf16* tile, int row, int ks) -> const bf16x8* {
    const int cg = (ks * 4 + fk) ^ (row & 7);
    return reinterpret_cast<const bf16x8*>(tile + row * BK + cg * 8);
  };

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j)
      acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < nsteps; ++t) {
    if (t + 1 < nsteps) stage(cur ^ 1, t + 1);
    bf16* la = lds_tile(cur, 0);
    bf16* lb = lds_tile(cur, 1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[MF], bfr[NF];
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
        af[mf] = *frag_ptr(la, wm + mf * 16 + fr, ks);
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
        bfr[nf] = *frag_ptr(lb, wn + nf * 16 + fr, ks);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mf], bfr[nf], acc[mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: D[row=(lane>>4)*4+reg][col=lane&15] per fragment ---
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const long row = bm + wm + mf * 16 + fk * 4 + reg;
      if (row >= p.M) continue;
      long out_row = row;
      if (MODE == 2) {
        // class row -> scattered dx pixel (n, 2*ii+a, 2*jj+b)
        const int hw = p.nh * p.nw;
        const int n = (int)(row / hw);
        const int rem = (int)(row - (long)n * hw);
        const int ii = rem / p.nw, jj = rem - (rem / p.nw) * p.nw;
        out_row = ((long)n * p.H + 2 * ii + p.cls_a) * p.W
                  + 2 * jj + p.cls_b;
      }
#pragma unroll
      for (int nf = 0; nf < NF; ++nf) {
        const long col = bn + wn + nf * 16 + fr;
        if (col < p.Nd)
          p.out[out_row * p.Nd + col] = from_f32<bf16>(acc[mf][nf][reg]);
      }
    }
  }
}

// ---- launchers --------------------------------------------------------
void launch_conv_igemm(const void* a, const void* b, void* out,
                       const void* zero, int N, int H, int W, int Cin,
                       int K, int OH, int OW, int R, int S, int stride,
                       int pad, int dgrad, hipStream_t stream) {
  ConvParams p;
  p.a = (const bf16*)a;
  p.b = (const bf16*)b;
  p.out = (bf16*)out;
  p.zero = (const bf16*)zero;
  p.N = N; p.H = H; p.W = W; p.Cin = Cin; p.K = K; p.OH = OH; p.OW = OW;
  p.R = R; p.S = S; p.stride = stride; p.pad = pad;
  if (!dgrad) {
    p.M = (long)N * OH * OW;
    p.Nd = K;
    p.Kd = (long)R * S * Cin;
  } else {
    p.M = (long)N * H * W;
    p.Nd = Cin;
    p.Kd = (long)R * S * K;
  }

#define LAUNCH(MODE, BM_, BN_, WR, WC)                                      \
  do {                                                                      \
    const long nbm = (p.M + (BM_) - 1) / (BM_);                             \
    const long nbn = (p.Nd + (BN_) - 1) / (BN_);                            \
    const size_t lds_bytes = 2 * ((BM_) + (BN_)) * BK * sizeof(bf16);       \
    hipLaunchKernelGGL((conv_igemm_kernel<MODE, BM_, BN_, WR, WC>),         \
                       dim3((unsigned)(nbm * nbn)), dim3(THREADS),          \
                       lds_bytes, stream, p);                               \
  } while (0)
#define LAUNCH_TILED(MODE)                                                  \
  do {                                                                      \
    if (p.Nd <= 64 && p.M >= 256) LAUNCH(MODE, 256, 64, 4, 1);              \
    else if (p.Nd <= 64) LAUNCH(MODE, 128, 64, 4, 1);                       \
    else LAUNCH(MODE, 128, 128, 2, 2);                                      \
  } while (0)

  // deep-pipeline structure (conv_mfma2.hip) first; DDLB_CONV_V2=0
  // forces the 128-tile structure (re-read per call so benchmarks can
  // A/B the two structures in one process)
  const char* v2e = getenv("DDLB_CONV_V2");
  const bool use_v2 = !(v2e && v2e[0] == '0');

  if (!dgrad) {
    if (!(use_v2 && launch_conv_igemm_v2(p, 0, stream))) LAUNCH_TILED(0);
  } else if (stride == 2 && K % 8 == 0) {
    // four parity classes, each a dense reduction over its valid taps
    for (int a_ = 0; a_ < 2; ++a_) {
      for (int b_ = 0; b_ < 2; ++b_) {
        p.cls_a = a_;
        p.cls_b = b_;
        // r valid iff (ih + pad - r) even  ->  r ≡ (a_ + pad) (mod 2)
        p.r0 = (a_ + pad) & 1;
        p.s0 = (b_ + pad) & 1;
        p.nr = (R - p.r0 + 1) / 2;
        p.ns = (S - p.s0 + 1) / 2;
        p.nh = (H - a_ + 1) / 2;
        p.nw = (W - b_ + 1) / 2;
        if (p.nr <= 0 || p.ns <= 0 || p.nh <= 0 || p.nw <= 0) continue;
        p.M = (long)N * p.nh * p.nw;
        p.Kd = (long)p.nr * p.ns * K;
        if (!(use_v2 && launch_conv_igemm_v2(p, 2, stream)))
          LAUNCH_TILED(2);
      }
    }
  } else {
    if (!(use_v2 && launch_conv_igemm_v2(p, 1, stream))) LAUNCH_TILED(1);
  }
#undef LAUNCH_TILED
#undef LAUNCH
  HIP_CHECK_LAST();
}
