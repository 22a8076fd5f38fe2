// Deep-pipelined NHWC implicit-GEMM convolution on MFMA (gfx950, bf16).
//
// Second-generation structure for the conv core the reference gets from
// cuDNN: the 256-row counted-vmcnt schedule from the
// CDNA4 guide's 8-phase GEMM template (cdna_hip_programming.md §5),
// adapted to implicit-GEMM staging:
//
//   * 256xBN block tile, BK=64, 8 waves (512 threads), 2 K-tile LDS
//     double buffer staged entirely by 16-byte global_load_lds.
//   * Each K-tile is computed in 4 phases (m-half x k-half); each phase
//     {ds_read fragment subtile; issue ONE half-slot prefetch (glds);
//      raw s_barrier; MFMA cluster under s_setprio(1); raw s_barrier}.
//   * vmcnt is COUNTED, never drained mid-loop: one s_waitcnt
//     vmcnt(GA+GB) per k-half boundary leaves the two most recent
//     half-slot prefetches in flight across the barriers (~2 phases
//     ~1100 cycles of slack > the ~900-cycle HBM latency). Raw
//     s_barrier (not __syncthreads) so the barrier itself never drains
//     the glds queue (the documented -16..-20% trap).
//   * LDS layout: per (buffer, k-half) a [rows][32] bf16 slot; the
//     16-byte chunk at (row, slot s) holds logical k-chunk
//     cg = s ^ sigma((row>>2)&3) with sigma = [0,2,3,1] — derived from
//     the measured ds_read_b128 lane groups {0-3,12-15,20-27} etc.
//     (MI355X_MICROARCH.md §LDS) to be conflict-free for the MFMA
//     fragment read (rows = lane&15, chunk = lane>>4). glds writes
//     lane-linear; the permutation is applied to the SOURCE address and
//     re-applied on the read (guide rule 21). The XOR argument reduces
//     to per-thread constants on both sides.
//   * A-operand im2col addressing is strength-reduced: the row->pixel
//     decomposition is computed once per staging slot and the
//     (r,s,c)-cursor advances by +32 with carries per stage call — no
//     divmod in the K-loop.
//
// Variants: BN=256 (waves 2x4, wave tile 128x64, 16 MFMA/phase,
// 128 KiB LDS) for Nd>=192; BN=128 (wave tile 128x32, 8 MFMA/phase,
// 96 KiB LDS) for Nd>=96. Narrower Nd falls back to conv_mfma.hip's
// 128-tile structure (launcher returns false).

#include "conv_igemm.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

using bf16 = conv_bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM2 256
#define BK2 64
#define THREADS2 512

namespace {

template <int N> DEV void wait_vmcnt() {
  static_assert(N >= 0 && N <= 8, "unsupported vmcnt");
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 1) asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else if constexpr (N == 3) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 7) asm volatile("s_waitcnt vmcnt(7)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
}

DEV void glds16(const bf16* src, bf16* dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)src,
      (__attribute__((address_space(3))) void*)dst, 16, 0, 0);
}

// sigma = [0,2,3,1]: stored slot s holds logical chunk s ^ sigma(rb)
DEV int sigma4(int x) { return (0x1320 >> (x * 4)) & 0xF; }

// ---- staging state ----------------------------------------------------
// Register-lean by construction (the BN=256 variant has ~96 VGPRs left
// after the 128-reg accumulator + 32-reg fragment set): the thread's GA
// A-slots and GB B-slots all carry the SAME k-chunk (cg depends only on
// the lane), so ONE (r,s,c) cursor per thread serves every slot;
// per-slot state is a 32-bit element offset plus a packed s16x2 (u,v)
// spatial base. The launcher guarantees every offset fits 32 bits.
//
// Cursor semantics per mode over kkg = cg*8 + 32*t:
//   MODE 0: kkg = (r*S + s)*Cin + c   (u,v) = (y*stride-pad, x*stride-pad)
//   MODE 1: kkg = (r*S + s)*K   + c   (u,v) = (y+pad, x+pad); stride==1
//   MODE 2: kkg = (r*ns + s)*K  + c   (u,v) = (ohb, owb)
struct Cursor { int r, s, c; };

template <int MODE>
DEV Cursor cursor_init(const ConvParams& p, int cg) {
  Cursor cu;
  const int kk = cg * 8;
  const int lim = (MODE == 0) ? p.Cin : p.K;
  const int send = (MODE == 2) ? p.ns : p.S;
  cu.c = kk % lim;
  const int rs = kk / lim;
  cu.r = rs / send;
  cu.s = rs - cu.r * send;
  return cu;
}

template <int MODE>
DEV void cursor_advance(const ConvParams& p, Cursor& cu) {
  cu.c += 32;
  const int lim = (MODE == 0) ? p.Cin : p.K;
  const int send = (MODE == 2) ? p.ns : p.S;
  while (cu.c >= lim) {
    cu.c -= lim;
    if (++cu.s == send) {
      cu.s = 0;
      ++cu.r;
    }
  }
}

DEV int pack_uv(int u, int v) { return (u << 16) | (v & 0xffff); }
DEV int uv_u(int uv) { return uv >> 16; }
DEV int uv_v(int uv) { return (int)(short)(uv & 0xffff); }

// per-A-slot init: 32-bit batch-base offset + packed spatial base;
// an invalid row poisons u so every bounds test fails
template <int MODE>
DEV void a_slot_init(const ConvParams& p, long m, unsigned& base,
                     int& uv) {
  const ConvRowCoords rc = conv_a_row_coords<MODE>(p, m);
  if (MODE == 0) {
    base = (unsigned)((long)rc.n * p.H * p.W * p.Cin);
    uv = pack_uv(rc.valid ? rc.y * p.stride - p.pad : -20000,
                 rc.x * p.stride - p.pad);
  } else if (MODE == 1) {
    base = (unsigned)((long)rc.n * p.OH * p.OW * p.K);
    uv = pack_uv(rc.valid ? rc.y + p.pad : -20000, rc.x + p.pad);
  } else {
    base = (unsigned)((long)rc.n * p.OH * p.OW * p.K);
    uv = pack_uv(rc.valid ? (rc.y + p.pad - p.r0) >> 1 : -20000,
                 (rc.x + p.pad - p.s0) >> 1);
  }
}

template <int MODE>
DEV const bf16* a_slot_addr(const ConvParams& p, const Cursor& cu,
                            unsigned base, int uv) {
  if (MODE == 0) {
    const int ih = uv_u(uv) + cu.r;
    const int iw = uv_v(uv) + cu.s;
    if (cu.r >= p.R || (unsigned)ih >= (unsigned)p.H ||
        (unsigned)iw >= (unsigned)p.W)
      return p.zero;
    return p.a + base + (unsigned)((ih * p.W + iw) * p.Cin + cu.c);
  } else {
    const int oh = uv_u(uv) - cu.r;
    const int ow = uv_v(uv) - cu.s;
    const int rend = (MODE == 2) ? p.nr : p.R;
    if (cu.r >= rend || (unsigned)oh >= (unsigned)p.OH ||
        (unsigned)ow >= (unsigned)p.OW)
      return p.zero;
    return p.a + base + (unsigned)((oh * p.OW + ow) * p.K + cu.c);
  }
}

// per-B-slot: modes 0/1 advance a flat offset; mode 2 recomputes from
// the shared cursor. kkg < Kd tail test == (cursor.r < R|nr).
template <int MODE>
DEV unsigned b_slot_init(const ConvParams& p, long col, int cg,
                         bool& colok) {
  colok = col < p.Nd;
  const long c0 = colok ? col : 0;
  if (MODE != 2) return (unsigned)(c0 * p.Kd + cg * 8);
  return (unsigned)(c0 * p.R * p.S * p.K);
}

template <int MODE>
DEV const bf16* b_slot_addr(const ConvParams& p, const Cursor& cu,
                            unsigned off, bool colok) {
  if (MODE != 2) {
    if (!colok || cu.r >= p.R) return p.zero;
    return p.b + off;
  }
  if (!colok || cu.r >= p.nr) return p.zero;
  return p.b + off +
         (unsigned)(((p.r0 + 2 * cu.r) * p.S + p.s0 + 2 * cu.s) * p.K +
                    cu.c);
}

// ---- the kernel -------------------------------------------------------
// COMB_B (BN_ == 64): the B slot is too small for one glds per thread
// per k-half, so one stage call covers BOTH k-halves (waves 0-3 fill
// k0, waves 4-7 fill k1) and B's cursor advances by 64.
// LIN: 1x1 stride-1 pad-0 conv — the implicit-GEMM A operand is the
// plain row-major tensor (no im2col arithmetic, no bounds tests)
template <int MODE, int BM_, int BN_, int WR, int WC, bool PH2 = false,
          bool LIN = false>
__global__ __launch_bounds__(THREADS2, 2)
void conv_igemm2_kernel(ConvParams p) {
  constexpr int WM = BM_ / WR;       // wave tile M
  constexpr int WN = BN_ / WC;       // wave tile N
  constexpr int MF = WM / 16;        // acc M fragments
  constexpr int NF = WN / 16;        // acc N fragments
  constexpr int MH = MF / 2;         // M fragments per phase
  constexpr int GA = BM_ * 32 / 8 / THREADS2;  // A glds per wave/slot
  constexpr bool COMB_B = (BN_ * 32 / 8) < THREADS2;
  constexpr int GB = COMB_B ? 1 : BN_ * 32 / 8 / THREADS2;
  static_assert(WR * WC == 8, "8 waves");
  static_assert(GA >= 1 && MH >= 1, "tile vs thread count");
  static_assert(!COMB_B || BN_ == 64, "combined-B staging needs BN=64");

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* lds = reinterpret_cast<bf16*>(smem);
  // slot(buf, kh): A rows then B rows, 32 bf16 per row
  auto aslot = [&](int buf, int kh) {
    return lds + ((buf * 2 + kh) * (BM_ + BN_)) * 32;
  };
  auto bslot = [&](int buf, int kh) {
    return aslot(buf, kh) + BM_ * 32;
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int fr = lane & 15;
  const int fk = lane >> 4;

  const int nbn = (int)((p.Nd + BN_ - 1) / BN_);
  int block = blockIdx.x;
  {  // bijective XCD-aware remap: contiguous chunk per XCD
    const int nwg = gridDim.x;
    const int q = nwg / 8, rmd = nwg % 8;
    const int xcd = block % 8, idx = block / 8;
    block = (xcd < rmd ? xcd * (q + 1) : rmd * (q + 1) + (xcd - rmd) * q)
            + idx;
  }
  const long bm = (long)(block / nbn) * BM_;
  const long bn = (long)(block % nbn) * BN_;

  const int nsteps = (int)((p.Kd + BK2 - 1) / BK2);

  // ---- staging slots ---------------------------------------------------
  // chunk index within a slot: ca = (wid*G + j)*64 + lane
  //   row = ca>>2, stored pos = ca&3, source cg = (ca&3)^sigma((ca>>4)&3)
  // Every slot of this thread shares the same cg. A and B each keep a
  // cursor: A advances +32 per stage_a (k-half), B +32 per stage_b —
  // or +64 when one combined call stages both halves.
  const int cg = (lane & 3) ^ sigma4((lane >> 4) & 3);
  Cursor acur = cursor_init<MODE>(p, cg);
  Cursor bcur = cursor_init<MODE>(p, cg + (COMB_B && wid >= 4 ? 4 : 0));
  unsigned abase[GA];
  int auv[GA];
  unsigned boff[GB];
  bool bok[GB];
#pragma unroll
  for (int j = 0; j < GA; ++j) {
    const int ca = (wid * GA + j) * 64 + lane;
    if (LIN) {
      const long m = bm + (ca >> 2);
      const long cols = (MODE == 0) ? p.Cin : p.K;
      abase[j] = (unsigned)((m < p.M ? m : 0) * cols);
      auv[j] = m < p.M;
    } else {
      a_slot_init<MODE>(p, bm + (ca >> 2), abase[j], auv[j]);
    }
  }
#pragma unroll
  for (int j = 0; j < GB; ++j) {
    const int cb = COMB_B ? (wid & 3) * 64 + lane
                          : (wid * GB + j) * 64 + lane;
    boff[j] = b_slot_init<MODE>(p, bn + (cb >> 2),
                                cg + (COMB_B && wid >= 4 ? 4 : 0),
                                bok[j]);
  }

  auto stage_a = [&](int buf, int kh) {
    bf16* la = aslot(buf, kh);
#pragma unroll
    for (int j = 0; j < GA; ++j) {
      const bf16* src;
      if (LIN)
        src = (auv[j] && acur.r == 0)
                  ? p.a + abase[j] + (unsigned)acur.c
                  : p.zero;
      else
        src = a_slot_addr<MODE>(p, acur, abase[j], auv[j]);
      glds16(src, la + ((wid * GA + j) * 64 + lane) * 8);
    }
    cursor_advance<MODE>(p, acur);
  };
  // non-combined: stages one k-half. combined: waves 0-3 fill k0 and
  // waves 4-7 fill k1 in ONE call (kh argument ignored).
  auto stage_b = [&](int buf, int kh) {
    bf16* lb = bslot(buf, COMB_B ? (wid >= 4 ? 1 : 0) : kh);
#pragma unroll
    for (int j = 0; j < GB; ++j) {
      glds16(b_slot_addr<MODE>(p, bcur, boff[j], bok[j]),
             lb + (((COMB_B ? (wid & 3) : wid * GB + j)) * 64 + lane) * 8);
      if (MODE != 2) boff[j] += COMB_B ? 64 : 32;
    }
    cursor_advance<MODE>(p, bcur);
    if (COMB_B) cursor_advance<MODE>(p, bcur);
  };

  // ---- fragment read geometry -----------------------------------------
  const int wr = wid / WC;           // wave M position
  const int wc = wid % WC;           // wave N position
  const int fsw = sigma4((fr >> 2) & 3);
  const int fchunk = (fk ^ fsw) * 8; // element offset of this lane's chunk
  const int awoff = wr * WM;         // wave A row base
  const int bwoff = wc * WN;         // wave B row base

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j)
      acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  bf16x8 bfr[NF];  // B fragments, retained across the two m-half phases

  // ---- prologue: stage K-tile 0 fully ----------------------------------
  stage_a(0, 0);
  stage_b(0, 0);  // combined form also covers k1 here
  stage_a(0, 1);
  if (!COMB_B) stage_b(0, 1);
  // A(k0),B(k0) landed; later halves may stay in flight
  wait_vmcnt<COMB_B ? GA : GA + GB>();
  __builtin_amdgcn_s_barrier();

  // ---- main loop: 4 phases per K-tile ----------------------------------
  // PH(msub, kh): ds_read fragments, issue one half-slot prefetch,
  // barrier, MFMA cluster, (counted vmcnt), barrier.
#define PH(M0, MC, KH, STAGE, VMW)                                       \
  do {                                                                    \
    bf16* la = aslot(cbuf, KH);                                           \
    bf16x8 af[MC];                                                        \
    _Pragma("unroll")                                                     \
    for (int mf = 0; mf < (MC); ++mf)                                     \
      af[mf] = *reinterpret_cast<const bf16x8*>(                          \
          la + (awoff + (M0) * (WM / 2) + mf * 16 + fr) * 32 + fchunk);   \
    if ((M0) == 0) {                                                      \
      bf16* lb = bslot(cbuf, KH);                                         \
      _Pragma("unroll")                                                   \
      for (int nf = 0; nf < NF; ++nf)                                     \
        bfr[nf] = *reinterpret_cast<const bf16x8*>(                       \
            lb + (bwoff + nf * 16 + fr) * 32 + fchunk);                   \
    }                                                                     \
    STAGE;                                                                \
    __builtin_amdgcn_s_barrier();                                         \
    __builtin_amdgcn_s_setprio(1);                                        \
    _Pragma("unroll")                                                     \
    for (int mf = 0; mf < (MC); ++mf)                                     \
      _Pragma("unroll")                                                   \
      for (int nf = 0; nf < NF; ++nf)                                     \
        acc[(M0) * MH + mf][nf] =                                         \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(                      \
                af[mf], bfr[nf], acc[(M0) * MH + mf][nf], 0, 0, 0);       \
    __builtin_amdgcn_s_setprio(0);                                        \
    VMW;                                                                  \
    __builtin_amdgcn_s_barrier();                                         \
  } while (0)

  // counted waits: end-of-ph1 guards this tile's k1 reads (allows the
  // two units issued at ph0/ph1 to stay in flight); end-of-ph3 guards
  // the next tile's k0 reads (allows ph2/ph3's units)
  constexpr int V1 = COMB_B ? GA + 1 : GA + GB;
  constexpr int V3 = COMB_B ? GA : GA + GB;
  for (int t = 0; t < nsteps; ++t) {
    const int cbuf = t & 1;
    const int nxt = cbuf ^ 1;
    const bool more = (t + 1 < nsteps);
    if (PH2) {
      // 2 phases per K-tile (full-M x k-half, 2x the MFMA per barrier
      // pair): both operand halves stage inside one phase, one counted
      // vmcnt per phase, 1-phase prefetch slack. COMB_B stages B (both
      // halves) once, in ph0.
      if (more && COMB_B) {
        PH(0, MF, 0, (stage_a(nxt, 0), stage_b(nxt, 0)),
           wait_vmcnt<GA + 1>());
        PH(0, MF, 1, stage_a(nxt, 1), wait_vmcnt<GA>());
      } else if (more) {
        PH(0, MF, 0, (stage_a(nxt, 0), stage_b(nxt, 0)),
           wait_vmcnt<V1>());
        PH(0, MF, 1, (stage_a(nxt, 1), stage_b(nxt, 1)),
           wait_vmcnt<V1>());
      } else {
        PH(0, MF, 0, , wait_vmcnt<0>());
        PH(0, MF, 1, , );
      }
    } else if (more) {
      PH(0, MH, 0, stage_a(nxt, 0), );
      PH(1, MH, 0, stage_b(nxt, 0), wait_vmcnt<V1>());
      PH(0, MH, 1, stage_a(nxt, 1), );
      if (COMB_B) {
        PH(1, MH, 1, , wait_vmcnt<V3>());
      } else {
        PH(1, MH, 1, stage_b(nxt, 1), wait_vmcnt<V3>());
      }
    } else {
      // last K-tile: nothing left to stage; the k1 halves may still be
      // in flight, so the mid-tile wait drains fully
      PH(0, MH, 0, , );
      PH(1, MH, 0, , wait_vmcnt<0>());
      PH(0, MH, 1, , );
      PH(1, MH, 1, , );
    }
  }
#undef PH

  // ---- epilogue: LDS transpose -> full-width coalesced stores ----------
  // The MFMA D layout (row = fk*4+reg per lane, col = fr across lanes)
  // would store 2 B/lane in 32-B row segments (~2x store-bandwidth
  // waste — these big-M/small-Kd convs are store-bound). Round-trip the
  // tile through the now-free LDS (row-major [BM_][BN_] bf16 image; its
  // 256-B rows make the b128 read-back groups conflict-free) and emit
  // 16-B/lane stores covering full rows.
  wait_vmcnt<0>();
  __builtin_amdgcn_s_barrier();
  static_assert((long)BM_ * BN_ * 2 <= 4L * (BM_ + BN_) * 32 * 2,
                "tile image must fit the staging LDS");
  {
    bf16* img = lds;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
      for (int nf = 0; nf < NF; ++nf)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg)
          img[(awoff + mf * 16 + fk * 4 + reg) * BN_ +
              bwoff + nf * 16 + fr] = from_f32<bf16>(acc[mf][nf][reg]);
    __syncthreads();
    constexpr int CPR = BN_ / 8;            // 16-B chunks per tile row
    constexpr int ITERS = BM_ * CPR / THREADS2;
    typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      const int idx = it * THREADS2 + tid;
      const int r = idx / CPR;
      const int ck = idx - r * CPR;
      const long row = bm + r;
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
      const long colbase = bn + ck * 8;
      const bf16* src = img + r * BN_ + ck * 8;
      if (colbase + 8 <= p.Nd) {
        *reinterpret_cast<u16x8*>(p.out + out_row * p.Nd + colbase) =
            *reinterpret_cast<const u16x8*>(src);
      } else if (colbase < p.Nd) {
        for (int j = 0; j < (int)(p.Nd - colbase); ++j)
          p.out[out_row * p.Nd + colbase + j] = src[j];
      }
    }
  }
}

template <int MODE>
bool dispatch_v2(const ConvParams& p, hipStream_t stream) {
  // DDLB_CONV_PH2=0 selects the 4-phase schedule for A/B (default: the
  // 2-phase full-M schedule — half the barriers per K-tile; the PMC
  // profile showed 37% of wave cycles parked on barriers at 4-phase)
  const char* phe = getenv("DDLB_CONV_PH2");
  const bool ph2 = !(phe && phe[0] == '0');
  const bool lin = (MODE != 2 && p.R == 1 && p.S == 1 && p.stride == 1 &&
                    p.pad == 0);
#define LAUNCH2(BM_, BN_, WR, WC, PH2_)                                     \
  do {                                                                      \
    const long nbm = (p.M + (BM_) - 1) / (BM_);                             \
    const long nbn = (p.Nd + (BN_) - 1) / (BN_);                            \
    const size_t lds_bytes = 4 * ((BM_) + (BN_)) * 32 * sizeof(bf16);       \
    if (lin)                                                                \
      hipLaunchKernelGGL(                                                   \
          (conv_igemm2_kernel<MODE, BM_, BN_, WR, WC, PH2_, true>),         \
          dim3((unsigned)(nbm * nbn)), dim3(THREADS2),                      \
          lds_bytes, stream, p);                                            \
    else                                                                    \
      hipLaunchKernelGGL(                                                   \
          (conv_igemm2_kernel<MODE, BM_, BN_, WR, WC, PH2_, false>),        \
          dim3((unsigned)(nbm * nbn)), dim3(THREADS2),                      \
          lds_bytes, stream, p);                                            \
  } while (0)
  // BN=256 (acc 128 regs/wave) cannot fit beside the im2col staging
  // state in the 256-VGPR/2-wave budget (measured 105-reg spill); the
  // 256x128 tile (acc 64, ~204 VGPRs clean) serves all Nd >= 96 with
  // column blocks. Nd in [48,96) gets a 512x64 tile (combined-B).
  // Default: the high-occupancy 128x128 tile (116-122 VGPR -> 4
  // waves/SIMD, 64 KiB LDS -> 2 blocks/CU): cross-block TLP hides the
  // staging latency the 1-block 256x128 tile paid in parked waves
  // (measured fwd 1.64->1.39 ms aggregate, dgrad 1.86->1.60 = 0.84x
  // MIOpen). DDLB_CONV_TILE=b restores the big tile for A/B.
  const char* te = getenv("DDLB_CONV_TILE");
  const bool big_tile = te && te[0] == 'b';
  if (p.Nd >= 96 && !big_tile) {
    if (ph2) LAUNCH2(128, 128, 2, 4, true);
    else LAUNCH2(128, 128, 2, 4, false);
  } else if (p.Nd >= 96) {
    if (ph2) LAUNCH2(256, 128, 2, 4, true);
    else LAUNCH2(256, 128, 2, 4, false);
  } else if (p.Nd >= 48 && !big_tile) {
    if (ph2) LAUNCH2(256, 64, 4, 2, true);
    else LAUNCH2(256, 64, 4, 2, false);
  } else if (p.Nd >= 48) {
    if (ph2) LAUNCH2(512, 64, 8, 1, true);
    else LAUNCH2(512, 64, 8, 1, false);
  } else {
    return false;
  }
#undef LAUNCH2
  return true;
}

}  // namespace

bool launch_conv_igemm_v2(const ConvParams& p, int mode,
                          hipStream_t stream) {
  // staging offsets are 32-bit; MODE 1 is stride-1 only (stride-2
  // dgrad goes through the MODE-2 parity classes)
  const long a_elems = (mode == 0)
                           ? (long)p.N * p.H * p.W * p.Cin
                           : (long)p.N * p.OH * p.OW * p.K;
  const long b_elems = (mode == 2)
                           ? (long)p.Nd * p.R * p.S * p.K
                           : p.Nd * p.Kd;
  if (a_elems >= (1L << 31) || b_elems >= (1L << 31)) return false;
  if (mode == 1 && p.stride != 1) return false;
  bool ok;
  if (mode == 0)
    ok = dispatch_v2<0>(p, stream);
  else if (mode == 1)
    ok = dispatch_v2<1>(p, stream);
  else
    ok = dispatch_v2<2>(p, stream);
  if (ok) HIP_CHECK_LAST();
  return ok;
}
