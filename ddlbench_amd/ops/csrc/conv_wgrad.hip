// NHWC bf16 convolution weight-grad on MFMA (gfx950) — completes the
// native conv triple (fprop/dgrad in conv_mfma.hip).
//
//   dw[k][r,s,c] = sum_{n,oh,ow} dy[n,oh,ow,k] * x[n, oh*st+r-p, ow*st+s-p, c]
//
// GEMM view: C[k][rsc] = A[k][p] * B[p][rsc] with the reduction over
// output pixels p. Both operands are p-major in memory (dy rows are
// [p][K], im2col rows are [p][rsc]), i.e. TRANSPOSED from the fragment
// layout, so both tiles stage into LDS row-major-in-p (16-byte
// global_load_lds chunks, the same XOR chunk swizzle as conv_mfma) and
// fragments gather with per-element transposed LDS reads. Split-P grid
// accumulates fp32 partials with atomics.
//
// Correctness-first structure (one LDS buffer pair, scalar transposed
// reads); the ds_read_b64_tr_b16 image is the planned next step
// (docs/ROADMAP.md). Constraints: bf16, K % 8 == 0, C % 8 == 0.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short bf16x8w;
typedef __attribute__((ext_vector_type(4))) float f32x4w;

#define WG_BKP 64     // p rows per step
#define WG_TK 64      // k tile
#define WG_TR 64      // rsc tile
#define WG_THREADS 256

struct WgradParams {
  const bf16* x;      // (N,H,W,C) memory
  const bf16* dy;     // (N,OH,OW,K) memory
  float* dw;          // (K, R*S*C) fp32, zero-initialized
  const bf16* zero;
  int N, H, W, C, K, OH, OW, R, S, stride, pad;
  long P;             // N*OH*OW
  long RSC;
  int split_p;        // p-chunks per (k,rsc) tile
};

__global__ __launch_bounds__(WG_THREADS, 2)
void conv_wgrad_kernel(WgradParams q) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* la = reinterpret_cast<bf16*>(smem);                 // [64p][64k]
  bf16* lb = la + WG_BKP * WG_TK;                           // [64p][64rsc]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  const int nk = (q.K + WG_TK - 1) / WG_TK;
  const int nr = (int)((q.RSC + WG_TR - 1) / WG_TR);
  const long block = blockIdx.x;
  const long kb = (block % nk) * WG_TK;
  const long rb = ((block / nk) % nr) * WG_TR;
  const int ps_idx = (int)(block / ((long)nk * nr));
  const long p_per = (q.P + q.split_p - 1) / q.split_p;
  const long p_begin = ps_idx * p_per;
  const long p_end = i64min(p_begin + p_per, q.P);

  // staging slots: 512 chunks per tile, 2 per thread per tile;
  // slot (prow = ca>>3, cg_store = ca&7) holds source chunk
  // cg = cg_store ^ (prow & 7)  (16B chunks stay contiguous)
  const int nsteps = (int)((p_end - p_begin + WG_BKP - 1) / WG_BKP);

  // wave tile: 64k x 16rsc (4 waves side by side in rsc)
  const int wn = wid * 16;
  f32x4w acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int t = 0; t < nsteps; ++t) {
    const long p0 = p_begin + (long)t * WG_BKP;
    // ---- stage A (dy) and B (im2col x) --------------------------------
#pragma unroll
    for (int l = 0; l < 2; ++l) {
      const int ca = l * WG_THREADS + tid;
      const int prow = ca >> 3;
      const int cg = (ca & 7) ^ (prow & 7);
      const long p = p0 + prow;
      const bf16* src = q.zero;
      if (p < p_end && kb + cg * 8 < q.K)
        src = q.dy + p * q.K + kb + cg * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(la + ca * 8), 16, 0,
          0);
    }
#pragma unroll
    for (int l = 0; l < 2; ++l) {
      const int ca = l * WG_THREADS + tid;
      const int prow = ca >> 3;
      const int cg = (ca & 7) ^ (prow & 7);
      const long p = p0 + prow;
      const long rsc = rb + cg * 8;
      const bf16* src = q.zero;
      if (p < p_end && rsc < q.RSC) {
        const int ohw = q.OH * q.OW;
        const int n = (int)(p / ohw);
        const int rem = (int)(p - (long)n * ohw);
        const int oh = rem / q.OW, ow = rem - (rem / q.OW) * q.OW;
        const int c0 = (int)(rsc % q.C);
        const int rs = (int)(rsc / q.C);
        const int r = rs / q.S, s = rs - (rs / q.S) * q.S;
        const int ih = oh * q.stride + r - q.pad;
        const int iw = ow * q.stride + s - q.pad;
        if (ih >= 0 && ih < q.H && iw >= 0 && iw < q.W)
          src = q.x + (((long)n * q.H + ih) * q.W + iw) * q.C + c0;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lb + ca * 8), 16, 0,
          0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    // ---- transposed fragment reads + MFMA -----------------------------
    // element (p, col) lives at p*64 + ((col>>3)^(p&7))*8 + (col&7)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8w bfrag;
      {
        const int col = wn + (lane & 15);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int p = ks * 32 + (lane >> 4) * 8 + e;
          bfrag[e] = reinterpret_cast<const short*>(lb)[
              p * WG_TR + ((col >> 3) ^ (p & 7)) * 8 + (col & 7)];
        }
      }
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        bf16x8w afrag;
        const int col = mf * 16 + (lane & 15);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int p = ks * 32 + (lane >> 4) * 8 + e;
          afrag[e] = reinterpret_cast<const short*>(la)[
              p * WG_TK + ((col >> 3) ^ (p & 7)) * 8 + (col & 7)];
        }
        acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mf], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- fp32 atomic accumulation into dw -------------------------------
  // D[row = (lane>>4)*4 + reg][col = lane&15]; row is the k index
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const long k = kb + mf * 16 + (lane >> 4) * 4 + reg;
      const long rsc = rb + wn + (lane & 15);
      if (k < q.K && rsc < q.RSC)
        atomicAdd(&q.dw[k * q.RSC + rsc], acc[mf][reg]);
    }
  }
}

void launch_conv_wgrad(const void* x, const void* dy, float* dw,
                       const void* zero, int N, int H, int W, int C, int K,
                       int OH, int OW, int R, int S, int stride, int pad,
                       hipStream_t stream) {
  WgradParams q;
  q.x = (const bf16*)x;
  q.dy = (const bf16*)dy;
  q.dw = dw;
  q.zero = (const bf16*)zero;
  q.N = N; q.H = H; q.W = W; q.C = C; q.K = K; q.OH = OH; q.OW = OW;
  q.R = R; q.S = S; q.stride = stride; q.pad = pad;
  q.P = (long)N * OH * OW;
  q.RSC = (long)R * S * C;
  const long nk = (K + WG_TK - 1) / WG_TK;
  const long nr = (q.RSC + WG_TR - 1) / WG_TR;
  long split = 2048 / i64max(nk * nr, 1);
  split = i64max(i64min(split, (q.P + WG_BKP - 1) / WG_BKP), 1);
  q.split_p = (int)split;
  const size_t lds_bytes = (WG_BKP * WG_TK + WG_BKP * WG_TR) * sizeof(bf16);
  hipLaunchKernelGGL(conv_wgrad_kernel,
                     dim3((unsigned)(nk * nr * q.split_p)),
                     dim3(WG_THREADS), lds_bytes, stream, q);
  HIP_CHECK_LAST();
}
