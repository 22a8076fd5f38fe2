// NHWC bf16 max-pool forward + backward (gfx950).
//
// The resnet stem's 3x3/s2 maxpool is the last torch-library kernel of
// any size on the flagship step (at::max_pool_backward_nhwc ~0.8 ms of
// a ~30 ms step). NHWC vectorized rewrite, bn_apply-style geometry:
// one thread owns an 8-channel group and walks output pixels, taps are
// 16-B channel-vector loads; forward stores a per-channel argmax byte
// (window slot 0..k*k-1), backward recomputes each INPUT pixel's
// covering windows and gathers dy where the argmax byte matches — no
// atomics, no zero-init scatter.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short short8m;
typedef __attribute__((ext_vector_type(8))) unsigned char u8x8m;

namespace {

DEV float b2f(short v) {
  __hip_bfloat16 h;
  unsigned short u = (unsigned short)v;
  __builtin_memcpy(&h, &u, 2);
  return __bfloat162float(h);
}

__global__ void maxpool_fwd_nhwc_kernel(
    const bf16* __restrict__ x, bf16* __restrict__ y,
    unsigned char* __restrict__ idx, int64_t N, int64_t C, int H, int W,
    int OH, int OW, int k, int stride, int pad, int CG8) {
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  const int64_t rows = N * OH * OW;
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  const short8m* xv = reinterpret_cast<const short8m*>(x);
  short8m* yv = reinterpret_cast<short8m*>(y);
  for (int64_t r = begin + rj; r < end; r += RG) {
    const int64_t n = r / (OH * OW);
    const int rem = (int)(r - n * OH * OW);
    const int oh = rem / OW, ow = rem - (rem / OW) * OW;
    float best[8];
    int barg[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      best[j] = -3.4e38f;
      barg[j] = 0;
    }
    const int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
    for (int kr = 0; kr < k; ++kr) {
      const int ih = ih0 + kr;
      if (ih < 0 || ih >= H) continue;
      for (int ks = 0; ks < k; ++ks) {
        const int iw = iw0 + ks;
        if (iw < 0 || iw >= W) continue;
        const short8m v =
            xv[(((n * H + ih) * W + iw) * C + c0) / 8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float f = b2f(v[j]);
          if (f > best[j]) {
            best[j] = f;
            barg[j] = kr * k + ks;
          }
        }
      }
    }
    short8m vy;
    u8x8m vi;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h = __float2bfloat16(best[j]);
      unsigned short u;
      __builtin_memcpy(&u, &h, 2);
      vy[j] = (short)u;
      vi[j] = (unsigned char)barg[j];
    }
    yv[(r * C + c0) / 8] = vy;
    *reinterpret_cast<u8x8m*>(idx + r * C + c0) = vi;
  }
}

__global__ void maxpool_bwd_nhwc_kernel(
    const bf16* __restrict__ dy, const unsigned char* __restrict__ idx,
    bf16* __restrict__ dx, int64_t N, int64_t C, int H, int W,
    int OH, int OW, int k, int stride, int pad, int CG8) {
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  const int64_t rows = N * H * W;
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  const short8m* dyv = reinterpret_cast<const short8m*>(dy);
  short8m* dxv = reinterpret_cast<short8m*>(dx);
  for (int64_t r = begin + rj; r < end; r += RG) {
    const int64_t n = r / (H * W);
    const int rem = (int)(r - n * H * W);
    const int ih = rem / W, iw = rem - (rem / W) * W;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    // covering output windows: oh*stride - pad <= ih < ... + k
    const int oh_lo = max(0, (ih + pad - k + stride) / stride);
    const int oh_hi = min(OH - 1, (ih + pad) / stride);
    const int ow_lo = max(0, (iw + pad - k + stride) / stride);
    const int ow_hi = min(OW - 1, (iw + pad) / stride);
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      const int kr = ih - (oh * stride - pad);
      if (kr < 0 || kr >= k) continue;
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        const int ks = iw - (ow * stride - pad);
        if (ks < 0 || ks >= k) continue;
        const int64_t orow = (n * OH + oh) * OW + ow;
        const u8x8m vi =
            *reinterpret_cast<const u8x8m*>(idx + orow * C + c0);
        const unsigned char want = (unsigned char)(kr * k + ks);
        bool any = false;
#pragma unroll
        for (int j = 0; j < 8; ++j) any |= (vi[j] == want);
        if (!any) continue;
        const short8m g = dyv[(orow * C + c0) / 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (vi[j] == want) acc[j] += b2f(g[j]);
      }
    }
    short8m vdx;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h = __float2bfloat16(acc[j]);
      unsigned short u;
      __builtin_memcpy(&u, &h, 2);
      vdx[j] = (short)u;
    }
    dxv[(r * C + c0) / 8] = vdx;
  }
}

}  // namespace

static void mp_geom(int64_t rows, int64_t C, int& CG8, int64_t& cblocks,
                    int64_t& S) {
  CG8 = (int)i64min(C / 8, 64);
  cblocks = (C / 8 + CG8 - 1) / CG8;
  S = i64min(i64max(rows / 16, 1), i64max(2048 / cblocks, 1));
}

void launch_maxpool_fwd(const void* x, void* y, unsigned char* idx,
                        int64_t N, int64_t C, int H, int W, int OH,
                        int OW, int k, int stride, int pad,
                        hipStream_t stream) {
  int CG8;
  int64_t cblocks, S;
  mp_geom(N * OH * OW, C, CG8, cblocks, S);
  hipLaunchKernelGGL(maxpool_fwd_nhwc_kernel, dim3(cblocks, S),
                     dim3(256), 0, stream, (const bf16*)x, (bf16*)y, idx,
                     N, C, H, W, OH, OW, k, stride, pad, CG8);
  HIP_CHECK_LAST();
}

void launch_maxpool_bwd(const void* dy, const unsigned char* idx,
                        void* dx, int64_t N, int64_t C, int H, int W,
                        int OH, int OW, int k, int stride, int pad,
                        hipStream_t stream) {
  int CG8;
  int64_t cblocks, S;
  mp_geom(N * H * W, C, CG8, cblocks, S);
  hipLaunchKernelGGL(maxpool_bwd_nhwc_kernel, dim3(cblocks, S),
                     dim3(256), 0, stream, (const bf16*)dy, idx,
                     (bf16*)dx, N, C, H, W, OH, OW, k, stride, pad, CG8);
  HIP_CHECK_LAST();
}
