// Depthwise 3x3 conv (stride 1/2, pad 1) forward + backward for gfx950.
//
// The MobileNetV2 hot op (groups == channels —
// /root/reference/benchmark/mnist/models/mnistmobilenetv2.py:28).
// Bandwidth-bound with zero inner-product depth, so no MFMA: one thread
// per output pixel, coalesced along W, 3x3 taps unrolled, weights via
// the read-only cache. NCHW.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

template <typename T, int S>
__global__ void dw3x3_fwd_kernel(const T* __restrict__ x,
                                 const float* __restrict__ w,
                                 T* __restrict__ y, int64_t N, int64_t C,
                                 int64_t H, int64_t W, int64_t OH,
                                 int64_t OW) {
  const int64_t total = N * C * OH * OW;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t ow = i % OW;
    const int64_t oh = (i / OW) % OH;
    const int64_t c = (i / (OW * OH)) % C;
    const int64_t n = i / (OW * OH * C);
    const float* wc = w + c * 9;
    const T* xp = x + (n * C + c) * H * W;
    float acc = 0.f;
    const int64_t ih0 = oh * S - 1, iw0 = ow * S - 1;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int64_t ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int64_t iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        acc = fmaf(to_f32(xp[ih * W + iw]), wc[kh * 3 + kw], acc);
      }
    }
    y[i] = from_f32<T>(acc);
  }
}

// dx: one thread per input pixel, gathering from dy
template <typename T, int S>
__global__ void dw3x3_bwd_dx_kernel(const T* __restrict__ dy,
                                    const float* __restrict__ w,
                                    T* __restrict__ dx, int64_t N, int64_t C,
                                    int64_t H, int64_t W, int64_t OH,
                                    int64_t OW) {
  const int64_t total = N * C * H * W;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t iw = i % W;
    const int64_t ih = (i / W) % H;
    const int64_t c = (i / (W * H)) % C;
    const int64_t n = i / (W * H * C);
    const float* wc = w + c * 9;
    const T* dyp = dy + (n * C + c) * OH * OW;
    float acc = 0.f;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int64_t t = ih + 1 - kh;  // oh*S = ih + pad - kh
      if (t < 0 || t % S) continue;
      const int64_t oh = t / S;
      if (oh >= OH) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int64_t u = iw + 1 - kw;
        if (u < 0 || u % S) continue;
        const int64_t ow = u / S;
        if (ow >= OW) continue;
        acc = fmaf(to_f32(dyp[oh * OW + ow]), wc[kh * 3 + kw], acc);
      }
    }
    dx[i] = from_f32<T>(acc);
  }
}

// dw: grid (C*9, SLICES) blocks reduce over N*OH*OW, atomic into f64
template <typename T, int S>
__global__ void dw3x3_bwd_dw_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    double* __restrict__ dw, int64_t N,
                                    int64_t C, int64_t H, int64_t W,
                                    int64_t OH, int64_t OW) {
  __shared__ double tmp[8];
  const int64_t tap = blockIdx.x;  // c*9 + kh*3 + kw
  const int64_t c = tap / 9;
  const int kh = (int)((tap % 9) / 3), kw = (int)(tap % 3);
  const int64_t total = N * OH * OW;
  const int64_t per = (total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, total);
  double acc = 0.0;
  for (int64_t i = begin + threadIdx.x; i < end; i += blockDim.x) {
    const int64_t ow = i % OW;
    const int64_t oh = (i / OW) % OH;
    const int64_t n = i / (OW * OH);
    const int64_t ih = oh * S - 1 + kh, iw = ow * S - 1 + kw;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
    acc += (double)to_f32(dy[(n * C + c) * OH * OW + oh * OW + ow]) *
           (double)to_f32(x[(n * C + c) * H * W + ih * W + iw]);
  }
  auto op = [](double v) { return wave_reduce_sum(v); };
  acc = block_reduce(acc, tmp, op, 0.0);
  if (threadIdx.x == 0) atomicAdd(&dw[tap], acc);
}

// ---- NHWC variants: fixed 8-channel group per thread -------------------
// Geometry mirrors the fused-BN NHWC kernels: thread owns channels
// c0..c0+7, walks output pixels; the 72 per-channel weights hoist into
// registers once, taps are 16-byte channel-vector loads.
typedef __attribute__((ext_vector_type(8))) short short8dw;

DEV float dw_bf16_at(const short8dw& v, int j) {
  __hip_bfloat16 h;
  unsigned short u = (unsigned short)v[j];
  __builtin_memcpy(&h, &u, 2);
  return __bfloat162float(h);
}

template <typename T, int S>
__global__ void dw3x3_fwd_nhwc_kernel(const T* __restrict__ x,
                                      const float* __restrict__ w,
                                      T* __restrict__ y, int64_t N,
                                      int64_t C, int64_t H, int64_t W,
                                      int64_t OH, int64_t OW, int CG8) {
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  float wr[9][8];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int j = 0; j < 8; ++j) wr[t][j] = w[(c0 + j) * 9 + t];
  const int64_t pix_total = N * OH * OW;
  const int64_t per = (pix_total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, pix_total);
  const short8dw* xv = reinterpret_cast<const short8dw*>(x);
  short8dw* yv = reinterpret_cast<short8dw*>(y);
  for (int64_t p = begin + rj; p < end; p += RG) {
    const int64_t ow = p % OW;
    const int64_t oh = (p / OW) % OH;
    const int64_t n = p / (OW * OH);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    const int64_t ih0 = oh * S - 1, iw0 = ow * S - 1;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int64_t ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int64_t iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        short8dw v = xv[(((n * H + ih) * W + iw) * C + c0) / 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] = fmaf(dw_bf16_at(v, j), wr[kh * 3 + kw][j], acc[j]);
      }
    }
    short8dw vy;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h = __float2bfloat16(acc[j]);
      unsigned short u;
      __builtin_memcpy(&u, &h, 2);
      vy[j] = (short)u;
    }
    yv[(p * C + c0) / 8] = vy;
  }
}

template <typename T, int S>
__global__ void dw3x3_bwd_dx_nhwc_kernel(const T* __restrict__ dy,
                                         const float* __restrict__ w,
                                         T* __restrict__ dx, int64_t N,
                                         int64_t C, int64_t H, int64_t W,
                                         int64_t OH, int64_t OW, int CG8) {
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  float wr[9][8];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int j = 0; j < 8; ++j) wr[t][j] = w[(c0 + j) * 9 + t];
  const int64_t pix_total = N * H * W;
  const int64_t per = (pix_total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, pix_total);
  const short8dw* dyv = reinterpret_cast<const short8dw*>(dy);
  short8dw* dxv = reinterpret_cast<short8dw*>(dx);
  for (int64_t p = begin + rj; p < end; p += RG) {
    const int64_t iw = p % W;
    const int64_t ih = (p / W) % H;
    const int64_t n = p / (W * H);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int64_t t = ih + 1 - kh;
      if (t < 0 || t % S) continue;
      const int64_t oh = t / S;
      if (oh >= OH) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int64_t u2 = iw + 1 - kw;
        if (u2 < 0 || u2 % S) continue;
        const int64_t ow = u2 / S;
        if (ow >= OW) continue;
        short8dw v = dyv[(((n * OH + oh) * OW + ow) * C + c0) / 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[j] = fmaf(dw_bf16_at(v, j), wr[kh * 3 + kw][j], acc[j]);
      }
    }
    short8dw vy;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h = __float2bfloat16(acc[j]);
      unsigned short u;
      __builtin_memcpy(&u, &h, 2);
      vy[j] = (short)u;
    }
    dxv[(p * C + c0) / 8] = vy;
  }
}

// dw: thread owns 8 channels, accumulates all 9 taps, atomics at end
template <typename T, int S>
__global__ void dw3x3_bwd_dw_nhwc_kernel(const T* __restrict__ x,
                                         const T* __restrict__ dy,
                                         double* __restrict__ dw,
                                         int64_t N, int64_t C, int64_t H,
                                         int64_t W, int64_t OH, int64_t OW,
                                         int CG8) {
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  float acc[9][8];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[t][j] = 0.f;
  const int64_t pix_total = N * OH * OW;
  const int64_t per = (pix_total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, pix_total);
  const short8dw* dyv = reinterpret_cast<const short8dw*>(dy);
  const short8dw* xv = reinterpret_cast<const short8dw*>(x);
  for (int64_t p = begin + rj; p < end; p += RG) {
    const int64_t ow = p % OW;
    const int64_t oh = (p / OW) % OH;
    const int64_t n = p / (OW * OH);
    short8dw vdy = dyv[(p * C + c0) / 8];
    const int64_t ih0 = oh * S - 1, iw0 = ow * S - 1;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int64_t ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int64_t iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        short8dw vx = xv[(((n * H + ih) * W + iw) * C + c0) / 8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[kh * 3 + kw][j] = fmaf(dw_bf16_at(vx, j),
                                     dw_bf16_at(vdy, j),
                                     acc[kh * 3 + kw][j]);
      }
    }
  }
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&dw[(c0 + j) * 9 + t], (double)acc[t][j]);
}

static inline int ew_grid(int64_t total, int block) {
  int64_t want = (total + block - 1) / block;
  return (int)i64min(want > 0 ? want : 1, 256 * 8);
}

template <typename T>
void launch_dw3x3_fwd(const T* x, const float* w, T* y, int64_t N, int64_t C,
                      int64_t H, int64_t W, int64_t OH, int64_t OW,
                      int stride_, int nhwc, hipStream_t stream) {
  const int block = 256;
  if (nhwc && sizeof(T) == 2 && C % 8 == 0) {
    const int CG8 = (int)i64min(C / 8, 64);
    const int64_t cblocks = (C / 8 + CG8 - 1) / CG8;
    int64_t S = i64min(i64max(N * OH * OW / 256, 1),
                       i64max(2048 / cblocks, 1));
    if (stride_ == 1)
      hipLaunchKernelGGL((dw3x3_fwd_nhwc_kernel<T, 1>), dim3(cblocks, S),
                         dim3(block), 0, stream, x, w, y, N, C, H, W, OH,
                         OW, CG8);
    else
      hipLaunchKernelGGL((dw3x3_fwd_nhwc_kernel<T, 2>), dim3(cblocks, S),
                         dim3(block), 0, stream, x, w, y, N, C, H, W, OH,
                         OW, CG8);
    HIP_CHECK_LAST();
    return;
  }
  const int grid = ew_grid(N * C * OH * OW, block);
  if (stride_ == 1)
    hipLaunchKernelGGL((dw3x3_fwd_kernel<T, 1>), dim3(grid), dim3(block), 0,
                       stream, x, w, y, N, C, H, W, OH, OW);
  else
    hipLaunchKernelGGL((dw3x3_fwd_kernel<T, 2>), dim3(grid), dim3(block), 0,
                       stream, x, w, y, N, C, H, W, OH, OW);
  HIP_CHECK_LAST();
}

template <typename T>
void launch_dw3x3_bwd_dx(const T* dy, const float* w, T* dx, int64_t N,
                         int64_t C, int64_t H, int64_t W, int64_t OH,
                         int64_t OW, int stride_, int nhwc,
                         hipStream_t stream) {
  const int block = 256;
  if (nhwc && sizeof(T) == 2 && C % 8 == 0) {
    const int CG8 = (int)i64min(C / 8, 64);
    const int64_t cblocks = (C / 8 + CG8 - 1) / CG8;
    int64_t S = i64min(i64max(N * H * W / 256, 1),
                       i64max(2048 / cblocks, 1));
    if (stride_ == 1)
      hipLaunchKernelGGL((dw3x3_bwd_dx_nhwc_kernel<T, 1>),
                         dim3(cblocks, S), dim3(block), 0, stream, dy, w,
                         dx, N, C, H, W, OH, OW, CG8);
    else
      hipLaunchKernelGGL((dw3x3_bwd_dx_nhwc_kernel<T, 2>),
                         dim3(cblocks, S), dim3(block), 0, stream, dy, w,
                         dx, N, C, H, W, OH, OW, CG8);
    HIP_CHECK_LAST();
    return;
  }
  const int grid = ew_grid(N * C * H * W, block);
  if (stride_ == 1)
    hipLaunchKernelGGL((dw3x3_bwd_dx_kernel<T, 1>), dim3(grid), dim3(block),
                       0, stream, dy, w, dx, N, C, H, W, OH, OW);
  else
    hipLaunchKernelGGL((dw3x3_bwd_dx_kernel<T, 2>), dim3(grid), dim3(block),
                       0, stream, dy, w, dx, N, C, H, W, OH, OW);
  HIP_CHECK_LAST();
}

template <typename T>
void launch_dw3x3_bwd_dw(const T* x, const T* dy, double* dw, int64_t N,
                         int64_t C, int64_t H, int64_t W, int64_t OH,
                         int64_t OW, int stride_, int nhwc,
                         hipStream_t stream) {
  const int block = 256;
  if (nhwc && sizeof(T) == 2 && C % 8 == 0) {
    const int CG8 = (int)i64min(C / 8, 64);
    const int64_t cblocks = (C / 8 + CG8 - 1) / CG8;
    int64_t Sg = i64min(i64max(N * OH * OW / 1024, 1),
                        i64max(1024 / cblocks, 1));
    if (stride_ == 1)
      hipLaunchKernelGGL((dw3x3_bwd_dw_nhwc_kernel<T, 1>),
                         dim3(cblocks, Sg), dim3(block), 0, stream, x, dy,
                         dw, N, C, H, W, OH, OW, CG8);
    else
      hipLaunchKernelGGL((dw3x3_bwd_dw_nhwc_kernel<T, 2>),
                         dim3(cblocks, Sg), dim3(block), 0, stream, x, dy,
                         dw, N, C, H, W, OH, OW, CG8);
    HIP_CHECK_LAST();
    return;
  }
  int64_t S = i64min((N * OH * OW + block - 1) / block,
                           i64max(2048 / (C * 9), 1));
  S = i64max(S, 1);
  if (stride_ == 1)
    hipLaunchKernelGGL((dw3x3_bwd_dw_kernel<T, 1>), dim3(C * 9, S),
                       dim3(block), 0, stream, x, dy, dw, N, C, H, W, OH, OW);
  else
    hipLaunchKernelGGL((dw3x3_bwd_dw_kernel<T, 2>), dim3(C * 9, S),
                       dim3(block), 0, stream, x, dy, dw, N, C, H, W, OH, OW);
  HIP_CHECK_LAST();
}

#define INSTANTIATE(T)                                                       \
  template void launch_dw3x3_fwd<T>(const T*, const float*, T*, int64_t,     \
                                    int64_t, int64_t, int64_t, int64_t,      \
                                    int64_t, int, int, hipStream_t);         \
  template void launch_dw3x3_bwd_dx<T>(const T*, const float*, T*, int64_t,  \
                                       int64_t, int64_t, int64_t, int64_t,   \
                                       int64_t, int, int, hipStream_t);      \
  template void launch_dw3x3_bwd_dw<T>(const T*, const T*, double*, int64_t, \
                                       int64_t, int64_t, int64_t, int64_t,   \
                                       int64_t, int, int, hipStream_t);
INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
#undef INSTANTIATE
