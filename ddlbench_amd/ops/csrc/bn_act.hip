// Fused BatchNorm2d + activation (+ residual add) for gfx950, NCHW.
//
// Replaces the reference's unfused BatchNorm2d -> ReLU (-> add) chains
// (SURVEY.md §2.6 "Key model ops") with single-pass kernels:
//   forward train:  stats reduce -> finalize -> apply(normalize+act[+add])
//   forward eval:   apply only (running stats)
//   backward:       masked-dy reduce -> finalize -> dx elementwise
// BN is HBM-bandwidth-bound on MI355X (≈8 TB/s HBM3E): fusing the
// activation and the residual add into the normalize pass removes one to
// two full tensor round-trips per block vs eager PyTorch.
//
// dtype: x/y/dy in T ∈ {float, bf16}; gamma/beta/stats always fp32;
// reductions accumulate in double (N*HW can exceed 1e7 elements).
// Activation codes: 0 = identity, 1 = relu, 2 = relu6.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

namespace {

template <int ACT> DEV float act_fwd(float v) {
  if (ACT == 1) return fmaxf(v, 0.f);
  if (ACT == 2) return fminf(fmaxf(v, 0.f), 6.f);
  return v;
}

// activation backward mask from the *output* y
template <int ACT> DEV float act_mask(float y) {
  if (ACT == 1) return y > 0.f ? 1.f : 0.f;
  if (ACT == 2) return (y > 0.f && y < 6.f) ? 1.f : 0.f;
  return 1.f;
}

}  // namespace

// BN kernel-variant override for A/B probes (set via bindings):
// 0 = auto, 1 = scalar-group reductions + grid-stride vector apply,
// 2 = vector reductions + fixed-channel apply
int g_bn_variant = 0;
void set_bn_variant(int v) { g_bn_variant = v; }

// ---- stats: per-channel sum / sumsq over N*H*W ------------------------
// grid = (C, S): block (c, s) covers slice s of channel c's N*HW space.
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, double* __restrict__ sums,
                                int64_t N, int64_t C, int64_t HW) {
  __shared__ double tmp[8];
  const int64_t c = blockIdx.x;
  const int64_t total = N * HW;
  const int64_t per = (total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, total);
  double s = 0.0, ss = 0.0;
  for (int64_t i = begin + threadIdx.x; i < end; i += blockDim.x) {
    const int64_t n = i / HW, r = i - n * HW;
    const float v = to_f32(x[(n * C + c) * HW + r]);
    s += v;
    ss += fma((double)v, (double)v, 0.0);
  }
  auto op = [](double v) { return wave_reduce_sum(v); };
  s = block_reduce(s, tmp, op, 0.0);
  __syncthreads();
  ss = block_reduce(ss, tmp, op, 0.0);
  if (threadIdx.x == 0) {
    sums[((int64_t)blockIdx.y * 2 + 0) * C + c] = s;
    sums[((int64_t)blockIdx.y * 2 + 1) * C + c] = ss;
  }
}

// ---- stats, NHWC (channels-last) --------------------------------------
// rows = N*H*W; element (row, c) at row*C + c. Block = CG channels x
// (256/CG) row-groups: every wave reads consecutive channels of one row
// (coalesced), the row-groups + a 4x unroll keep >=16 rows in flight per
// block; per-channel partials combine across row-groups in LDS, one f64
// atomic per channel per block.
template <typename T>
__global__ void bn_stats_nhwc_kernel(const T* __restrict__ x,
                                     double* __restrict__ sums,
                                     int64_t rows, int64_t C, int CG) {
  __shared__ double tmp[2 * 256];
  const int ci = threadIdx.x % CG;
  const int rj = threadIdx.x / CG;
  const int RG = blockDim.x / CG;
  const int64_t c = (int64_t)blockIdx.x * CG + ci;
  const bool active = (c < C) && (rj < RG);
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  double s = 0.0, ss = 0.0;
  if (active) {
    int64_t r = begin + rj;
    const int64_t step = RG;
    for (; r + 3 * step < end; r += 4 * step) {
      float v0 = to_f32(x[(r + 0 * step) * C + c]);
      float v1 = to_f32(x[(r + 1 * step) * C + c]);
      float v2 = to_f32(x[(r + 2 * step) * C + c]);
      float v3 = to_f32(x[(r + 3 * step) * C + c]);
      s += (double)v0 + v1 + v2 + v3;
      ss += fma((double)v0, v0, fma((double)v1, v1,
                fma((double)v2, v2, (double)v3 * v3)));
    }
    for (; r < end; r += step) {
      const float v = to_f32(x[r * C + c]);
      s += v;
      ss += fma((double)v, (double)v, 0.0);
    }
  }
  tmp[threadIdx.x] = s;
  tmp[256 + threadIdx.x] = ss;
  __syncthreads();
  if (rj == 0 && c < C) {
    for (int j = 1; j < RG; ++j) {
      s += tmp[j * CG + ci];
      ss += tmp[256 + j * CG + ci];
    }
    sums[((int64_t)blockIdx.y * 2 + 0) * C + c] = s;
    sums[((int64_t)blockIdx.y * 2 + 1) * C + c] = ss;
  }
}

// ---- stats, NHWC vectorized: 8 channels (16 B) per lane --------------
// Each thread covers channels c0..c0+7 of CG8*8-channel groups; f32
// per-thread partials over a short row slice (<= a few hundred values)
// combine in LDS across row-groups, then one f64 atomic per channel.
template <typename T>
__global__ void bn_stats_nhwc_vec_kernel(const T* __restrict__ x,
                                         double* __restrict__ sums,
                                         int64_t rows, int64_t C, int CG8) {
  typedef __attribute__((ext_vector_type(8))) short short8x;
  __shared__ float tmp[2][256][8];
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  const bool active = (c0 < C) && (rj < RG);
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float ss[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (active) {
    const short8x* xv = reinterpret_cast<const short8x*>(x);
    for (int64_t r = begin + rj; r < end; r += RG) {
      short8x v = xv[(r * C + c0) / 8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __hip_bfloat16 h;
        unsigned short u = (unsigned short)v[j];
        __builtin_memcpy(&h, &u, 2);
        const float f = __bfloat162float(h);
        s[j] += f;
        ss[j] = fmaf(f, f, ss[j]);
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    tmp[0][threadIdx.x][j] = s[j];
    tmp[1][threadIdx.x][j] = ss[j];
  }
  __syncthreads();
  if (rj == 0 && c0 < C) {
    for (int g = 1; g < RG; ++g)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        s[j] += tmp[0][g * CG8 + ci][j];
        ss[j] += tmp[1][g * CG8 + ci][j];
      }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sums[((int64_t)blockIdx.y * 2 + 0) * C + c0 + j] = (double)s[j];
      sums[((int64_t)blockIdx.y * 2 + 1) * C + c0 + j] = (double)ss[j];
    }
  }
}

// fp32 NHWC vec stats (4 channels / 16 B)
__global__ void bn_stats_nhwc_vec_f32_kernel(const float* __restrict__ x,
                                             double* __restrict__ sums,
                                             int64_t rows, int64_t C,
                                             int CG4) {
  typedef __attribute__((ext_vector_type(4))) float float4x;
  __shared__ float tmp[2][256][4];
  const int ci = threadIdx.x % CG4;
  const int rj = threadIdx.x / CG4;
  const int RG = blockDim.x / CG4;
  const int64_t c0 = ((int64_t)blockIdx.x * CG4 + ci) * 4;
  const bool active = (c0 < C) && (rj < RG);
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  float s[4] = {0, 0, 0, 0}, ss[4] = {0, 0, 0, 0};
  if (active) {
    const float4x* xv = reinterpret_cast<const float4x*>(x);
    for (int64_t r = begin + rj; r < end; r += RG) {
      float4x v = xv[(r * C + c0) / 4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        s[j] += v[j];
        ss[j] = fmaf(v[j], v[j], ss[j]);
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    tmp[0][threadIdx.x][j] = s[j];
    tmp[1][threadIdx.x][j] = ss[j];
  }
  __syncthreads();
  if (rj == 0 && c0 < C) {
    for (int g = 1; g < RG; ++g)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        s[j] += tmp[0][g * CG4 + ci][j];
        ss[j] += tmp[1][g * CG4 + ci][j];
      }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      sums[((int64_t)blockIdx.y * 2 + 0) * C + c0 + j] = (double)s[j];
      sums[((int64_t)blockIdx.y * 2 + 1) * C + c0 + j] = (double)ss[j];
    }
  }
}

// ---- finalize: mean/invstd + running-stat update ----------------------
// sums = per-(row-slice) partial slabs [S][2][C] (the reduce kernels
// write plain stores — a C=64 layer at S~1500 slices would otherwise
// serialize ~1500 f64 atomics per channel address)
// one block per channel; 1024 threads block-reduce the S slabs (the
// C-bounded grid is the parallelism bottleneck at C=64/S~2048 — a
// 256-thread block left the chip at 1 wave/CU reading 2 MB)
__global__ void bn_finalize_kernel(const double* __restrict__ sums,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int64_t C, int64_t S, double count,
                                   float eps, float momentum) {
  __shared__ double tmp[16];
  const int64_t c = blockIdx.x;
  double s = 0.0, ss = 0.0;
  for (int64_t b = threadIdx.x; b < S; b += blockDim.x) {
    s += sums[(b * 2 + 0) * C + c];
    ss += sums[(b * 2 + 1) * C + c];
  }
  auto op = [](double v) { return wave_reduce_sum(v); };
  s = block_reduce(s, tmp, op, 0.0);
  __syncthreads();
  ss = block_reduce(ss, tmp, op, 0.0);
  if (threadIdx.x != 0) return;
  const double m = s / count;
  double var = ss / count - m * m;
  var = var < 0.0 ? 0.0 : var;
  mean[c] = (float)m;
  invstd[c] = (float)rsqrt(var + (double)eps);
  if (running_mean != nullptr) {
    const double unbiased = count > 1.0 ? var * count / (count - 1.0) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * (float)m;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * (float)unbiased;
  }
}

// ---- apply: y = act(gamma*(x-mean)*invstd + beta [+ res]) -------------
template <typename T, int ACT, bool ADD>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const T* __restrict__ res,
                                T* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                int64_t C, int64_t cdiv, int64_t total) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t c = (i / cdiv) % C;
    float v = (to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (ADD) v += to_f32(res[i]);
    y[i] = from_f32<T>(act_fwd<ACT>(v));
  }
}

// ---- vectorized apply / dx: 8 elements (16 B) per thread -------------
// NHWC: 8 consecutive elements are 8 consecutive channels (chunks are
// 8-aligned in c when C %% 8 == 0); NCHW: 8 consecutive elements share
// one channel when HW %% 8 == 0. Dispatcher falls back to the scalar
// kernels otherwise.
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

template <typename T> struct vec8;
template <> struct vec8<float> {
  using type = __attribute__((ext_vector_type(8))) float;
};
template <> struct vec8<__hip_bfloat16> { using type = short8v; };

DEV float elt_f32(const short8v& v, int j) {
  __hip_bfloat16 h;
  unsigned short u = (unsigned short)v[j];
  __builtin_memcpy(&h, &u, 2);
  return __bfloat162float(h);
}
DEV float elt_f32(const __attribute__((ext_vector_type(8))) float& v,
                  int j) { return v[j]; }

template <typename T>
DEV void set_elt(short8v& v, int j, float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  unsigned short u;
  __builtin_memcpy(&u, &h, 2);
  v[j] = (short)u;
}
template <typename T>
DEV void set_elt(__attribute__((ext_vector_type(8))) float& v, int j,
                 float f) { v[j] = f; }

template <typename T, int ACT, bool ADD, bool NHWC>
__global__ void bn_apply_vec_kernel(const T* __restrict__ x,
                                    const T* __restrict__ res,
                                    T* __restrict__ y,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    int64_t C, int64_t cdiv,
                                    int64_t total8) {
  using V = typename vec8<T>::type;
  const V* xv = reinterpret_cast<const V*>(x);
  const V* rv = reinterpret_cast<const V*>(res);
  V* yv = reinterpret_cast<V*>(y);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total8; i += stride) {
    const int64_t e0 = i * 8;
    V vx = xv[i];
    V vr;
    if (ADD) vr = rv[i];
    V vy;
    if (NHWC) {
      const int64_t c0 = e0 % C;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int64_t c = c0 + j;
        float v = (elt_f32(vx, j) - mean[c]) * invstd[c] * gamma[c]
                  + beta[c];
        if (ADD) v += elt_f32(vr, j);
        set_elt<T>(vy, j, act_fwd<ACT>(v));
      }
    } else {
      const int64_t c = (e0 / cdiv) % C;
      const float mu = mean[c], is = invstd[c], g = gamma[c], b = beta[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = (elt_f32(vx, j) - mu) * is * g + b;
        if (ADD) v += elt_f32(vr, j);
        set_elt<T>(vy, j, act_fwd<ACT>(v));
      }
    }
    yv[i] = vy;
  }
}

template <typename T, int ACT, bool ADD, bool NHWC>
__global__ void bn_bwd_dx_vec_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ y,
                                     const T* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     const float* __restrict__ k,
                                     T* __restrict__ dx,
                                     T* __restrict__ dres, int64_t C,
                                     int64_t cdiv, int64_t total8) {
  using V = typename vec8<T>::type;
  const V* dyv = reinterpret_cast<const V*>(dy);
  const V* yv = reinterpret_cast<const V*>(y);
  const V* xv = reinterpret_cast<const V*>(x);
  V* dxv = reinterpret_cast<V*>(dx);
  V* drv = reinterpret_cast<V*>(dres);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total8; i += stride) {
    const int64_t e0 = i * 8;
    V vdy = dyv[i], vy = yv[i], vx = xv[i];
    V vdx, vdr;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int64_t c = NHWC ? (e0 % C + j) : ((e0 / cdiv) % C);
      const float g = elt_f32(vdy, j) * act_mask<ACT>(elt_f32(vy, j));
      const float xhat = (elt_f32(vx, j) - mean[c]) * invstd[c];
      set_elt<T>(vdx, j, k[c] * (g - k[C + c] - xhat * k[2 * C + c]));
      if (ADD) set_elt<T>(vdr, j, g);
    }
    dxv[i] = vdx;
    if (ADD) drv[i] = vdr;
  }
}

// ---- NHWC apply / dx, fixed-channel geometry ---------------------------
// Same (CG8 x RG) block shape as the reductions: each thread owns 8
// channels and walks rows, so per-channel params load ONCE per thread
// instead of once per element (the grid-stride variant is issue-bound
// on the 32 extra scalar loads per 48 B of traffic).
template <typename T, int ACT, bool ADD>
__global__ void bn_apply_nhwc_kernel2(const T* __restrict__ x,
                                      const T* __restrict__ res,
                                      T* __restrict__ y,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      unsigned char* __restrict__ msk,
                                      int64_t rows, int64_t C, int CG8) {
  typedef __attribute__((ext_vector_type(8))) short short8x;
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = invstd[c0 + j] * gamma[c0 + j];
    sh[j] = beta[c0 + j] - mean[c0 + j] * sc[j];
  }
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  const short8x* xv = reinterpret_cast<const short8x*>(x);
  const short8x* rv = reinterpret_cast<const short8x*>(res);
  short8x* yv = reinterpret_cast<short8x*>(y);
  for (int64_t r = begin + rj; r < end; r += RG) {
    const int64_t i8 = (r * C + c0) / 8;
    short8x vx = xv[i8];
    short8x vr;
    if (ADD) vr = rv[i8];
    short8x vy;
    unsigned mbits = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h;
      unsigned short u = (unsigned short)vx[j];
      __builtin_memcpy(&h, &u, 2);
      float v = fmaf(__bfloat162float(h), sc[j], sh[j]);
      if (ADD) {
        u = (unsigned short)vr[j];
        __builtin_memcpy(&h, &u, 2);
        v += __bfloat162float(h);
      }
      if (ACT != 0 && act_mask<ACT>(v) > 0.f) mbits |= 1u << j;
      h = __float2bfloat16(act_fwd<ACT>(v));
      __builtin_memcpy(&u, &h, 2);
      vy[j] = (short)u;
    }
    yv[i8] = vy;
    // 1-bit activation mask: backward re-reads this byte instead of y
    // (docs/ROADMAP.md item 7: ~2 of 7 memory passes saved)
    if (ACT != 0 && msk != nullptr) msk[i8] = (unsigned char)mbits;
  }
}

// MASKED: the activation gradient gate comes from the 1-bit mask the
// forward apply wrote (one byte per 8 channels); y is never read.
template <typename T, int ACT, bool ADD, bool MASKED = false>
__global__ void bn_bwd_dx_nhwc_kernel2(const T* __restrict__ dy,
                                       const T* __restrict__ y,
                                       const T* __restrict__ x,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ invstd,
                                       const float* __restrict__ k,
                                       T* __restrict__ dx,
                                       T* __restrict__ dres,
                                       const unsigned char* __restrict__ msk,
                                       int64_t rows,
                                       int64_t C, int CG8) {
  typedef __attribute__((ext_vector_type(8))) short short8x;
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  if (c0 >= C || rj >= RG) return;
  float k1[8], k2[8], k3[8], mu[8], is[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    k1[j] = k[c0 + j];
    k2[j] = k[C + c0 + j];
    k3[j] = k[2 * C + c0 + j];
    mu[j] = mean[c0 + j];
    is[j] = invstd[c0 + j];
  }
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  const short8x* dyv = reinterpret_cast<const short8x*>(dy);
  const short8x* yv = reinterpret_cast<const short8x*>(y);
  const short8x* xv = reinterpret_cast<const short8x*>(x);
  short8x* dxv = reinterpret_cast<short8x*>(dx);
  short8x* drv = reinterpret_cast<short8x*>(dres);
  for (int64_t r = begin + rj; r < end; r += RG) {
    const int64_t i8 = (r * C + c0) / 8;
    short8x vdy = dyv[i8], vx = xv[i8];
    short8x vy;
    unsigned mb = 0;
    if (MASKED) mb = msk[i8];
    else vy = yv[i8];
    short8x vdx, vdr;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 h;
      unsigned short u;
      u = (unsigned short)vdy[j];
      __builtin_memcpy(&h, &u, 2);
      const float fdy = __bfloat162float(h);
      float gate;
      if (MASKED) {
        gate = (float)((mb >> j) & 1u);
      } else {
        u = (unsigned short)vy[j];
        __builtin_memcpy(&h, &u, 2);
        gate = act_mask<ACT>(__bfloat162float(h));
      }
      u = (unsigned short)vx[j];
      __builtin_memcpy(&h, &u, 2);
      const float fx = __bfloat162float(h);
      const float g = fdy * gate;
      const float xhat = (fx - mu[j]) * is[j];
      h = __float2bfloat16(k1[j] * (g - k2[j] - xhat * k3[j]));
      __builtin_memcpy(&u, &h, 2);
      vdx[j] = (short)u;
      if (ADD) {
        h = __float2bfloat16(g);
        __builtin_memcpy(&u, &h, 2);
        vdr[j] = (short)u;
      }
    }
    dxv[i8] = vdx;
    if (ADD) drv[i8] = vdr;
  }
}

// ---- backward reduce: per-channel Σdy', Σdy'*xhat ---------------------
template <typename T, int ACT>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ y,
                                     const T* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     double* __restrict__ sums,
                                     int64_t N, int64_t C, int64_t HW) {
  __shared__ double tmp[8];
  const int64_t c = blockIdx.x;
  const float mu = mean[c], is = invstd[c];
  const int64_t total = N * HW;
  const int64_t per = (total + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, total);
  double sdy = 0.0, sdyx = 0.0;
  for (int64_t i = begin + threadIdx.x; i < end; i += blockDim.x) {
    const int64_t n = i / HW, r = i - n * HW;
    const int64_t idx = (n * C + c) * HW + r;
    const float g = to_f32(dy[idx]) * act_mask<ACT>(to_f32(y[idx]));
    const float xhat = (to_f32(x[idx]) - mu) * is;
    sdy += g;
    sdyx += fma((double)g, (double)xhat, 0.0);
  }
  auto op = [](double v) { return wave_reduce_sum(v); };
  sdy = block_reduce(sdy, tmp, op, 0.0);
  __syncthreads();
  sdyx = block_reduce(sdyx, tmp, op, 0.0);
  if (threadIdx.x == 0) {
    sums[((int64_t)blockIdx.y * 2 + 0) * C + c] = sdy;
    sums[((int64_t)blockIdx.y * 2 + 1) * C + c] = sdyx;
  }
}

// ---- backward reduce, NHWC (same block geometry as the stats) ---------
template <typename T, int ACT, bool MASKED = false>
__global__ void bn_bwd_reduce_nhwc_kernel(const T* __restrict__ dy,
                                          const T* __restrict__ y,
                                          const T* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          double* __restrict__ sums,
                                          const unsigned char* __restrict__
                                              msk,
                                          int64_t rows, int64_t C, int CG) {
  __shared__ double tmp[2 * 256];
  const int ci = threadIdx.x % CG;
  const int rj = threadIdx.x / CG;
  const int RG = blockDim.x / CG;
  const int64_t c = (int64_t)blockIdx.x * CG + ci;
  const bool active = (c < C) && (rj < RG);
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  double sdy = 0.0, sdyx = 0.0;
  if (active) {
    const float mu = mean[c], is = invstd[c];
    const int cbit = (int)(c & 7);
    for (int64_t r = begin + rj; r < end; r += RG) {
      const int64_t idx = r * C + c;
      float gate;
      if (MASKED)
        gate = (float)((msk[idx >> 3] >> cbit) & 1u);
      else
        gate = act_mask<ACT>(to_f32(y[idx]));
      const float g = to_f32(dy[idx]) * gate;
      const float xhat = (to_f32(x[idx]) - mu) * is;
      sdy += g;
      sdyx += fma((double)g, (double)xhat, 0.0);
    }
  }
  tmp[threadIdx.x] = sdy;
  tmp[256 + threadIdx.x] = sdyx;
  __syncthreads();
  if (rj == 0 && c < C) {
    for (int j = 1; j < RG; ++j) {
      sdy += tmp[j * CG + ci];
      sdyx += tmp[256 + j * CG + ci];
    }
    sums[((int64_t)blockIdx.y * 2 + 0) * C + c] = sdy;
    sums[((int64_t)blockIdx.y * 2 + 1) * C + c] = sdyx;
  }
}

template <typename T, int ACT, bool MASKED = false>
__global__ void bn_bwd_reduce_nhwc_vec_kernel(
    const T* __restrict__ dy, const T* __restrict__ y,
    const T* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, double* __restrict__ sums,
    const unsigned char* __restrict__ msk,
    int64_t rows, int64_t C, int CG8) {
  typedef __attribute__((ext_vector_type(8))) short short8x;
  __shared__ float tmp[2][256][8];
  const int ci = threadIdx.x % CG8;
  const int rj = threadIdx.x / CG8;
  const int RG = blockDim.x / CG8;
  const int64_t c0 = ((int64_t)blockIdx.x * CG8 + ci) * 8;
  const bool active = (c0 < C) && (rj < RG);
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  float sdy[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float sdyx[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  float mu[8], is[8];
  if (active) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[c0 + j];
      is[j] = invstd[c0 + j];
    }
    const short8x* dyv = reinterpret_cast<const short8x*>(dy);
    const short8x* yv = reinterpret_cast<const short8x*>(y);
    const short8x* xv = reinterpret_cast<const short8x*>(x);
    for (int64_t r = begin + rj; r < end; r += RG) {
      const int64_t i8 = (r * C + c0) / 8;
      short8x vdy = dyv[i8], vx = xv[i8];
      short8x vy;
      unsigned mb = 0;
      if (MASKED) mb = msk[i8];
      else vy = yv[i8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __hip_bfloat16 h;
        unsigned short u;
        u = (unsigned short)vdy[j];
        __builtin_memcpy(&h, &u, 2);
        const float fdy = __bfloat162float(h);
        float gate;
        if (MASKED) {
          gate = (float)((mb >> j) & 1u);
        } else {
          u = (unsigned short)vy[j];
          __builtin_memcpy(&h, &u, 2);
          gate = act_mask<ACT>(__bfloat162float(h));
        }
        u = (unsigned short)vx[j];
        __builtin_memcpy(&h, &u, 2);
        const float fx = __bfloat162float(h);
        const float g = fdy * gate;
        sdy[j] += g;
        sdyx[j] = fmaf(g, (fx - mu[j]) * is[j], sdyx[j]);
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    tmp[0][threadIdx.x][j] = sdy[j];
    tmp[1][threadIdx.x][j] = sdyx[j];
  }
  __syncthreads();
  if (rj == 0 && c0 < C) {
    for (int g = 1; g < RG; ++g)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sdy[j] += tmp[0][g * CG8 + ci][j];
        sdyx[j] += tmp[1][g * CG8 + ci][j];
      }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sums[((int64_t)blockIdx.y * 2 + 0) * C + c0 + j] = (double)sdy[j];
      sums[((int64_t)blockIdx.y * 2 + 1) * C + c0 + j] = (double)sdyx[j];
    }
  }
}

__global__ void bn_bwd_finalize_kernel(const double* __restrict__ sums,
                                       const float* __restrict__ gamma,
                                       const float* __restrict__ invstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ k,  // [3][C]
                                       int64_t C, int64_t S, double count,
                                       int training) {
  __shared__ double tmp[16];
  const int64_t c = blockIdx.x;
  double sdy = 0.0, sdyx = 0.0;
  for (int64_t b = threadIdx.x; b < S; b += blockDim.x) {
    sdy += sums[(b * 2 + 0) * C + c];
    sdyx += sums[(b * 2 + 1) * C + c];
  }
  auto op = [](double v) { return wave_reduce_sum(v); };
  sdy = block_reduce(sdy, tmp, op, 0.0);
  __syncthreads();
  sdyx = block_reduce(sdyx, tmp, op, 0.0);
  if (threadIdx.x != 0) return;
  dgamma[c] = (float)sdyx;
  dbeta[c] = (float)sdy;
  k[c] = gamma[c] * invstd[c];                       // k1
  k[C + c] = training ? (float)(sdy / count) : 0.f;  // k2 (mean of dy')
  k[2 * C + c] = training ? (float)(sdyx / count) : 0.f;  // k3
}

// dx = k1 * (dy' - k2 - xhat*k3); optionally dres = dy'
template <typename T, int ACT, bool ADD>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ y,
                                 const T* __restrict__ x,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ k,
                                 T* __restrict__ dx, T* __restrict__ dres,
                                 int64_t C, int64_t cdiv, int64_t total) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t c = (i / cdiv) % C;
    const float g = to_f32(dy[i]) * act_mask<ACT>(to_f32(y[i]));
    const float xhat = (to_f32(x[i]) - mean[c]) * invstd[c];
    dx[i] = from_f32<T>(k[c] * (g - k[C + c] - xhat * k[2 * C + c]));
    if (ADD) dres[i] = from_f32<T>(g);
  }
}

// ---- launchers --------------------------------------------------------
static inline int elementwise_grid(int64_t total, int block) {
  int64_t want = (total + block - 1) / block;
  return (int)i64min(want > 0 ? want : 1, 256 * 8);
}


// Reduction-kernel dispatch geometry, shared by the launchers AND the
// workspace sizing in the bindings (partial slab = [S][2][C] doubles).
// g_bn_variant: 0 = shape-adaptive (vec reductions for C <= 512 where
// the probe shows them ahead), 1 = scalar, 2 = vec.
struct BnGeom { bool vec; int64_t cblocks, S; };

static BnGeom bn_reduce_geom(int64_t rows, int64_t C, int elsize) {
  BnGeom g;
  const bool vec_ok = (elsize == 2 && C % 8 == 0) ||
                      (elsize == 4 && C % 4 == 0);
  g.vec = vec_ok && (g_bn_variant == 2 ||
                     (g_bn_variant == 0 && C <= 512 && elsize == 2));
  if (g.vec) {
    const int64_t lanes = elsize == 2 ? C / 8 : C / 4;
    const int64_t CG = i64min(lanes, 64);
    g.cblocks = (lanes + CG - 1) / CG;
  } else {
    const int64_t CG = C >= 64 ? 64 : C;
    g.cblocks = (C + CG - 1) / CG;
  }
  // rows/16 (not /512): a block still covers >=16 rows per row-group,
  // and the old cap starved the small-HW layers (C2048@7x7 ran 196
  // blocks on a 256-CU chip)
  g.S = i64min(i64max(rows / 16, 1), i64max(2048 / g.cblocks, 1));
  return g;
}

int64_t bn_reduce_gridS(int64_t N, int64_t C, int64_t HW, int nhwc,
                        int elsize) {
  if (nhwc) return bn_reduce_geom(N * HW, C, elsize).S;
  int64_t S = i64min((N * HW + 255) / 256, i64max(2048 / C, 1));
  return i64max(S, 1);
}

template <typename T>
void launch_bn_stats(const T* x, double* sums, int64_t N, int64_t C,
                     int64_t HW, int64_t S, int nhwc, hipStream_t stream) {
  const int block = 256;
  if (nhwc) {
    const int64_t rows = N * HW;
    const BnGeom g = bn_reduce_geom(rows, C, (int)sizeof(T));
    if (g.vec && sizeof(T) == 2) {
      const int CG8 = (int)i64min(C / 8, 64);
      hipLaunchKernelGGL((bn_stats_nhwc_vec_kernel<T>),
                         dim3(g.cblocks, S),
                         dim3(block), 0, stream, x, sums, rows, C, CG8);
    } else if (g.vec && sizeof(T) == 4) {
      const int CG4 = (int)i64min(C / 4, 64);
      hipLaunchKernelGGL(bn_stats_nhwc_vec_f32_kernel, dim3(g.cblocks, S),
                         dim3(block), 0, stream, (const float*)x, sums,
                         rows, C, CG4);
    } else {
      const int CG = C >= 64 ? 64 : (int)C;
      hipLaunchKernelGGL((bn_stats_nhwc_kernel<T>), dim3(g.cblocks, S),
                         dim3(block), 0, stream, x, sums, rows, C, CG);
    }
  } else {
    hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(C, S), dim3(block), 0,
                       stream, x, sums, N, C, HW);
  }
  HIP_CHECK_LAST();
}

void launch_bn_finalize(double* sums, float* mean, float* invstd,
                        float* rm, float* rv, int64_t C, int64_t S,
                        double count,
                        float eps, float momentum, hipStream_t stream) {
  const int block = 1024;
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((unsigned)C),
                     dim3(block), 0, stream, sums, mean, invstd, rm, rv, C,
                     S, count, eps, momentum);
  HIP_CHECK_LAST();
}

template <typename T>
bool launch_bn_apply(const T* x, const T* res, T* y, const float* mean,
                     const float* invstd, const float* gamma,
                     const float* beta, unsigned char* msk, int64_t C,
                     int64_t HW, int64_t total,
                     int act, int nhwc, hipStream_t stream) {
  const int block = 256;
  const int64_t cdiv = nhwc ? 1 : HW;
  const bool vec8ok = (total % 8 == 0) &&
                      (nhwc ? (C % 8 == 0) : (HW % 8 == 0));
  const int64_t rows_nhwc = nhwc ? total / C : 0;
  // kernel2 starves below ~32k rows (too few blocks); grid-stride vec
  // covers that regime — unless a mask must be written
  if (nhwc && sizeof(T) == 2 && C % 8 == 0 && g_bn_variant != 1 &&
      (rows_nhwc >= 32768 || (msk != nullptr && act != 0) ||
       g_bn_variant == 2)) {
    const int64_t rows = total / C;
    const int CG8 = (int)i64min(C / 8, 64);
    const int64_t cblocks = (C / 8 + CG8 - 1) / CG8;
    int64_t S = i64min(i64max(rows / 16, 1), i64max(2048 / cblocks, 1));
    const bool add = res != nullptr;
#define KCASE(ACT, ADD)                                                     \
    hipLaunchKernelGGL((bn_apply_nhwc_kernel2<T, ACT, ADD>),                \
                       dim3(cblocks, S), dim3(block), 0, stream, x, res,    \
                       y, mean, invstd, gamma, beta, msk, rows, C, CG8)
    if (act == 0) { if (add) KCASE(0, true); else KCASE(0, false); }
    else if (act == 1) { if (add) KCASE(1, true); else KCASE(1, false); }
    else { if (add) KCASE(2, true); else KCASE(2, false); }
#undef KCASE
    HIP_CHECK_LAST();
    return act != 0 && msk != nullptr;  // mask written on this path
  }
  if (vec8ok) {
    const int64_t total8 = total / 8;
    const int grid8 = elementwise_grid(total8, block);
    const bool add = res != nullptr;
#define VCASE(ACT, ADD, NHWC)                                               \
    hipLaunchKernelGGL((bn_apply_vec_kernel<T, ACT, ADD, NHWC>),            \
                       dim3(grid8), dim3(block), 0, stream, x, res, y,      \
                       mean, invstd, gamma, beta, C, cdiv, total8)
#define VSEL(ACT)                                                           \
    do { if (nhwc) { if (add) VCASE(ACT, true, true);                       \
                     else VCASE(ACT, false, true); }                        \
         else { if (add) VCASE(ACT, true, false);                           \
                else VCASE(ACT, false, false); } } while (0)
    if (act == 0) VSEL(0);
    else if (act == 1) VSEL(1);
    else VSEL(2);
#undef VSEL
#undef VCASE
    HIP_CHECK_LAST();
    return false;
  }
  const int grid = elementwise_grid(total, block);
#define CASE(ACT, ADD)                                                       \
  hipLaunchKernelGGL((bn_apply_kernel<T, ACT, ADD>), dim3(grid), dim3(block), \
                     0, stream, x, res, y, mean, invstd, gamma, beta, C,    \
                     cdiv, total)
  const bool add = res != nullptr;
  if (act == 0 && !add) CASE(0, false);
  else if (act == 0 && add) CASE(0, true);
  else if (act == 1 && !add) CASE(1, false);
  else if (act == 1 && add) CASE(1, true);
  else if (act == 2 && !add) CASE(2, false);
  else CASE(2, true);
#undef CASE
  HIP_CHECK_LAST();
  return false;
}

template <typename T>
void launch_bn_bwd_reduce(const T* dy, const T* y, const T* x,
                          const float* mean, const float* invstd,
                          double* sums, const unsigned char* msk,
                          int64_t N, int64_t C, int64_t HW, int64_t S,
                          int act, int nhwc, hipStream_t stream) {
  const int block = 256;
  if (nhwc) {
    const int64_t rows = N * HW;
    const bool masked = msk != nullptr && act != 0;
    const BnGeom g = bn_reduce_geom(rows, C, (int)sizeof(T));
    if (g.vec && sizeof(T) == 2) {
      const int CG8 = (int)i64min(C / 8, 64);
#define CASE(ACT, MSK)                                                      \
      hipLaunchKernelGGL((bn_bwd_reduce_nhwc_vec_kernel<T, ACT, MSK>),      \
                         dim3(g.cblocks, S), dim3(block), 0, stream, dy,    \
                         y, x, mean, invstd, sums, msk, rows, C, CG8)
      if (act == 0) CASE(0, false);
      else if (act == 1) { if (masked) CASE(1, true); else CASE(1, false); }
      else { if (masked) CASE(2, true); else CASE(2, false); }
#undef CASE
    } else {
      const int CG = C >= 64 ? 64 : (int)C;
#define CASE(ACT, MSK)                                                      \
      hipLaunchKernelGGL((bn_bwd_reduce_nhwc_kernel<T, ACT, MSK>),          \
                         dim3(g.cblocks, S), dim3(block), 0, stream, dy,    \
                         y, x, mean, invstd, sums, msk, rows, C, CG)
      if (act == 0) CASE(0, false);
      else if (act == 1) { if (masked) CASE(1, true); else CASE(1, false); }
      else { if (masked) CASE(2, true); else CASE(2, false); }
#undef CASE
    }
  } else {
#define CASE(ACT)                                                          \
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, ACT>), dim3(C, S),         \
                       dim3(block), 0, stream, dy, y, x, mean, invstd,     \
                       sums, N, C, HW)
    if (act == 0) CASE(0);
    else if (act == 1) CASE(1);
    else CASE(2);
#undef CASE
  }
  HIP_CHECK_LAST();
}

void launch_bn_bwd_finalize(double* sums, const float* gamma,
                            const float* invstd, float* dgamma, float* dbeta,
                            float* k, int64_t C, int64_t S, double count,
                            int training, hipStream_t stream) {
  const int block = 1024;
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((unsigned)C),
                     dim3(block), 0, stream, sums, gamma, invstd, dgamma,
                     dbeta, k, C, S, count, training);
  HIP_CHECK_LAST();
}

template <typename T>
void launch_bn_bwd_dx(const T* dy, const T* y, const T* x, const float* mean,
                      const float* invstd, const float* k, T* dx, T* dres,
                      const unsigned char* msk,
                      int64_t C, int64_t HW, int64_t total, int act,
                      int nhwc, hipStream_t stream) {
  const int block = 256;
  const int64_t cdiv = nhwc ? 1 : HW;
  const bool vec8ok = (total % 8 == 0) &&
                      (nhwc ? (C % 8 == 0) : (HW % 8 == 0));
  // a non-null mask FORCES the masked-capable kernel2 path (the vec
  // fallback would read the null y)
  const int64_t rows_nhwc = nhwc ? total / C : 0;
  if (nhwc && sizeof(T) == 2 && C % 8 == 0 &&
      ((g_bn_variant != 1 && rows_nhwc >= 32768) || msk != nullptr ||
       g_bn_variant == 2)) {
    const int64_t rows = total / C;
    const int CG8 = (int)i64min(C / 8, 64);
    const int64_t cblocks = (C / 8 + CG8 - 1) / CG8;
    int64_t S = i64min(i64max(rows / 16, 1), i64max(2048 / cblocks, 1));
    const bool add = dres != nullptr;
    const bool masked = msk != nullptr && act != 0;
#define KCASE(ACT, ADD, MSK)                                                \
    hipLaunchKernelGGL((bn_bwd_dx_nhwc_kernel2<T, ACT, ADD, MSK>),          \
                       dim3(cblocks, S), dim3(block), 0, stream, dy, y, x,  \
                       mean, invstd, k, dx, dres, msk, rows, C, CG8)
#define KSEL(ACT)                                                           \
    do { if (add) { if (masked) KCASE(ACT, true, true);                     \
                    else KCASE(ACT, true, false); }                         \
         else { if (masked) KCASE(ACT, false, true);                        \
                else KCASE(ACT, false, false); } } while (0)
    if (act == 0) KSEL(0);
    else if (act == 1) KSEL(1);
    else KSEL(2);
#undef KSEL
#undef KCASE
    HIP_CHECK_LAST();
    return;
  }
  if (vec8ok) {
    const int64_t total8 = total / 8;
    const int grid8 = elementwise_grid(total8, block);
    const bool add = dres != nullptr;
#define VCASE(ACT, ADD, NHWC)                                               \
    hipLaunchKernelGGL((bn_bwd_dx_vec_kernel<T, ACT, ADD, NHWC>),           \
                       dim3(grid8), dim3(block), 0, stream, dy, y, x,       \
                       mean, invstd, k, dx, dres, C, cdiv, total8)
#define VSEL(ACT)                                                           \
    do { if (nhwc) { if (add) VCASE(ACT, true, true);                       \
                     else VCASE(ACT, false, true); }                        \
         else { if (add) VCASE(ACT, true, false);                           \
                else VCASE(ACT, false, false); } } while (0)
    if (act == 0) VSEL(0);
    else if (act == 1) VSEL(1);
    else VSEL(2);
#undef VSEL
#undef VCASE
    HIP_CHECK_LAST();
    return;
  }
  const int grid = elementwise_grid(total, block);
#define CASE(ACT, ADD)                                                    \
  hipLaunchKernelGGL((bn_bwd_dx_kernel<T, ACT, ADD>), dim3(grid),         \
                     dim3(block), 0, stream, dy, y, x, mean, invstd, k,   \
                     dx, dres, C, cdiv, total)
  const bool add = dres != nullptr;
  if (act == 0 && !add) CASE(0, false);
  else if (act == 0 && add) CASE(0, true);
  else if (act == 1 && !add) CASE(1, false);
  else if (act == 1 && add) CASE(1, true);
  else if (act == 2 && !add) CASE(2, false);
  else CASE(2, true);
#undef CASE
  HIP_CHECK_LAST();
}

#define INSTANTIATE(T)                                                        \
  template void launch_bn_stats<T>(const T*, double*, int64_t, int64_t,       \
                                   int64_t, int64_t, int, hipStream_t);       \
  template bool launch_bn_apply<T>(const T*, const T*, T*, const float*,      \
                                   const float*, const float*, const float*,  \
                                   unsigned char*,                            \
                                   int64_t, int64_t, int64_t, int, int,       \
                                   hipStream_t);                              \
  template void launch_bn_bwd_reduce<T>(const T*, const T*, const T*,         \
                                        const float*, const float*, double*,  \
                                        const unsigned char*,                 \
                                        int64_t, int64_t, int64_t, int64_t,   \
                                        int, int, hipStream_t);               \
  template void launch_bn_bwd_dx<T>(const T*, const T*, const T*,             \
                                    const float*, const float*, const float*, \
                                    T*, T*, const unsigned char*,             \
                                    int64_t, int64_t, int64_t, int,           \
                                    int, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
#undef INSTANTIATE
