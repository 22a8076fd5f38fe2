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
    atomicAdd(&sums[c], s);
    atomicAdd(&sums[C + c], ss);
  }
}

// ---- stats, NHWC (channels-last): thread-per-channel, coalesced -------
// rows = N*H*W; element (row, c) at row*C + c. Consecutive lanes read
// consecutive channels -> every row access is one coalesced segment.
template <typename T>
__global__ void bn_stats_nhwc_kernel(const T* __restrict__ x,
                                     double* __restrict__ sums,
                                     int64_t rows, int64_t C) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  double s = 0.0, ss = 0.0;
  for (int64_t r = begin; r < end; ++r) {
    const float v = to_f32(x[r * C + c]);
    s += v;
    ss += fma((double)v, (double)v, 0.0);
  }
  atomicAdd(&sums[c], s);
  atomicAdd(&sums[C + c], ss);
}

// ---- finalize: mean/invstd + running-stat update ----------------------
__global__ void bn_finalize_kernel(const double* __restrict__ sums,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int64_t C, double count, float eps,
                                   float momentum) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const double m = sums[c] / count;
  double var = sums[C + c] / count - m * m;
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
    atomicAdd(&sums[c], sdy);
    atomicAdd(&sums[C + c], sdyx);
  }
}

// ---- backward reduce, NHWC: thread-per-channel, coalesced -------------
template <typename T, int ACT>
__global__ void bn_bwd_reduce_nhwc_kernel(const T* __restrict__ dy,
                                          const T* __restrict__ y,
                                          const T* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ invstd,
                                          double* __restrict__ sums,
                                          int64_t rows, int64_t C) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mu = mean[c], is = invstd[c];
  const int64_t per = (rows + gridDim.y - 1) / gridDim.y;
  const int64_t begin = (int64_t)blockIdx.y * per;
  const int64_t end = i64min(begin + per, rows);
  double sdy = 0.0, sdyx = 0.0;
  for (int64_t r = begin; r < end; ++r) {
    const int64_t idx = r * C + c;
    const float g = to_f32(dy[idx]) * act_mask<ACT>(to_f32(y[idx]));
    const float xhat = (to_f32(x[idx]) - mu) * is;
    sdy += g;
    sdyx += fma((double)g, (double)xhat, 0.0);
  }
  atomicAdd(&sums[c], sdy);
  atomicAdd(&sums[C + c], sdyx);
}

__global__ void bn_bwd_finalize_kernel(const double* __restrict__ sums,
                                       const float* __restrict__ gamma,
                                       const float* __restrict__ invstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ k,  // [3][C]
                                       int64_t C, double count,
                                       int training) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const double sdy = sums[c], sdyx = sums[C + c];
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

template <typename T>
void launch_bn_stats(const T* x, double* sums, int64_t N, int64_t C,
                     int64_t HW, int nhwc, hipStream_t stream) {
  const int block = 256;
  if (nhwc) {
    const int64_t rows = N * HW;
    const int64_t cblocks = (C + block - 1) / block;
    int64_t S = i64min(i64max(rows / 1024, 1), i64max(2048 / cblocks, 1));
    hipLaunchKernelGGL((bn_stats_nhwc_kernel<T>), dim3(cblocks, S),
                       dim3(block), 0, stream, x, sums, rows, C);
  } else {
    int64_t S = i64min((N * HW + block - 1) / block, i64max(2048 / C, 1));
    S = i64max(S, 1);
    hipLaunchKernelGGL((bn_stats_kernel<T>), dim3(C, S), dim3(block), 0,
                       stream, x, sums, N, C, HW);
  }
  HIP_CHECK_LAST();
}

void launch_bn_finalize(const double* sums, float* mean, float* invstd,
                        float* rm, float* rv, int64_t C, double count,
                        float eps, float momentum, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + block - 1) / block),
                     dim3(block), 0, stream, sums, mean, invstd, rm, rv, C,
                     count, eps, momentum);
  HIP_CHECK_LAST();
}

template <typename T>
void launch_bn_apply(const T* x, const T* res, T* y, const float* mean,
                     const float* invstd, const float* gamma,
                     const float* beta, int64_t C, int64_t HW, int64_t total,
                     int act, int nhwc, hipStream_t stream) {
  const int block = 256;
  const int grid = elementwise_grid(total, block);
  const int64_t cdiv = nhwc ? 1 : HW;
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
}

template <typename T>
void launch_bn_bwd_reduce(const T* dy, const T* y, const T* x,
                          const float* mean, const float* invstd,
                          double* sums, int64_t N, int64_t C, int64_t HW,
                          int act, int nhwc, hipStream_t stream) {
  const int block = 256;
  if (nhwc) {
    const int64_t rows = N * HW;
    const int64_t cblocks = (C + block - 1) / block;
    int64_t S = i64min(i64max(rows / 1024, 1), i64max(2048 / cblocks, 1));
#define CASE(ACT)                                                           \
    hipLaunchKernelGGL((bn_bwd_reduce_nhwc_kernel<T, ACT>),                 \
                       dim3(cblocks, S), dim3(block), 0, stream, dy, y, x,  \
                       mean, invstd, sums, rows, C)
    if (act == 0) CASE(0);
    else if (act == 1) CASE(1);
    else CASE(2);
#undef CASE
  } else {
    int64_t S = i64min((N * HW + block - 1) / block, i64max(2048 / C, 1));
    S = i64max(S, 1);
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

void launch_bn_bwd_finalize(const double* sums, const float* gamma,
                            const float* invstd, float* dgamma, float* dbeta,
                            float* k, int64_t C, double count, int training,
                            hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((C + block - 1) / block),
                     dim3(block), 0, stream, sums, gamma, invstd, dgamma,
                     dbeta, k, C, count, training);
  HIP_CHECK_LAST();
}

template <typename T>
void launch_bn_bwd_dx(const T* dy, const T* y, const T* x, const float* mean,
                      const float* invstd, const float* k, T* dx, T* dres,
                      int64_t C, int64_t HW, int64_t total, int act,
                      int nhwc, hipStream_t stream) {
  const int block = 256;
  const int grid = elementwise_grid(total, block);
  const int64_t cdiv = nhwc ? 1 : HW;
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
                                   int64_t, int, hipStream_t);                \
  template void launch_bn_apply<T>(const T*, const T*, T*, const float*,      \
                                   const float*, const float*, const float*,  \
                                   int64_t, int64_t, int64_t, int, int,       \
                                   hipStream_t);                              \
  template void launch_bn_bwd_reduce<T>(const T*, const T*, const T*,         \
                                        const float*, const float*, double*,  \
                                        int64_t, int64_t, int64_t, int, int,  \
                                        hipStream_t);                         \
  template void launch_bn_bwd_dx<T>(const T*, const T*, const T*,             \
                                    const float*, const float*, const float*, \
                                    T*, T*, int64_t, int64_t, int64_t, int,   \
                                    int, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
#undef INSTANTIATE
