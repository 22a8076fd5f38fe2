#include "hip/hip_runtime.h"
// Fused softmax cross-entropy (mean reduction) for gfx950.
//
// Replaces F.cross_entropy in the benchmark loss path
// (/root/reference/benchmark/mnist/mnist_pytorch.py uses F.cross_entropy;
// the gpipe scripts use F.nll_loss on log-softmax output). Small-vocab
// rows (CNN classifiers, K <= 2048) use one wave per row; large-vocab
// rows (GNMT K = 32320) use one 256-thread block per row with 16-byte
// vector loads. Backward is one vectorized elementwise kernel — no
// materialized softmax tensor.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

typedef __attribute__((ext_vector_type(8))) short short8v;

namespace {

DEV float bf16_at(const short8v& v, int j) {
  __hip_bfloat16 h;
  unsigned short u = (unsigned short)v[j];
  __builtin_memcpy(&h, &u, 2);
  return __bfloat162float(h);
}

}  // namespace

// ---- small K: one wave per row ---------------------------------------
template <typename T>
__global__ void ce_fwd_wave_kernel(const T* __restrict__ logits,
                                   const int64_t* __restrict__ target,
                                   float* __restrict__ lse,
                                   float* __restrict__ loss_sum,
                                   int64_t B, int64_t K) {
  const int64_t b = blockIdx.x;
  if (b >= B) return;
  const T* row = logits + b * K;
  const int lane = threadIdx.x;
  float m = -INFINITY;
  for (int64_t k = lane; k < K; k += WAVE) m = fmaxf(m, to_f32(row[k]));
  m = wave_reduce_max(m);
  m = __shfl(m, 0, WAVE);
  float s = 0.f;
  for (int64_t k = lane; k < K; k += WAVE) s += __expf(to_f32(row[k]) - m);
  s = wave_reduce_sum(s);
  if (lane == 0) {
    const float l = m + __logf(s);
    lse[b] = l;
    atomicAdd(loss_sum, l - to_f32(row[target[b]]));
  }
}

// ---- large K: one 256-thread block per row, bf16x8 vector loads ------
template <typename T, bool VEC>
__global__ void ce_fwd_block_kernel(const T* __restrict__ logits,
                                    const int64_t* __restrict__ target,
                                    float* __restrict__ lse,
                                    float* __restrict__ loss_sum,
                                    int64_t B, int64_t K) {
  __shared__ float tmp[8];
  const int64_t b = blockIdx.x;
  if (b >= B) return;
  const T* row = logits + b * K;
  const int tid = threadIdx.x;
  float m = -INFINITY;
  if (VEC) {
    const short8v* rv = reinterpret_cast<const short8v*>(row);
    for (int64_t k8 = tid; k8 < K / 8; k8 += blockDim.x) {
      short8v v = rv[k8];
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, bf16_at(v, j));
    }
  } else {
    for (int64_t k = tid; k < K; k += blockDim.x)
      m = fmaxf(m, to_f32(row[k]));
  }
  auto maxop = [](float v) { return wave_reduce_max(v); };
  m = block_reduce(m, tmp, maxop, -INFINITY);
  if (tid == 0) tmp[0] = m;
  __syncthreads();
  m = tmp[0];
  __syncthreads();
  float s = 0.f;
  if (VEC) {
    const short8v* rv = reinterpret_cast<const short8v*>(row);
    for (int64_t k8 = tid; k8 < K / 8; k8 += blockDim.x) {
      short8v v = rv[k8];
#pragma unroll
      for (int j = 0; j < 8; ++j) s += __expf(bf16_at(v, j) - m);
    }
  } else {
    for (int64_t k = tid; k < K; k += blockDim.x)
      s += __expf(to_f32(row[k]) - m);
  }
  auto sumop = [](float v) { return wave_reduce_sum(v); };
  s = block_reduce(s, tmp, sumop, 0.f);
  if (tid == 0) {
    const float l = m + __logf(s);
    lse[b] = l;
    atomicAdd(loss_sum, l - to_f32(row[target[b]]));
  }
}

// dx = (softmax - onehot) * gscale   (gscale = dloss / B for mean)
template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ gscale,
                              T* __restrict__ dx, int64_t B, int64_t K) {
  const int64_t total = B * K;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t b = i / K, k = i - b * K;
    float p = __expf(to_f32(logits[i]) - lse[b]);
    if (k == target[b]) p -= 1.f;
    dx[i] = from_f32<T>(p * gscale[0]);
  }
}

template <typename T>
void launch_ce_fwd(const T* logits, const int64_t* target, float* lse,
                   float* loss_sum, int64_t B, int64_t K,
                   hipStream_t stream) {
  if (K >= 4096) {
    const bool vec = (K % 8 == 0) && sizeof(T) == 2;
    if (vec)
      hipLaunchKernelGGL((ce_fwd_block_kernel<T, true>),
                         dim3((uint32_t)B), dim3(256), 0, stream, logits,
                         target, lse, loss_sum, B, K);
    else
      hipLaunchKernelGGL((ce_fwd_block_kernel<T, false>),
                         dim3((uint32_t)B), dim3(256), 0, stream, logits,
                         target, lse, loss_sum, B, K);
  } else {
    hipLaunchKernelGGL((ce_fwd_wave_kernel<T>), dim3((uint32_t)B),
                       dim3(WAVE), 0, stream, logits, target, lse,
                       loss_sum, B, K);
  }
  HIP_CHECK_LAST();
}

template <typename T>
void launch_ce_bwd(const T* logits, const int64_t* target, const float* lse,
                   const float* gscale, T* dx, int64_t B, int64_t K,
                   hipStream_t stream) {
  const int block = 256;
  int64_t want = (B * K + block - 1) / block;
  const int grid = (int)i64min(want > 0 ? want : 1, 256 * 8);
  hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(grid), dim3(block), 0, stream,
                     logits, target, lse, gscale, dx, B, K);
  HIP_CHECK_LAST();
}

#define INSTANTIATE(T)                                                     \
  template void launch_ce_fwd<T>(const T*, const int64_t*, float*, float*, \
                                 int64_t, int64_t, hipStream_t);           \
  template void launch_ce_bwd<T>(const T*, const int64_t*, const float*,   \
                                 const float*, T*, int64_t, int64_t,       \
                                 hipStream_t);
INSTANTIATE(float)
INSTANTIATE(__hip_bfloat16)
#undef INSTANTIATE
