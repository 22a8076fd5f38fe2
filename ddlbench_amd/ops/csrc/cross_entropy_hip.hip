#include "hip/hip_runtime.h"
// Fused softmax cross-entropy (mean reduction) for gfx950.
//
// Replaces F.cross_entropy in the benchmark loss path
// (/root/reference/benchmark/mnist/mnist_pytorch.py uses F.cross_entropy;
// the gpipe scripts use F.nll_loss on log-softmax output). One wave per
// row computes max + log-sum-exp in a single pass over the logits; the
// backward is one elementwise kernel — no materialized softmax tensor.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

// one wave per row: lse[b], per-row loss atomically summed into loss[0]
template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
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
  hipLaunchKernelGGL((ce_fwd_kernel<T>), dim3((uint32_t)B), dim3(WAVE), 0,
                     stream, logits, target, lse, loss_sum, B, K);
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
