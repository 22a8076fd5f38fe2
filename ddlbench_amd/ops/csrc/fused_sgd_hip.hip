#include "hip/hip_runtime.h"
// Fused multi-tensor SGD (+momentum, +weight-decay) step for gfx950.
//
// Replaces what the reference gets from torch.optim.SGD's per-tensor
// Python loop (/root/reference/benchmark/mnist/mnist_pytorch.py uses plain
// SGD): ONE kernel updates every parameter of the model. Memory-bound —
// the design goal is one coalesced pass over (param, grad, momentum).
//
// Semantics (match torch.optim.SGD exactly):
//   d = g + wd * p
//   m = mu * m + d            (on the first step: m = d)
//   p = p - lr * m
//
// Mixed precision: param/grad may be bf16 while the momentum buffer stays
// fp32 (full-bf16 training keeps fp32 optimizer state). All math in fp32.
//
// Tensor list encoding: device arrays of pointers plus an exclusive
// prefix-sum of numels; each thread binary-searches its global element
// index into (tensor, offset). log2(#tensors) <= 8 extra SALU/VALU per
// element against 12-20 bytes of traffic — noise for a memory-bound op.

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

template <typename T>
__global__ void fused_sgd_kernel(
    uintptr_t* __restrict__ params, uintptr_t* __restrict__ grads,
    uintptr_t* __restrict__ moms, const int64_t* __restrict__ prefix,
    int n_tensors, int64_t total, float lr, float momentum, float wd,
    int first_step, int use_momentum) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += stride) {
    // binary search: largest t with prefix[t] <= e
    int lo = 0, hi = n_tensors - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= e) lo = mid; else hi = mid - 1;
    }
    const int64_t off = e - prefix[lo];
    T* p = reinterpret_cast<T*>(params[lo]) + off;
    const T* g = reinterpret_cast<const T*>(grads[lo]) + off;
    float pv = to_f32(*p);
    float d = to_f32(*g) + wd * pv;
    if (use_momentum) {
      float* m = reinterpret_cast<float*>(moms[lo]) + off;
      float mv = first_step ? d : fmaf(momentum, *m, d);
      *m = mv;
      d = mv;
    }
    *p = from_f32<T>(fmaf(-lr, d, pv));
  }
}

// Fused multi-tensor Adam / AdamW (fp32 m,v state; bias correction
// folded into host-computed scalars). decoupled_wd=1 gives AdamW.
template <typename T>
__global__ void fused_adam_kernel(
    uintptr_t* __restrict__ params, uintptr_t* __restrict__ grads,
    uintptr_t* __restrict__ ms, uintptr_t* __restrict__ vs,
    const int64_t* __restrict__ prefix, int n_tensors, int64_t total,
    float lr, float beta1, float beta2, float eps, float wd,
    float bc1, float bc2, int decoupled_wd) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += stride) {
    int lo = 0, hi = n_tensors - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= e) lo = mid; else hi = mid - 1;
    }
    const int64_t off = e - prefix[lo];
    T* p = reinterpret_cast<T*>(params[lo]) + off;
    const T* g = reinterpret_cast<const T*>(grads[lo]) + off;
    float* m = reinterpret_cast<float*>(ms[lo]) + off;
    float* v = reinterpret_cast<float*>(vs[lo]) + off;
    float pv = to_f32(*p);
    float gv = to_f32(*g);
    if (wd != 0.f && !decoupled_wd) gv = fmaf(wd, pv, gv);
    const float mv = fmaf(beta1, *m, (1.f - beta1) * gv);
    const float vv = fmaf(beta2, *v, (1.f - beta2) * gv * gv);
    *m = mv;
    *v = vv;
    const float mhat = mv / bc1;
    const float vhat = vv / bc2;
    float upd = mhat / (sqrtf(vhat) + eps);
    if (wd != 0.f && decoupled_wd) upd = fmaf(wd, pv, upd);
    *p = from_f32<T>(fmaf(-lr, upd, pv));
  }
}

template <typename T>
void launch_fused_adam(uintptr_t* params, uintptr_t* grads, uintptr_t* ms,
                       uintptr_t* vs, const int64_t* prefix, int n_tensors,
                       int64_t total, float lr, float beta1, float beta2,
                       float eps, float wd, float bc1, float bc2,
                       int decoupled_wd, hipStream_t stream) {
  const int block = 256;
  int64_t want = (total + block - 1) / block;
  int grid = (int)i64min(want, 256 * 8);
  if (grid == 0) return;
  hipLaunchKernelGGL((fused_adam_kernel<T>), dim3(grid), dim3(block), 0,
                     stream, params, grads, ms, vs, prefix, n_tensors,
                     total, lr, beta1, beta2, eps, wd, bc1, bc2,
                     decoupled_wd);
  HIP_CHECK_LAST();
}

template void launch_fused_adam<float>(uintptr_t*, uintptr_t*, uintptr_t*,
                                       uintptr_t*, const int64_t*, int,
                                       int64_t, float, float, float, float,
                                       float, float, float, int,
                                       hipStream_t);
template void launch_fused_adam<__hip_bfloat16>(
    uintptr_t*, uintptr_t*, uintptr_t*, uintptr_t*, const int64_t*, int,
    int64_t, float, float, float, float, float, float, float, int,
    hipStream_t);

template <typename T>
void launch_fused_sgd(uintptr_t* params, uintptr_t* grads, uintptr_t* moms,
                      const int64_t* prefix, int n_tensors, int64_t total,
                      float lr, float momentum, float wd, int first_step,
                      hipStream_t stream) {
  const int block = 256;
  int64_t want = (total + block - 1) / block;
  // >> 256 workgroups to fill 256 CUs / 8 XCDs; cap and grid-stride.
  int grid = (int)i64min(want, 256 * 8);
  if (grid == 0) return;
  hipLaunchKernelGGL((fused_sgd_kernel<T>), dim3(grid), dim3(block), 0,
                     stream, params, grads, moms, prefix, n_tensors, total,
                     lr, momentum, wd, first_step, momentum != 0.f);
  HIP_CHECK_LAST();
}

template void launch_fused_sgd<float>(uintptr_t*, uintptr_t*, uintptr_t*,
                                      const int64_t*, int, int64_t, float,
                                      float, float, int, hipStream_t);
template void launch_fused_sgd<__hip_bfloat16>(uintptr_t*, uintptr_t*,
                                               uintptr_t*, const int64_t*,
                                               int, int64_t, float, float,
                                               float, int, hipStream_t);
