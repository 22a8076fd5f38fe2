// Common helpers for ddlbench_amd CDNA4 (gfx950) kernels.
// Wave width on CDNA4 is 64 (MI355X_MICROARCH.md) — hard-coded throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

__host__ __device__ inline long long i64min(long long a, long long b) {
  return a < b ? a : b;
}
__host__ __device__ inline long long i64max(long long a, long long b) {
  return a > b ? a : b;
}

// ---- dtype conversion -------------------------------------------------
DEV float to_f32(float v) { return v; }
DEV float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T> DEV T from_f32(float v);
template <> DEV float from_f32<float>(float v) { return v; }
template <> DEV __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// ---- wave + block reductions -----------------------------------------
DEV float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0 of the wave
}

DEV double wave_reduce_sum(double v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;
}

DEV float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// Block reduction via LDS; `tmp` must hold >= blockDim.x/WAVE entries.
// Result valid in thread 0; broadcast left to caller.
template <typename R, typename F>
DEV R block_reduce(R v, R* tmp, F wave_op, R ident) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_op(v);
  if (lane == 0) tmp[wid] = v;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  v = (threadIdx.x < nw) ? tmp[threadIdx.x] : ident;
  if (wid == 0) v = wave_op(v);
  return v;
}

#define HIP_CHECK_LAST()                                                  \
  do {                                                                    \
    hipError_t err_ = hipGetLastError();                                  \
    if (err_ != hipSuccess)                                               \
      throw std::runtime_error(std::string("HIP kernel launch failed: ") + \
                               hipGetErrorString(err_));                  \
  } while (0)
