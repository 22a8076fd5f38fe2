#include "hip/hip_runtime.h"
// Sequence utilities for the GNMT workload (gfx950).
//
// Parity with the reference's only custom native kernel
// (/root/reference/pipedream-fork/runtime/translation/seq2seq/csrc/
// pack_utils_kernel.cu — revert_varlen_tensor): reverse each batch
// element's valid prefix along time (emulated bidirectional LSTM),
// zero-fill the padding. Re-designed for CDNA4: one wave-coalesced
// grid-stride loop over (t, b, f) instead of a block per (t, b) row —
// the op is pure bandwidth.
//
// Also: valid-timestep mask build on device (the reference computes it
// on the CPU per batch — pack_utils.cpp:13-31; keeping it on-GPU avoids
// a host round trip per minibatch).

#include "common.h"
#include <stdint.h>
#include <stdexcept>
#include <string>

// out[t][b][f] = in[len[b]-1-t][b][f] if t < len[b] else 0
template <typename T>
__global__ void revert_varlen_kernel(const T* __restrict__ in,
                                     T* __restrict__ out,
                                     const int64_t* __restrict__ lengths,
                                     int64_t Tm, int64_t B, int64_t F) {
  const int64_t total = Tm * B * F;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int64_t f = i % F;
    const int64_t b = (i / F) % B;
    const int64_t t = i / (F * B);
    const int64_t len = lengths[b];
    out[i] = (t < len) ? in[((len - 1 - t) * B + b) * F + f]
                       : from_f32<T>(0.f);
  }
}

// mask[t][b] = t < len[b]  (uint8)
__global__ void varlen_mask_kernel(const int64_t* __restrict__ lengths,
                                   uint8_t* __restrict__ mask, int64_t Tm,
                                   int64_t B) {
  const int64_t total = Tm * B;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += stride) {
    const int64_t b = i % B;
    const int64_t t = i / B;
    mask[i] = t < lengths[b];
  }
}

static inline int seq_grid(int64_t total, int block) {
  int64_t want = (total + block - 1) / block;
  return (int)i64min(want > 0 ? want : 1, 256 * 8);
}

template <typename T>
void launch_revert_varlen(const T* in, T* out, const int64_t* lengths,
                          int64_t Tm, int64_t B, int64_t F,
                          hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL((revert_varlen_kernel<T>),
                     dim3(seq_grid(Tm * B * F, block)), dim3(block), 0,
                     stream, in, out, lengths, Tm, B, F);
  HIP_CHECK_LAST();
}

void launch_varlen_mask(const int64_t* lengths, uint8_t* mask, int64_t Tm,
                        int64_t B, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(varlen_mask_kernel, dim3(seq_grid(Tm * B, block)),
                     dim3(block), 0, stream, lengths, mask, Tm, B);
  HIP_CHECK_LAST();
}

template void launch_revert_varlen<float>(const float*, float*,
                                          const int64_t*, int64_t, int64_t,
                                          int64_t, hipStream_t);
template void launch_revert_varlen<__hip_bfloat16>(
    const __hip_bfloat16*, __hip_bfloat16*, const int64_t*, int64_t,
    int64_t, int64_t, hipStream_t);
