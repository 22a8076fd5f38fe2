// Python bindings for the ddlbench_amd CDNA4 kernels (gfx950).
// Compiled by hipcc via torch.utils.cpp_extension (PYTORCH_ROCM_ARCH=gfx950);
// HIP-native throughout — no CUDA names, no hipify.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <algorithm>
#include <stdexcept>
#include <unordered_map>
#include <vector>

// ---- launcher declarations (defined in the .hip kernel files) ---------
template <typename T>
void launch_fused_sgd(uintptr_t*, uintptr_t*, uintptr_t*, const int64_t*,
                      int, int64_t, float, float, float, int, hipStream_t);
template <typename T>
void launch_fused_adam(uintptr_t*, uintptr_t*, uintptr_t*, uintptr_t*,
                       const int64_t*, int, int64_t, float, float, float,
                       float, float, float, float, int, hipStream_t);

template <typename T>
void launch_bn_stats(const T*, double*, int64_t, int64_t, int64_t, int64_t,
                     int, hipStream_t);
void set_bn_variant(int v);
int64_t bn_reduce_gridS(int64_t N, int64_t C, int64_t HW, int nhwc,
                        int elsize);
void launch_bn_finalize(double*, float*, float*, float*, float*,
                        int64_t, int64_t, double, float, float,
                        hipStream_t);
template <typename T>
bool launch_bn_apply(const T*, const T*, T*, const float*, const float*,
                     const float*, const float*, unsigned char*, int64_t,
                     int64_t, int64_t, int, int, hipStream_t);
template <typename T>
void launch_bn_bwd_reduce(const T*, const T*, const T*, const float*,
                          const float*, double*, const unsigned char*,
                          int64_t, int64_t, int64_t, int64_t,
                          int, int, hipStream_t);
void launch_bn_bwd_finalize(double*, const float*, const float*,
                            float*, float*, float*, int64_t, int64_t,
                            double, int, hipStream_t);
template <typename T>
void launch_bn_bwd_dx(const T*, const T*, const T*, const float*,
                      const float*, const float*, T*, T*,
                      const unsigned char*, int64_t, int64_t,
                      int64_t, int, int, hipStream_t);

template <typename T>
void launch_ce_fwd(const T*, const int64_t*, float*, float*, int64_t,
                   int64_t, hipStream_t);
template <typename T>
void launch_ce_bwd(const T*, const int64_t*, const float*, const float*, T*,
                   int64_t, int64_t, hipStream_t);

void launch_conv_igemm(const void*, const void*, void*, const void*, int,
                       int, int, int, int, int, int, int, int, int, int,
                       int, hipStream_t);
void launch_conv_wgrad(const void*, const void*, float*, float*, long,
                       const void*, int,
                       int, int, int, int, int, int, int, int, int, int,
                       hipStream_t);

void launch_maxpool_fwd(const void*, void*, unsigned char*, int64_t,
                        int64_t, int, int, int, int, int, int, int,
                        hipStream_t);
void launch_maxpool_bwd(const void*, const unsigned char*, void*,
                        int64_t, int64_t, int, int, int, int, int, int,
                        int, hipStream_t);

template <typename T>
void launch_revert_varlen(const T*, T*, const int64_t*, int64_t, int64_t,
                          int64_t, hipStream_t);
void launch_varlen_mask(const int64_t*, uint8_t*, int64_t, int64_t,
                        hipStream_t);

template <typename T>
void launch_dw3x3_fwd(const T*, const float*, T*, int64_t, int64_t, int64_t,
                      int64_t, int64_t, int64_t, int, int, hipStream_t);
template <typename T>
void launch_dw3x3_bwd_dx(const T*, const float*, T*, int64_t, int64_t,
                         int64_t, int64_t, int64_t, int64_t, int, int,
                         hipStream_t);
template <typename T>
void launch_dw3x3_bwd_dw(const T*, const T*, double*, int64_t, int64_t,
                         int64_t, int64_t, int64_t, int64_t, int, int,
                         hipStream_t);

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_gpu_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the HIP device");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == at::kBFloat16;
}

template <typename T>
T* dptr(const torch::Tensor& t) {
  return reinterpret_cast<T*>(t.data_ptr());
}

// Cached per-device 16-byte zero page for the conv kernels' padding
// redirect (a fresh torch::zeros per call cost ~160 fill launches per
// training step).
torch::Tensor zero_page(const torch::Tensor& like) {
  static std::unordered_map<int, torch::Tensor> zp;
  const int dev = (int)like.get_device();
  auto it = zp.find(dev);
  if (it == zp.end()) {
    zp[dev] = torch::zeros({16}, like.options().dtype(at::kBFloat16));
    it = zp.find(dev);
  }
  return it->second;
}

// Per-device BN reduction workspace: [S][2][C] double partial slabs.
// Reduce-style kernels write plain per-block stores and the finalize
// kernels sum over S — no cross-block atomics (a C=64 layer at S~1500
// slices serialized ~1500 f64 atomics per channel address), and no
// zeroing (every slab slot is written every call).
torch::Tensor bn_sums_workspace(const torch::Tensor& like, int64_t C,
                                int64_t S) {
  static std::unordered_map<int, torch::Tensor> ws;
  const int dev = (int)like.get_device();
  const int64_t need = 2 * C * S;
  auto it = ws.find(dev);
  if (it == ws.end() || it->second.numel() < need) {
    const int64_t cap = std::max<int64_t>(need, 1 << 18);
    ws[dev] = torch::empty({cap}, like.options().dtype(at::kDouble));
    it = ws.find(dev);
  }
  return it->second;
}

}  // namespace

// ---- fused SGD --------------------------------------------------------
void fused_sgd(torch::Tensor ptr_params, torch::Tensor ptr_grads,
               torch::Tensor ptr_moms, torch::Tensor prefix, int64_t total,
               double lr, double momentum, double weight_decay,
               bool first_step, bool bf16) {
  check_gpu_contig(ptr_params, "ptr_params");
  auto s = cur_stream();
  if (bf16)
    launch_fused_sgd<__hip_bfloat16>(
        dptr<uintptr_t>(ptr_params), dptr<uintptr_t>(ptr_grads),
        dptr<uintptr_t>(ptr_moms), dptr<int64_t>(prefix),
        (int)ptr_params.numel(), total, (float)lr, (float)momentum,
        (float)weight_decay, first_step, s);
  else
    launch_fused_sgd<float>(
        dptr<uintptr_t>(ptr_params), dptr<uintptr_t>(ptr_grads),
        dptr<uintptr_t>(ptr_moms), dptr<int64_t>(prefix),
        (int)ptr_params.numel(), total, (float)lr, (float)momentum,
        (float)weight_decay, first_step, s);
}

void fused_adam(torch::Tensor ptr_params, torch::Tensor ptr_grads,
                torch::Tensor ptr_ms, torch::Tensor ptr_vs,
                torch::Tensor prefix, int64_t total, double lr,
                double beta1, double beta2, double eps,
                double weight_decay, double bc1, double bc2,
                bool decoupled_wd, bool bf16) {
  check_gpu_contig(ptr_params, "ptr_params");
  auto s = cur_stream();
  if (bf16)
    launch_fused_adam<__hip_bfloat16>(
        dptr<uintptr_t>(ptr_params), dptr<uintptr_t>(ptr_grads),
        dptr<uintptr_t>(ptr_ms), dptr<uintptr_t>(ptr_vs),
        dptr<int64_t>(prefix), (int)ptr_params.numel(), total, (float)lr,
        (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
        (float)bc1, (float)bc2, decoupled_wd, s);
  else
    launch_fused_adam<float>(
        dptr<uintptr_t>(ptr_params), dptr<uintptr_t>(ptr_grads),
        dptr<uintptr_t>(ptr_ms), dptr<uintptr_t>(ptr_vs),
        dptr<int64_t>(prefix), (int)ptr_params.numel(), total, (float)lr,
        (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
        (float)bc1, (float)bc2, decoupled_wd, s);
}

// ---- fused BN + act (+residual) ---------------------------------------
std::vector<torch::Tensor> bn_act_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> res, torch::Tensor gamma,
    torch::Tensor beta, c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var, bool training, double momentum,
    double eps, int64_t act, bool nhwc) {
  TORCH_CHECK(x.is_cuda(), "x must be on the HIP device");
  TORCH_CHECK(x.dim() == 4, "x must be 4-D");
  TORCH_CHECK(nhwc ? x.is_contiguous(at::MemoryFormat::ChannelsLast)
                   : x.is_contiguous(), "x layout mismatch");
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto s = cur_stream();
  auto fopt = x.options().dtype(at::kFloat);
  torch::Tensor mean = torch::empty({C}, fopt);
  torch::Tensor invstd = torch::empty({C}, fopt);
  if (training) {
    const int64_t S =
        bn_reduce_gridS(N, C, HW, nhwc, (int)x.element_size());
    torch::Tensor sums = bn_sums_workspace(x, C, S);
    if (is_bf16(x))
      launch_bn_stats<__hip_bfloat16>(dptr<__hip_bfloat16>(x),
                                      dptr<double>(sums), N, C, HW, S,
                                      nhwc, s);
    else
      launch_bn_stats<float>(dptr<float>(x), dptr<double>(sums), N, C, HW,
                             S, nhwc, s);
    launch_bn_finalize(
        dptr<double>(sums), dptr<float>(mean), dptr<float>(invstd),
        running_mean ? dptr<float>(*running_mean) : nullptr,
        running_var ? dptr<float>(*running_var) : nullptr, C, S,
        (double)(N * HW), (float)eps, (float)momentum, s);
  } else {
    TORCH_CHECK(running_mean && running_var,
                "eval mode needs running stats");
    mean.copy_(*running_mean);
    invstd.copy_((running_var->to(at::kFloat) + eps).rsqrt());
  }
  torch::Tensor y = torch::empty_like(x);
  const int64_t total = x.numel();
  // 1-bit activation mask for the backward (bf16 NHWC training path):
  // bit (i & 7) of byte i/8 gates element i's activation gradient.
  torch::Tensor msk;
  const bool want_mask =
      training && act != 0 && nhwc && is_bf16(x) && C % 8 == 0;
  if (want_mask)
    msk = torch::empty({total / 8}, x.options().dtype(at::kByte));
  bool mask_written = false;
  if (is_bf16(x))
    mask_written = launch_bn_apply<__hip_bfloat16>(
        dptr<__hip_bfloat16>(x),
        res ? dptr<__hip_bfloat16>(*res) : nullptr, dptr<__hip_bfloat16>(y),
        dptr<float>(mean), dptr<float>(invstd), dptr<float>(gamma),
        dptr<float>(beta),
        want_mask ? msk.data_ptr<unsigned char>() : nullptr,
        C, HW, total, (int)act, nhwc, s);
  else
    launch_bn_apply<float>(dptr<float>(x),
                           res ? dptr<float>(*res) : nullptr, dptr<float>(y),
                           dptr<float>(mean), dptr<float>(invstd),
                           dptr<float>(gamma), dptr<float>(beta), nullptr,
                           C, HW, total, (int)act, nhwc, s);
  if (mask_written) return {y, mean, invstd, msk};
  return {y, mean, invstd};
}

std::vector<torch::Tensor> bn_act_bwd(torch::Tensor dy,
                                      c10::optional<torch::Tensor> y_opt,
                                      torch::Tensor x, torch::Tensor mean,
                                      torch::Tensor invstd,
                                      torch::Tensor gamma, int64_t act,
                                      bool training, bool need_dres,
                                      bool nhwc,
                                      c10::optional<torch::Tensor> msk) {
  TORCH_CHECK(dy.is_cuda(), "dy must be on the HIP device");
  TORCH_CHECK(y_opt || msk,
              "bn_act_bwd needs y or the 1-bit activation mask");
  const int64_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto s = cur_stream();
  auto fopt = x.options().dtype(at::kFloat);
  const int64_t S = bn_reduce_gridS(N, C, HW, nhwc, (int)x.element_size());
  torch::Tensor sums = bn_sums_workspace(x, C, S);
  torch::Tensor dgamma = torch::empty({C}, fopt);
  torch::Tensor dbeta = torch::empty({C}, fopt);
  torch::Tensor k = torch::empty({3, C}, fopt);
  torch::Tensor dx = torch::empty_like(x);
  torch::Tensor dres =
      need_dres ? torch::empty_like(x) : torch::Tensor();
  const int64_t total = x.numel();
  const unsigned char* mp =
      msk ? msk->data_ptr<unsigned char>() : nullptr;
  if (is_bf16(x)) {
    TORCH_CHECK(mp || y_opt, "masked path needs the mask tensor");
    const __hip_bfloat16* yp =
        y_opt ? dptr<__hip_bfloat16>(*y_opt) : nullptr;
    launch_bn_bwd_reduce<__hip_bfloat16>(
        dptr<__hip_bfloat16>(dy), yp,
        dptr<__hip_bfloat16>(x), dptr<float>(mean), dptr<float>(invstd),
        dptr<double>(sums), mp, N, C, HW, S, (int)act, nhwc, s);
    launch_bn_bwd_finalize(dptr<double>(sums), dptr<float>(gamma),
                           dptr<float>(invstd), dptr<float>(dgamma),
                           dptr<float>(dbeta), dptr<float>(k), C, S,
                           (double)(N * HW), training, s);
    launch_bn_bwd_dx<__hip_bfloat16>(
        dptr<__hip_bfloat16>(dy), yp,
        dptr<__hip_bfloat16>(x), dptr<float>(mean), dptr<float>(invstd),
        dptr<float>(k), dptr<__hip_bfloat16>(dx),
        need_dres ? dptr<__hip_bfloat16>(dres) : nullptr, mp, C, HW, total,
        (int)act, nhwc, s);
  } else {
    TORCH_CHECK(y_opt, "fp32 backward needs y");
    launch_bn_bwd_reduce<float>(dptr<float>(dy), dptr<float>(*y_opt),
                                dptr<float>(x), dptr<float>(mean),
                                dptr<float>(invstd), dptr<double>(sums),
                                nullptr, N, C, HW, S, (int)act, nhwc, s);
    launch_bn_bwd_finalize(dptr<double>(sums), dptr<float>(gamma),
                           dptr<float>(invstd), dptr<float>(dgamma),
                           dptr<float>(dbeta), dptr<float>(k), C, S,
                           (double)(N * HW), training, s);
    launch_bn_bwd_dx<float>(dptr<float>(dy), dptr<float>(*y_opt),
                            dptr<float>(x),
                            dptr<float>(mean), dptr<float>(invstd),
                            dptr<float>(k), dptr<float>(dx),
                            need_dres ? dptr<float>(dres) : nullptr,
                            nullptr, C, HW,
                            total, (int)act, nhwc, s);
  }
  if (need_dres) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

// ---- cross entropy ----------------------------------------------------
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits,
                                  torch::Tensor target) {
  check_gpu_contig(logits, "logits");
  const int64_t B = logits.size(0), K = logits.size(1);
  auto s = cur_stream();
  auto fopt = logits.options().dtype(at::kFloat);
  torch::Tensor lse = torch::empty({B}, fopt);
  torch::Tensor loss_sum = torch::zeros({1}, fopt);
  if (is_bf16(logits))
    launch_ce_fwd<__hip_bfloat16>(dptr<__hip_bfloat16>(logits),
                                  dptr<int64_t>(target), dptr<float>(lse),
                                  dptr<float>(loss_sum), B, K, s);
  else
    launch_ce_fwd<float>(dptr<float>(logits), dptr<int64_t>(target),
                         dptr<float>(lse), dptr<float>(loss_sum), B, K, s);
  return {loss_sum, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor target,
                     torch::Tensor lse, torch::Tensor gscale) {
  const int64_t B = logits.size(0), K = logits.size(1);
  auto s = cur_stream();
  torch::Tensor dx = torch::empty_like(logits);
  if (is_bf16(logits))
    launch_ce_bwd<__hip_bfloat16>(dptr<__hip_bfloat16>(logits),
                                  dptr<int64_t>(target), dptr<float>(lse),
                                  dptr<float>(gscale),
                                  dptr<__hip_bfloat16>(dx), B, K, s);
  else
    launch_ce_bwd<float>(dptr<float>(logits), dptr<int64_t>(target),
                         dptr<float>(lse), dptr<float>(gscale),
                         dptr<float>(dx), B, K, s);
  return dx;
}

// ---- MFMA implicit-GEMM conv (NHWC bf16) ------------------------------
// x: (N,C,H,W) logical, channels_last memory; w: (K,C,R,S) logical,
// channels_last memory (= (K,R,S,C) image). Returns channels_last y.
torch::Tensor conv_igemm_fwd(torch::Tensor x, torch::Tensor w,
                             int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_igemm: bf16 HIP tensors only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels_last");
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "w must be channels_last");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(C % 8 == 0, "conv_igemm fwd needs C % 8 == 0");
  const int OH = (H + 2 * (int)pad - R) / (int)stride + 1;
  const int OW = (W + 2 * (int)pad - S) / (int)stride + 1;
  auto y = torch::empty({N, K, OH, OW},
                        x.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto zero = zero_page(x);
  launch_conv_igemm(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                    zero.data_ptr(), N, H, W, C, K, OH, OW, R, S,
                    (int)stride, (int)pad, /*dgrad=*/0,
                    cur_stream());
  return y;
}

// dy: (N,K,OH,OW) channels_last; w_perm: (C,R,S,K) plain-contiguous
// memory (host permutes once per backward). Returns channels_last dx.
torch::Tensor conv_igemm_dgrad(torch::Tensor dy, torch::Tensor w_perm,
                               int64_t N, int64_t C, int64_t H, int64_t W,
                               int64_t stride, int64_t pad) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "conv_igemm: bf16 HIP tensors only");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "dy must be channels_last");
  TORCH_CHECK(w_perm.is_contiguous(), "w_perm must be contiguous");
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  const int R = w_perm.size(1), S = w_perm.size(2);
  TORCH_CHECK(K % 8 == 0, "conv_igemm dgrad needs K % 8 == 0");
  // stride-2 parity classes with R==1 or S==1 have pixel classes no tap
  // reaches — those dx entries must be zeros, so zero-init in that case
  // (zero_() after empty: torch::zeros does not honour the
  // channels_last memory_format request)
  auto dx = torch::empty({N, C, H, W},
                         dy.options().memory_format(
                             at::MemoryFormat::ChannelsLast));
  if (stride == 2 && (R == 1 || S == 1)) dx.zero_();
  auto zero = zero_page(dy);
  launch_conv_igemm(dy.data_ptr(), w_perm.data_ptr(), dx.data_ptr(),
                    zero.data_ptr(), (int)N, (int)H, (int)W, (int)C, K,
                    OH, OW, R, S, (int)stride, (int)pad, /*dgrad=*/1,
                    cur_stream());
  return dx;
}

// x: (N,C,H,W) channels_last; dy: (N,K,OH,OW) channels_last.
// Returns fp32 dw in (K, R*S*C) memory = logical (K,C,R,S) channels_last
// after the python-side permute.
torch::Tensor conv_igemm_wgrad(torch::Tensor x, torch::Tensor dy,
                               int64_t R, int64_t S, int64_t stride,
                               int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "conv_wgrad: bf16 HIP tensors only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels_last");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "dy must be channels_last");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = dy.size(1), OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(C % 8 == 0 && K % 8 == 0,
              "conv_wgrad needs C %% 8 == 0 and K %% 8 == 0");
  auto dw = torch::empty({K, R * S * C},
                         x.options().dtype(at::kFloat));
  auto zero = zero_page(x);
  // split-P partial-slab workspace (cached per device, 32 MB fp32)
  static std::unordered_map<int, torch::Tensor> wgws;
  const int devi = (int)x.get_device();
  auto wit = wgws.find(devi);
  if (wit == wgws.end()) {
    wgws[devi] = torch::empty({8 << 20},
                              x.options().dtype(at::kFloat));
    wit = wgws.find(devi);
  }
  launch_conv_wgrad(x.data_ptr(), dy.data_ptr(),
                    dw.data_ptr<float>(), wit->second.data_ptr<float>(),
                    (long)wit->second.numel(), zero.data_ptr(),
                    N, H, W, C, K,
                    OH, OW, (int)R, (int)S, (int)stride, (int)pad,
                    cur_stream());
  return dw;
}

// ---- NHWC max-pool ----------------------------------------------------
std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k,
                                       int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16,
              "maxpool: bf16 HIP tensors only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "maxpool: x must be channels_last");
  const int64_t N = x.size(0), C = x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  TORCH_CHECK(C % 8 == 0, "maxpool needs C %% 8 == 0");
  const int OH = (H + 2 * (int)pad - (int)k) / (int)stride + 1;
  const int OW = (W + 2 * (int)pad - (int)k) / (int)stride + 1;
  auto y = torch::empty({N, C, OH, OW},
                        x.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto idx = torch::empty({N * OH * OW * C},
                          x.options().dtype(at::kByte));
  launch_maxpool_fwd(x.data_ptr(), y.data_ptr(),
                     idx.data_ptr<unsigned char>(), N, C, H, W, OH, OW,
                     (int)k, (int)stride, (int)pad, cur_stream());
  return {y, idx};
}

torch::Tensor maxpool_bwd(torch::Tensor dy, torch::Tensor idx,
                          int64_t H, int64_t W, int64_t k,
                          int64_t stride, int64_t pad) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "maxpool: bf16 HIP tensors only");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "maxpool: dy must be channels_last");
  const int64_t N = dy.size(0), C = dy.size(1);
  const int OH = (int)dy.size(2), OW = (int)dy.size(3);
  auto dx = torch::empty({N, C, H, W},
                         dy.options().memory_format(
                             at::MemoryFormat::ChannelsLast));
  launch_maxpool_bwd(dy.data_ptr(), idx.data_ptr<unsigned char>(),
                     dx.data_ptr(), N, C, (int)H, (int)W, OH, OW,
                     (int)k, (int)stride, (int)pad, cur_stream());
  return dx;
}

// ---- depthwise 3x3 ----------------------------------------------------
torch::Tensor dw3x3_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                        bool nhwc) {
  TORCH_CHECK(x.is_cuda(), "x must be on the HIP device");
  check_gpu_contig(w, "w");
  const int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int64_t OH = (H + 2 - 3) / stride + 1, OW = (W + 2 - 3) / stride + 1;
  auto s = cur_stream();
  torch::Tensor y = nhwc
      ? torch::empty({N, C, OH, OW}, x.options().memory_format(
            at::MemoryFormat::ChannelsLast))
      : torch::empty({N, C, OH, OW}, x.options());
  if (is_bf16(x))
    launch_dw3x3_fwd<__hip_bfloat16>(dptr<__hip_bfloat16>(x), dptr<float>(w),
                                     dptr<__hip_bfloat16>(y), N, C, H, W, OH,
                                     OW, (int)stride, nhwc, s);
  else
    launch_dw3x3_fwd<float>(dptr<float>(x), dptr<float>(w), dptr<float>(y),
                            N, C, H, W, OH, OW, (int)stride, nhwc, s);
  return y;
}

std::vector<torch::Tensor> dw3x3_bwd(torch::Tensor x, torch::Tensor w,
                                     torch::Tensor dy, int64_t stride,
                                     bool need_dx, bool need_dw,
                                     bool nhwc) {
  const int64_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int64_t OH = dy.size(2), OW = dy.size(3);
  auto s = cur_stream();
  torch::Tensor dx, dwt;
  if (need_dx) {
    dx = torch::empty_like(x);
    if (is_bf16(x))
      launch_dw3x3_bwd_dx<__hip_bfloat16>(
          dptr<__hip_bfloat16>(dy), dptr<float>(w), dptr<__hip_bfloat16>(dx),
          N, C, H, W, OH, OW, (int)stride, nhwc, s);
    else
      launch_dw3x3_bwd_dx<float>(dptr<float>(dy), dptr<float>(w),
                                 dptr<float>(dx), N, C, H, W, OH, OW,
                                 (int)stride, nhwc, s);
  }
  if (need_dw) {
    torch::Tensor acc = torch::zeros({C * 9}, x.options().dtype(at::kDouble));
    if (is_bf16(x))
      launch_dw3x3_bwd_dw<__hip_bfloat16>(dptr<__hip_bfloat16>(x),
                                          dptr<__hip_bfloat16>(dy),
                                          dptr<double>(acc), N, C, H, W, OH,
                                          OW, (int)stride, nhwc, s);
    else
      launch_dw3x3_bwd_dw<float>(dptr<float>(x), dptr<float>(dy),
                                 dptr<double>(acc), N, C, H, W, OH, OW,
                                 (int)stride, nhwc, s);
    dwt = acc.to(at::kFloat).view({C, 1, 3, 3});
  }
  return {dx, dwt};
}

// ---- sequence utils (GNMT) --------------------------------------------
torch::Tensor revert_varlen(torch::Tensor x, torch::Tensor lengths) {
  check_gpu_contig(x, "x");
  TORCH_CHECK(x.dim() == 3, "x must be (T, B, F)");
  TORCH_CHECK(lengths.scalar_type() == at::kLong, "lengths must be int64");
  const int64_t T = x.size(0), B = x.size(1), F = x.size(2);
  auto out = torch::empty_like(x);
  auto s = cur_stream();
  if (is_bf16(x))
    launch_revert_varlen<__hip_bfloat16>(
        dptr<__hip_bfloat16>(x), dptr<__hip_bfloat16>(out),
        dptr<int64_t>(lengths), T, B, F, s);
  else
    launch_revert_varlen<float>(dptr<float>(x), dptr<float>(out),
                                dptr<int64_t>(lengths), T, B, F, s);
  return out;
}

torch::Tensor varlen_mask(torch::Tensor lengths, int64_t T) {
  TORCH_CHECK(lengths.is_cuda(), "lengths must be on the HIP device");
  const int64_t B = lengths.size(0);
  auto mask = torch::empty({T, B}, lengths.options().dtype(at::kByte));
  launch_varlen_mask(dptr<int64_t>(lengths), dptr<uint8_t>(mask), T, B,
                     cur_stream());
  return mask;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "ddlbench_amd gfx950 HIP kernels";
  m.def("fused_sgd", &fused_sgd, "fused multi-tensor SGD step");
  m.def("fused_adam", &fused_adam, "fused multi-tensor Adam/AdamW step");
  m.def("bn_act_fwd", &bn_act_fwd, "fused BN+act(+res) forward");
  m.def("set_bn_variant", &set_bn_variant,
        "BN kernel variant for A/B probes (0 auto, 1 scalar-group, 2 vec)");
  m.def("bn_act_bwd", &bn_act_bwd, "fused BN+act(+res) backward");
  m.def("maxpool_fwd", &maxpool_fwd, "NHWC bf16 max-pool forward");
  m.def("maxpool_bwd", &maxpool_bwd, "NHWC bf16 max-pool backward");
  m.def("conv_igemm_fwd", &conv_igemm_fwd,
        "NHWC bf16 MFMA implicit-GEMM conv forward");
  m.def("conv_igemm_dgrad", &conv_igemm_dgrad,
        "NHWC bf16 MFMA implicit-GEMM conv data-grad");
  m.def("conv_igemm_wgrad", &conv_igemm_wgrad,
        "NHWC bf16 MFMA implicit-GEMM conv weight-grad (fp32 out)");
  m.def("ce_fwd", &ce_fwd, "cross-entropy forward");
  m.def("ce_bwd", &ce_bwd, "cross-entropy backward");
  m.def("revert_varlen", &revert_varlen,
        "reverse each batch element's valid time prefix (T,B,F)");
  m.def("varlen_mask", &varlen_mask, "valid-timestep mask (T,B) uint8");
  m.def("dw3x3_fwd", &dw3x3_fwd, "depthwise 3x3 forward");
  m.def("dw3x3_bwd", &dw3x3_bwd, "depthwise 3x3 backward");
}
