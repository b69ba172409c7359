// Fused focal-L2 multi-task loss (forward + backward).
//
// The reference materialises mask.expand().clone(), st, factor and out as full
// (nstack, N, C, H, W) temporaries (reference models/loss_model.py:134-161).
// Here forward is ONE pass producing per-stack partial sums, and backward is
// ONE elementwise pass that re-derives the focal factor analytically — no
// temporaries, ~6x less HBM traffic at the 128^2 scale.
//
//   st     = gt >= 0.01 ? s - alpha : 1 - s - beta
//   factor = |1 - st|^gamma                      (gamma = 1 or 2)
//   loss   = sum over elements of (s - gt)^2 * factor * w_c * mask
//   w_c    = keypoint_task_weight  for heat channels,
//            multi_task_weight     for the person-mask channel (C-2), else 1
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

__device__ __forceinline__ float chan_weight(int c, int C, int heat_start,
                                             int bkg_start, float mtw, float ktw) {
  if (c == C - 2) return mtw;
  if (c >= heat_start && c < bkg_start) return ktw;
  return 1.0f;
}

template <typename T, int GAMMA>
__global__ void focal_l2_fwd_kernel(
    const T* __restrict__ pred,   // [S, N, C, H, W]
    const T* __restrict__ gt,     // [N, C, H, W]
    const T* __restrict__ mask,   // [N, 1, H, W]
    float* __restrict__ stack_sums,  // [S]
    long long per_stack, int N, int C, long long HW,
    int heat_start, int bkg_start, float mtw, float ktw,
    float alpha, float beta) {
  __shared__ float lds[16];
  int s_idx = blockIdx.y;
  const T* p = pred + (long long)s_idx * per_stack;
  float acc = 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < per_stack; i += (long long)gridDim.x * blockDim.x) {
    long long hw = i % HW;
    long long nc = i / HW;
    int c = (int)(nc % C);
    int n = (int)(nc / C);
    float sv = ldf(p + i);
    float gv = ldf(gt + i);
    float mv = ldf(mask + (long long)n * HW + hw);
    float st = gv >= 0.01f ? sv - alpha : 1.f - sv - beta;
    float u = 1.f - st;
    float factor = GAMMA == 1 ? fabsf(u) : u * u;
    float d = sv - gv;
    acc += d * d * factor * mv * chan_weight(c, C, heat_start, bkg_start, mtw, ktw);
  }
  float total = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(&stack_sums[s_idx], total);
}

template <typename T, int GAMMA>
__global__ void focal_l2_bwd_kernel(
    const T* __restrict__ pred, const T* __restrict__ gt,
    const T* __restrict__ mask, T* __restrict__ dpred,
    const float* __restrict__ stack_gscale,  // [S] upstream * nw[j]/sum(nw)
    long long per_stack, int N, int C, long long HW,
    int heat_start, int bkg_start, float mtw, float ktw,
    float alpha, float beta) {
  int s_idx = blockIdx.y;
  const T* p = pred + (long long)s_idx * per_stack;
  T* dp = dpred + (long long)s_idx * per_stack;
  float gscale = stack_gscale[s_idx];
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < per_stack; i += (long long)gridDim.x * blockDim.x) {
    long long hw = i % HW;
    long long nc = i / HW;
    int c = (int)(nc % C);
    int n = (int)(nc / C);
    float sv = ldf(p + i);
    float gv = ldf(gt + i);
    float mv = ldf(mask + (long long)n * HW + hw);
    bool pos = gv >= 0.01f;
    float st = pos ? sv - alpha : 1.f - sv - beta;
    float u = 1.f - st;
    float d = sv - gv;
    float factor, dfactor_ds;
    if (GAMMA == 1) {
      factor = fabsf(u);
      float sgn = u > 0.f ? 1.f : (u < 0.f ? -1.f : 0.f);
      // du/ds = -dst/ds ; dst/ds = pos ? 1 : -1
      dfactor_ds = sgn * (pos ? -1.f : 1.f);
    } else {
      factor = u * u;
      dfactor_ds = 2.f * u * (pos ? -1.f : 1.f);
    }
    float w = mv * chan_weight(c, C, heat_start, bkg_start, mtw, ktw);
    float g = w * (2.f * d * factor + d * d * dfactor_ds) * gscale;
    stf(dp + i, g);
  }
}

}  // namespace ibp

using torch::Tensor;
static inline hipStream_t cur_stream3() {
  return at::hip::getCurrentHIPStream().stream();
}

// returns per-stack sums [S] (fp32); host applies nstack weights
Tensor focal_l2_fwd(const Tensor& pred, const Tensor& gt, const Tensor& mask,
                    int64_t heat_start, int64_t bkg_start, int64_t gamma,
                    double mtw, double ktw, double alpha, double beta) {
  TORCH_CHECK(pred.dim() == 5 && pred.is_contiguous());
  int S = (int)pred.size(0), N = (int)pred.size(1), C = (int)pred.size(2);
  long long HW = (long long)pred.size(3) * pred.size(4);
  long long per_stack = (long long)N * C * HW;
  Tensor sums = torch::zeros({S}, pred.options().dtype(torch::kFloat32));
  dim3 block(256), grid(ibp::grid_1d(per_stack, 256, 2048), S);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      pred.scalar_type(), "focal_l2_fwd", [&] {
    using T = scalar_t;
    auto launch = [&](auto gamma_tag) {
      constexpr int G = decltype(gamma_tag)::value;
      hipLaunchKernelGGL((ibp::focal_l2_fwd_kernel<T, G>), grid, block, 0,
                         cur_stream3(),
                         reinterpret_cast<const T*>(pred.data_ptr()),
                         reinterpret_cast<const T*>(gt.data_ptr()),
                         reinterpret_cast<const T*>(mask.data_ptr()),
                         sums.data_ptr<float>(), per_stack, N, C, HW,
                         (int)heat_start, (int)bkg_start, (float)mtw, (float)ktw,
                         (float)alpha, (float)beta);
    };
    if (gamma == 1) launch(std::integral_constant<int, 1>{});
    else launch(std::integral_constant<int, 2>{});
  });
  return sums;
}

Tensor focal_l2_bwd(const Tensor& pred, const Tensor& gt, const Tensor& mask,
                    const Tensor& stack_gscale, int64_t heat_start,
                    int64_t bkg_start, int64_t gamma, double mtw, double ktw,
                    double alpha, double beta) {
  int S = (int)pred.size(0), N = (int)pred.size(1), C = (int)pred.size(2);
  long long HW = (long long)pred.size(3) * pred.size(4);
  long long per_stack = (long long)N * C * HW;
  Tensor dpred = torch::empty_like(pred);
  dim3 block(256), grid(ibp::grid_1d(per_stack, 256, 2048), S);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      pred.scalar_type(), "focal_l2_bwd", [&] {
    using T = scalar_t;
    auto launch = [&](auto gamma_tag) {
      constexpr int G = decltype(gamma_tag)::value;
      hipLaunchKernelGGL((ibp::focal_l2_bwd_kernel<T, G>), grid, block, 0,
                         cur_stream3(),
                         reinterpret_cast<const T*>(pred.data_ptr()),
                         reinterpret_cast<const T*>(gt.data_ptr()),
                         reinterpret_cast<const T*>(mask.data_ptr()),
                         reinterpret_cast<T*>(dpred.data_ptr()),
                         stack_gscale.data_ptr<float>(), per_stack, N, C, HW,
                         (int)heat_start, (int)bkg_start, (float)mtw, (float)ktw,
                         (float)alpha, (float)beta);
    };
    if (gamma == 1) launch(std::integral_constant<int, 1>{});
    else launch(std::integral_constant<int, 2>{});
  });
  return dpred;
}
