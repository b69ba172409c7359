// Fused focal-L2 multi-task loss (forward + backward) with on-the-fly
// ground-truth pyramid.
//
// The reference materialises mask.expand().clone(), st, factor and out as full
// (nstack, N, C, H, W) temporaries AND a downsampled GT/mask pair per scale
// (reference models/loss_model.py:52-56, 134-161). Here each scale's forward
// is ONE pass producing per-stack partial sums, backward is ONE elementwise
// pass re-deriving the focal factor analytically, and the GT pyramid never
// exists: the kernel average-pools the full-res GT window and bilinearly
// samples + thresholds mask_miss per element (ratio r = H0/H, integer for the
// 128 -> 64/32/16/8 supervision pyramid) — no adaptive_avg_pool2d /
// interpolate launches, no (N,C,Hs,Ws) temporaries (round-1 weak #7).
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

// average of the r x r full-res window backing low-res pixel (y, x)
// (== adaptive_avg_pool2d with an integer ratio)
template <typename T>
__device__ __forceinline__ float gt_window_avg(const T* gt, int W0, int y,
                                               int x, int r) {
  if (r == 1) return ldf(gt + (long long)y * W0 + x);
  float s = 0.f;
  const T* row = gt + (long long)y * r * W0 + x * r;
  for (int dy = 0; dy < r; ++dy, row += W0)
    for (int dx = 0; dx < r; ++dx) s += ldf(row + dx);
  return s / (float)(r * r);
}

// bilinear (align_corners=False) sample of the full-res mask at low-res
// (y, x), then the reference's >= 0.5 threshold that keeps the value
// (loss_model.py:55-56 semantics as implemented in models/loss.py)
template <typename T>
__device__ __forceinline__ float mask_bilinear_thr(const T* mask, int H0,
                                                   int W0, int y, int x,
                                                   int r) {
  float v;
  if (r == 1) {
    v = ldf(mask + (long long)y * W0 + x);
  } else {
    const float sy = (y + 0.5f) * r - 0.5f;
    const float sx = (x + 0.5f) * r - 0.5f;
    int y0 = (int)floorf(sy), x0 = (int)floorf(sx);
    const float wy = sy - y0, wx = sx - x0;
    const int y0c = min(max(y0, 0), H0 - 1), y1c = min(max(y0 + 1, 0), H0 - 1);
    const int x0c = min(max(x0, 0), W0 - 1), x1c = min(max(x0 + 1, 0), W0 - 1);
    const float v00 = ldf(mask + (long long)y0c * W0 + x0c);
    const float v01 = ldf(mask + (long long)y0c * W0 + x1c);
    const float v10 = ldf(mask + (long long)y1c * W0 + x0c);
    const float v11 = ldf(mask + (long long)y1c * W0 + x1c);
    v = (1 - wy) * ((1 - wx) * v00 + wx * v01) +
        wy * ((1 - wx) * v10 + wx * v11);
  }
  return v >= 0.5f ? v : 0.f;
}

template <typename T, int GAMMA>
__global__ void focal_l2_fwd_kernel(
    const T* __restrict__ pred,   // [S, N, C, H, W]
    const T* __restrict__ gt,     // [N, C, H0, W0]  (H0 = r*H)
    const T* __restrict__ mask,   // [N, 1, H0, W0]
    float* __restrict__ stack_sums,  // [S]
    long long per_stack, int N, int C, int H, int W, int H0, int W0, int r,
    int heat_start, int bkg_start, float mtw, float ktw,
    float alpha, float beta) {
  __shared__ float lds[16];
  int s_idx = blockIdx.y;
  const T* p = pred + (long long)s_idx * per_stack;
  const long long HW = (long long)H * W;
  const long long HW0 = (long long)H0 * W0;
  float acc = 0.f;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < per_stack; i += (long long)gridDim.x * blockDim.x) {
    long long hw = i % HW;
    long long nc = i / HW;
    int c = (int)(nc % C);
    int n = (int)(nc / C);
    int y = (int)(hw / W), x = (int)(hw % W);
    float sv = ldf(p + i);
    float gv = gt_window_avg(gt + nc * HW0, W0, y, x, r);
    float mv = mask_bilinear_thr(mask + (long long)n * HW0, H0, W0, y, x, r);
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
    long long per_stack, int N, int C, int H, int W, int H0, int W0, int r,
    int heat_start, int bkg_start, float mtw, float ktw,
    float alpha, float beta) {
  int s_idx = blockIdx.y;
  const T* p = pred + (long long)s_idx * per_stack;
  T* dp = dpred + (long long)s_idx * per_stack;
  float gscale = stack_gscale[s_idx];
  const long long HW = (long long)H * W;
  const long long HW0 = (long long)H0 * W0;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < per_stack; i += (long long)gridDim.x * blockDim.x) {
    long long hw = i % HW;
    long long nc = i / HW;
    int c = (int)(nc % C);
    int n = (int)(nc / C);
    int y = (int)(hw / W), x = (int)(hw % W);
    float sv = ldf(p + i);
    float gv = gt_window_avg(gt + nc * HW0, W0, y, x, r);
    float mv = mask_bilinear_thr(mask + (long long)n * HW0, H0, W0, y, x, r);
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

static int pyramid_ratio(const Tensor& pred, const Tensor& gt) {
  int H = (int)pred.size(3), W = (int)pred.size(4);
  int H0 = (int)gt.size(2), W0 = (int)gt.size(3);
  TORCH_CHECK(H0 % H == 0 && W0 % W == 0 && H0 / H == W0 / W,
              "focal_l2: GT size ", H0, "x", W0,
              " must be an integer multiple of prediction size ", H, "x", W);
  return H0 / H;
}

// returns per-stack sums [S] (fp32); host applies nstack weights.
// gt/mask may be FULL resolution: the kernel pools/samples on the fly.
Tensor focal_l2_fwd(const Tensor& pred, const Tensor& gt, const Tensor& mask,
                    int64_t heat_start, int64_t bkg_start, int64_t gamma,
                    double mtw, double ktw, double alpha, double beta) {
  TORCH_CHECK(pred.dim() == 5 && pred.is_contiguous());
  TORCH_CHECK(gt.is_contiguous() && mask.is_contiguous());
  int S = (int)pred.size(0), N = (int)pred.size(1), C = (int)pred.size(2);
  int H = (int)pred.size(3), W = (int)pred.size(4);
  int H0 = (int)gt.size(2), W0 = (int)gt.size(3);
  int r = pyramid_ratio(pred, gt);
  long long per_stack = (long long)N * C * H * W;
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
                         sums.data_ptr<float>(), per_stack, N, C, H, W, H0, W0,
                         r, (int)heat_start, (int)bkg_start, (float)mtw,
                         (float)ktw, (float)alpha, (float)beta);
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
  int H = (int)pred.size(3), W = (int)pred.size(4);
  int H0 = (int)gt.size(2), W0 = (int)gt.size(3);
  int r = pyramid_ratio(pred, gt);
  long long per_stack = (long long)N * C * H * W;
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
                         stack_gscale.data_ptr<float>(), per_stack, N, C, H, W,
                         H0, W0, r, (int)heat_start, (int)bkg_start, (float)mtw,
                         (float)ktw, (float)alpha, (float)beta);
    };
    if (gamma == 1) launch(std::integral_constant<int, 1>{});
    else launch(std::integral_constant<int, 2>{});
  });
  return dpred;
}
