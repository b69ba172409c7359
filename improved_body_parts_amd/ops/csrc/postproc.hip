// On-device keypoint post-processing primitives.
//
// Replaces the reference's CPU-bound post-process stages (5.2 FPS headline
// bottleneck, reference README.md:68): peak NMS (utils/util.py:177-183),
// sub-pixel centroid refinement (:186-211) and the 20-point limb line-integral
// scoring of find_connections (evaluate.py:206-276). The tiny greedy assembly
// stays on the host, consuming the device-scored candidates.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

// 3x3 max-pool-equality NMS with reflect border + threshold; NCHW layout
// (post-processing operates on the resized NCHW heatmaps).
template <typename T>
__global__ void nms3x3_kernel(const T* __restrict__ heat, T* __restrict__ out,
                              int NC, int H, int W, float thre) {
  long long total = (long long)NC * H * W;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    int w = (int)(i % W);
    long long q = i / W;
    int h = (int)(q % H);
    int nc = (int)(q / H);
    const T* plane = heat + (long long)nc * H * W;
    float v = ldf(plane + (long long)h * W + w);
    float m = -1e30f;
    #pragma unroll
    for (int dy = -1; dy <= 1; ++dy) {
      #pragma unroll
      for (int dx = -1; dx <= 1; ++dx) {
        int hh = h + dy, ww = w + dx;
        hh = hh < 0 ? -hh : (hh >= H ? 2 * H - hh - 2 : hh);  // reflect
        ww = ww < 0 ? -ww : (ww >= W ? 2 * W - ww - 2 : ww);
        m = fmaxf(m, ldf(plane + (long long)hh * W + ww));
      }
    }
    stf(out + i, (v == m && v >= thre) ? v : 0.f);
  }
}

// collect peaks (nonzero after NMS) with sub-pixel centroid refinement.
// out rows: [channel, x_refined, y_refined, score_box_mean, peak_score]
template <typename T>
__global__ void collect_peaks_kernel(const T* __restrict__ nmsed,
                                     const T* __restrict__ smoothed,
                                     float* __restrict__ out, int* __restrict__ cnt,
                                     int C, int H, int W, int radius, int max_peaks) {
  long long total = (long long)C * H * W;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    float v = ldf(nmsed + i);
    if (v <= 0.f) continue;
    int w = (int)(i % W);
    long long q = i / W;
    int h = (int)(q % H);
    int c = (int)(q / H);
    int slot = atomicAdd(cnt, 1);
    if (slot >= max_peaks) continue;
    // weighted-centroid refinement over (2r+1)^2 (reference util.py:186-211)
    float xr = (float)w, yr = (float)h, boxmean = v;
    const T* plane = smoothed + (long long)c * H * W;
    if (h - radius >= 0 && h + radius < H && w - radius >= 0 && w + radius < W) {
      float s = 0.f, sx = 0.f, sy = 0.f;
      for (int dy = -radius; dy <= radius; ++dy)
        for (int dx = -radius; dx <= radius; ++dx) {
          float sv = ldf(plane + (long long)(h + dy) * W + (w + dx));
          s += sv; sx += sv * dx; sy += sv * dy;
        }
      if (s > 0.f) {
        xr = w + sx / s;
        yr = h + sy / s;
        boxmean = s / ((2 * radius + 1) * (2 * radius + 1));
      }
    }
    float* row = out + (long long)slot * 5;
    row[0] = (float)c; row[1] = xr; row[2] = yr; row[3] = boxmean; row[4] = v;
  }
}

// Score every candidate limb: sample mid_num points along each A->B segment on
// the limb channel; emit [mean_score_with_dist_prior, pass_ratio, length].
// grid: one thread per (limb_type, a, b) candidate triple (flattened).
template <typename T>
__global__ void limb_score_kernel(
    const T* __restrict__ paf,          // [Cpaf, H, W]
    const float* __restrict__ peaks,    // [P][5] rows from collect_peaks
    const int* __restrict__ cand_idx,   // [ncand][3]: limb_id, peakA, peakB
    float* __restrict__ scores,         // [ncand][3]
    int ncand, int H, int W, int mid_num, float thre2) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < ncand;
       i += gridDim.x * blockDim.x) {
    int limb = cand_idx[i * 3 + 0];
    const float* pa = peaks + (long long)cand_idx[i * 3 + 1] * 5;
    const float* pb = peaks + (long long)cand_idx[i * 3 + 2] * 5;
    float ax = pa[1], ay = pa[2], bx = pb[1], by = pb[2];
    float dx = bx - ax, dy = by - ay;
    float len = sqrtf(dx * dx + dy * dy) + 1e-9f;
    const T* plane = paf + (long long)limb * H * W;
    float sum = 0.f;
    int pass = 0;
    // short limbs sample fewer points (reference evaluate.py:228:
    // mid_num = min(round(norm)+1, param mid_num))
    const int mn = min((int)roundf(len) + 1, mid_num);
    for (int k = 0; k < mn; ++k) {
      float t = mn == 1 ? 0.f : (float)k / (mn - 1);
      int x = (int)roundf(ax + t * dx);
      int y = (int)roundf(ay + t * dy);
      x = min(max(x, 0), W - 1);
      y = min(max(y, 0), H - 1);
      float v = ldf(plane + (long long)y * W + x);
      sum += v;
      if (v > thre2) ++pass;
    }
    float mean = sum / mn;
    // distance prior of the reference (evaluate.py:240): penalise limbs longer
    // than half the image height
    float prior = fminf(0.5f * H / len - 1.f, 0.f);
    scores[i * 3 + 0] = mean + prior;
    scores[i * 3 + 1] = (float)pass / mn;
    scores[i * 3 + 2] = len;
  }
}

}  // namespace ibp

using torch::Tensor;
static inline hipStream_t cur_stream4() {
  return at::hip::getCurrentHIPStream().stream();
}

Tensor heatmap_nms(const Tensor& heat, double thre) {
  TORCH_CHECK(heat.is_cuda() && heat.is_contiguous());
  int W = (int)heat.size(-1), H = (int)heat.size(-2);
  int NC = (int)(heat.numel() / ((long long)H * W));
  Tensor out = torch::empty_like(heat);
  dim3 block(256), grid(ibp::grid_1d(heat.numel(), 256, 4096));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      heat.scalar_type(), "heatmap_nms", [&] {
    hipLaunchKernelGGL(ibp::nms3x3_kernel<scalar_t>, grid, block, 0, cur_stream4(),
                       reinterpret_cast<const scalar_t*>(heat.data_ptr()),
                       reinterpret_cast<scalar_t*>(out.data_ptr()),
                       NC, H, W, (float)thre);
  });
  return out;
}

std::vector<Tensor> collect_peaks(const Tensor& nmsed, const Tensor& smoothed,
                                  int64_t radius, int64_t max_peaks) {
  int W = (int)nmsed.size(-1), H = (int)nmsed.size(-2);
  int C = (int)(nmsed.numel() / ((long long)H * W));
  Tensor out = torch::zeros({max_peaks, 5},
                            nmsed.options().dtype(torch::kFloat32));
  Tensor cnt = torch::zeros({1}, nmsed.options().dtype(torch::kInt32));
  dim3 block(256), grid(ibp::grid_1d(nmsed.numel(), 256, 4096));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      nmsed.scalar_type(), "collect_peaks", [&] {
    hipLaunchKernelGGL(ibp::collect_peaks_kernel<scalar_t>, grid, block, 0,
                       cur_stream4(),
                       reinterpret_cast<const scalar_t*>(nmsed.data_ptr()),
                       reinterpret_cast<const scalar_t*>(smoothed.data_ptr()),
                       out.data_ptr<float>(), cnt.data_ptr<int>(),
                       C, H, W, (int)radius, (int)max_peaks);
  });
  return {out, cnt};
}

Tensor limb_scores(const Tensor& paf, const Tensor& peaks, const Tensor& cand_idx,
                   int64_t mid_num, double thre2) {
  int W = (int)paf.size(-1), H = (int)paf.size(-2);
  int ncand = (int)cand_idx.size(0);
  Tensor scores = torch::zeros({std::max(ncand, 1), 3},
                               paf.options().dtype(torch::kFloat32));
  if (ncand == 0) return scores;
  dim3 block(256), grid(ibp::grid_1d(ncand, 256, 2048));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      paf.scalar_type(), "limb_scores", [&] {
    hipLaunchKernelGGL(ibp::limb_score_kernel<scalar_t>, grid, block, 0,
                       cur_stream4(),
                       reinterpret_cast<const scalar_t*>(paf.data_ptr()),
                       peaks.data_ptr<float>(), cand_idx.data_ptr<int>(),
                       scores.data_ptr<float>(), ncand, H, W, (int)mid_num,
                       (float)thre2);
  });
  return scores;
}
