// On-device ground-truth heatmap generation (CDNA4).
//
// Device twin of the numpy oracle in data/heatmapper.py (itself the behavior
// of reference py_cocodata_server/py_data_heatmapper.py:50-240): Gaussian
// keypoint maps on the original-resolution grid sampled every `stride` px,
// PAF-as-Gaussian limb maps from perpendicular distance-to-segment with
// per-pixel hit-count averaging, eroded-mask + reverse-keypoint backgrounds.
// This keeps the whole 512^2 GT pipeline on the GPU (north-star requirement),
// replacing the reference's ~40 samples/s/process CPU path (README.md:35).
//
// Layout: one thread per output pixel of one sample; the thread walks all
// persons once for the 18 keypoint channels and once per limb type, keeping
// every intermediate in registers — each of the 50 CHW planes is written
// exactly once, so the kernel is store-bound (~50 floats/pixel) rather than
// bound by the P*J joint reads (which come from a tiny L2-resident table).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace ibp {

constexpr int MAX_PARTS = 32;   // canonical config: 18 (+ headroom)
constexpr int MAX_LIMBS = 64;   // canonical 30 / dense 49

struct GtParams {
  const float* joints;   // [N][P][J][3] (x, y, vis) original-res coords
  const float* mask_all; // [N][h][w] stride-grid all-person mask (or null)
  float* out;            // [N][C][h][w]
  int N, P, J, h, w, C;
  int stride;
  int paf_layers, heat_start, bkg_start;
  int limb_from[MAX_LIMBS], limb_to[MAX_LIMBS];
  float double_sigma2;       // 2*sigma^2 (keypoints)
  float paf_sigma;
  float limb_thre;           // limb gaussian threshold (quirk: -> 0.01)
  float paf_thre;            // bbox dilation in original px
  int half;                  // gaussian_size/2 in grid cells
};

__global__ void heatmap_gt_kernel(GtParams p) {
  long long total = (long long)p.N * p.h * p.w;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const int x = (int)(i % p.w);
    long long q = i / p.w;
    const int y = (int)(q % p.h);
    const int n = (int)(q / p.h);
    // original-resolution coordinate of this grid cell centre
    const float gx = x * p.stride + p.stride * 0.5f - 0.5f;
    const float gy = y * p.stride + p.stride * 0.5f - 0.5f;
    const float* joints = p.joints + ((long long)n * p.P) * p.J * 3;
    float* out = p.out + ((long long)n * p.C) * p.h * p.w;
    const long long pix = (long long)y * p.w + x;
    const long long plane = (long long)p.h * p.w;

    const int n_parts = p.bkg_start - p.heat_start;
    float heat[MAX_PARTS];
    for (int c = 0; c < n_parts; ++c) heat[c] = 0.f;

    // ---- keypoint channels: max over persons, truncated window ----------
    for (int pe = 0; pe < p.P; ++pe) {
      const float* jrow = joints + (long long)pe * p.J * 3;
      for (int part = 0; part < n_parts; ++part) {
        const float vis = jrow[part * 3 + 2];
        if (vis >= 2.f) continue;
        const float jx = jrow[part * 3 + 0];
        const float jy = jrow[part * 3 + 1];
        // window: +-half grid cells around the rounded centre (oracle
        // put_gaussian_maps) — same truncation, not just the analytic tail
        const int cx = (int)roundf(jx / p.stride);
        const int cy = (int)roundf(jy / p.stride);
        if (x < cx - p.half || x > cx + p.half ||
            y < cy - p.half || y > cy + p.half) continue;
        const float dx = gx - jx, dy = gy - jy;
        const float v = __expf(-(dx * dx + dy * dy) / p.double_sigma2);
        heat[part] = fmaxf(heat[part], v);
      }
    }

    // ---- limb channels: sum/hit-count average over persons ---------------
    for (int k = 0; k < p.paf_layers; ++k) {
      const int fr = p.limb_from[k], to = p.limb_to[k];
      float acc = 0.f;
      int count = 0;
      for (int pe = 0; pe < p.P; ++pe) {
        const float* jrow = joints + (long long)pe * p.J * 3;
        if (jrow[fr * 3 + 2] >= 2.f || jrow[to * 3 + 2] >= 2.f) continue;
        const float x1 = jrow[fr * 3 + 0], y1 = jrow[fr * 3 + 1];
        const float x2 = jrow[to * 3 + 0], y2 = jrow[to * 3 + 1];
        const float xD = x2 - x1, yD = y2 - y1;
        const float d2 = xD * xD + yD * yD;
        if (d2 == 0.f) continue;
        // bbox in grid cells, dilated by paf_thre original px (oracle
        // put_limb_gaussian_maps rounding included)
        const int min_sx = (int)roundf((fminf(x1, x2) - p.paf_thre) / p.stride);
        const int min_sy = (int)roundf((fminf(y1, y2) - p.paf_thre) / p.stride);
        const int max_sx = (int)roundf((fmaxf(x1, x2) + p.paf_thre) / p.stride);
        const int max_sy = (int)roundf((fmaxf(y1, y2) + p.paf_thre) / p.stride);
        if (x < min_sx || x > max_sx || y < min_sy || y > max_sy) continue;
        const float norm = sqrtf(d2);
        const float dist = fabsf((xD * (y1 - gy) - (x1 - gx) * yD) / (norm + 1e-6f));
        float g = __expf(-(dist * dist) / (2.f * p.paf_sigma * p.paf_sigma));
        if (g <= p.limb_thre) g = 0.01f;  // oracle limb_gaussian quirk
        acc += g;
        count += 1;
      }
      const float v = count > 0 ? acc / count : 0.f;
      out[(long long)k * plane + pix] = fminf(fmaxf(v, 0.f), 1.f);
    }

    // ---- write keypoint channels + backgrounds ---------------------------
    float kmax = 0.f;
    for (int c = 0; c < n_parts; ++c) {
      const float v = fminf(heat[c], 1.f);
      out[(long long)(p.heat_start + c) * plane + pix] = v;
      kmax = fmaxf(kmax, v);
    }
    // bkg 0: 3x3-eroded all-person mask with edge replication
    float m = 1.f;
    if (p.mask_all) {
      const float* mk = p.mask_all + (long long)n * plane;
      m = 1e30f;
      for (int dy2 = -1; dy2 <= 1; ++dy2)
        for (int dx2 = -1; dx2 <= 1; ++dx2) {
          int yy = min(max(y + dy2, 0), p.h - 1);
          int xx = min(max(x + dx2, 0), p.w - 1);
          m = fminf(m, mk[(long long)yy * p.w + xx]);
        }
    }
    out[(long long)p.bkg_start * plane + pix] = fminf(fmaxf(m, 0.f), 1.f);
    out[(long long)(p.bkg_start + 1) * plane + pix] = kmax;
  }
}

}  // namespace ibp

using torch::Tensor;

Tensor heatmap_gt(const Tensor& joints, const c10::optional<Tensor>& mask_all,
                  const Tensor& limb_pairs, int64_t h, int64_t w,
                  int64_t stride, int64_t heat_start, int64_t bkg_start,
                  int64_t num_layers, double sigma, double paf_sigma,
                  double keypoint_thre, double limb_thre, double paf_thre) {
  TORCH_CHECK(joints.is_cuda() && joints.dim() == 4 && joints.size(3) == 3,
              "joints must be [N][P][J][3] on device");
  TORCH_CHECK(joints.scalar_type() == at::ScalarType::Float);
  auto jc = joints.contiguous();
  ibp::GtParams p;
  p.joints = jc.data_ptr<float>();
  Tensor mk;
  if (mask_all.has_value()) {
    mk = mask_all->contiguous().to(torch::kFloat32);
    TORCH_CHECK(mk.dim() == 3 && mk.size(1) == h && mk.size(2) == w);
    p.mask_all = mk.data_ptr<float>();
  } else {
    p.mask_all = nullptr;
  }
  p.N = (int)joints.size(0);
  p.P = (int)joints.size(1);
  p.J = (int)joints.size(2);
  p.h = (int)h; p.w = (int)w;
  p.C = (int)num_layers;
  p.stride = (int)stride;
  p.paf_layers = (int)heat_start;
  p.heat_start = (int)heat_start;
  p.bkg_start = (int)bkg_start;
  auto lp = limb_pairs.contiguous().cpu();
  TORCH_CHECK(lp.size(0) == heat_start && lp.size(0) <= ibp::MAX_LIMBS);
  TORCH_CHECK(bkg_start - heat_start <= ibp::MAX_PARTS);
  auto lpa = lp.accessor<int, 2>();
  for (int k = 0; k < lp.size(0); ++k) {
    p.limb_from[k] = lpa[k][0];
    p.limb_to[k] = lpa[k][1];
  }
  p.double_sigma2 = (float)(2.0 * sigma * sigma);
  p.paf_sigma = (float)paf_sigma;
  p.limb_thre = (float)limb_thre;
  p.paf_thre = (float)paf_thre;
  p.half = (int)(std::ceil(std::sqrt(-2.0 * sigma * sigma *
                                     std::log(keypoint_thre)) / stride));
  Tensor out = torch::empty({joints.size(0), num_layers, h, w},
                            joints.options().dtype(torch::kFloat32));
  p.out = out.data_ptr<float>();
  long long total = (long long)p.N * h * w;
  dim3 block(256), grid(ibp::grid_1d(total, 256, 8192));
  auto stream = at::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(ibp::heatmap_gt_kernel, grid, block, 0, stream, p);
  return out;
}
