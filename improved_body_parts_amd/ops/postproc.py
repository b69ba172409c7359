"""Device post-processing wrappers (peaks, limb scoring)."""
from __future__ import annotations


from ._backend import hip_extension


def heatmap_nms_hip(heat, threshold=0.1):
    ext = hip_extension()
    return ext.heatmap_nms(heat.contiguous(), float(threshold))


def find_peaks_device(heat, threshold=0.1, radius=2, max_peaks=512):
    """heat: (C, H, W) keypoint channels on device. Returns fp32 [P][5] rows of
    (channel, x, y, refined_score, peak_score) on the host."""
    ext = hip_extension()
    heat = heat.contiguous()
    nmsed = ext.heatmap_nms(heat, float(threshold))
    out, cnt = ext.collect_peaks(nmsed, heat, radius, max_peaks)
    n = int(cnt.item())
    return out[:min(n, max_peaks)].cpu()


def limb_scores_device(paf, peaks_dev, cand_idx_dev, mid_num=20, thre2=0.1):
    ext = hip_extension()
    return ext.limb_scores(paf.contiguous(), peaks_dev, cand_idx_dev,
                           mid_num, float(thre2))
