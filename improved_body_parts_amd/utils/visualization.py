"""Visualization & model-inspection utilities.

Capability parity with the reference's L6 layer: PAF vector-field HSV
rendering (reference demo_image.py:64-101 ``show_color_vector``), skeleton /
keypoint drawing on images (demo_image.py:561-596), and model statistics
(parameter/FLOP counting, the role of visulizatoin/draw_net.py's thop and
graphviz dumps). Implemented numpy-first — this image has no cv2/matplotlib,
so rasterisation is done directly and matplotlib/graphviz are optional.
"""
from __future__ import annotations

import numpy as np
import torch


# ---------------------------------------------------------------------------
# colour tools
# ---------------------------------------------------------------------------

def hsv_to_rgb(h, s, v):
    """Vectorised HSV->RGB, h/s/v in [0,1] arrays; returns float arrays."""
    h = np.asarray(h) % 1.0
    s = np.asarray(s)
    v = np.asarray(v)
    i = np.floor(h * 6.0).astype(int)
    f = h * 6.0 - i
    p = v * (1.0 - s)
    q = v * (1.0 - s * f)
    t = v * (1.0 - s * (1.0 - f))
    i = i % 6
    r = np.choose(i, [v, q, p, p, t, v])
    g = np.choose(i, [t, v, v, q, p, p])
    b = np.choose(i, [p, p, t, v, v, q])
    return r, g, b


def person_colors(n):
    """n visually distinct RGB (0-255) colours (reference demo colour wheel)."""
    h = np.arange(n) / max(n, 1)
    r, g, b = hsv_to_rgb(h, np.ones(n), np.ones(n))
    return (np.stack([r, g, b], axis=1) * 255).astype(np.uint8)


def show_color_vector(paf_x, paf_y):
    """Render a 2-channel PAF vector field as an HSV image (hue=direction,
    value=magnitude), the role of reference demo_image.py:64-101.

    paf_x/paf_y: (H, W) arrays. Returns (H, W, 3) uint8 RGB."""
    paf_x = np.asarray(paf_x, dtype=np.float32)
    paf_y = np.asarray(paf_y, dtype=np.float32)
    mag = np.sqrt(paf_x ** 2 + paf_y ** 2)
    ang = (np.arctan2(paf_y, paf_x) + np.pi) / (2 * np.pi)
    mmax = mag.max() if mag.max() > 0 else 1.0
    r, g, b = hsv_to_rgb(ang, np.ones_like(ang), mag / mmax)
    return (np.stack([r, g, b], axis=-1) * 255).astype(np.uint8)


def heatmap_overlay(image, heatmap, alpha=0.5):
    """Blend a (H, W) heatmap over an RGB uint8 image (debug display the
    reference keeps as commented matplotlib blocks, train.py:188-200)."""
    image = np.asarray(image)
    hm = np.asarray(heatmap, dtype=np.float32)
    hm = (hm - hm.min()) / (np.ptp(hm) + 1e-8)
    r, g, b = hsv_to_rgb(0.66 * (1.0 - hm), np.ones_like(hm), hm)
    color = np.stack([r, g, b], axis=-1) * 255
    out = (1 - alpha * hm[..., None]) * image + alpha * hm[..., None] * color
    return out.astype(np.uint8)


# ---------------------------------------------------------------------------
# numpy rasterisation (no cv2 in this image)
# ---------------------------------------------------------------------------

def draw_line(img, x0, y0, x1, y1, color, thickness=2):
    """Draw a line segment on (H, W, 3) uint8 in place (dense sampling)."""
    H, W = img.shape[:2]
    n = int(max(abs(x1 - x0), abs(y1 - y0), 1)) * 2 + 1
    xs = np.linspace(x0, x1, n)
    ys = np.linspace(y0, y1, n)
    r = max(int(thickness) // 2, 0)
    for dx in range(-r, r + 1):
        for dy in range(-r, r + 1):
            xi = np.clip(np.round(xs + dx).astype(int), 0, W - 1)
            yi = np.clip(np.round(ys + dy).astype(int), 0, H - 1)
            img[yi, xi] = color
    return img


def draw_circle(img, x, y, radius, color):
    H, W = img.shape[:2]
    y0, y1 = max(int(y - radius), 0), min(int(y + radius) + 1, H)
    x0, x1 = max(int(x - radius), 0), min(int(x + radius) + 1, W)
    if y1 <= y0 or x1 <= x0:
        return img
    yy, xx = np.mgrid[y0:y1, x0:x1]
    mask = (yy - y) ** 2 + (xx - x) ** 2 <= radius ** 2
    img[y0:y1, x0:x1][mask] = color
    return img


# COCO-17 skeleton edges for drawing assembled people
_COCO_EDGES = [(0, 1), (0, 2), (1, 3), (2, 4), (5, 6), (5, 7), (7, 9), (6, 8),
               (8, 10), (5, 11), (6, 12), (11, 12), (11, 13), (13, 15),
               (12, 14), (14, 16)]


def draw_people(image, keypoints, point_radius=4, line_thickness=3):
    """Draw assembled people (the output of engine.inference.process) onto an
    RGB uint8 image copy (reference demo_image.py:561-596).

    keypoints: list of (coco_17_points, score) where each point is (x, y) or
    None / (0, 0) for missing."""
    canvas = np.ascontiguousarray(np.asarray(image)).copy()
    if canvas.dtype != np.uint8:
        canvas = (np.clip(canvas, 0, 1) * 255).astype(np.uint8)
    colors = person_colors(max(len(keypoints), 1))
    for pi, (pts, _score) in enumerate(keypoints):
        color = colors[pi % len(colors)].tolist()
        def ok(p):
            return p is not None and not (p[0] == 0 and p[1] == 0)
        for a, b in _COCO_EDGES:
            if a < len(pts) and b < len(pts) and ok(pts[a]) and ok(pts[b]):
                draw_line(canvas, pts[a][0], pts[a][1], pts[b][0], pts[b][1],
                          color, line_thickness)
        for p in pts:
            if ok(p):
                draw_circle(canvas, p[0], p[1], point_radius, color)
    return canvas


# ---------------------------------------------------------------------------
# model statistics (role of thop/graphviz in visulizatoin/draw_net.py)
# ---------------------------------------------------------------------------

def count_parameters(model):
    return sum(p.numel() for p in model.parameters())


@torch.no_grad()
def count_conv_flops(model, input_shape=(1, 512, 512, 3)):
    """Forward-hook FLOP count of conv/linear layers for one NHWC input
    (reference draw_net.py:113-117 used thop; unavailable offline)."""
    flops = [0]
    hooks = []

    def conv_hook(mod, inp, out):
        k = mod.kernel_size[0] * mod.kernel_size[1]
        flops[0] += 2 * out.numel() * (inp[0].shape[1] // mod.groups) * k

    def linear_hook(mod, inp, out):
        flops[0] += 2 * out.numel() * mod.in_features

    for m in model.modules():
        if isinstance(m, torch.nn.Conv2d):
            hooks.append(m.register_forward_hook(conv_hook))
        elif isinstance(m, torch.nn.Linear):
            hooks.append(m.register_forward_hook(linear_hook))
    was_training = model.training
    model.eval()
    try:
        model(torch.zeros(*input_shape))
    finally:
        for h in hooks:
            h.remove()
        model.train(was_training)
    return flops[0]


def model_summary(model, input_shape=(1, 512, 512, 3)):
    """One-line-per-module text summary + totals (draw_net.py's role)."""
    lines = []
    for name, mod in model.named_modules():
        n = sum(p.numel() for p in mod.parameters(recurse=False))
        if n:
            lines.append(f"{name:60s} {type(mod).__name__:16s} {n:>12,d}")
    total = count_parameters(model)
    lines.append(f"{'TOTAL':60s} {'':16s} {total:>12,d}")
    return "\n".join(lines)


def export_onnx(model, path, input_shape=(1, 512, 512, 3)):  # pragma: no cover
    """ONNX export (reference draw_net.py:89-93). Needs the onnx package."""
    model.eval()
    torch.onnx.export(model, torch.zeros(*input_shape), path,
                      opset_version=17)
    return path
