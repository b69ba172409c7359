"""Visualization utilities tests (capability of reference demo_image.py:64-101,
561-596 and visulizatoin/draw_net.py)."""
import numpy as np
import torch

from improved_body_parts_amd.utils.visualization import (
    count_conv_flops, count_parameters, draw_people, heatmap_overlay,
    hsv_to_rgb, model_summary, person_colors, show_color_vector)


def test_hsv_to_rgb_primaries():
    r, g, b = hsv_to_rgb(np.array([0.0, 1 / 3, 2 / 3]), np.ones(3), np.ones(3))
    rgb = np.stack([r, g, b], 1)
    np.testing.assert_allclose(rgb, np.eye(3), atol=1e-6)


def test_show_color_vector_shape_and_range():
    px = np.random.randn(32, 40).astype(np.float32)
    py = np.random.randn(32, 40).astype(np.float32)
    img = show_color_vector(px, py)
    assert img.shape == (32, 40, 3) and img.dtype == np.uint8


def test_draw_people_marks_pixels():
    img = np.zeros((64, 64, 3), np.uint8)
    pts = [None] * 17
    pts[5], pts[7], pts[9] = (10.0, 10.0), (30.0, 30.0), (50.0, 20.0)
    canvas = draw_people(img, [(pts, 0.9)])
    assert canvas.sum() > 0
    assert img.sum() == 0  # input untouched


def test_heatmap_overlay():
    img = np.zeros((16, 16, 3), np.uint8)
    hm = np.zeros((16, 16), np.float32)
    hm[8, 8] = 1.0
    out = heatmap_overlay(img, hm)
    assert out[8, 8].sum() > 0


def test_model_stats():
    from improved_body_parts_amd.models import PoseNet
    net = PoseNet(1, 64, 50, bn=True, increase=32)
    n = count_parameters(net)
    assert n > 1e5
    fl = count_conv_flops(net, (1, 64, 64, 3))
    assert fl > n  # FLOPs dominate params for convs
    assert "TOTAL" in model_summary(net)


def test_person_colors_distinct():
    cols = person_colors(8)
    assert len({tuple(c) for c in cols}) == 8
