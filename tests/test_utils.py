"""CPU tests for utils/util.py (padding, smoothing, NMS, centroid, BN-eval)."""
import numpy as np
import torch

from improved_body_parts_amd.utils import (
    GaussianSmoothing, center_pad, keypoint_heatmap_nms, padRightDownCorner,
    refine_centroid, set_bn_eval, set_bn_eval_fp32)


def test_pad_right_down_corner_multiple():
    img = np.random.RandomState(0).rand(100, 130, 3).astype(np.float32)
    padded, pad = padRightDownCorner(img, 64, 0.5)
    assert padded.shape[0] % 64 == 0 and padded.shape[1] % 64 == 0
    # original content preserved at the top-left
    np.testing.assert_array_equal(padded[:100, :130], img)
    # pad = [top, left, bottom, right]
    assert pad[0] == 0 and pad[1] == 0
    assert pad[2] == padded.shape[0] - 100 and pad[3] == padded.shape[1] - 130
    assert float(padded[100:, :, :].max()) == 0.5


def test_center_pad_and_unpad_roundtrip():
    img = np.random.RandomState(1).rand(50, 70, 3).astype(np.float32)
    padded, pad = center_pad(img, 64, 0.0)
    assert padded.shape[0] % 64 == 0 and padded.shape[1] % 64 == 0
    top, left = pad[0], pad[1]
    np.testing.assert_array_equal(padded[top:top + 50, left:left + 70], img)


def test_gaussian_smoothing_preserves_mass_location():
    sm = GaussianSmoothing(channels=1, kernel_size=7, sigma=2.0)
    x = torch.zeros(1, 1, 21, 21)
    x[0, 0, 10, 10] = 1.0
    y = sm(x)
    # valid convolution (reference semantics): shrinks by kernel_size - 1
    assert y.shape == (1, 1, 15, 15)
    # peak stays at the impulse (shifted by the valid-conv crop)
    iy, ix = divmod(int(y.argmax()), 15)
    assert (iy, ix) == (7, 7)
    assert float(y.max()) < 1.0 and float(y.sum()) > 0.5


def test_keypoint_heatmap_nms_isolates_peaks():
    h = torch.zeros(1, 1, 16, 16)
    h[0, 0, 4, 4] = 0.9
    h[0, 0, 4, 5] = 0.6   # neighbour suppressed
    h[0, 0, 12, 12] = 0.5
    out = keypoint_heatmap_nms(h, thre=0.1)
    nz = (out[0, 0] > 0).nonzero().tolist()
    assert [4, 4] in nz and [12, 12] in nz and [4, 5] not in nz


def test_refine_centroid_subpixel():
    heat = np.zeros((16, 16), np.float32)
    yy, xx = np.mgrid[0:16, 0:16]
    heat += np.exp(-((xx - 7.4) ** 2 + (yy - 8.6) ** 2) / 4.0)
    x, y, score = refine_centroid(heat, (7, 9), radius=3)
    assert abs(x - 7.4) < 0.35 and abs(y - 8.6) < 0.35
    assert score > 0.2  # score = box mean (reference semantics)


def test_set_bn_eval_freezes_stats():
    net = torch.nn.Sequential(torch.nn.Conv2d(3, 4, 3), torch.nn.BatchNorm2d(4))
    net.train()
    net.apply(set_bn_eval_fp32)
    bn = net[1]
    assert not bn.training
    rm = bn.running_mean.clone()
    net(torch.randn(2, 3, 8, 8))
    assert torch.equal(bn.running_mean, rm)  # eval BN: stats untouched
    # the SWA variant additionally drops BN to bf16 (reference used fp16)
    bn2 = torch.nn.BatchNorm2d(4)
    set_bn_eval(bn2)
    assert not bn2.training and bn2.weight.dtype == torch.bfloat16
