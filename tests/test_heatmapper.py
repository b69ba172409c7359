import numpy as np
import pytest

from improved_body_parts_amd.config import CanonicalConfig
from improved_body_parts_amd.data import Heatmapper, limb_gaussian


@pytest.fixture(scope="module")
def cfg():
    return CanonicalConfig(128, 128, 4)


@pytest.fixture(scope="module")
def hm(cfg):
    return Heatmapper(cfg)


def _one_person(cfg, visible_parts):
    joints = np.zeros((1, cfg.num_parts, 3), np.float32)
    joints[:, :, 2] = 3  # never marked
    for part, (x, y) in visible_parts.items():
        pid = cfg.parts_dict[part]
        joints[0, pid] = [x, y, 1]
    return joints


def test_keypoint_gaussian_peak(cfg, hm):
    joints = _one_person(cfg, {"nose": (64.0, 64.0)})
    maps = hm.create_heatmaps(joints, np.zeros(cfg.mask_shape, np.float32))
    ch = cfg.heat_start + cfg.parts_dict["nose"]
    m = maps[ch]
    # grid cell centre nearest to (64, 64): cell 16 has centre 65.5, cell 15 -> 61.5
    iy, ix = np.unravel_index(m.argmax(), m.shape)
    assert (ix, iy) in [(15, 15), (16, 16), (15, 16), (16, 15)]
    # peak value = exp(-(d^2+d^2)/2sigma^2) at the nearest cell centre
    gx = ix * cfg.stride + cfg.stride / 2 - 0.5
    expected = np.exp(-2 * (gx - 64.0) ** 2 / (2 * 9 ** 2))
    assert m.max() == pytest.approx(expected, rel=1e-4)
    # truncation: far away must be exactly 0
    assert m[0, 0] == 0.0


def test_keypoint_max_combining_two_people(cfg, hm):
    joints = np.zeros((2, cfg.num_parts, 3), np.float32)
    joints[:, :, 2] = 3
    joints[0, 0] = [40.0, 40.0, 1]
    joints[1, 0] = [48.0, 40.0, 1]   # 8 px apart, overlapping gaussians
    maps = hm.create_heatmaps(joints, np.zeros(cfg.mask_shape, np.float32))
    ch = cfg.heat_start
    single = Heatmapper(cfg)
    m1 = single.create_heatmaps(joints[:1], np.zeros(cfg.mask_shape, np.float32))[ch]
    m2 = single.create_heatmaps(joints[1:], np.zeros(cfg.mask_shape, np.float32))[ch]
    np.testing.assert_allclose(maps[ch], np.maximum(m1, m2), atol=1e-6)


def test_invisible_joints_ignored(cfg, hm):
    joints = _one_person(cfg, {"nose": (64.0, 64.0)})
    joints[0, cfg.parts_dict["nose"], 2] = 2  # not marked
    maps = hm.create_heatmaps(joints, np.zeros(cfg.mask_shape, np.float32))
    assert maps[cfg.heat_start + 0].max() == 0.0


def test_limb_channel_response(cfg, hm):
    joints = _one_person(cfg, {"neck": (40.0, 64.0), "nose": (90.0, 64.0)})
    maps = hm.create_heatmaps(joints, np.zeros(cfg.mask_shape, np.float32))
    # limb 0 is neck->nose
    m = maps[cfg.paf_start + 0]
    # on the segment midline the perpendicular distance is ~0 -> response ~1
    row = int(round((64.0 + cfg.stride / 2 - 0.5) / cfg.stride))
    assert m[16, 16] > 0.9  # cell (16,16) centre (65.5, 65.5), 1.5 px off the line
    # response decays with perpendicular distance
    assert m[22, 16] < m[17, 16]
    # both keypoint channels present too
    assert maps[cfg.heat_start + cfg.parts_dict["neck"]].max() > 0.9


def test_limb_count_averaging(cfg, hm):
    # two identical limbs (two people, same joints) must average, not sum
    joints = np.zeros((2, cfg.num_parts, 3), np.float32)
    joints[:, :, 2] = 3
    for p in range(2):
        joints[p, cfg.parts_dict["neck"]] = [40.0, 64.0, 1]
        joints[p, cfg.parts_dict["nose"]] = [90.0, 64.0, 1]
    maps2 = hm.create_heatmaps(joints, np.zeros(cfg.mask_shape, np.float32))
    maps1 = hm.create_heatmaps(joints[:1], np.zeros(cfg.mask_shape, np.float32))
    np.testing.assert_allclose(maps2[cfg.paf_start], maps1[cfg.paf_start], atol=1e-6)


def test_background_channels(cfg, hm):
    joints = _one_person(cfg, {"nose": (64.0, 64.0)})
    mask_all = np.zeros(cfg.mask_shape, np.float32)
    mask_all[10:20, 10:20] = 1.0
    maps = hm.create_heatmaps(joints, mask_all)
    # channel bkg_start = eroded mask_all
    bkg = maps[cfg.bkg_start]
    assert bkg[15, 15] == 1.0
    assert bkg[10, 10] == 0.0      # eroded border
    assert bkg[5, 5] == 0.0
    # channel bkg_start+1 = max over keypoint channels
    rev = maps[cfg.bkg_start + 1]
    kp = maps[cfg.heat_start:cfg.heat_start + cfg.heat_layers].max(axis=0)
    np.testing.assert_allclose(rev, kp, atol=1e-6)


def test_output_range_and_shape(cfg, hm):
    joints = _one_person(cfg, {"nose": (64.0, 64.0), "neck": (64.0, 80.0),
                               "Rsho": (50.0, 80.0)})
    maps = hm.create_heatmaps(joints, np.ones(cfg.mask_shape, np.float32))
    assert maps.shape == (cfg.num_layers,) + cfg.mask_shape
    assert maps.min() >= 0.0 and maps.max() <= 1.0
    assert maps.dtype == np.float32


def test_limb_gaussian_floor():
    # reference distances() writes 0.01 where the response <= threshold
    X, Y = np.meshgrid(np.arange(0, 100, 4.0), np.arange(0, 100, 4.0))
    g = limb_gaussian(X, Y, 7.0, 10, 10, 30, 10, thresh=0.015)
    assert g.min() == pytest.approx(0.01)
    assert g.max() <= 1.0


def test_offset_maps(cfg, hm):
    joints = _one_person(cfg, {"nose": (64.0, 64.0)})
    off, mask = hm.put_offset(joints)
    assert off.shape == (2,) + cfg.mask_shape
    assert mask.shape == (2,) + cfg.mask_shape
    assert mask.max() == 1.0
    nz = mask[0] > 0
    assert np.abs(off[0][nz]).max() <= 1.0


@pytest.mark.gpu
def test_device_gt_matches_oracle():
    """HIP heatmap_gt kernel vs the numpy oracle on random skeletons."""
    import torch
    from improved_body_parts_amd.data import sample_people, create_heatmaps_device
    from improved_body_parts_amd.config import GetConfig

    config = GetConfig("Canonical")
    hm = Heatmapper(config)
    rng = np.random.default_rng(7)
    h, w = config.height // config.stride, config.width // config.stride
    N, P = 4, 5
    joints = np.full((N, P, config.num_parts, 3), 2.0, dtype=np.float32)
    masks = np.ones((N, h, w), dtype=np.float32)
    want = []
    for n in range(N):
        people = sample_people(rng, config.width, config.height, max_people=P)
        joints[n, :len(people)] = people
        masks[n, rng.integers(0, h // 2):rng.integers(h // 2, h),
              rng.integers(0, w // 2):rng.integers(w // 2, w)] = 0.0
        want.append(hm.create_heatmaps(people, masks[n]))
    want = np.stack(want)
    got = create_heatmaps_device(joints, masks, config).cpu().numpy()
    assert got.shape == want.shape == (N, config.num_layers, h, w)
    np.testing.assert_allclose(got, want, atol=2e-5, rtol=1e-4)


@pytest.mark.gpu
def test_device_gt_loader_stream():
    import torch
    from improved_body_parts_amd.config import GetConfig
    from improved_body_parts_amd.data import DeviceGTSyntheticLoader

    config = GetConfig("Canonical")
    loader = DeviceGTSyntheticLoader(config, batch_size=3, steps_per_epoch=2,
                                     dtype=torch.bfloat16)
    batches = list(loader)
    assert len(batches) == 2
    img, mm, hm = batches[0]
    assert img.shape == (3, 512, 512, 3) and img.is_cuda
    assert hm.shape == (3, config.num_layers, 128, 128)
    assert mm.shape == (3, 1, 128, 128)
    assert torch.isfinite(hm.float()).all()
    assert float(hm.float().max()) <= 1.0 + 1e-3


@pytest.mark.gpu
def test_device_gt_stress_config_768():
    """Device GT generator at the 768^2 stress-config geometry vs oracle."""
    import torch
    from improved_body_parts_amd.config import GetConfig
    from improved_body_parts_amd.data import sample_people, create_heatmaps_device

    config = GetConfig("Canonical768")
    hm = Heatmapper(config)
    rng = np.random.default_rng(3)
    h = w = 768 // 4
    people = sample_people(rng, 768, 768, max_people=3)
    joints = np.full((1, 3, config.num_parts, 3), 2.0, dtype=np.float32)
    joints[0, :len(people)] = people
    mask = np.ones((1, h, w), dtype=np.float32)
    want = hm.create_heatmaps(people, mask[0])
    got = create_heatmaps_device(joints, mask, config).cpu().numpy()[0]
    np.testing.assert_allclose(got, want, atol=2e-5, rtol=1e-4)
