import numpy as np
import pytest
import torch

from improved_body_parts_amd.config import CanonicalConfig
from improved_body_parts_amd.data import (
    AugmentSelection, SyntheticPoseDataset, Transformer, sample_people,
)


@pytest.fixture(scope="module")
def cfg():
    return CanonicalConfig(128, 128, 4)


def test_synthetic_dataset_contract(cfg):
    ds = SyntheticPoseDataset(cfg, length=8, seed=3)
    img, mm, hm = ds[0]
    assert img.shape == (128, 128, 3) and img.dtype == torch.float32
    assert mm.shape == (1, 32, 32)
    assert hm.shape == (50, 32, 32)
    assert 0.0 <= float(img.min()) and float(img.max()) <= 1.0
    # determinism per index
    img2, mm2, hm2 = ds[0]
    assert torch.equal(img, img2) and torch.equal(hm, hm2)
    # different index -> different sample
    img3, _, _ = ds[1]
    assert not torch.equal(img, img3)


def test_synthetic_people_plausible(cfg):
    rng = np.random.default_rng(0)
    ppl = sample_people(rng, 128, 128, max_people=3)
    assert ppl.ndim == 3 and ppl.shape[1:] == (18, 3)
    assert set(np.unique(ppl[:, :, 2])) <= {1.0, 2.0}


def test_transform_identity_keeps_center_joint(cfg):
    tr = Transformer(cfg)
    img = np.random.rand(200, 160, 3).astype(np.float32)
    joints = np.zeros((1, 18, 3), np.float32)
    joints[0, :, 2] = 3
    joints[0, 0] = [80, 100, 1]   # at objpos
    meta = {"objpos": [80, 100], "scale_provided": cfg.transform_params.target_dist,
            "joints": joints}
    aug = AugmentSelection.unrandom()
    img_t, mm, ma, meta_t = tr.transform(img, np.ones((200, 160)), np.ones((200, 160)),
                                         meta, aug=aug)
    assert img_t.shape == (128, 128, 3)
    assert mm.shape == (32, 32) and ma.shape == (32, 32)
    # the main-person centre lands on the crop centre
    np.testing.assert_allclose(meta_t["joints"][0, 0, :2], [64, 64], atol=1e-3)


def test_transform_flip_swaps_left_right(cfg):
    tr = Transformer(cfg)
    img = np.random.rand(128, 128, 3).astype(np.float32)
    joints = np.zeros((1, 18, 3), np.float32)
    joints[0, :, 2] = 3
    rsho, lsho = cfg.parts_dict["Rsho"], cfg.parts_dict["Lsho"]
    joints[0, rsho] = [40, 64, 1]
    joints[0, lsho] = [88, 64, 1]
    meta = {"objpos": [64, 64], "scale_provided": cfg.transform_params.target_dist,
            "joints": joints}
    aug = AugmentSelection(flip=True, degree=0.0, crop=(0, 0), scale=1.0)
    _, _, _, meta_t = tr.transform(img, np.ones((128, 128)), np.ones((128, 128)),
                                   meta, aug=aug)
    out = meta_t["joints"]
    # after mirroring x and swapping ids, Rsho should still be on the right-ish
    assert out[0, rsho, 0] == pytest.approx(128 - 88, abs=1e-3)
    assert out[0, lsho, 0] == pytest.approx(128 - 40, abs=1e-3)


def test_transform_rotation_preserves_distances(cfg):
    tr = Transformer(cfg)
    joints = np.zeros((1, 18, 3), np.float32)
    joints[0, :, 2] = 3
    joints[0, 0] = [64, 64, 1]
    joints[0, 1] = [64, 84, 1]
    meta = {"objpos": [64, 64], "scale_provided": cfg.transform_params.target_dist,
            "joints": joints}
    aug = AugmentSelection(flip=False, degree=30.0, crop=(0, 0), scale=1.0)
    _, _, _, meta_t = tr.transform(np.zeros((128, 128, 3), np.float32),
                                   np.ones((128, 128)), np.ones((128, 128)),
                                   meta, aug=aug)
    d = np.linalg.norm(meta_t["joints"][0, 0, :2] - meta_t["joints"][0, 1, :2])
    assert d == pytest.approx(20.0, rel=1e-3)


def test_dataloader_integration(cfg):
    ds = SyntheticPoseDataset(cfg, length=6)
    loader = torch.utils.data.DataLoader(ds, batch_size=3, num_workers=0)
    img, mm, hm = next(iter(loader))
    assert img.shape == (3, 128, 128, 3)
    assert hm.shape == (3, 50, 32, 32)


def test_transform_image_joint_alignment(cfg):
    """A bright delta painted at a joint must land at the transformed joint
    coordinate after the SAME affine warps the image (full random aug)."""
    import numpy as np
    from improved_body_parts_amd.data import Transformer, AugmentSelection
    import random as _random
    rng = _random.Random(5)
    t = Transformer(cfg)
    for trial in range(5):
        jx, jy = rng.uniform(100, 400), rng.uniform(100, 400)
        img = np.zeros((512, 512, 3), np.float32)
        y0, x0 = int(jy), int(jx)
        img[y0 - 2:y0 + 3, x0 - 2:x0 + 3] = 1.0
        joints = np.zeros((1, cfg.num_parts, 3), np.float32)
        joints[:, :, 2] = 2.0
        joints[0, 0] = (jx, jy, 1.0)
        meta = {"joints": joints.copy(), "objpos": [jx, jy],
                "scale_provided": 150.0 / 368}
        mask = np.ones((512, 512), np.float32)
        aug = AugmentSelection.random(cfg.transform_params, rng=rng)
        out_img, _, _, out_meta = t.transform(img, mask.copy(), mask.copy(),
                                              dict(meta), aug=aug)
        nx, ny = out_meta["joints"][0, 0, :2]
        if not (4 <= nx < cfg.width - 4 and 4 <= ny < cfg.height - 4):
            continue  # joint warped out of frame — nothing to check
        window = out_img[int(ny) - 4:int(ny) + 5, int(nx) - 4:int(nx) + 5]
        assert window.max() > 0.2, \
            f"trial {trial}: no bright pixel near transformed joint ({nx},{ny})"
