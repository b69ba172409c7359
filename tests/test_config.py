import numpy as np
import pytest

from improved_body_parts_amd.config import (
    CanonicalConfig, COCOSourceConfig, GetConfig, InferenceParams, TrainingOpt,
    config_reader,
)


def test_channel_layout():
    cfg = GetConfig("Canonical")
    assert cfg.num_parts == 18
    assert cfg.paf_layers == 30
    assert cfg.heat_layers == 18
    assert cfg.num_layers == 50
    assert cfg.paf_start == 0
    assert cfg.heat_start == 30
    assert cfg.bkg_start == 48
    assert cfg.stride == 4
    assert cfg.width == cfg.height == 512
    assert cfg.mask_shape == (128, 128)
    assert cfg.parts_shape == (128, 128, 50)


def test_flip_tables_match_reference():
    # hard-coded tables of reference config/config.py:121-124
    cfg = GetConfig("Canonical")
    ref_heat = np.array([0, 1, 5, 6, 7, 2, 3, 4, 11, 12, 13, 8, 9, 10, 15, 14, 17, 16, 18, 19])
    ref_paf = np.array([0, 2, 1, 4, 3, 6, 5, 8, 7, 12, 13, 14, 9, 10, 11, 18, 19, 20,
                        15, 16, 17, 22, 21, 25, 26, 23, 24, 28, 27, 29])
    np.testing.assert_array_equal(cfg.flip_heat_ord, ref_heat)
    np.testing.assert_array_equal(cfg.flip_paf_ord, ref_paf)


def test_limb_tables_match_reference():
    cfg = GetConfig("Canonical")
    assert cfg.limb_from == [1, 1, 1, 1, 1, 0, 0, 14, 15, 1, 2, 3, 1, 5, 6, 1, 8, 9,
                             1, 11, 12, 0, 0, 2, 8, 5, 11, 16, 17, 8]
    assert cfg.limb_to == [0, 14, 15, 16, 17, 14, 15, 16, 17, 2, 3, 4, 5, 6, 7, 8, 9,
                           10, 11, 12, 13, 2, 5, 8, 12, 11, 9, 2, 5, 11]


def test_config_variants():
    c384 = GetConfig("Canonical384")
    assert c384.width == 384 and c384.mask_shape == (96, 96)
    dense = GetConfig("DenseSkeleton")
    assert dense.paf_layers == 49
    assert dense.num_layers == 49 + 18 + 2
    # flip table still a permutation
    assert sorted(dense.flip_paf_ord.tolist()) == list(range(49))


def test_coco_adapter_neck_synthesis():
    cfg = GetConfig("Canonical")
    src = COCOSourceConfig("x.h5")
    joints = np.zeros((1, 17, 3), np.float32)
    joints[:, :, 2] = 2  # nothing marked
    joints[0, src.parts_dict["Rsho"]] = [10, 20, 1]
    joints[0, src.parts_dict["Lsho"]] = [30, 40, 0]
    meta = src.convert({"joints": joints}, cfg)
    out = meta["joints"]
    assert out.shape == (1, 18, 3)
    neck = out[0, cfg.parts_dict["neck"]]
    assert neck[0] == 20 and neck[1] == 30
    assert neck[2] == 0  # min visibility of the two shoulders
    # unmarked parts are 2/3
    assert out[0, cfg.parts_dict["nose"], 2] >= 2


def test_training_opt_overrides():
    opt = TrainingOpt(nstack=2, batch_size=8)
    assert opt.nstack == 2 and opt.batch_size == 8
    assert opt.nstack_weight == [1, 1]
    with pytest.raises(AttributeError):
        TrainingOpt(not_a_field=1)


def test_inference_params_reader(tmp_path):
    params, model_params = config_reader()
    assert params["thre1"] == pytest.approx(0.1)
    assert params["mid_num"] == 20
    assert model_params["stride"] == 4
    assert model_params["max_downsample"] == 64
    ini = tmp_path / "config"
    ini.write_text("[param]\nthre1 = 0.25\nscale_search = [0.5, 1.0, 1.5]\nboxsize = 320\n")
    params, model_params = config_reader(str(ini))
    assert params["thre1"] == pytest.approx(0.25)
    assert params["scale_search"] == [0.5, 1.0, 1.5]
    assert model_params["boxsize"] == 320


def test_slim384_variant():
    """24-limb 44-channel @384 preset (reference config2.py capability)."""
    c = GetConfig("Slim384")
    assert c.paf_layers == 24
    assert c.num_layers == 24 + 18 + 2
    assert (c.width, c.height) == (384, 384)
    assert len(c.flip_paf_ord) == 24


@pytest.mark.parametrize("name,side,paf", [("Canonical384", 384, 30),
                                           ("Canonical768", 768, 30),
                                           ("DenseSkeleton", 512, 49),
                                           ("Slim384", 384, 24)])
def test_config_variants_end_to_end_shapes(name, side, paf):
    """Every config preset drives the GT generator + loss at its own geometry."""
    import torch
    from improved_body_parts_amd.config import TrainingOpt
    from improved_body_parts_amd.data import Heatmapper, sample_people
    from improved_body_parts_amd.models import MultiTaskLoss
    cfg = GetConfig(name)
    assert (cfg.width, cfg.paf_layers) == (side, paf)
    grid = side // cfg.stride
    rng = np.random.default_rng(0)
    people = sample_people(rng, cfg.width, cfg.height, max_people=2)
    maps = Heatmapper(cfg).create_heatmaps(people, np.ones((grid, grid), np.float32))
    assert maps.shape == (cfg.num_layers, grid, grid)
    opt = TrainingOpt(nstack=1, batch_size=1, nstack_weight=[1])
    crit = MultiTaskLoss(opt, cfg)
    preds = [[torch.rand(1, cfg.num_layers, grid // 2 ** s, grid // 2 ** s) * 0.1
              for s in range(5)]]
    loss = crit(preds, (torch.ones(1, 1, grid, grid),
                        torch.from_numpy(maps)[None]))
    assert torch.isfinite(loss)
