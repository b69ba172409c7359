"""Direct numerical parity against the UPSTREAM reference code.

These tests import the reference repo itself (mounted read-only at
/root/reference in the build environment) and compare our implementations
against it on random data — the strongest form of behavioral parity checking.
They skip automatically where the reference is not mounted (GPU boxes, CI).
"""
import os
import sys

import numpy as np
import pytest
import torch

REF = "/root/reference"
pytestmark = [
    pytest.mark.skipif(not os.path.isdir(REF),
                       reason="reference repo not mounted"),
    # the reference triggers scipy/torch deprecation warnings on import
    pytest.mark.filterwarnings("ignore::DeprecationWarning"),
    pytest.mark.filterwarnings("ignore::UserWarning"),
]

if os.path.isdir(REF):
    sys.path.insert(0, REF)


def _stub_ref_modules():
    """Stub the reference's unused heavy imports (cv2/torchvision absent here)."""
    import types
    if "cv2" not in sys.modules:
        from improved_body_parts_amd.data.heatmapper import _erode3x3
        fake = types.ModuleType("cv2")
        fake.erode = lambda m, kernel: _erode3x3(np.asarray(m, np.float32))
        sys.modules["cv2"] = fake
    if "torchvision" not in sys.modules:
        sys.modules["torchvision"] = types.ModuleType("torchvision")


@pytest.fixture(scope="module")
def setups():
    from improved_body_parts_amd.config import GetConfig, TrainingOpt
    config = GetConfig("Canonical")
    opt = TrainingOpt(nstack=2, batch_size=2, nstack_weight=[1, 1])
    return config, opt


def test_config_tables_match_reference():
    """Channel layout, limb table, flip orders vs reference config/config.py."""
    from config.config import GetConfig as RefGetConfig
    from improved_body_parts_amd.config import GetConfig
    ref = RefGetConfig("Canonical")
    ours = GetConfig("Canonical")
    assert ours.num_layers == ref.num_layers == 50
    assert ours.paf_layers == ref.paf_layers
    assert ours.limb_from == ref.limb_from
    assert ours.limb_to == ref.limb_to
    assert ours.heat_start == ref.heat_start
    assert ours.bkg_start == ref.bkg_start
    np.testing.assert_array_equal(np.asarray(ours.flip_heat_ord),
                                  np.asarray(ref.flip_heat_ord))
    np.testing.assert_array_equal(np.asarray(ours.flip_paf_ord),
                                  np.asarray(ref.flip_paf_ord))
    assert ours.dt_gt_mapping == ref.dt_gt_mapping


def test_focal_l2_loss_matches_reference(setups):
    """MultiTaskLoss forward (focal L2, 5 scales, task weights) vs the
    reference models/loss_model.py on identical random predictions/targets."""
    from models.loss_model import MultiTaskLoss as RefLoss
    from improved_body_parts_amd.models import MultiTaskLoss
    config, opt = setups
    torch.manual_seed(0)
    n, c = 2, config.num_layers
    pred = [[torch.randn(n, c, 128 // (2 ** s), 128 // (2 ** s)) * 0.1
             for s in range(5)] for _ in range(opt.nstack)]
    mask = (torch.rand(n, 1, 128, 128) > 0.2).float()
    gt = torch.rand(n, c, 128, 128)
    ours = MultiTaskLoss(opt, config)(
        [[t.clone().requires_grad_(False) for t in st] for st in pred],
        (mask, gt))
    ref = RefLoss(opt, config)(pred, (mask, gt))
    assert torch.allclose(ours.float(), ref.float(), rtol=1e-4), \
        f"ours {float(ours)} ref {float(ref)}"


def test_heatmapper_matches_reference(setups):
    """GT heatmap generation vs reference py_cocodata_server/py_data_heatmapper.py
    on the same skeletons. The reference only needs cv2.erode — stub it so the
    comparison runs in this cv2-free image."""
    _stub_ref_modules()
    from config.config import GetConfig as RefGetConfig
    from py_cocodata_server.py_data_heatmapper import Heatmapper as RefHeatmapper
    from improved_body_parts_amd.config import GetConfig
    from improved_body_parts_amd.data import Heatmapper, sample_people
    ref_cfg = RefGetConfig("Canonical")
    cfg = GetConfig("Canonical")
    rng = np.random.default_rng(5)
    people = sample_people(rng, cfg.width, cfg.height, max_people=3)
    mask_all = np.ones(cfg.mask_shape, dtype=np.float32)
    ours = Heatmapper(cfg).create_heatmaps(people.copy(), mask_all)
    theirs = RefHeatmapper(ref_cfg).create_heatmaps(
        people.astype(float).copy(), mask_all)
    # reference returns HWC pre-transpose? its create_heatmaps returns CHW too
    theirs = np.asarray(theirs, dtype=np.float32)
    if theirs.shape != ours.shape and theirs.shape[-1] == ours.shape[0]:
        theirs = theirs.transpose(2, 0, 1)
    assert ours.shape == theirs.shape
    np.testing.assert_allclose(ours, theirs, atol=2e-3, rtol=1e-3)


def test_util_pad_matches_reference():
    _stub_ref_modules()
    from utils import util as ref_util
    from improved_body_parts_amd.utils import padRightDownCorner
    img = np.random.RandomState(0).randint(0, 255, (37, 53, 3)).astype(np.uint8)
    ours, pad_o = padRightDownCorner(img, 64, 128)
    theirs, pad_r = ref_util.padRightDownCorner(img, 64, 128)
    np.testing.assert_array_equal(ours, theirs)
    assert list(pad_o) == list(pad_r)


def test_refine_centroid_matches_reference():
    _stub_ref_modules()
    from utils import util as ref_util
    from improved_body_parts_amd.utils import refine_centroid
    rs = np.random.RandomState(1)
    fmap = rs.rand(40, 40).astype(np.float32)
    for anchor in [(5, 7), (0, 0), (39, 39), (20, 20)]:
        ours = refine_centroid(fmap, anchor, 2)
        theirs = ref_util.refine_centroid(fmap, anchor, 2)
        # the reference's meshgrid axes are swapped (acknowledged no-op in its
        # own comment); compare score and the distance of refined points
        assert abs(ours[2] - theirs[2]) < 1e-6
        assert abs(ours[0] - theirs[0]) < 0.51 and abs(ours[1] - theirs[1]) < 0.51


def _import_reference_evaluate():
    """Import reference evaluate.py with its unused heavy deps stubbed."""
    import types
    _stub_ref_modules()
    tv = sys.modules["torchvision"]
    tv.__path__ = []
    tvm = types.ModuleType("torchvision.models")
    tvm.densenet = types.ModuleType("densenet")
    tv.models = tvm
    sys.modules["torchvision.models"] = tvm
    for name in ["tqdm", "matplotlib", "matplotlib.pyplot", "pycocotools",
                 "pycocotools.coco", "pycocotools.cocoeval", "configobj"]:
        sys.modules.setdefault(name, types.ModuleType(name))
    sys.modules["pycocotools.coco"].COCO = object
    sys.modules["pycocotools.cocoeval"].COCOeval = object
    sys.modules["tqdm"].tqdm = lambda x: x

    class _ConfigObj(dict):
        def __init__(self, path):
            super().__init__()
            section = None
            for line in open(path):
                line = line.split("#")[0].strip()
                if not line:
                    continue
                if line.startswith("["):
                    section = line.strip("[]")
                    self[section] = {}
                    continue
                if "=" in line and section:
                    k, v = (s.strip() for s in line.split("=", 1))
                    self[section][k] = v
    sys.modules["configobj"].ConfigObj = _ConfigObj
    argv = sys.argv
    sys.argv = ["evaluate.py"]
    try:
        import evaluate as ref_eval
    finally:
        sys.argv = argv
    return ref_eval


def _synth_scene(config, people, H=256, W=256, sigma=3.0):
    n_heat = config.num_layers - config.paf_layers
    heat = np.zeros((H, W, n_heat), dtype=np.float32)
    paf = np.zeros((H, W, config.paf_layers), dtype=np.float32)
    yy, xx = np.mgrid[0:H, 0:W].astype(np.float32)
    for person in people:
        for part, (x, y) in person.items():
            g = np.exp(-((xx - x) ** 2 + (yy - y) ** 2) / (2 * sigma ** 2))
            heat[..., part] = np.maximum(heat[..., part], g)
        for k, (a, b) in enumerate(config.limbs_conn):
            if a in person and b in person:
                ax, ay = person[a]
                bx, by = person[b]
                vx, vy = bx - ax, by - ay
                L2 = vx * vx + vy * vy + 1e-9
                t = np.clip(((xx - ax) * vx + (yy - ay) * vy) / L2, 0, 1)
                d2 = (xx - (ax + t * vx)) ** 2 + (yy - (ay + t * vy)) ** 2
                paf[..., k] = np.maximum(paf[..., k],
                                         np.exp(-d2 / (2 * sigma ** 2)))
    return heat, paf


def test_grouping_matches_reference_evaluate():
    """find_connections + find_people vs the reference's own implementations
    (evaluate.py:206-498) on the same peaks and PAF maps. All limbs in the
    scene are >= 20 px so both samplers use the full mid_num=20 grid and the
    comparison is exact."""
    ref_eval = _import_reference_evaluate()
    from improved_body_parts_amd.config import GetConfig, InferenceParams
    from improved_body_parts_amd.engine.inference import (find_connections,
                                                          find_peaks,
                                                          find_people)
    config = GetConfig("Canonical")
    params, _ = InferenceParams().as_params_dict()
    params["remove_recon"] = 0
    people = [
        {0: (60.0, 40.0), 1: (60.0, 70.0), 2: (35.0, 72.0), 3: (30.0, 112.0),
         5: (85.0, 72.0), 6: (90.0, 112.0), 8: (45.0, 135.0), 11: (75.0, 135.0)},
        {0: (180.0, 50.0), 1: (180.0, 80.0), 2: (155.0, 82.0), 3: (150.0, 122.0),
         5: (205.0, 82.0), 6: (210.0, 122.0), 8: (165.0, 145.0), 11: (195.0, 145.0)},
    ]
    heat, paf = _synth_scene(config, people)
    all_peaks = find_peaks(torch.from_numpy(heat), params, config)

    ours_conn, ours_special = find_connections(
        all_peaks, torch.from_numpy(paf), 256, params, config)
    ref_conn, ref_special = ref_eval.find_connections(all_peaks, paf, 256,
                                                      dict(params))
    assert ours_special == ref_special
    for k in range(len(ref_conn)):
        r = np.asarray(ref_conn[k], dtype=np.float64).reshape(-1, 6)
        o = np.asarray(ours_conn[k], dtype=np.float64).reshape(-1, 6)
        assert o.shape == r.shape, f"limb {k}: {o.shape} vs {r.shape}"
        if len(r):
            # same (idA, idB) pairs with same scores/lengths
            o = o[np.lexsort((o[:, 1], o[:, 0]))]
            r = r[np.lexsort((r[:, 1], r[:, 0]))]
            np.testing.assert_array_equal(o[:, :2], r[:, :2])
            np.testing.assert_allclose(o[:, 2], r[:, 2], rtol=1e-5)
            np.testing.assert_allclose(o[:, 5], r[:, 5], rtol=1e-5)

    ours_subset, ours_cand = find_people(ours_conn, ours_special, all_peaks,
                                         params, config)
    ref_subset, ref_cand = ref_eval.find_people(ref_conn, ref_special,
                                                all_peaks, dict(params))
    assert len(ours_subset) == len(ref_subset) == 2
    np.testing.assert_allclose(np.asarray(ours_cand), np.asarray(ref_cand))
    o = np.asarray(ours_subset)
    r = np.asarray(ref_subset)
    # row order can differ: sort by first present part id
    o = o[np.argsort(o[:, :, 0].max(axis=1))]
    r = r[np.argsort(r[:, :, 0].max(axis=1))]
    np.testing.assert_allclose(o[:, :18, 0], r[:, :18, 0])   # part assignment
    np.testing.assert_allclose(o[:, -1, 0], r[:, -1, 0])     # part counts
    np.testing.assert_allclose(o[:, -2, 0], r[:, -2, 0], rtol=1e-5)  # scores


@pytest.mark.parametrize("seed", [123, 77, 2024])
@pytest.mark.parametrize("remove_recon", [0, 1])
def test_grouping_matches_reference_crowded(remove_recon, seed):
    """Crowded random scenes (overlapping people, short limbs, missing parts)
    through both greedy-assembly implementations, including the competition-
    resolution branch (remove_recon=1)."""
    ref_eval = _import_reference_evaluate()
    from improved_body_parts_amd.config import GetConfig, InferenceParams
    from improved_body_parts_amd.data import sample_people
    from improved_body_parts_amd.engine.inference import (find_connections,
                                                          find_peaks,
                                                          find_people)
    config = GetConfig("Canonical")
    params, _ = InferenceParams().as_params_dict()
    params["remove_recon"] = remove_recon
    rng = np.random.default_rng(seed + remove_recon)
    H = W = 320
    ppl = sample_people(rng, W, H, max_people=5)
    people = []
    for p in ppl:
        d = {j: (float(p[j, 0]), float(p[j, 1])) for j in range(18)
             if p[j, 2] < 2 and 0 <= p[j, 0] < W and 0 <= p[j, 1] < H}
        if d:
            people.append(d)
    heat, paf = _synth_scene(config, people, H=H, W=W)
    all_peaks = find_peaks(torch.from_numpy(heat), params, config)
    o_conn, o_sp = find_connections(all_peaks, torch.from_numpy(paf), H,
                                    params, config)
    r_conn, r_sp = ref_eval.find_connections(all_peaks, paf, H, dict(params))
    assert o_sp == r_sp
    for k in range(len(r_conn)):
        r = np.asarray(r_conn[k], dtype=np.float64).reshape(-1, 6)
        o = np.asarray(o_conn[k], dtype=np.float64).reshape(-1, 6)
        assert o.shape == r.shape, f"limb {k}: {o.shape} vs {r.shape}"
        if len(r):
            o = o[np.lexsort((o[:, 1], o[:, 0]))]
            r = r[np.lexsort((r[:, 1], r[:, 0]))]
            np.testing.assert_array_equal(o[:, :2], r[:, :2])
            np.testing.assert_allclose(o[:, 2], r[:, 2], rtol=1e-5)
    o_sub, o_cand = find_people(o_conn, o_sp, all_peaks, params, config)
    r_sub, r_cand = ref_eval.find_people(r_conn, r_sp, all_peaks, dict(params))
    assert len(o_sub) == len(r_sub)
    if len(r_sub):
        o = np.asarray(o_sub)
        r = np.asarray(r_sub)
        o = o[np.argsort(o[:, :18, 0].max(axis=1))]
        r = r[np.argsort(r[:, :18, 0].max(axis=1))]
        np.testing.assert_allclose(o[:, :18, 0], r[:, :18, 0])
        np.testing.assert_allclose(o[:, -2, 0], r[:, -2, 0], rtol=1e-5)


def test_posenet_checkpoint_and_forward_match_reference():
    """Build the REFERENCE PoseNet (models/posenet.py) with random weights,
    load its state_dict into ours (strict), and compare full forward outputs
    on the same input — proves parameter-layout AND architecture parity."""
    _import_reference_evaluate()  # installs the stubs evaluate's deps need
    from models.posenet import PoseNet as RefPoseNet
    from improved_body_parts_amd.models import PoseNet
    torch.manual_seed(0)
    # inp_dim must be 256 here: the reference Backbone hard-codes its 64/128
    # channel plan (layers_transposed.py:167-180, nFeat unused), so any other
    # width breaks the reference itself; ours scales the plan with nFeat.
    ref = RefPoseNet(2, 256, 50, bn=True, increase=32)
    ours = PoseNet(2, 256, 50, bn=True, increase=32)
    missing, unexpected = ours.load_state_dict(ref.state_dict(), strict=False)
    assert not missing, f"missing keys: {missing[:8]}"
    assert not unexpected, f"unexpected keys: {unexpected[:8]}"
    ref.eval()
    ours.eval()
    x = torch.rand(1, 64, 64, 3)
    with torch.no_grad():
        out_ref = ref(x)
        out_ours = ours(x)
    assert len(out_ref) == len(out_ours) == 2
    for s in range(5):
        a, b = out_ours[1][s], out_ref[1][s]
        assert a.shape == b.shape
        rel = float((a - b).norm() / (b.norm() + 1e-12))
        assert rel < 1e-5, f"scale {s}: rel {rel}"


def test_parallel_loss_matches_reference(setups):
    """MultiTaskLossParallel (plain-L2 default, no batch division) vs the
    reference models/loss_model_parallel.py."""
    _stub_ref_modules()
    from models.loss_model_parallel import MultiTaskLossParallel as RefLoss
    from improved_body_parts_amd.models import MultiTaskLossParallel
    config, opt = setups
    torch.manual_seed(3)
    n, c = 2, config.num_layers
    pred = [[torch.randn(n, c, 128 // (2 ** s), 128 // (2 ** s)) * 0.1
             for s in range(5)] for _ in range(opt.nstack)]
    mask = (torch.rand(n, 1, 128, 128) > 0.2).float()
    gt = torch.rand(n, c, 128, 128)
    ours = MultiTaskLossParallel(opt, config)(pred, (mask, gt))
    ref = RefLoss(opt, config)(pred, (mask, gt))
    assert torch.allclose(ours.float(), ref.float(), rtol=1e-4), \
        f"ours {float(ours)} ref {float(ref)}"


def test_coco_source_convert_matches_reference():
    """COCO joint-order -> canonical conversion incl. neck synthesis and
    visibility flags vs reference config/config.py::COCOSourceConfig.convert."""
    from config.config import (COCOSourceConfig as RefSrc,
                               GetConfig as RefGetConfig)
    from improved_body_parts_amd.config import COCOSourceConfig, GetConfig
    ref_cfg = RefGetConfig("Canonical")
    cfg = GetConfig("Canonical")
    rs = np.random.RandomState(4)
    joints = rs.rand(5, 17, 3).astype(np.float64) * 100
    joints[:, :, 2] = rs.randint(0, 4, (5, 17))
    meta = {"joints": joints.copy(), "image": "x.jpg"}
    ours = COCOSourceConfig("d.h5").convert(dict(meta), cfg)
    theirs = RefSrc("d.h5").convert({"joints": joints.copy(), "image": "x.jpg"},
                                    ref_cfg)
    np.testing.assert_allclose(np.asarray(ours["joints"]),
                               np.asarray(theirs["joints"]))


@pytest.mark.parametrize("refmod,ourname", [
    ("models.posenet_final", "PoseNetFinal"),
    ("models.posenet2", "PoseNetAttention"),
    ("models.posenet3", "PoseNetLight"),
    ("models.posenet_independent", "PoseNetIndependent"),
])
def test_variant_checkpoint_and_forward_match_reference(refmod, ourname):
    """Every model-variant family loads the corresponding reference variant's
    state_dict STRICT and reproduces its forward outputs."""
    import importlib
    _import_reference_evaluate()
    import improved_body_parts_amd.models as M
    rm = importlib.import_module(refmod)
    torch.manual_seed(0)
    ref = rm.PoseNet(2, 256, 50, bn=True, increase=32)
    ours = getattr(M, ourname)(2, 256, 50, bn=True, increase=32)
    ours.load_state_dict(ref.state_dict())  # strict
    ref.eval()
    ours.eval()
    x = torch.rand(1, 64, 64, 3)
    with torch.no_grad():
        a = ref(x)
        b = ours(x)
    for s in range(5):
        rel = float((b[1][s] - a[1][s]).norm() / (a[1][s].norm() + 1e-12))
        assert rel < 1e-5, f"{ourname} scale {s}: rel {rel}"


def test_ae_variant_checkpoint_and_forward_match_reference():
    """AEPoseNet loads the reference ae_pose state_dict strict and reproduces
    its forward (single-scale [nstack][1] output nesting included)."""
    import importlib
    _import_reference_evaluate()
    import improved_body_parts_amd.models as M
    rm = importlib.import_module("models.ae_pose")
    torch.manual_seed(0)
    ref = rm.PoseNet(2, 256, 50, bn=True)
    ours = M.AEPoseNet(2, 256, 50, bn=True)
    ours.load_state_dict(ref.state_dict())
    ref.eval()
    ours.eval()
    x = torch.rand(1, 64, 64, 3)
    with torch.no_grad():
        a = ref(x)
        b = ours(x)
    for i in range(2):
        rel = float((b[i][0] - a[i][0]).norm() / (a[i][0].norm() + 1e-12))
        assert rel < 1e-5, f"stack {i}: {rel}"


def test_our_checkpoint_loads_into_reference():
    """Inverse interop direction: a checkpoint written by OUR trainer loads
    strict into the reference PoseNet."""
    import tempfile
    _import_reference_evaluate()
    from models.posenet import PoseNet as RefPoseNet
    from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
    from improved_body_parts_amd.engine import save_checkpoint
    from improved_body_parts_amd.engine.optimizer import FusedSGD
    from improved_body_parts_amd.models import Network
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=256, batch_size=1,
                      nstack_weight=[1, 1])
    net = Network(opt, cfg, bn=True, dist=True)
    sgd = FusedSGD(net.parameters(), lr=1e-5)
    with tempfile.TemporaryDirectory() as d:
        path = save_checkpoint(net, sgd, 1.23, 7, d)
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
    assert set(ckpt) == {"weights", "optimizer_weight", "train_loss", "epoch"}
    ref = RefPoseNet(2, 256, 50, bn=True, increase=128)
    # our Network prefixes the model under 'posenet.'
    weights = {k[len("posenet."):]: v for k, v in ckpt["weights"].items()
               if k.startswith("posenet.")}
    ref.load_state_dict(weights)  # strict


def test_offset_maps_reference_path_is_broken_upstream():
    """The reference's put_offset cannot run under its own Canonical config:
    config.offset_layers is 2 (shared offsets) while put_offset asserts a
    2*num_parts depth (py_data_heatmapper.py:290) — dead code upstream
    (disabled in the default pipeline, py_data_iterator.py:64). Our
    implementation keeps the documented shared-offset semantics and is
    covered by tests/test_heatmapper.py::test_offset_maps; this test pins the
    upstream defect so the divergence is deliberate."""
    _stub_ref_modules()
    from config.config import GetConfig as RefGetConfig
    from py_cocodata_server.py_data_heatmapper import Heatmapper as RefHeatmapper
    ref_hm = RefHeatmapper(RefGetConfig("Canonical"))
    joints = np.zeros((1, 18, 3), np.float32)
    with pytest.raises(AssertionError):
        ref_hm.put_offset(joints)
