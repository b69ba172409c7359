import re

import pytest
import torch

from improved_body_parts_amd.models import (
    Network, NetworkEval, PoseNet, SELayer, Residual, Backbone, Hourglass,
)


def test_posenet_output_structure(small_config, small_opt):
    net = PoseNet(small_opt.nstack, small_opt.hourglass_inp_dim,
                  small_config.num_layers, bn=True, increase=small_opt.increase)
    x = torch.rand(2, 128, 128, 3)
    out = net(x)
    assert len(out) == small_opt.nstack
    for stack in out:
        assert len(stack) == 5
        for s, t in enumerate(stack):
            assert t.shape == (2, 50, 32 // (2 ** s), 32 // (2 ** s))


def test_posenet_backward(small_config, small_opt):
    net = PoseNet(small_opt.nstack, small_opt.hourglass_inp_dim,
                  small_config.num_layers, bn=True, increase=small_opt.increase)
    out = net(torch.rand(1, 128, 128, 3))
    out[0][0].sum().backward()
    assert net.pre.conv1.weight.grad is not None


def test_state_dict_matches_reference_layout():
    """Reference checkpoint key layout (models/posenet.py + layers_transposed.py):
    verified once against the real reference model (1848 keys, 129,206,792
    elements at the default 4x256 config); here we pin the invariants."""
    net = PoseNet(4, 256, 50, bn=True, increase=128, init_weights=False)
    sd = net.state_dict()
    n = sum(v.numel() for v in sd.values())
    assert len(sd) == 1848
    assert n == 129206792
    for probe in [
        "pre.conv1.weight", "pre.res1.convBlock.0.weight",
        "pre.res1.skipConv.0.weight", "pre.dilation.0.conv.weight",
        "hourglass.0.hg.0.0.convBlock.3.weight", "hourglass.3.hg.3.4.convBlock.6.weight",
        "features.0.before_regress.0.0.conv.weight",
        "features.0.before_regress.0.2.fc.0.weight",
        "outs.0.0.conv.weight", "merge_preds.0.0.conv.conv.weight",
        "merge_features.2.4.conv.conv.weight",
    ]:
        assert probe in sd, probe
    assert sd["outs.0.0.conv.weight"].shape == (50, 256, 1, 1)
    assert sd["hourglass.0.hg.3.4.convBlock.0.weight"].shape == (384, 768, 1, 1)


def test_weight_init_statistics():
    net = PoseNet(1, 64, 50, bn=True, increase=32)
    w = net.pre.conv1.weight
    assert w.std().item() == pytest.approx(0.001, rel=0.3)
    for m in net.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            assert torch.all(m.weight == 1) and torch.all(m.bias == 0)
            break


def test_network_train_eval_modes(small_config, small_opt):
    net = Network(small_opt, small_config, bn=True, dist=True)
    img = torch.rand(1, 128, 128, 3)
    mm = torch.ones(1, 1, 32, 32)
    hm = torch.rand(1, 50, 32, 32)
    net.train()
    loss = net((img, mm, hm))
    assert loss.dim() == 0 and torch.isfinite(loss)
    net.eval()
    with torch.no_grad():
        outs, loss2 = net((img, mm, hm))
    assert len(outs) == small_opt.nstack and torch.isfinite(loss2)


def test_network_swa_mode(small_config, small_opt):
    net = Network(small_opt, small_config, bn=True, dist=True, swa=True)
    net.train()
    out = net((torch.rand(1, 128, 128, 3), torch.ones(1, 1, 32, 32),
               torch.rand(1, 50, 32, 32)))
    assert isinstance(out, list)  # swa mode returns raw outputs


def test_network_eval_wrapper(small_config, small_opt):
    net = NetworkEval(small_opt, small_config, bn=True)
    net.eval()
    out = net(torch.rand(1, 128, 128, 3))
    assert len(out) == small_opt.nstack
    net.train()
    with pytest.raises(ValueError):
        net(torch.rand(1, 128, 128, 3))


def test_selayer_shape():
    se = SELayer(64)
    x = torch.rand(2, 64, 8, 8)
    y = se(x)
    assert y.shape == x.shape
    y.sum().backward()


def test_hourglass_scales():
    hg = Hourglass(4, 64, increase=16, bn=True)
    outs = hg(torch.rand(1, 64, 32, 32))
    assert [o.shape[-1] for o in outs] == [32, 16, 8, 4, 2]
    # scale s carries nFeat + s*increase channels (consumed by Features'
    # Conv(inp_dim + i*increase, inp_dim) — reference posenet.py:31-36)
    assert [o.shape[1] for o in outs] == [64, 80, 96, 112, 128]


def test_residual_channel_change():
    r = Residual(32, 64, bn=True)
    y = r(torch.rand(1, 32, 16, 16))
    assert y.shape == (1, 64, 16, 16)
    r2 = Residual(64, 64, bn=True)
    assert not hasattr(r2, "skipConv")


@pytest.mark.parametrize("name", ["final", "attention", "light", "independent"])
def test_variant_forward_backward(name):
    """Every 5-scale variant runs fwd+bwd on the CPU plumbing config
    (reference-variant smoke pattern, e.g. posenet_final.py:211-228)."""
    from improved_body_parts_amd.models import build_posenet
    torch.manual_seed(0)
    net = build_posenet(name, nstack=2, inp_dim=64, oup_dim=50, bn=True,
                        increase=32)
    out = net(torch.rand(1, 128, 128, 3))
    assert len(out) == 2 and len(out[0]) == 5
    for s, t in enumerate(out[0]):
        assert t.shape == (1, 50, 32 // (2 ** s), 32 // (2 ** s))
    out[0][0].sum().backward()
    grads = [p.grad for p in net.parameters() if p.grad is not None]
    assert len(grads) > 0 and all(torch.isfinite(g).all() for g in grads)


def test_ae_variant_forward_backward():
    from improved_body_parts_amd.models import build_posenet
    net = build_posenet("ae", nstack=2, inp_dim=64, oup_dim=50, bn=True,
                        increase=32)
    out = net(torch.rand(1, 128, 128, 3))
    assert len(out) == 2 and len(out[0]) == 1
    assert out[0][0].shape == (1, 50, 32, 32)
    out[-1][0].sum().backward()


def test_variant_registry_rejects_unknown():
    from improved_body_parts_amd.models import build_posenet
    with pytest.raises(ValueError):
        build_posenet("nope", 1, 64, 50)


def test_network_with_variant():
    """Network honours opt.model_variant (driver-level variant selection)."""
    from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
    from improved_body_parts_amd.models import Network
    from improved_body_parts_amd.models.variants import PoseNetLight
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=64, increase=32, batch_size=1,
                      nstack_weight=[1, 1], model_variant="light")
    net = Network(opt, cfg, bn=True, dist=True)
    assert isinstance(net.posenet, PoseNetLight)
    loss = net((torch.rand(1, 128, 128, 3), torch.ones(1, 1, 32, 32),
                torch.rand(1, 50, 32, 32)))
    assert torch.isfinite(loss)
