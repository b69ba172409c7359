import os

import pytest
import torch

from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
from improved_body_parts_amd.data import SyntheticPoseDataset
from improved_body_parts_amd.engine import (
    FusedSGD, Trainer, SWATrainer, load_checkpoint, save_checkpoint,
)
from improved_body_parts_amd.models import Network


@pytest.fixture()
def tiny_setup(tmp_path):
    cfg = CanonicalConfig(64, 64, 4)
    opt = TrainingOpt(nstack=1, hourglass_inp_dim=32, increase=16, batch_size=2,
                      nstack_weight=[1])
    ds = SyntheticPoseDataset(cfg, length=4, seed=7)
    return cfg, opt, ds, tmp_path


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(0)
    w1 = torch.nn.Parameter(torch.randn(17, 5))
    w2 = torch.nn.Parameter(torch.randn(17, 5))
    with torch.no_grad():
        w2.copy_(w1)
    opt1 = FusedSGD([w1], lr=0.1, momentum=0.9, weight_decay=0.01)
    opt2 = torch.optim.SGD([w2], lr=0.1, momentum=0.9, weight_decay=0.01)
    for step in range(5):
        g = torch.randn(17, 5)
        for w, o in ((w1, opt1), (w2, opt2)):
            o.zero_grad()
            w.grad = g.clone()
            o.step()
    assert torch.allclose(w1, w2, atol=1e-6)


def test_checkpoint_roundtrip(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    net = Network(opt, cfg, bn=True, dist=True)
    sgd = FusedSGD(net.parameters(), lr=1e-3)
    path = save_checkpoint(net, sgd, 1.23, 7, directory=str(tmp))
    assert os.path.basename(path) == "PoseNet_7_epoch.pth"
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    assert set(ckpt.keys()) == {"weights", "optimizer_weight", "train_loss", "epoch"}
    assert not any(k.startswith("module.") for k in ckpt["weights"])

    net2 = Network(opt, cfg, bn=True, dist=True)
    epoch, loss = load_checkpoint(net2, path)
    assert epoch == 7 and loss == pytest.approx(1.23)
    for (k1, v1), (k2, v2) in zip(net.state_dict().items(), net2.state_dict().items()):
        assert k1 == k2
        assert torch.allclose(v1.float(), v2.float())


def test_checkpoint_accepts_module_prefix(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    net = Network(opt, cfg, bn=True, dist=True)
    state = {"weights": {f"module.{k}": v for k, v in net.state_dict().items()},
             "optimizer_weight": None, "train_loss": 0.5, "epoch": 3}
    p = tmp / "PoseNet_3_epoch.pth"
    torch.save(state, p)
    net2 = Network(opt, cfg, bn=True, dist=True)
    epoch, _ = load_checkpoint(net2, str(p))
    assert epoch == 3
    assert torch.allclose(net2.posenet.pre.conv1.weight, net.posenet.pre.conv1.weight)


def test_trainer_step_reduces_loss(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    torch.manual_seed(0)
    tr = Trainer(opt, cfg, ds, rank=0, world_size=1, num_workers=0,
                 checkpoint_dir=str(tmp), device=torch.device("cpu"))
    batch = next(iter(tr.train_loader))
    l0 = tr.train_step(batch)
    for g in tr.optimizer.param_groups:
        g["lr"] = 1e-3
    losses = [tr.train_step(batch) for _ in range(4)]
    assert losses[-1] < l0  # optimisation makes progress on a fixed batch


def test_trainer_loss_explosion_guard(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    opt.loss_explosion_thre = 1e-8  # everything "explodes"
    tr = Trainer(opt, cfg, ds, rank=0, world_size=1, num_workers=0,
                 checkpoint_dir=str(tmp), device=torch.device("cpu"))
    batch = next(iter(tr.train_loader))
    assert tr.train_step(batch) is None


def test_trainer_fit_writes_checkpoint_and_log(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    tr = Trainer(opt, cfg, ds, val_dataset=SyntheticPoseDataset(cfg, length=2),
                 rank=0, world_size=1, num_workers=0, checkpoint_dir=str(tmp),
                 device=torch.device("cpu"))
    tr.fit(1)
    assert (tmp / "PoseNet_0_epoch.pth").exists()
    assert (tmp / "log").exists()


def test_swa_trainer_averaging(tiny_setup):
    cfg, opt, ds, tmp = tiny_setup
    tr = SWATrainer(opt, cfg, ds, rank=0, world_size=1, num_workers=0,
                    checkpoint_dir=str(tmp), device=torch.device("cpu"),
                    swa_freq=1, lr_max=1e-2, lr_min=1e-3)
    key = "posenet.outs.0.0.conv.weight"  # head conv: large gradients
    tr.train_epoch(0, max_iters=1)
    assert tr.swa_count == 1
    w_e0 = tr.model.state_dict()[key].detach().clone()
    tr.train_epoch(1, max_iters=1)
    assert tr.swa_count == 2
    w_e1 = tr.model.state_dict()[key].detach().clone()
    assert not torch.equal(w_e0, w_e1)
    tr.swap_swa_weights()
    w_swa = tr.model.state_dict()[key].detach()
    assert torch.allclose(w_swa, (w_e0 + w_e1) / 2, atol=1e-7)


def test_lr_schedule():
    from improved_body_parts_amd.utils import adjust_learning_rate
    w = torch.nn.Parameter(torch.zeros(1))
    o = torch.optim.SGD([w], lr=1.0)
    lr0 = adjust_learning_rate(o, epoch=0, iters_done=0, iters_per_epoch=100,
                               base_lr=1.0, warmup_epochs=3)
    assert lr0 < 0.01  # warm-up start
    lr_mid = adjust_learning_rate(o, epoch=1, iters_done=50, iters_per_epoch=100,
                                  base_lr=1.0, warmup_epochs=3)
    assert lr_mid == pytest.approx(0.5, rel=0.01)
    lr_full = adjust_learning_rate(o, epoch=5, iters_done=0, iters_per_epoch=100,
                                   base_lr=1.0, warmup_epochs=3)
    assert lr_full == pytest.approx(1.0)
    lr_decay = adjust_learning_rate(o, epoch=15, iters_done=0, iters_per_epoch=100,
                                    base_lr=1.0, warmup_epochs=3, decay_every=15)
    assert lr_decay == pytest.approx(0.2)


def test_trainer_resume_continues_epoch(tiny_setup):
    """save -> resume in a fresh Trainer -> epochs continue where they left
    off with weights and optimizer state restored (reference
    train_distributed.py:149-197)."""
    cfg, opt, ds, tmp = tiny_setup
    tr = Trainer(opt, cfg, ds, rank=0, world_size=1, num_workers=0,
                 checkpoint_dir=str(tmp), device=torch.device("cpu"))
    tr.fit(1)
    w0 = tr.model.posenet.pre.conv1.weight.detach().clone()

    tr2 = Trainer(opt, cfg, ds, rank=0, world_size=1, num_workers=0,
                  checkpoint_dir=str(tmp), device=torch.device("cpu"))
    tr2.resume(str(tmp / "PoseNet_0_epoch.pth"))
    assert tr2.start_epoch == 1
    assert torch.equal(tr2.model.posenet.pre.conv1.weight.detach(), w0)
    # optimizer momentum state came back too
    states = [s for s in tr2.optimizer.state.values() if s]
    assert states, "optimizer state not restored"
    tr2.fit(2)  # runs epoch 1 only
    assert (tmp / "PoseNet_1_epoch.pth").exists()
