import torch
import pytest

from improved_body_parts_amd import ops
from improved_body_parts_amd.models import MultiTaskLoss, MultiTaskLossParallel


def _reference_focal_l2(s, sxing, mask_miss, heat_start, bkg_start, gamma=1,
                        multi_task_weight=0.1, keypoint_task_weight=3,
                        nstack_weight=(1, 1)):
    """Direct transliteration of reference loss_model.py:134-161 as the oracle."""
    mask = mask_miss.expand_as(sxing).clone()
    mask[:, :, -2, :, :] *= multi_task_weight
    mask[:, :, heat_start:bkg_start, :, :] *= keypoint_task_weight
    st = torch.where(torch.ge(sxing, 0.01), s, 1 - s)
    factor = torch.abs(1.0 - st) if gamma == 1 else (1.0 - st) ** gamma
    out = (s - sxing) ** 2 * factor * mask
    loss_nstack = out.sum(dim=(1, 2, 3, 4))
    w = [loss_nstack[i] * nstack_weight[i] for i in range(len(nstack_weight))]
    return sum(w) / sum(nstack_weight)


@pytest.mark.parametrize("gamma", [1, 2])
def test_focal_l2_matches_reference_formula(gamma):
    torch.manual_seed(0)
    nstack, n, c, h, w = 2, 3, 50, 16, 16
    pred = torch.rand(nstack, n, c, h, w)
    gt = torch.rand(n, c, h, w) * (torch.rand(n, c, h, w) > 0.7)
    mask = (torch.rand(n, 1, h, w) > 0.2).float()
    ours = ops.focal_l2_loss(pred, gt, mask, heat_start=30, bkg_start=48,
                             gamma=gamma, multi_task_weight=0.1,
                             keypoint_task_weight=3, nstack_weight=(1, 1))
    ref = _reference_focal_l2(pred, gt.unsqueeze(0), mask.unsqueeze(0),
                              30, 48, gamma=gamma, nstack_weight=(1, 1))
    assert torch.allclose(ours, ref, rtol=1e-5, atol=1e-5)


def test_focal_l2_gradients_flow():
    pred = torch.rand(1, 1, 50, 8, 8, requires_grad=True)
    gt = torch.rand(1, 50, 8, 8)
    mask = torch.ones(1, 1, 8, 8)
    loss = ops.focal_l2_loss(pred, gt, mask, heat_start=30, bkg_start=48,
                             nstack_weight=(1,))
    loss.backward()
    assert pred.grad is not None and torch.isfinite(pred.grad).all()


def test_multitask_loss_scales(small_config, small_opt):
    crit = MultiTaskLoss(small_opt, small_config)
    nstack = small_opt.nstack
    preds = [[torch.rand(2, 50, 32 // 2 ** s, 32 // 2 ** s) for s in range(5)]
             for _ in range(nstack)]
    mask = torch.ones(2, 1, 32, 32)
    gt = torch.rand(2, 50, 32, 32)
    loss = crit(preds, (mask, gt))
    assert loss.dim() == 0 and torch.isfinite(loss)
    # zero mask => only the scale-resize of GT contributes nothing: loss must be 0
    loss0 = crit(preds, (torch.zeros(2, 1, 32, 32), gt))
    assert float(loss0) == 0.0


def test_mask_miss_zero_region_excluded(small_config, small_opt):
    crit = MultiTaskLoss(small_opt, small_config)
    nstack = small_opt.nstack
    gt = torch.zeros(1, 50, 32, 32)
    base = [[torch.zeros(1, 50, 32 // 2 ** s, 32 // 2 ** s) for s in range(5)]
            for _ in range(nstack)]
    # error only in the left half, mask kills the left half
    wrong = [[t.clone() for t in stack] for stack in base]
    for stack in wrong:
        stack[0][..., :16] = 1.0
    mask = torch.ones(1, 1, 32, 32)
    mask[..., :16] = 0
    loss = crit(wrong, (mask, gt))
    assert float(loss) == pytest.approx(0.0, abs=1e-6)


def test_parallel_loss_reference_semantics(small_config, small_opt):
    """MultiTaskLossParallel follows the reference's DataParallel-path
    semantics exactly (loss_model_parallel.py): plain L2 by default with the
    channel-broadcast unthresholded mask and NO batch division; focal option
    uses gamma=2 without alpha/beta/task weights."""
    import torch.nn.functional as F
    crit = MultiTaskLossParallel(small_opt, small_config)
    nstack = small_opt.nstack
    torch.manual_seed(1)
    preds = [[torch.rand(2, 50, 32 // 2 ** s, 32 // 2 ** s) for s in range(5)]
             for _ in range(nstack)]
    mask = (torch.rand(2, 1, 32, 32) > 0.3).float()
    gt = torch.rand(2, 50, 32, 32)
    lp = crit(preds, (mask, gt))
    # manual oracle
    acc = 0.0
    for i in range(5):
        pred = torch.stack([preds[j][i] for j in range(nstack)], 0)
        size = pred.shape[-2:]
        m = F.interpolate(mask, size=size, mode="bilinear", align_corners=False)
        g = F.adaptive_avg_pool2d(gt, output_size=size)
        out = (pred - g[None]) ** 2 * m[None]
        per_stack = out.sum(dim=(1, 2, 3, 4))
        acc = acc + per_stack.mean() * small_opt.scale_weight[i]
    want = acc / sum(small_opt.scale_weight)
    assert torch.allclose(lp, want, rtol=1e-5)
    # focal option differs from plain L2
    lf = MultiTaskLossParallel(small_opt, small_config, use_focal=True)(
        preds, (mask, gt))
    assert not torch.allclose(lf, lp, rtol=1e-3)


def test_plain_l2_and_l1_losses():
    s = torch.rand(2, 1, 50, 8, 8)
    gt = torch.rand(1, 1, 50, 8, 8)
    mask = torch.ones(1, 1, 1, 8, 8)
    l2 = MultiTaskLoss.l2_loss(s, gt, mask, 30, 48, nstack_weight=(1, 1))
    assert torch.isfinite(l2)
    l1 = MultiTaskLoss.l1_loss(s, gt, torch.ones_like(s), nstack_weight=(1, 1))
    assert torch.isfinite(l1)
