"""The 4-stage IMHN PoseNet and its Network / NetworkEval wrappers.

Capability parity with reference models/posenet.py (Merge :13-21, Features :24-40,
PoseNet :43-139, Network :142-173, NetworkEval :175-193). Same constructor
signature, same output structure ([nstack][5] heatmap tensors), same state-dict
key layout (reference checkpoints load unchanged), same weight init
(normal(0, 0.001) convs, BN weight=1/bias=0, Linear normal(0, 0.01)).

Re-designed for MI355X: every hot op dispatches through
:mod:`improved_body_parts_amd.ops` (HIP/CDNA4 kernels on GPU, eager PyTorch on CPU),
and the cross-stack merge chain reuses one fused 1x1-conv-sum kernel.
"""
from __future__ import annotations

import torch
from torch import nn

from .layers import Conv, Hourglass, SELayer, Backbone
from .loss import MultiTaskLoss, MultiTaskLossParallel


class Merge(nn.Module):
    """1x1 conv changing channel count (reference models/posenet.py:13-21).

    ``residual`` (optional) is summed into the conv's epilogue on the HIP
    path — the cross-stack ``merge_preds(pred) + merge_features(feat)`` chain
    becomes two conv kernels with zero separate add launches."""

    def __init__(self, x_dim, y_dim, bn=False):
        super().__init__()
        self.conv = Conv(x_dim, y_dim, 1, relu=False, bn=bn)

    def forward(self, x, residual=None):
        return self.conv(x, residual_post=residual)


class Features(nn.Module):
    """Two 3x3 Convs + SE per scale, 5 scales (reference models/posenet.py:24-40)."""

    def __init__(self, inp_dim, increase=128, bn=False):
        super().__init__()
        self.before_regress = nn.ModuleList([
            nn.Sequential(
                Conv(inp_dim + i * increase, inp_dim, 3, bn=bn, dropout=False),
                Conv(inp_dim, inp_dim, 3, bn=bn, dropout=False),
                SELayer(inp_dim),
            ) for i in range(5)
        ])

    def forward(self, fms):
        assert len(fms) == 5, f"hourglass produced {len(fms)} scales, expected 5"
        return [self.before_regress[i](fms[i]) for i in range(5)]


class PoseNet(nn.Module):
    """Stacked ("identity-mapping") hourglass network with 5-scale supervision.

    :param nstack: number of stacked hourglasses (4 in the headline config)
    :param inp_dim: hourglass channel width (256)
    :param oup_dim: regressed channels (50 = 30 paf + 18 keypoint + 2 background)
    :param bn: use batch normalisation
    :param increase: channel increase per hourglass down-step (128)

    Input is NHWC float in [0,1] (shape (N, H, W, 3)); output is
    ``[nstack][5]`` tensors of shape (N, oup_dim, H/4 / 2^s, W/4 / 2^s).
    """

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__()
        self.pre = Backbone(nFeat=inp_dim)
        self.hourglass = nn.ModuleList([Hourglass(4, inp_dim, increase, bn=bn)
                                        for _ in range(nstack)])
        self.features = nn.ModuleList([Features(inp_dim, increase=increase, bn=bn)
                                       for _ in range(nstack)])
        self.outs = nn.ModuleList([
            nn.ModuleList([Conv(inp_dim, oup_dim, 1, relu=False, bn=False)
                           for _ in range(5)]) for _ in range(nstack)])
        self.merge_features = nn.ModuleList([
            nn.ModuleList([Merge(inp_dim, inp_dim + j * increase, bn=bn)
                           for j in range(5)]) for _ in range(nstack - 1)])
        self.merge_preds = nn.ModuleList([
            nn.ModuleList([Merge(oup_dim, inp_dim + j * increase, bn=bn)
                           for j in range(5)]) for _ in range(nstack - 1)])
        self.nstack = nstack
        if init_weights:
            self._initialize_weights()

    def forward(self, imgs):
        # NHWC in [0,1] -> NCHW (the ops layer keeps channels_last memory format on GPU)
        x = imgs.permute(0, 3, 1, 2)
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        x = self.pre(x)
        pred = []
        features_cache = None
        for i in range(self.nstack):
            # the scale-0 feature-cache add rides the hourglass's top join
            # conv (post_add); scales 1-4 stay separate adds because the
            # hourglass's internal up-path consumes them cache-free
            post0 = features_cache[0] if i > 0 else None
            hourglass_feature = self.hourglass[i](x, post_add=post0)
            if i == 0:
                features_cache = [None] * 5
            else:
                # residual feature cache across stacks (reference posenet.py:93-98)
                hourglass_feature = [hourglass_feature[0]] + \
                    [hourglass_feature[s] + features_cache[s]
                     for s in range(1, 5)]
            features_instack = self.features[i](hourglass_feature)
            preds_instack = []
            for j in range(5):
                preds_instack.append(self.outs[i][j](features_instack[j]))
                if i != self.nstack - 1:
                    mf = self.merge_features[i][j](features_instack[j])
                    merged = self.merge_preds[i][j](preds_instack[j],
                                                    residual=mf)
                    if j == 0:
                        x = x + merged
                    features_cache[j] = merged
            pred.append(preds_instack)
        return pred

    def _initialize_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                m.weight.data.normal_(0, 0.001)
                if m.bias is not None:
                    m.bias.data.zero_()
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight.data, 0, 0.01)
                m.bias.data.zero_()


def _build(opt, config, bn, init_weights):
    """Model factory honouring opt.model_variant (reference: one Network
    wrapper per variant file; here one wrapper over the variant registry)."""
    variant = getattr(opt, "model_variant", "imhn")
    if variant in (None, "imhn"):
        return PoseNet(opt.nstack, opt.hourglass_inp_dim, config.num_layers,
                       bn=bn, increase=opt.increase, init_weights=init_weights)
    from .variants import build_posenet
    return build_posenet(variant, opt.nstack, opt.hourglass_inp_dim,
                         config.num_layers, bn=bn, increase=opt.increase,
                         init_weights=init_weights)


class Network(nn.Module):
    """Model + loss fused in one module so every rank/replica computes its own loss
    (reference models/posenet.py:142-173)."""

    def __init__(self, opt, config, bn=False, dist=False, swa=False):
        super().__init__()
        self.posenet = _build(opt, config, bn, init_weights=True)
        self.criterion = MultiTaskLoss(opt, config) if dist else MultiTaskLossParallel(opt, config)
        self.swa = swa

    def forward(self, input_all):
        inp_imgs = input_all[0]
        target_tuple = input_all[1:]
        output_tuple = self.posenet(inp_imgs)
        if not self.training:
            loss = self.criterion(output_tuple, target_tuple)
            return output_tuple, loss
        if not self.swa:
            return self.criterion(output_tuple, target_tuple)
        return output_tuple


class NetworkEval(nn.Module):
    """Inference-only wrapper (reference models/posenet.py:175-193)."""

    def __init__(self, opt, config, bn=False):
        super().__init__()
        self.posenet = _build(opt, config, bn, init_weights=False)

    def forward(self, inp_imgs):
        if self.training:
            raise ValueError("Only eval mode is available!")
        return self.posenet(inp_imgs)
