"""Architecture variants of the IMHN PoseNet.

Capability parity with the reference's model-ablation files, expressed as
variant classes over the shared HIP-dispatched blocks instead of whole-file
forks. State-dict layouts match the reference variants KEY-FOR-KEY (verified
by strict checkpoint loads + forward comparison in
tests/test_reference_parity.py):

  * ``PoseNetFinal``      — reference models/posenet_final.py (+
    layers_transposed_final.py): identity-mapping hourglass built from plain
    Conv blocks (two refinements after each upsample, residual-add THEN
    LeakyReLU), residual-chain backbone, SE attention on every scale of every
    stack, channel-compress 1x1 + two 3x3 before regression.
  * ``PoseNetAttention``  — reference models/posenet2.py: SE attention on all
    scales/stacks, heads + merges on the UNCOMPRESSED per-scale channels,
    bn-free merge convs.
  * ``PoseNetLight``      — reference models/posenet3.py: plain conv stem, SE
    attention on all scales/stacks, a single 3x3 Conv before each head, heads
    + merges on uncompressed channels.
  * ``PoseNetIndependent``— reference models/posenet_independent.py: built on
    the CLASSIC hourglass blocks (models/layers.py: plain up1+up2 merge, no
    post-upsample refine), plain conv stem, no cross-stack feature cache.
  * ``AEPoseNet``         — reference models/ae_pose.py: single-scale
    Associative-Embedding-style stacked hourglass.

All accept ``(nstack, inp_dim, oup_dim, bn, increase)`` like ``PoseNet`` and
run through the same ops layer (HIP kernels on MI355X, eager on CPU).
"""
from __future__ import annotations

import torch
from torch import nn

from .. import ops
from .layers import LEAKY_SLOPE, Conv, Hourglass, Residual, SELayer
from .posenet import Merge, PoseNet


# ---------------------------------------------------------------------------
# "final" variant blocks (reference layers_transposed_final.py)
# ---------------------------------------------------------------------------

class BackboneFinal(nn.Module):
    """Residual-chain stem: 7x7 s2 -> res(64->128) -> pool -> res(128->128)
    -> res(128->nFeat) (reference layers_transposed_final.py:82-107)."""

    def __init__(self, nFeat=256, inplanes=3, resBlock=Residual):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True)
        self.res1 = resBlock(64, 128)
        self.pool = nn.MaxPool2d(2, 2)
        self.res2 = resBlock(128, 128)
        self.res3 = resBlock(128, nFeat)

    def forward(self, x):
        x = ops.conv_bn_act(x, self.conv1, self.bn1, act=True,
                            training=self.training)
        x = self.res1(x)
        x = ops.maxpool2x2(x)
        x = self.res2(x)
        return self.res3(x)


class HourglassFinal(nn.Module):
    """Identity-mapping hourglass from plain Conv blocks: two refinements
    after each upsample, residual add BEFORE the LeakyReLU
    (reference layers_transposed_final.py:110-189). Returns 5 scales.

    Slot layout per depth (matches the reference's ModuleList indices):
      0 skip Conv (relu=False) | 1 down Conv (+increase) | 2 pre-upsample Conv
      (-increase) | 3, 4 refine Convs (4 relu=False) | 5 LeakyReLU |
      6 innermost Conv (deepest depth only)."""

    def __init__(self, depth, nFeat, increase=128, bn=False, convBlock=Conv):
        super().__init__()
        self.depth = depth
        hg = []
        for d in range(depth):
            c0 = nFeat + increase * d
            c1 = nFeat + increase * (d + 1)
            mods = [
                convBlock(c0, c0, bn=bn, relu=False),
                convBlock(c0, c1, bn=bn),
                convBlock(c1, c0, bn=bn),
                convBlock(c0, c0, bn=bn),
                convBlock(c0, c0, bn=bn, relu=False),
                nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True),
            ]
            if d == depth - 1:
                mods.append(convBlock(c1, c1, bn=bn))
            hg.append(nn.ModuleList(mods))
        self.hg = nn.ModuleList(hg)
        self.downsample = nn.MaxPool2d(2, 2)
        self.upsample = nn.Upsample(scale_factor=2, mode="nearest")

    def _forward(self, d, x, up_fms):
        up1 = self.hg[d][0](x)
        low1 = ops.maxpool2x2(x)
        low1 = self.hg[d][1](low1)
        if d == self.depth - 1:
            low2 = self.hg[d][6](low1)
        else:
            low2 = self._forward(d + 1, low1, up_fms)
        low3 = self.hg[d][2](low2)
        up_fms.append(low2)
        up2 = ops.upsample2x_nearest(low3)
        deconv2 = self.hg[d][4](self.hg[d][3](up2))
        return self.hg[d][5](up1 + deconv2)  # add THEN activate

    def forward(self, x):
        up_fms = []
        top = self._forward(0, x, up_fms)
        return [top] + up_fms[::-1]


class FeaturesCompress(nn.Module):
    """Channel-compress 1x1 then two 3x3 before regression
    (reference posenet_final.py:24-48)."""

    def __init__(self, inp_dim, increase=128, bn=False):
        super().__init__()
        self.before_regress = nn.ModuleList([
            nn.Sequential(
                Conv(inp_dim + i * increase, inp_dim, 1, bn=bn, dropout=False),
                Conv(inp_dim, inp_dim, 3, bn=bn, dropout=False),
                Conv(inp_dim, inp_dim, 3, bn=bn, dropout=False),
            ) for i in range(5)
        ])

    def forward(self, fms):
        assert len(fms) == 5
        return [self.before_regress[i](fms[i]) for i in range(5)]


class _AttentionStacksMixin:
    """Shared forward for the variants that SE-attend every hourglass scale
    before regression (reference posenet2.py/posenet3.py/posenet_final.py
    forward bodies are identical in structure)."""

    def forward(self, imgs):
        x = imgs.permute(0, 3, 1, 2)
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        x = self.pre(x)
        pred = []
        features_cache = [None] * 5
        for i in range(self.nstack):
            hourglass_feature = self.hourglass[i](x)
            hourglass_feature = [self.channel_attention[i][s](hourglass_feature[s])
                                 for s in range(5)]
            if i > 0:
                hourglass_feature = [hourglass_feature[s] + features_cache[s]
                                     for s in range(5)]
            features_instack = self.features[i](hourglass_feature)
            preds_instack = []
            for j in range(5):
                preds_instack.append(self.outs[i][j](features_instack[j]))
                if i != self.nstack - 1:
                    merged = (self.merge_preds[i][j](preds_instack[j])
                              + self.merge_features[i][j](features_instack[j]))
                    if j == 0:
                        x = x + merged
                    features_cache[j] = merged
            pred.append(preds_instack)
        return pred


class PoseNetFinal(_AttentionStacksMixin, PoseNet):
    """Reference models/posenet_final.py: final IMHN ablation."""

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__(nstack, inp_dim, oup_dim, bn=bn, increase=increase,
                         init_weights=False, **kwargs)
        self.pre = BackboneFinal(nFeat=inp_dim)
        self.hourglass = nn.ModuleList([
            HourglassFinal(4, inp_dim, increase, bn=bn) for _ in range(nstack)])
        self.features = nn.ModuleList([
            FeaturesCompress(inp_dim, increase=increase, bn=bn)
            for _ in range(nstack)])
        self.channel_attention = nn.ModuleList([
            nn.ModuleList([SELayer(inp_dim + j * increase) for j in range(5)])
            for _ in range(nstack)])
        if init_weights:
            self._initialize_weights()


# ---------------------------------------------------------------------------
# attention variant (reference posenet2.py)
# ---------------------------------------------------------------------------

class FeaturesWide(nn.Module):
    """Two 3x3 Convs per scale on the uncompressed channels
    (reference posenet2.py Features)."""

    def __init__(self, inp_dim, increase=128, bn=False):
        super().__init__()
        self.before_regress = nn.ModuleList([
            nn.Sequential(
                Conv(inp_dim + i * increase, inp_dim + i * increase, 3, bn=bn),
                Conv(inp_dim + i * increase, inp_dim + i * increase, 3, bn=bn),
            ) for i in range(5)
        ])

    def forward(self, fms):
        assert len(fms) == 5
        return [self.before_regress[i](fms[i]) for i in range(5)]


class PoseNetAttention(_AttentionStacksMixin, PoseNet):
    """Reference models/posenet2.py: SE attention on every scale of every
    stack; heads + merges on the uncompressed per-scale channels; the merge
    convs carry a bias and no BN (reference posenet2.py:18)."""

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__(nstack, inp_dim, oup_dim, bn=bn, increase=increase,
                         init_weights=False, **kwargs)
        self.features = nn.ModuleList([
            FeaturesWide(inp_dim, increase=increase, bn=bn)
            for _ in range(nstack)])
        self.channel_attention = nn.ModuleList([
            nn.ModuleList([SELayer(inp_dim + j * increase) for j in range(5)])
            for _ in range(nstack)])
        self.outs = nn.ModuleList([
            nn.ModuleList([Conv(inp_dim + j * increase, oup_dim, 1, relu=False,
                                bn=False) for j in range(5)])
            for _ in range(nstack)])
        self.merge_features = nn.ModuleList([
            nn.ModuleList([Merge(inp_dim + j * increase, inp_dim + j * increase,
                                 bn=False) for j in range(5)])
            for _ in range(nstack - 1)])
        self.merge_preds = nn.ModuleList([
            nn.ModuleList([Merge(oup_dim, inp_dim + j * increase, bn=False)
                           for j in range(5)]) for _ in range(nstack - 1)])
        if init_weights:
            self._initialize_weights()


# ---------------------------------------------------------------------------
# light variant (reference posenet3.py)
# ---------------------------------------------------------------------------

class FeaturesLight(nn.Module):
    """A single 3x3 Conv per scale on the uncompressed channels
    (reference posenet3.py Features)."""

    def __init__(self, inp_dim, increase=128, bn=False):
        super().__init__()
        self.before_regress = nn.ModuleList([
            nn.Sequential(Conv(inp_dim + i * increase, inp_dim + i * increase,
                               3, bn=bn)) for i in range(5)
        ])

    def forward(self, fms):
        assert len(fms) == 5
        return [self.before_regress[i](fms[i]) for i in range(5)]


class PoseNetLight(_AttentionStacksMixin, PoseNet):
    """Reference models/posenet3.py: plain conv stem, SE attention on all
    scales/stacks, single pre-regress conv, heads + merges on the uncompressed
    channels (merges keep BN, reference posenet3.py:21)."""

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__(nstack, inp_dim, oup_dim, bn=bn, increase=increase,
                         init_weights=False, **kwargs)
        self.pre = nn.Sequential(
            Conv(3, 64, 7, 2, bn=bn),
            Conv(64, 128, bn=bn),
            nn.MaxPool2d(2, 2),
            Conv(128, 128, bn=bn),
            Conv(128, inp_dim, bn=bn),
        )
        self.features = nn.ModuleList([
            FeaturesLight(inp_dim, increase=increase, bn=bn)
            for _ in range(nstack)])
        self.channel_attention = nn.ModuleList([
            nn.ModuleList([SELayer(inp_dim + j * increase) for j in range(5)])
            for _ in range(nstack)])
        self.outs = nn.ModuleList([
            nn.ModuleList([Conv(inp_dim + j * increase, oup_dim, 1, relu=False,
                                bn=False) for j in range(5)])
            for _ in range(nstack)])
        self.merge_features = nn.ModuleList([
            nn.ModuleList([Merge(inp_dim + j * increase, inp_dim + j * increase,
                                 bn=bn) for j in range(5)])
            for _ in range(nstack - 1)])
        self.merge_preds = nn.ModuleList([
            nn.ModuleList([Merge(oup_dim, inp_dim + j * increase, bn=bn)
                           for j in range(5)]) for _ in range(nstack - 1)])
        if init_weights:
            self._initialize_weights()


# ---------------------------------------------------------------------------
# independent-stack variant (reference posenet_independent.py + models/layers.py)
# ---------------------------------------------------------------------------

class HourglassClassic(nn.Module):
    """The original (Associative-Embedding-style) hourglass: 3 Conv slots per
    depth + innermost, plain up1 + up2 merge with NO post-upsample refinement
    (reference models/layers.py:81-169). Returns 5 scales."""

    def __init__(self, depth, nFeat, increase=128, bn=False, convBlock=Conv):
        super().__init__()
        self.depth = depth
        hg = []
        for d in range(depth):
            c0 = nFeat + increase * d
            c1 = nFeat + increase * (d + 1)
            mods = [
                convBlock(c0, c0, bn=bn),
                convBlock(c0, c1, bn=bn),
                convBlock(c1, c0, bn=bn),
            ]
            if d == depth - 1:
                mods.append(convBlock(c1, c1, bn=bn))
            hg.append(nn.ModuleList(mods))
        self.hg = nn.ModuleList(hg)
        self.downsample = nn.MaxPool2d(2, 2)
        self.upsample = nn.Upsample(scale_factor=2, mode="nearest")

    def _forward(self, d, x, up_fms):
        up1 = self.hg[d][0](x)
        low1 = ops.maxpool2x2(x)
        low1 = self.hg[d][1](low1)
        if d == self.depth - 1:
            low2 = self.hg[d][3](low1)
        else:
            low2 = self._forward(d + 1, low1, up_fms)
        low3 = self.hg[d][2](low2)
        up_fms.append(low2)
        up2 = ops.upsample2x_nearest(low3)
        return up1 + up2

    def forward(self, x):
        up_fms = []
        top = self._forward(0, x, up_fms)
        return [top] + up_fms[::-1]


class PoseNetIndependent(nn.Module):
    """Reference models/posenet_independent.py: classic hourglass blocks,
    plain conv stem, two uncompressed pre-regress convs, bias-carrying merge
    convs, and NO cross-stack residual feature cache."""

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__()
        self.nstack = nstack
        self.pre = nn.Sequential(
            Conv(3, 64, 7, 2, bn=bn),
            Conv(64, 128, bn=bn),
            nn.MaxPool2d(2, 2),
            Conv(128, 128, bn=bn),
            Conv(128, inp_dim, bn=bn),
        )
        self.hourglass = nn.ModuleList([
            HourglassClassic(4, inp_dim, increase, bn=bn)
            for _ in range(nstack)])
        self.features = nn.ModuleList([
            FeaturesWide(inp_dim, increase=increase, bn=bn)
            for _ in range(nstack)])
        self.outs = nn.ModuleList([
            nn.ModuleList([Conv(inp_dim + j * increase, oup_dim, 1, relu=False,
                                bn=False) for j in range(5)])
            for _ in range(nstack)])
        self.merge_features = nn.ModuleList([
            nn.ModuleList([Merge(inp_dim + j * increase, inp_dim + j * increase,
                                 bn=False) for j in range(5)])
            for _ in range(nstack - 1)])
        self.merge_preds = nn.ModuleList([
            nn.ModuleList([Merge(oup_dim, inp_dim + j * increase, bn=False)
                           for j in range(5)]) for _ in range(nstack - 1)])
        if init_weights:
            PoseNet._initialize_weights(self)

    def forward(self, imgs):
        x = imgs.permute(0, 3, 1, 2)
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        x = self.pre(x)
        pred = []
        for i in range(self.nstack):
            hourglass_feature = self.hourglass[i](x)
            features_instack = self.features[i](hourglass_feature)
            preds_instack = []
            for j in range(5):
                preds_instack.append(self.outs[i][j](features_instack[j]))
                if i != self.nstack - 1 and j == 0:
                    x = x + self.merge_preds[i][j](preds_instack[j]) \
                        + self.merge_features[i][j](features_instack[j])
            pred.append(preds_instack)
        return pred


# ---------------------------------------------------------------------------
# Associative-Embedding-style single-scale stack (reference ae_pose.py +
# ae_layer.py — "Copied from Associative Embedding pytorch project")
# ---------------------------------------------------------------------------

class AEConv(nn.Module):
    """AE-style conv block: conv ALWAYS carries a bias, plain ReLU (not
    leaky), and the activation comes BEFORE the BN (reference ae_layer.py
    Conv)."""

    def __init__(self, inp_dim, out_dim, kernel_size=3, stride=1, bn=False,
                 relu=True):
        super().__init__()
        self.inp_dim = inp_dim
        self.conv = nn.Conv2d(inp_dim, out_dim, kernel_size, stride,
                              padding=(kernel_size - 1) // 2, bias=True)
        self.relu = nn.ReLU() if relu else None
        self.bn = nn.BatchNorm2d(out_dim) if bn else None

    def forward(self, x):
        assert x.size(1) == self.inp_dim
        x = self.conv(x)
        if self.relu is not None:
            x = self.relu(x)
        if self.bn is not None:
            x = self.bn(x)
        return x


class AEHourglass(nn.Module):
    """Recursive attribute-style hourglass, single-scale output
    (reference ae_layer.py Hourglass): up1 + upsample(low3(low2(low1(pool))))."""

    def __init__(self, n, f, bn=None, increase=128):
        super().__init__()
        nf = f + increase
        self.up1 = AEConv(f, f, 3, bn=bn)
        self.pool1 = nn.MaxPool2d(2, 2)
        self.low1 = AEConv(f, nf, 3, bn=bn)
        if n > 1:
            self.low2 = AEHourglass(n - 1, nf, bn=bn, increase=increase)
        else:
            self.low2 = AEConv(nf, nf, 3, bn=bn)
        self.low3 = AEConv(nf, f, 3)
        self.up2 = nn.UpsamplingNearest2d(scale_factor=2)

    def forward(self, x):
        up1 = self.up1(x)
        low1 = self.low1(ops.maxpool2x2(x))
        low2 = self.low2(low1)
        low3 = self.low3(low2)
        return up1 + ops.upsample2x_nearest(low3)


class AEMerge(nn.Module):
    """1x1 channel adapter (reference ae_pose.py Merge)."""

    def __init__(self, x_dim, y_dim):
        super().__init__()
        self.conv = AEConv(x_dim, y_dim, 1, relu=False, bn=False)

    def forward(self, x):
        return self.conv(x)


class AEPoseNet(nn.Module):
    """Single-scale stacked hourglass (reference models/ae_pose.py:20-58):
    AE conv stem, per-stack [hourglass -> 2 feature convs -> 1x1 head],
    inter-stack merge at one scale. Output: [nstack] x (N, oup_dim, H/4, W/4).
    State-dict layout matches the reference exactly."""

    def __init__(self, nstack, inp_dim, oup_dim, bn=False, increase=128,
                 init_weights=True, **kwargs):
        super().__init__()
        self.nstack = nstack
        self.pre = nn.Sequential(
            AEConv(3, 64, 7, 2, bn=bn),
            AEConv(64, 128, bn=bn),
            nn.MaxPool2d(2, 2),
            AEConv(128, 128, bn=bn),
            AEConv(128, inp_dim, bn=bn),
        )
        self.features = nn.ModuleList([
            nn.Sequential(
                AEHourglass(4, inp_dim, bn, increase),
                AEConv(inp_dim, inp_dim, 3, bn=False),
                AEConv(inp_dim, inp_dim, 3, bn=False),
            ) for _ in range(nstack)])
        self.outs = nn.ModuleList([
            AEConv(inp_dim, oup_dim, 1, relu=False, bn=False)
            for _ in range(nstack)])
        self.merge_features = nn.ModuleList([AEMerge(inp_dim, inp_dim)
                                             for _ in range(nstack - 1)])
        self.merge_preds = nn.ModuleList([AEMerge(oup_dim, inp_dim)
                                          for _ in range(nstack - 1)])
        if init_weights:
            PoseNet._initialize_weights(self)

    def forward(self, imgs):
        x = imgs.permute(0, 3, 1, 2)
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        x = self.pre(x)
        preds = []
        for i in range(self.nstack):
            feature = self.features[i](x)
            pred = self.outs[i](feature)
            if i != self.nstack - 1:
                x = x + self.merge_preds[i](pred) \
                    + self.merge_features[i](feature)
            # [nstack][1] nesting mirrors the reference (ae_pose.py:49-57) so
            # the Network/loss wrappers see the usual [stack][scale] shape
            preds.append([pred])
        return preds


VARIANTS = {
    "imhn": PoseNet,
    "final": PoseNetFinal,
    "attention": PoseNetAttention,
    "light": PoseNetLight,
    "independent": PoseNetIndependent,
    "ae": AEPoseNet,
}


def build_posenet(name, nstack, inp_dim, oup_dim, bn=False, increase=128,
                  **kwargs):
    """Factory over all model variants (replaces the reference's per-file
    driver imports)."""
    try:
        cls = VARIANTS[name]
    except KeyError:
        raise ValueError(f"unknown posenet variant '{name}'; "
                         f"one of {sorted(VARIANTS)}") from None
    return cls(nstack, inp_dim, oup_dim, bn=bn, increase=increase, **kwargs)
