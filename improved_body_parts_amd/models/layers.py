"""Building blocks of the IMHN stacked hourglass.

Capability parity with reference models/layers_transposed.py (Conv :90-120,
Residual :12-48, BasicResidual :51-87, DilatedConv :123-155, Backbone :158-194,
Hourglass :197-282, SELayer :285-306) — re-implemented, not translated:

  * Parameters live in standard ``nn.Conv2d`` / ``nn.BatchNorm2d`` containers with
    the SAME attribute nesting as the reference, so reference checkpoints load
    unchanged (checkpoint-format parity is a north-star requirement).
  * ``forward`` dispatches through :mod:`improved_body_parts_amd.ops`: on an MI355X
    the hot ops run as hand-written CDNA4 HIP kernels (MFMA implicit-GEMM conv,
    fused BN+LeakyReLU, fused pool/upsample, fused SE); on CPU the same ops run
    as eager PyTorch and serve as the numerics oracle for the kernels.
  * LeakyReLU slope is 0.01 everywhere, matching the reference.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops

LEAKY_SLOPE = 0.01


class Conv(nn.Module):
    """3x3 / 1x1 / 7x7 conv + optional BN + optional LeakyReLU (reference layers_transposed.py:90-120)."""

    def __init__(self, inp_dim, out_dim, kernel_size=3, stride=1, bn=True, relu=True,
                 dropout=False, dialated=1):
        super().__init__()
        self.inp_dim = inp_dim
        self.dropout = dropout
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True) if relu else None
        pad = (kernel_size - 1) // 2
        if bn:
            self.conv = nn.Conv2d(inp_dim, out_dim, kernel_size, stride, padding=pad, bias=False)
            self.bn = nn.BatchNorm2d(out_dim)
        else:
            self.conv = nn.Conv2d(inp_dim, out_dim, kernel_size, stride, padding=pad, bias=True)
            self.bn = None

    def forward(self, x, residual_post=None, residual_post2=None):
        assert x.size(1) == self.inp_dim, \
            f"input channel {x.size(1)} does not fit kernel channel {self.inp_dim}"
        if self.dropout:
            x = F.dropout(x, p=0.2, training=self.training, inplace=False)
        return ops.conv_bn_act(x, self.conv, self.bn, act=self.relu is not None,
                               training=self.training,
                               residual_post=residual_post,
                               residual_post2=residual_post2)


class DilatedConv(nn.Module):
    """Dilated 3x3 conv + BN + LeakyReLU, stride 1 (reference layers_transposed.py:123-155)."""

    def __init__(self, inp_dim, out_dim, kernel_size=3, stride=1, bn=True, relu=True,
                 dropout=False, dialation=3):
        super().__init__()
        assert stride == 1, "DilatedConv supports stride=1 only"
        self.inp_dim = inp_dim
        self.dropout = dropout
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True) if relu else None
        if bn:
            self.conv = nn.Conv2d(inp_dim, out_dim, kernel_size, stride, padding=dialation,
                                  bias=False, dilation=dialation)
            self.bn = nn.BatchNorm2d(out_dim)
        else:
            self.conv = nn.Conv2d(inp_dim, out_dim, kernel_size, stride, padding=dialation,
                                  bias=True, dilation=dialation)
            self.bn = None

    def forward(self, x):
        assert x.size(1) == self.inp_dim
        if self.dropout:
            x = F.dropout(x, p=0.2, training=self.training, inplace=False)
        return ops.conv_bn_act(x, self.conv, self.bn, act=self.relu is not None,
                               training=self.training)


class Residual(nn.Module):
    """Bottleneck residual 1x1 -> 3x3 -> 1x1 + skip (reference layers_transposed.py:12-48).

    Parameter layout matches the reference's ``convBlock`` Sequential indices
    (0=conv1x1, 1=bn, 3=conv3x3, 4=bn, 6=conv1x1, 7=bn) and ``skipConv`` (0=conv, 1=bn).
    """

    def __init__(self, ins, outs, bn=True, relu=True):
        super().__init__()
        self.relu_flag = relu
        mid = outs // 2
        self.convBlock = nn.Sequential(
            nn.Conv2d(ins, mid, 1, bias=False),
            nn.BatchNorm2d(mid),
            nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True),
            nn.Conv2d(mid, mid, 3, 1, 1, bias=False),
            nn.BatchNorm2d(mid),
            nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True),
            nn.Conv2d(mid, outs, 1, bias=False),
            nn.BatchNorm2d(outs),
        )
        if ins != outs:
            self.skipConv = nn.Sequential(
                nn.Conv2d(ins, outs, 1, bias=False),
                nn.BatchNorm2d(outs),
            )
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True)
        self.ins = ins
        self.outs = outs

    def forward(self, x):
        cb = self.convBlock
        y = ops.conv_bn_act(x, cb[0], cb[1], act=True, training=self.training)
        y = ops.conv_bn_act(y, cb[3], cb[4], act=True, training=self.training)
        # last conv+bn fuses the residual add (+ optional relu) on the HIP path
        if self.ins != self.outs:
            residual = ops.conv_bn_act(x, self.skipConv[0], self.skipConv[1], act=False,
                                       training=self.training)
        else:
            residual = x
        return ops.conv_bn_add_act(y, cb[6], cb[7], residual, act=self.relu_flag,
                                   training=self.training)


class BasicResidual(nn.Module):
    """Two 3x3 convs + skip (reference layers_transposed.py:51-87)."""

    def __init__(self, inp_dim, out_dim, stride=1, bn=True, relu=True):
        super().__init__()
        self.relu_flag = relu
        self.conv1 = nn.Conv2d(inp_dim, out_dim, 3, padding=1, stride=stride, bias=False)
        self.bn1 = nn.BatchNorm2d(out_dim)
        self.relu1 = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True)
        self.conv2 = nn.Conv2d(out_dim, out_dim, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(out_dim)
        self.skip = nn.Sequential(
            nn.Conv2d(inp_dim, out_dim, 1, stride=stride, bias=False),
            nn.BatchNorm2d(out_dim),
        ) if stride != 1 or inp_dim != out_dim else nn.Sequential()
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True)

    def forward(self, x):
        y = ops.conv_bn_act(x, self.conv1, self.bn1, act=True, training=self.training)
        if len(self.skip) > 0:
            skip = ops.conv_bn_act(x, self.skip[0], self.skip[1], act=False,
                                   training=self.training)
        else:
            skip = x
        return ops.conv_bn_add_act(y, self.conv2, self.bn2, skip, act=self.relu_flag,
                                   training=self.training)


class Backbone(nn.Module):
    """Stem: 7x7 s2 conv -> residual -> maxpool -> residual -> 6 dilated convs -> concat
    (reference layers_transposed.py:158-194). Output channels = 256 (128 + 128)."""

    def __init__(self, nFeat=256, inplanes=3, resBlock=Residual, dilatedBlock=DilatedConv):
        super().__init__()
        self.nFeat = nFeat
        self.inplanes = inplanes
        # channel plan scales with nFeat; at nFeat=256 it reproduces the
        # reference exactly (64 -> 128 -> 128 || 128 -> concat 256)
        c1, c2 = nFeat // 4, nFeat // 2
        self.conv1 = nn.Conv2d(inplanes, c1, kernel_size=7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(c1)
        self.relu = nn.LeakyReLU(negative_slope=LEAKY_SLOPE, inplace=True)
        self.res1 = resBlock(c1, c2)
        self.pool = nn.MaxPool2d(2, 2)
        self.res2 = resBlock(c2, c2)
        self.dilation = nn.Sequential(
            dilatedBlock(c2, c2, dialation=3),
            dilatedBlock(c2, c2, dialation=3),
            dilatedBlock(c2, c2, dialation=4),
            dilatedBlock(c2, c2, dialation=4),
            dilatedBlock(c2, c2, dialation=5),
            dilatedBlock(c2, c2, dialation=5),
        )

    def forward(self, x):
        x = ops.conv_bn_act(x, self.conv1, self.bn1, act=True, training=self.training)
        x = self.res1(x)
        x = ops.maxpool2x2(x)
        x = self.res2(x)
        x1 = self.dilation(x)
        return ops.channel_concat(x, x1)


class Hourglass(nn.Module):
    """Recursive 4-depth hourglass returning 5 scales (reference layers_transposed.py:197-282).

    Per depth d the module list holds [up-residual, down-residual(+increase),
    up-path residual(-increase), post-upsample refine Conv, (innermost residual)].
    Up-sampling is nearest x2 followed by a 3x3 Conv refine; the skip join is an add.
    """

    def __init__(self, depth, nFeat, increase=128, bn=False, resBlock=Residual, convBlock=Conv):
        super().__init__()
        self.depth = depth
        self.nFeat = nFeat
        self.increase = increase
        self.bn = bn
        self.resBlock = resBlock
        self.convBlock = convBlock
        hg = []
        for d in range(depth):
            c0 = nFeat + increase * d
            c1 = nFeat + increase * (d + 1)
            mods = [
                resBlock(c0, c0, bn=bn),            # 0: skip path at this scale
                resBlock(c0, c1, bn=bn),            # 1: after downsample
                resBlock(c1, c0, bn=bn),            # 2: before upsample
                convBlock(c0, c0, bn=bn),           # 3: refine after upsample
            ]
            if d == depth - 1:
                mods.append(resBlock(c1, c1, bn=bn))  # 4: innermost
            hg.append(nn.ModuleList(mods))
        self.hg = nn.ModuleList(hg)
        self.downsample = nn.MaxPool2d(2, 2)
        self.upsample = nn.Upsample(scale_factor=2, mode="nearest")

    def _forward(self, d, x, up_fms, post_add=None):
        up1 = self.hg[d][0](x)
        low1 = ops.maxpool2x2(x)
        low1 = self.hg[d][1](low1)
        if d == self.depth - 1:
            low2 = self.hg[d][4](low1)
        else:
            low2 = self._forward(d + 1, low1, up_fms)
        low3 = self.hg[d][2](low2)
        up_fms.append(low2)
        up2 = ops.upsample2x_nearest(low3)
        # the up1 + deconv1 skip join rides the refine conv's epilogue
        # (post-act residual) instead of a separate elementwise kernel;
        # at d=0 the caller's cross-stack feature-cache add joins too
        # (legal only for the top scale — deeper low2 outputs feed the
        # internal up-path, which must see them WITHOUT the cache add)
        return self.hg[d][3](up2, residual_post=up1, residual_post2=post_add)

    def forward(self, x, post_add=None):
        up_fms = []
        top = self._forward(0, x, up_fms, post_add)
        return [top] + up_fms[::-1]


class SELayer(nn.Module):
    """Squeeze-and-Excitation channel attention (reference layers_transposed.py:285-306)."""

    def __init__(self, inp_dim, reduction=16):
        super().__init__()
        assert inp_dim > reduction
        self.avg_pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Sequential(
            nn.Linear(inp_dim, inp_dim // reduction),
            nn.LeakyReLU(inplace=True),
            nn.Linear(inp_dim // reduction, inp_dim),
            nn.Sigmoid(),
        )

    def forward(self, x):
        return ops.se_layer(x, self.fc[0], self.fc[2])
