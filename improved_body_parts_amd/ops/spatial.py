"""Autograd wrappers for the NHWC spatial kernels (maxpool / upsample / SE)."""
from __future__ import annotations

import torch
import torch.nn.functional as F

from ._backend import hip_extension

LEAKY_SLOPE = 0.01
_CL = torch.channels_last


class MaxPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_extension()
        x = x.contiguous(memory_format=_CL)
        n, c, h, w = x.shape
        y, arg = ext.maxpool2x2_fwd(x, n, h, w, c)
        ctx.save_for_backward(arg)
        ctx.dims = (n, h, w, c)
        return y.permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        ext = hip_extension()
        (arg,) = ctx.saved_tensors
        n, h, w, c = ctx.dims
        dx = ext.maxpool2x2_bwd(dy.contiguous(memory_format=_CL), arg, n, h, w, c)
        return dx.permute(0, 3, 1, 2)


class Upsample2xFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_extension()
        x = x.contiguous(memory_format=_CL)
        n, c, h, w = x.shape
        ctx.dims = (n, h, w, c)
        y = ext.upsample2x_fwd(x, n, h, w, c)
        return y.permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        ext = hip_extension()
        n, h, w, c = ctx.dims
        dx = ext.upsample2x_bwd(dy.contiguous(memory_format=_CL), n, h, w, c)
        return dx.permute(0, 3, 1, 2)


def maxpool2x2_hip(x):
    return MaxPool2x2Fn.apply(x)


def upsample2x_hip(x):
    return Upsample2xFn.apply(x)


# ---------------------------------------------------------------------------
# squeeze-excitation: custom kernels for the HW-sized work (GAP + channel scale
# + their backward reductions); the tiny C<->C/16 FCs run as library GEMMs.
# ---------------------------------------------------------------------------

class _SeGapFn(torch.autograd.Function):
    """pooled[n][c] = mean_hw x — forward reduce kernel, backward broadcast."""

    @staticmethod
    def forward(ctx, x):
        ext = hip_extension()
        x = x.contiguous(memory_format=_CL)
        n, c, h, w = x.shape
        ctx.dims = (n, h, w, c)
        ctx.xdtype = x.dtype
        pooled = ext.se_reduce(x, None, n, h * w, c) / float(h * w)
        return pooled  # fp32 [N, C]

    @staticmethod
    def backward(ctx, dp):
        n, h, w, c = ctx.dims
        # dx[n,c,h,w] = dp[n,c] / HW — a broadcast copy
        dx = (dp.float() / (h * w)).to(ctx.xdtype).view(n, c, 1, 1)
        return dx.expand(n, c, h, w).contiguous(memory_format=_CL)


class _SeScaleFn(torch.autograd.Function):
    """y = x * s[n][c] with s from the sigmoid gate."""

    @staticmethod
    def forward(ctx, x, s):
        ext = hip_extension()
        x = x.contiguous(memory_format=_CL)
        n, c, h, w = x.shape
        s32 = s.float().contiguous()
        # se_scale allocates like x (NCHW shape, channels_last strides)
        y = ext.se_scale(x, s32, None, n, h * w, c)
        ctx.save_for_backward(x, s32)
        ctx.dims = (n, h, w, c)
        ctx.sdtype = s.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_extension()
        x, s32 = ctx.saved_tensors
        n, h, w, c = ctx.dims
        dy = dy.contiguous(memory_format=_CL)
        dx = ext.se_scale(dy, s32, None, n, h * w, c)
        ds = ext.se_reduce(dy, x, n, h * w, c)
        return dx, ds.to(ctx.sdtype)


def se_layer_hip(x, fc1, fc2):
    if not torch.is_grad_enabled():
        # inference: GAP + one fused gate kernel + scale (3 launches instead
        # of ~7; the 256x16 FCs are far below useful GEMM-library sizes)
        ext = hip_extension()
        xc = x.contiguous(memory_format=_CL)
        n, c, h_, w_ = xc.shape
        pooled = ext.se_reduce(xc, None, n, h_ * w_, c)
        # the GAP division rides the gate kernel (pool_scale) — one less launch
        s = ext.se_gate(pooled, fc1.weight, fc1.bias, fc2.weight, fc2.bias,
                        LEAKY_SLOPE, 1.0 / float(h_ * w_))
        return ext.se_scale(xc, s, None, n, h_ * w_, c)
    pooled = _SeGapFn.apply(x)                       # [N, C] fp32
    h = F.leaky_relu(fc1(pooled.to(fc1.weight.dtype)), LEAKY_SLOPE)
    s = torch.sigmoid(fc2(h))
    return _SeScaleFn.apply(x, s)
