"""Fused conv + BatchNorm + LeakyReLU (+ residual) on the HIP path.

The convolution itself goes through :mod:`.conv_kernels` (hand-written MFMA
implicit-GEMM for the supported shapes); BN statistics, the scale/shift+act
epilogue and the whole BN backward run as the fused CDNA4 kernels of
bn_act.hip. Tensors stay NHWC (torch channels_last) end to end.

Replaces: nn.Conv2d -> nn.BatchNorm2d -> nn.LeakyReLU chains
(reference models/layers_transposed.py:90-120) and their autograd.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from ._backend import hip_extension

LEAKY_SLOPE = 0.01
_CL = torch.channels_last


def _to_cl(t):
    return t.contiguous(memory_format=_CL)


def _conv_forward(x, weight, bias, stride, padding, dilation):
    """Convolution forward dispatch: MFMA implicit-GEMM kernel when the shape is
    supported, library fallback otherwise (first iterations / odd shapes)."""
    from . import conv_kernels
    y = conv_kernels.conv_fwd(x, weight, stride, padding, dilation)
    if y is None:
        y = F.conv2d(x, weight, None, stride, padding, dilation)
        y = _to_cl(y)
    if bias is not None:
        y = y + bias.view(1, -1, 1, 1)
    return y


def _conv_dgrad(dy, weight, x_shape, stride, padding, dilation):
    from . import conv_kernels
    dx = conv_kernels.conv_dgrad(dy, weight, x_shape, stride, padding, dilation)
    if dx is None:
        dx = torch.nn.grad.conv2d_input(x_shape, weight, dy, stride, padding,
                                        dilation)
        dx = _to_cl(dx)
    return dx


def _conv_wgrad(x, dy, w_shape, stride, padding, dilation):
    from . import conv_kernels
    dw = conv_kernels.conv_wgrad(x, dy, w_shape, stride, padding, dilation)
    if dw is None:
        dw = torch.nn.grad.conv2d_weight(x, w_shape, dy, stride, padding,
                                         dilation)
    return dw


_BIAS_FOLD = {}


def _bias_fold(weight, bias):
    """(ones, bias.float()) for the bias-only epilogue, cached against the
    bias version — a fresh pair per call was ~60 FillFunctor/copy launches
    per inference step (the 1x1 heads carry biases)."""
    import weakref
    key = id(bias)
    entry = _BIAS_FOLD.get(key)
    ver = bias._version
    if entry is not None and entry[0] == ver:
        return entry[1], entry[2]
    scale = torch.ones(weight.shape[0], device=bias.device, dtype=torch.float32)
    shift = bias.detach().float()
    if entry is None:
        weakref.finalize(bias, _BIAS_FOLD.pop, key, None)
    _BIAS_FOLD[key] = (ver, scale, shift)
    return scale, shift


def _tick_running_stats(bn_mod):
    """The finalize kernel updates running stats in place without going
    through ATen — tick their version counters so the eval-path folded-BN
    cache (keyed on them) does not serve stale scale/shift."""
    torch.autograd.graph.increment_version(bn_mod.running_mean)
    torch.autograd.graph.increment_version(bn_mod.running_var)


class ConvBnActFn(torch.autograd.Function):
    """y = leaky( bn( conv(x, w) ) (+ residual) ) (+ residual_post [+2]),
    all fused on device. ``residual`` joins BEFORE the activation (bottleneck
    skip); ``residual_post``/``residual_post2`` join AFTER it (hourglass
    up1+deconv1, cross-stack feature-cache) — their backward is identity."""

    @staticmethod
    def forward(ctx, x, weight, bias, gamma, beta, residual,
                residual_post, residual_post2,
                stride, padding, dilation, act, training, bn_mod, inference):
        ext = hip_extension()
        x = _to_cl(x)

        # inference fast path: fold BN into the conv epilogue -> ONE kernel.
        # `inference` is computed OUTSIDE apply(): grad mode is always off
        # inside Function.forward, so torch.is_grad_enabled() can't be used here.
        if inference and x.dtype == torch.bfloat16:
            from . import conv_kernels
            if gamma is not None:
                # cache the folded scale/shift on the BN module
                ver = (gamma._version, bn_mod.running_mean._version,
                       bn_mod.running_var._version)
                cached = getattr(bn_mod, "_ibp_folded", None)
                if cached is not None and cached[0] == ver:
                    scale, shift = cached[1], cached[2]
                else:
                    invstd = torch.rsqrt(bn_mod.running_var.float() + bn_mod.eps)
                    scale = gamma.float() * invstd
                    shift = beta.float() - bn_mod.running_mean.float() * scale
                    bn_mod._ibp_folded = (ver, scale, shift)
            elif bias is not None:
                scale, shift = _bias_fold(weight, bias)
            if scale is not None or act or residual is not None \
                    or residual_post is not None:
                y = conv_kernels.conv_fwd(x, weight, stride, padding, dilation,
                                          scale=scale, shift=shift,
                                          residual=residual, act=act,
                                          residual_post=residual_post,
                                          residual_post2=residual_post2)
                if y is not None:
                    ctx.conf = None
                    return y

        y_conv = _conv_forward(x, weight, bias if gamma is None else None,
                               stride, padding, dilation)
        C = y_conv.shape[1]

        sync_group, sync_world = _sync_info(bn_mod, training)
        if gamma is not None:
            if training:
                mom = bn_mod.momentum if bn_mod.momentum is not None else 0.1
                if sync_world > 1:
                    # SyncBN: all-reduce (sums, sumsq) so the fused path keeps
                    # cross-rank statistics (module forward is bypassed here)
                    import torch.distributed as dist
                    sums, sumsq = ext.bn_stats(y_conv, C)
                    flat = torch.cat([sums, sumsq])
                    dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=sync_group)
                    M = (y_conv.numel() // C) * sync_world
                    mean, invstd, scale, shift = ext.bn_finalize(
                        flat[:C], flat[C:], M, gamma.float(), beta.float(),
                        bn_mod.running_mean, bn_mod.running_var,
                        bn_mod.num_batches_tracked, mom, bn_mod.eps)
                    _tick_running_stats(bn_mod)
                else:
                    # single fused kernel chain: stats + colsum + per-channel
                    # epilogue + running-stat update (the Python mean/var/rsqrt
                    # chain was ~6 tiny launches per conv layer)
                    mean, invstd, scale, shift = ext.bn_stats_finalize(
                        y_conv, C, gamma.float(), beta.float(),
                        bn_mod.running_mean, bn_mod.running_var,
                        bn_mod.num_batches_tracked, mom, bn_mod.eps)
                    _tick_running_stats(bn_mod)
            else:
                mean = bn_mod.running_mean.float()
                var = bn_mod.running_var.float()
                invstd = torch.rsqrt(var + bn_mod.eps)
                scale = (gamma.float() * invstd)
                shift = (beta.float() - mean * scale)
        elif act or residual is not None:
            mean = invstd = None
            scale = torch.ones(C, device=x.device, dtype=torch.float32)
            shift = torch.zeros(C, device=x.device, dtype=torch.float32)
        else:
            # plain conv (the 1x1 heads): no epilogue pass needed
            mean = invstd = scale = shift = None

        if scale is not None:
            res_cl = _to_cl(residual) if residual is not None else None
            y = ext.bn_act_fwd(y_conv, scale, shift, res_cl, LEAKY_SLOPE, act)
        else:
            y = y_conv

        # y (pre post-add) is what backward needs to reconstruct the act region
        ctx.save_for_backward(x, weight, gamma, y_conv, y, mean, invstd)
        ctx.conf = (stride, padding, dilation, act, training,
                    residual is not None, bias is not None and gamma is None,
                    residual_post is not None, residual_post2 is not None)
        ctx.sync = (sync_group, sync_world)
        if residual_post is not None:
            y = y + _to_cl(residual_post)
        if residual_post2 is not None:
            y = y + _to_cl(residual_post2)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_extension()
        x, weight, gamma, y_conv, y, mean, invstd = ctx.saved_tensors
        (stride, padding, dilation, act, training, has_res, has_bias,
         has_post, has_post2) = ctx.conf
        C = y_conv.shape[1]
        dy = _to_cl(dy)
        has_bn = gamma is not None
        # post-act residuals are pure adds: their gradient is dy itself
        dpost = dy if has_post else None
        dpost2 = dy if has_post2 else None

        if not has_bn and not act and not has_res and not has_bias:
            # plain conv fast path (the 1x1 heads without bias)
            dx = _conv_dgrad(dy, weight, x.shape, stride, padding, dilation) \
                if ctx.needs_input_grad[0] else None
            dw = _conv_wgrad(x, dy, weight.shape, stride, padding, dilation) \
                if ctx.needs_input_grad[1] else None
            return (dx, dw, None, None, None, None, dpost, dpost2,
                    None, None, None, None, None, None, None)

        dpre, sum_dpre, sum_dxhat = ext.bn_act_bwd(
            dy, y, y_conv, mean if has_bn else None, invstd if has_bn else None,
            LEAKY_SLOPE, act, has_bn, C)

        sync_group, sync_world = getattr(ctx, "sync", (None, 1))
        if has_bn and training and sync_world > 1:
            # SyncBN backward: dx needs GLOBAL sum_dpre/sum_dxhat over the
            # global count; averaging (sum / world) makes the local-M division
            # inside bn_act_bwd_apply equal the global-M division, and the
            # per-rank dgamma/dbeta then DDP-average to the correct value.
            import torch.distributed as dist
            flat = torch.cat([sum_dpre, sum_dxhat])
            dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=sync_group)
            flat = flat / sync_world
            sum_dpre, sum_dxhat = flat[:C], flat[C:]

        dres = dpre if has_res else None
        if has_bn:
            if training:
                dconv = ext.bn_act_bwd_apply(dpre, y_conv, mean, invstd,
                                             gamma.float(), sum_dpre, sum_dxhat, C)
            else:
                dconv = ext.bn_act_bwd_apply(dpre, y_conv, mean, invstd,
                                             gamma.float(), None, None, C)
            dgamma = sum_dxhat.to(gamma.dtype)
            dbeta = sum_dpre.to(gamma.dtype)
        else:
            dconv = dpre
            dgamma = dbeta = None

        dbias = sum_dpre.to(weight.dtype) if has_bias else None
        dx = _conv_dgrad(dconv, weight, x.shape, stride, padding, dilation) \
            if ctx.needs_input_grad[0] else None
        dw = _conv_wgrad(x, dconv, weight.shape, stride, padding, dilation) \
            if ctx.needs_input_grad[1] else None
        return (dx, dw, dbias, dgamma, dbeta, dres, dpost, dpost2,
                None, None, None, None, None, None, None)


def _sync_info(bn_mod, training):
    """(process_group, world_size) when bn_mod is a SyncBatchNorm2d in a
    multi-rank training run; (None, 1) otherwise."""
    if not training or bn_mod is None or not hasattr(bn_mod, "process_group"):
        return None, 1
    import torch.distributed as dist
    if not dist.is_initialized():
        return None, 1
    group = bn_mod.process_group
    world = dist.get_world_size(group)
    return group, world


def conv_bn_act_hip(x, conv, bn, act: bool, residual=None, training: bool = False,
                    residual_post=None, residual_post2=None):
    """Module-level entry used by models.layers: pulls parameters out of the
    nn.Conv2d / nn.BatchNorm2d containers and runs the fused function."""
    gamma = bn.weight if bn is not None else None
    beta = bn.bias if bn is not None else None
    bn_training = training and (bn is not None and bn.training)
    inference = not bn_training and not torch.is_grad_enabled()
    return ConvBnActFn.apply(
        x, conv.weight, conv.bias, gamma, beta, residual,
        residual_post, residual_post2,
        conv.stride, conv.padding, conv.dilation, act,
        bn_training, bn, inference)
