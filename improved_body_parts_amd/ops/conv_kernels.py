"""MFMA implicit-GEMM convolution dispatch.

The CDNA4 kernels (csrc/conv_mfma.hip) implement NHWC bf16 convolution as an
implicit GEMM on the bf16 matrix cores:
    C[M=N*Ho*Wo][N=Cout] = A[M][K=KH*KW*Cin] @ B[K][Cout]
with fwd, dgrad (stride-1: conv with 180-rotated transposed weights) and
split-K wgrad. Weights are re-packed to [KH*KW*Cin][Cout] (Cout contiguous)
per forward; the pack is cached against the weight tensor's version counter.

Each entry point returns ``None`` when the shape is outside the kernel's
envelope, and the caller falls back to the library conv.
"""
from __future__ import annotations

import os

import torch

from ._backend import hip_extension

_CL = torch.channels_last

# shapes supported by the MFMA path: stride-1 any dilation, and stride-2 fwd
DISABLE = os.environ.get("IBP_AMD_DISABLE_MFMA_CONV") == "1"


def _supported(x, weight, stride, padding, dilation, for_grad=False):
    """Envelope of the MFMA kernels: bf16, square stride. Any Cin (subpieces
    crossing filter-tap boundaries — the 7x7 Cin=3 stem — take the kernel's
    per-element gather path), any padding (Ho/Wo are passed explicitly)."""
    if DISABLE:
        return False
    if x.dtype != torch.bfloat16:
        return False
    if stride[0] != stride[1]:
        return False
    return True


import weakref

_pack_cache = {}


def _cached_pack(weight: torch.Tensor, tag: str, pack_fn) -> torch.Tensor:
    """Cache packs per weight tensor, invalidated by the autograd version
    counter; a weakref finalizer drops the entry when the weight dies (a bare
    id() key would alias recycled ids)."""
    key = (id(weight), tag)
    entry = _pack_cache.get(key)
    ver = weight._version
    if entry is not None and entry[0] == ver:
        return entry[1]
    packed = pack_fn(weight.detach().to(torch.bfloat16))
    if entry is None:
        weakref.finalize(weight, _pack_cache.pop, key, None)
    _pack_cache[key] = (ver, packed)
    return packed


def _pack_pair(weight: torch.Tensor):
    """Both packs of a weight, regenerated when its version changes.

    On GPU bf16 weights this is ONE pack_weight_kernel launch into two
    persistent buffers; weights change every optimizer step, so the previous
    per-pack eager flip/permute/reshape chains re-ran ~500 small kernels per
    training step (~1.5% in launch overhead alone)."""
    key = (id(weight), "pair")
    entry = _pack_cache.get(key)
    ver = weight._version
    if entry is not None and entry[0] == ver:
        return entry[1], entry[2]
    cout, cin, kh, kw = weight.shape
    w = weight.detach()
    use_kernel = (w.is_cuda and w.dtype == torch.bfloat16
                  and w.is_contiguous())
    if use_kernel:
        if entry is not None and entry[3]:
            fwd_buf, dgr_buf = entry[1], entry[2]
        else:
            fwd_buf = torch.empty(cout, kh * kw * cin, dtype=torch.bfloat16,
                                  device=w.device)
            dgr_buf = torch.empty(cin, kh * kw * cout, dtype=torch.bfloat16,
                                  device=w.device)
        hip_extension().pack_conv_weight(w, fwd_buf, dgr_buf)
    else:
        wd = w.to(torch.bfloat16)
        fwd_buf = wd.permute(0, 2, 3, 1).reshape(cout, -1).contiguous()
        dgr_buf = torch.flip(wd, dims=(2, 3)).permute(1, 2, 3, 0) \
            .reshape(cin, -1).contiguous()
    if entry is None:
        weakref.finalize(weight, _pack_cache.pop, key, None)
    _pack_cache[key] = (ver, fwd_buf, dgr_buf, use_kernel)
    return fwd_buf, dgr_buf


def packed_weight(weight: torch.Tensor) -> torch.Tensor:
    """[Cout, Cin, kh, kw] -> bf16 [Cout][kh*kw*Cin] contiguous (N-major for
    the kernel's B-tile row loads)."""
    return _pack_pair(weight)[0]


def packed_weight_dgrad(weight: torch.Tensor) -> torch.Tensor:
    """Weights for dgrad-as-conv: rotate 180° spatially, swap Cin/Cout ->
    [Cin][kh*kw*Cout]."""
    return _pack_pair(weight)[1]


# ---------------------------------------------------------------------------
# 7x7 s2 stem as a space-to-depth 4x4 s1 conv
# ---------------------------------------------------------------------------
# The Cin=3 stem forces the kernel's per-element tap gather (Cin % 8 != 0).
# Rewriting x (N,3,H,W) as S (N,12->16,H/2,W/2) with channel
# c' = (py*2+px)*3 + c turns it into a 4x4 stride-1 conv over 16 channels
# (kh = 2*kh' + py - 1, effective pad (2 left, 1 right)) — every subpiece is
# tap-uniform and 16-B vectorizable. Same trick for wgrad; dgrad is never
# needed (the stem is the input layer).

def _is_stem(x, weight, stride, padding, dilation):
    return (weight.shape[2] == 7 and weight.shape[3] == 7
            and stride[0] == 2 and stride[1] == 2 and x.shape[1] == 3
            and padding[0] == 3 and padding[1] == 3
            and dilation[0] == 1 and dilation[1] == 1
            and x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0)


def _stem_s2d_input(x):
    """(N,3,H,W) channels_last -> (N,16,H/2,W/2) channels_last (padded c')."""
    import torch.nn.functional as F
    n, _, h, w = x.shape
    v = x.contiguous(memory_format=_CL).permute(0, 2, 3, 1)  # N,H,W,3
    v = v.reshape(n, h // 2, 2, w // 2, 2, 3).permute(0, 1, 3, 2, 4, 5)
    v = v.reshape(n, h // 2, w // 2, 12)
    v = F.pad(v, (0, 4))                                     # c' 12..15 = 0
    return v.permute(0, 3, 1, 2)  # NCHW view over NHWC storage


_STEM_LUT = {}


def _stem_weight_lut(device):
    """Flat source indices: W'[c'][kh'][kw'] position backing W[c][kh][kw]."""
    lut = _STEM_LUT.get(device)
    if lut is not None:
        return lut
    import numpy as np
    src = np.zeros(3 * 7 * 7, np.int64)
    for c in range(3):
        for kh in range(7):
            for kw in range(7):
                py, kh2 = (kh + 1) & 1, (kh + 1) >> 1
                px, kw2 = (kw + 1) & 1, (kw + 1) >> 1
                cp = (py * 2 + px) * 3 + c
                src[(c * 7 + kh) * 7 + kw] = (cp * 4 + kh2) * 4 + kw2
    lut = torch.from_numpy(src).to(device)
    _STEM_LUT[device] = lut
    return lut


def _stem_pack_weight(weight):
    """[64,3,7,7] -> packed bf16 [64][K=4*4*16] for the s2d conv."""
    def pack(w):
        cout = w.shape[0]
        wp = torch.zeros(cout, 16, 4, 4, dtype=w.dtype, device=w.device)
        lut = _stem_weight_lut(w.device)
        wp.view(cout, -1)[:, lut] = w.reshape(cout, -1)
        # -> [cout][kh'*kw'*c'] (tap-major, channel fastest) like packed_weight
        return wp.permute(0, 2, 3, 1).reshape(cout, -1).contiguous()
    return _cached_pack(weight, "s2d", pack)


def _stem_fwd(x, weight, scale, shift, residual, act,
              residual_post, residual_post2):
    ext = hip_extension()
    s = _stem_s2d_input(x).contiguous(memory_format=_CL)
    n, _, hs, ws = s.shape
    cout = weight.shape[0]
    if residual is not None:
        residual = residual.contiguous(memory_format=_CL)
    if residual_post is not None:
        residual_post = residual_post.contiguous(memory_format=_CL)
    if residual_post2 is not None:
        residual_post2 = residual_post2.contiguous(memory_format=_CL)
    y = ext.conv_mfma_fwd(s, _stem_pack_weight(weight), n, hs, ws, 16, cout,
                          4, 4, 1, 2, 2, 1, 1, hs, ws,
                          scale, shift, residual, act,
                          residual_post, residual_post2, 1)
    return y.permute(0, 3, 1, 2)


def _stem_wgrad(x, dy, w_shape):
    ext = hip_extension()
    s = _stem_s2d_input(x).contiguous(memory_format=_CL)
    dy = dy.contiguous(memory_format=_CL)
    n, _, hs, ws = s.shape
    cout = w_shape[0]
    dwp = ext.conv_mfma_wgrad(s, dy, n, hs, ws, 16, cout, 4, 4,
                              1, 2, 2, 1, 1, dy.shape[2], dy.shape[3])
    lut = _stem_weight_lut(x.device)
    return dwp.reshape(cout, -1)[:, lut].reshape(cout, 3, 7, 7)


def conv_fwd(x, weight, stride, padding, dilation,
             scale=None, shift=None, residual=None, act=False,
             residual_post=None, residual_post2=None):
    """MFMA conv forward; with scale/shift/residual/act set, the folded-BN
    (+residual +leaky) epilogue runs inside the conv kernel — the whole
    Conv+BN+LeakyReLU module is ONE kernel on the inference path.
    ``residual_post``/``residual_post2`` are added AFTER the activation (the
    hourglass up1+deconv1 join and the cross-stack feature-cache add)."""
    if not _supported(x, weight, stride, padding, dilation):
        return None
    ext = hip_extension()
    if not hasattr(ext, "conv_mfma_fwd"):
        return None
    if _is_stem(x, weight, stride, padding, dilation):
        return _stem_fwd(x, weight, scale, shift, residual, act,
                         residual_post, residual_post2)
    x = x.contiguous(memory_format=_CL)
    n, cin, h, w_ = x.shape
    cout, _, kh, kw = weight.shape
    ho = (h + 2 * padding[0] - dilation[0] * (kh - 1) - 1) // stride[0] + 1
    wo = (w_ + 2 * padding[1] - dilation[1] * (kw - 1) - 1) // stride[1] + 1
    if residual is not None:
        residual = residual.contiguous(memory_format=_CL)
    if residual_post is not None:
        residual_post = residual_post.contiguous(memory_format=_CL)
    if residual_post2 is not None:
        residual_post2 = residual_post2.contiguous(memory_format=_CL)
    y = ext.conv_mfma_fwd(x, packed_weight(weight), n, h, w_, cin, cout,
                          kh, kw, stride[0], padding[0], padding[1],
                          dilation[0], dilation[1], ho, wo,
                          scale, shift, residual, act,
                          residual_post, residual_post2, 1)
    return y.permute(0, 3, 1, 2)  # NHWC buffer -> NCHW view (channels_last)


def conv_dgrad(dy, weight, x_shape, stride, padding, dilation):
    """dx = stride-1 conv of the (virtually) zero-dilated dy with 180-rotated
    transposed weights; pad' = dil*(K-1) - pad, gather stride zs = stride."""
    if not _supported(dy, weight, stride, padding, dilation, for_grad=True):
        return None
    ext = hip_extension()
    if not hasattr(ext, "conv_mfma_fwd"):
        return None
    dy = dy.contiguous(memory_format=_CL)
    n, cout, h, w_ = dy.shape
    cin, hx, wx = x_shape[1], x_shape[2], x_shape[3]
    kh, kw = weight.shape[2], weight.shape[3]
    pad_h = dilation[0] * (kh - 1) - padding[0]
    pad_w = dilation[1] * (kw - 1) - padding[1]
    if pad_h < 0 or pad_w < 0:
        return None
    dx = ext.conv_mfma_fwd(dy, packed_weight_dgrad(weight), n, h, w_, cout, cin,
                           kh, kw, 1, pad_h, pad_w,
                           dilation[0], dilation[1], hx, wx,
                           None, None, None, False, None, None, stride[0])
    return dx.permute(0, 3, 1, 2)


def conv_wgrad(x, dy, w_shape, stride, padding, dilation):
    """Hand-written MFMA weight gradient (csrc/conv_wgrad.hip).

    The contraction runs over M = N*Ho*Wo — the strided dimension of both
    NHWC operands — so fragments are staged through LDS [64 m][16 ch]
    subtiles with a permuted row order and consumed with gfx950's hardware
    transpose read ``ds_read_b64_tr_b16`` (round 1 left this on MIOpen
    igemm_wrw; the builtin ``__builtin_amdgcn_ds_read_tr16_b64_v4i16``
    makes the layout tractable). Any KHxKW/stride/dilation; bf16 only."""
    if DISABLE or x.dtype != torch.bfloat16 or dy.dtype != torch.bfloat16:
        return None
    if stride[0] != stride[1]:
        return None
    ext = hip_extension()
    if not hasattr(ext, "conv_mfma_wgrad"):
        return None
    if w_shape[2] == 7 and w_shape[3] == 7 and x.shape[1] == 3 \
            and stride[0] == 2 and padding[0] == 3 and dilation[0] == 1 \
            and x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0:
        return _stem_wgrad(x, dy, w_shape)
    x = x.contiguous(memory_format=_CL)
    dy = dy.contiguous(memory_format=_CL)
    n, cin, h, w_ = x.shape
    cout, _, kh, kw = w_shape
    ho, wo = dy.shape[2], dy.shape[3]
    return ext.conv_mfma_wgrad(x, dy, n, h, w_, cin, cout, kh, kw,
                               stride[0], padding[0], padding[1],
                               dilation[0], dilation[1], ho, wo)
