#!/usr/bin/env python3
"""Cross-check the v8 BN kernels against the scalar path on many shapes.

Runs both code paths in one process by toggling the cached env flag through
re-import is impossible — instead we compare v8 against a pure-torch oracle.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from improved_body_parts_amd.ops._backend import hip_extension  # noqa: E402

SHAPES = [
    (2, 64, 32, 32), (2, 96, 32, 32), (1, 96, 17, 17), (2, 128, 16, 16),
    (2, 160, 16, 16), (3, 192, 8, 8), (2, 64, 128, 128), (16, 256, 128, 128),
    (2, 160, 33, 33), (1, 64, 1, 1), (5, 512, 7, 7), (2, 768, 8, 8),
]


def rel(a, b):
    return ((a - b).norm() / (b.norm() + 1e-12)).item()


def main():
    ext = hip_extension()
    bad = 0
    for (n, c, h, w) in SHAPES:
        torch.manual_seed(n * 1000 + c)
        x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        dy = torch.randn_like(x)
        y = torch.randn_like(x)
        M = n * h * w
        xf = x.float()

        # stats oracle
        sums, sumsq = ext.bn_stats(x, c)
        s_ref = xf.sum(dim=(0, 2, 3))
        q_ref = (xf * xf).sum(dim=(0, 2, 3))
        e1, e2 = rel(sums, s_ref), rel(sumsq, q_ref)

        # bwd reduce oracle (act=True, need_xhat=True)
        mean = s_ref / M
        var = (q_ref / M - mean * mean).clamp(min=0)
        invstd = torch.rsqrt(var + 1e-5)
        dpre, sd, sx = ext.bn_act_bwd(dy, y, x, mean, invstd, 0.01, True, True, c)
        dpre_ref = torch.where(y.float() > 0, dy.float(), dy.float() * 0.01)
        sd_ref = dpre_ref.sum(dim=(0, 2, 3))
        xhat = (xf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
        sx_ref = (dpre_ref * xhat).sum(dim=(0, 2, 3))
        e3 = rel(dpre.float(), dpre_ref)
        e4, e5 = rel(sd, sd_ref), rel(sx, sx_ref)

        # apply oracle
        gamma = torch.randn(c, device="cuda")
        dx = ext.bn_act_bwd_apply(dpre, x, mean, invstd, gamma, sd, sx, c)
        g = dpre.float() - sd_ref.view(1, -1, 1, 1) / M \
            - xhat * sx_ref.view(1, -1, 1, 1) / M
        dx_ref = gamma.view(1, -1, 1, 1) * invstd.view(1, -1, 1, 1) * g
        e6 = rel(dx.float(), dx_ref)

        # fwd oracle
        scale = torch.randn(c, device="cuda")
        shift = torch.randn(c, device="cuda")
        yv = ext.bn_act_fwd(x, scale, shift, None, 0.01, True)
        y_ref = torch.nn.functional.leaky_relu(
            xf * scale.view(1, -1, 1, 1) + shift.view(1, -1, 1, 1), 0.01)
        e7 = rel(yv.float(), y_ref)

        errs = [e1, e2, e3, e4, e5, e6, e7]
        flag = "OK " if max(errs) < 2e-2 else "BAD"
        if flag == "BAD":
            bad += 1
        print(f"{flag} {str((n,c,h,w)):>20} stats={e1:.1e}/{e2:.1e} "
              f"dpre={e3:.1e} sums={e4:.1e}/{e5:.1e} apply={e6:.1e} fwd={e7:.1e}")
    print("BAD_SHAPES:", bad)


if __name__ == "__main__":
    main()
