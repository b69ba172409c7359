#!/usr/bin/env python3
"""Hunt for out-of-bounds writes / races in the MFMA conv path."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from improved_body_parts_amd.ops import conv_kernels

CL = torch.channels_last
shapes = [
    (8, 512, 256, 32, 3, 1),    # features.2 conv at scale 2
    (8, 256, 50, 32, 1, 1),     # outs head at scale 2
    (8, 50, 768, 8, 1, 1),      # merge_preds scale 4
    (8, 256, 256, 16, 3, 1),
    (8, 768, 768, 8, 3, 1),
    (8, 640, 640, 16, 3, 1),
]
bad = 0
for (n, cin, cout, hw, k, d) in shapes:
    torch.manual_seed(0)
    pre = torch.full((1 << 20,), 777.0, device="cuda")
    x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    mid = torch.full((1 << 20,), 888.0, device="cuda")
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
    post = torch.full((1 << 20,), 999.0, device="cuda")
    pad = (k - 1) // 2 * d
    ref = F.conv2d(x.float(), w.float(), None, 1, pad, d)
    outs = []
    for it in range(30):
        y = conv_kernels.conv_fwd(x, w, (1, 1), (pad, pad), (d, d))
        assert y is not None
        outs.append(y.float())
    torch.cuda.synchronize()
    e_pairwise = max(float((outs[i] - outs[0]).abs().max()) for i in range(1, 30))
    e_ref = float((outs[0] - ref).norm() / ref.norm())
    c1 = float((pre - 777).abs().max())
    c2 = float((mid - 888).abs().max())
    c3 = float((post - 999).abs().max())
    # dgrad too
    dy = torch.randn(n, cout, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    dgr = [conv_kernels.conv_dgrad(dy, w, (n, cin, hw, hw), (1, 1), (pad, pad), (d, d))
           for _ in range(10)]
    dref = torch.nn.grad.conv2d_input((n, cin, hw, hw), w.float(), dy.float(), 1, pad, d)
    if dgr[0] is None:
        e_dg = e_dgp = -1.0
    else:
        e_dg = float((dgr[0].float() - dref).norm() / dref.norm())
        e_dgp = max(float((dgr[i].float() - dgr[0].float()).abs().max()) for i in range(1, 10))
    status = "OK "
    if e_pairwise > 0 or e_ref > 2e-2 or max(c1, c2, c3) > 0 or e_dg > 2e-2 or e_dgp > 0:
        status = "BAD"
        bad += 1
    print(f"{status} {(n,cin,cout,hw,k,d)} run2run={e_pairwise:.2e} vs_lib={e_ref:.2e} "
          f"canary={max(c1,c2,c3):.1f} dgrad={e_dg:.2e} dgrad_r2r={e_dgp:.2e}")
print("BAD:", bad)
