#!/usr/bin/env python3
"""Targeted sweep: dgrad of the Merge (50 -> C) convs across scales."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from improved_body_parts_amd.ops import conv_kernels

CL = torch.channels_last
print("dgrad of Merge convs: dx(50ch) = dgrad(dy(Cch), w(C,50,1,1))")
for (cout, hw) in [(256, 128), (384, 64), (512, 32), (640, 16), (768, 8)]:
    torch.manual_seed(1)
    n = 8
    w = (torch.randn(cout, 50, 1, 1, device="cuda") * 0.05).bfloat16()
    dy = torch.randn(n, cout, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    dx = conv_kernels.conv_dgrad(dy, w, (n, 50, hw, hw), (1, 1), (0, 0), (1, 1))
    ref = torch.nn.grad.conv2d_input((n, 50, hw, hw), w.float(), dy.float(), 1, 0, 1)
    if dx is None:
        print(f"C={cout} hw={hw}: fallback"); continue
    rel = float((dx.float() - ref).norm() / ref.norm())
    mx = float(dx.float().abs().max())
    print(f"C={cout} hw={hw}: rel={rel:.3e} max|dx|={mx:.3e} (ref max {float(ref.abs().max()):.3e})")
