#!/usr/bin/env python3
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from improved_body_parts_amd.ops import conv_kernels
CL = torch.channels_last

def probe(n, cin, cout, hw):
    torch.manual_seed(1)
    x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    w = (torch.randn(cout, cin, 1, 1, device="cuda") * 0.05).bfloat16()
    y = conv_kernels.conv_fwd(x, w, (1, 1), (0, 0), (1, 1))
    ref = F.conv2d(x.float(), w.float())
    if y is None:
        print(f"cin={cin} cout={cout} hw={hw}: fallback"); return
    d = (y.float() - ref).abs()
    rel = float(d.norm() / ref.norm())
    M = n * hw * hw
    BN = 128 if (cout > 64 and ((cout+127)//128*128) <= ((cout+63)//64*64)) else 64
    ntiles = ((M + 127)//128) * ((cout + BN - 1)//BN)
    nk = (cin + 63)//64
    sk = 1
    if ntiles < 384 and nk > 1:
        sk = min(nk, (384 + ntiles - 1)//ntiles)
        ch = (nk + sk - 1)//sk
        sk = (nk + ch - 1)//ch
    print(f"cin={cin:5d} cout={cout:3d} hw={hw:3d} M={M:6d} BN={BN} nk={nk:2d} splitk={sk} rel={rel:.3e}")
    if rel > 1e-2:
        bad = (d > 0.1).float()  # [n, cout, hw, hw]
        bad_m = bad.sum(dim=1).flatten()          # per output pixel
        mt = bad_m.reshape(-1, 128 if bad_m.numel() % 128 == 0 else 1).sum(dim=1)
        nz = (mt > 0).nonzero().flatten()
        print(f"   bad m-tiles: {nz[:10].tolist()} of {mt.numel()} (count {len(nz)})")
        bad_c = bad.sum(dim=(0, 2, 3))
        print(f"   bad cols: first/last {bad_c.nonzero().flatten()[:5].tolist()} ... total {int((bad_c>0).sum())}")

for args in [(8,512,50,32), (8,512,64,32), (8,512,50,45), (8,512,50,64),
             (8,1024,50,32), (8,256,50,32), (8,512,33,32), (4,512,50,32),
             (8,512,128,32), (16,512,50,32)]:
    probe(*args)
