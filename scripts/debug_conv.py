import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, sys
import torch.nn.functional as F
from improved_body_parts_amd.ops import conv_kernels
CL = torch.channels_last
cases = [
    (2, 64, 64, 16, 1, 1, 1),
    (2, 64, 128, 32, 3, 1, 1),
    (1, 128, 128, 32, 3, 1, 3),
    (2, 256, 50, 32, 1, 1, 1),
    (1, 50, 256, 16, 1, 1, 1),
    (2, 64, 64, 64, 1, 2, 1),
    (1, 256, 77, 16, 1, 1, 1),
    (3, 192, 320, 20, 3, 1, 1),
    (2, 384, 384, 8, 3, 1, 1),
]
for c in cases:
    n, cin, cout, hw, k, s, d = c
    torch.manual_seed(0)
    x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
    pad = (k - 1) // 2 * d
    print("case", c, flush=True)
    y = conv_kernels.conv_fwd(x, w, (s, s), (pad, pad), (d, d))
    torch.cuda.synchronize()
    assert y is not None, c
    ref = F.conv2d(x.float(), w.float(), None, s, pad, d)
    err = (y.float() - ref).norm().item() / (ref.norm().item() + 1e-9)
    print("  rel err", err, flush=True)
print("ALL OK")

# region diagnosis for the Cout=77 case
n, cin, cout, hw, k, s, d = (1, 256, 77, 16, 1, 1, 1)
torch.manual_seed(0)
x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
y = conv_kernels.conv_fwd(x, w, (s, s), (0, 0), (d, d)).float()
ref = F.conv2d(x.float(), w.float(), None, s, 0, d)
err = (y - ref).abs()
print("err by col block:", [round(err[:, c0:c0+16].max().item(), 3) for c0 in range(0, 77, 16)])
e2 = err.permute(0, 2, 3, 1).reshape(-1, 77)
print("err by m half:", err.shape, round(e2[:128].max().item(), 3), round(e2[128:].max().item(), 3))
bad = (e2.max(dim=1).values > 0.5).nonzero().flatten()
print("bad rows:", bad[:20].tolist(), "count", len(bad))
badc = (e2.max(dim=0).values > 0.5).nonzero().flatten()
print("bad cols:", badc[:30].tolist(), "count", len(badc))
