import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
import torch.nn.functional as F
from improved_body_parts_amd.ops import conv_kernels
CL = torch.channels_last

def bench(fn, iters=30, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

shapes = [  # (n, cin, cout, hw, k, s, d) — hottest IMHN layers at batch 4 & 16
    (16, 128, 128, 128, 3, 1, 1),
    (16, 256, 256, 128, 3, 1, 1),
    (16, 128, 128, 128, 3, 1, 5),
    (16, 256, 128, 128, 1, 1, 1),
    (16, 384, 192, 64, 1, 1, 1),
    (16, 192, 192, 64, 3, 1, 1),
    (16, 768, 384, 8, 1, 1, 1),
    (16, 384, 384, 8, 3, 1, 1),
    (4, 256, 256, 128, 3, 1, 1),
    (16, 3, 64, 512, 7, 2, 1),   # the stem (space-to-depth path)
]
for (n, cin, cout, hw, k, s, d) in shapes:
    x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
    pad = (k - 1) // 2 * d
    y1 = conv_kernels.conv_fwd(x, w, (s, s), (pad, pad), (d, d))
    assert y1 is not None
    t_mfma = bench(lambda: conv_kernels.conv_fwd(x, w, (s, s), (pad, pad), (d, d)))
    wcl = w.contiguous(memory_format=CL)
    t_lib = bench(lambda: F.conv2d(x, wcl, None, s, pad, d))
    flops = 2 * n * hw * hw / (s * s) * cout * cin * k * k
    print(f"{(n,cin,cout,hw,k,s,d)}: mfma {t_mfma:.3f} ms ({flops/t_mfma/1e9:.0f} TF) "
          f"| miopen {t_lib:.3f} ms ({flops/t_lib/1e9:.0f} TF) | ratio {t_lib/t_mfma:.2f}x")
