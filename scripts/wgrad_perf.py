"""wgrad kernel vs MIOpen igemm_wrw across the IMHN's training shapes."""
import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from improved_body_parts_amd.ops import conv_kernels
CL = torch.channels_last

def bench(fn, iters=30, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

shapes = [  # (n, cin, cout, hw, k, s, d)
    (16, 128, 128, 128, 3, 1, 1),
    (16, 256, 256, 128, 3, 1, 1),
    (16, 128, 128, 128, 3, 1, 5),
    (16, 256, 128, 128, 1, 1, 1),
    (16, 384, 192, 64, 1, 1, 1),
    (16, 192, 192, 64, 3, 1, 1),
    (16, 768, 384, 8, 1, 1, 1),
    (16, 384, 384, 8, 3, 1, 1),
    (16, 3, 64, 512, 7, 2, 1),    # the stem
    (16, 256, 50, 128, 1, 1, 1),  # head
    (16, 50, 256, 128, 1, 1, 1),  # merge
]
for (n, cin, cout, hw, k, s, d) in shapes:
    x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16().contiguous(memory_format=CL)
    pad = (k - 1) // 2 * d
    ho = (hw + 2 * pad - d * (k - 1) - 1) // s + 1
    dy = (torch.randn(n, cout, ho, ho, device="cuda") * 0.1).bfloat16().contiguous(memory_format=CL)
    wshape = (cout, cin, k, k)
    dw = conv_kernels.conv_wgrad(x, dy, wshape, (s, s), (pad, pad), (d, d))
    assert dw is not None
    t_us = bench(lambda: conv_kernels.conv_wgrad(x, dy, wshape, (s, s), (pad, pad), (d, d)))
    t_lib = bench(lambda: torch.nn.grad.conv2d_weight(x, wshape, dy, s, pad, d))
    flops = 2 * n * ho * ho * cout * cin * k * k
    print(f"{(n,cin,cout,hw,k,s,d)}: wgrad {t_us:.3f} ms ({flops/t_us/1e9:.0f} TF) "
          f"| miopen {t_lib:.3f} ms ({flops/t_lib/1e9:.0f} TF) | ratio {t_lib/t_us:.2f}x",
          flush=True)
