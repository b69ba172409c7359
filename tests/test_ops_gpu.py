"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 oracle.

Run on an MI355X: python -m pytest tests -m gpu -x -q
"""
import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from improved_body_parts_amd import ops
    from improved_body_parts_amd.ops import _backend

CL = torch.channels_last


def _assert_close(got, ref, rtol, atol, what=""):
    got = got.detach().float().cpu()
    ref = ref.detach().float().cpu()
    err = (got - ref).abs().max().item()
    denom = ref.abs().max().item() + 1e-8
    assert torch.allclose(got, ref, rtol=rtol, atol=atol), \
        f"{what}: max abs err {err:.3e} (ref max {denom:.3e})"


def _assert_rel(got, ref, rel, what=""):
    """Relative-L2 comparison — the right metric for deep-net tensors where a
    few elements of a large reduction legitimately differ in low precision."""
    got = got.detach().float().cpu()
    ref = ref.detach().float().cpu()
    num = (got - ref).norm().item()
    den = ref.norm().item() + 1e-12
    assert num / den < rel, f"{what}: rel L2 err {num / den:.3e} (limit {rel})"


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    assert _backend.hip_available(), \
        "HIP extension must be built and loaded on a GPU box"


# ---------------------------------------------------------------------------
# fused conv+bn+act vs eager fp32
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-4), (torch.bfloat16, 5e-2)])
@pytest.mark.parametrize("training", [True, False])
def test_conv_bn_act_forward_backward(dtype, tol, training):
    torch.manual_seed(0)
    n, cin, cout, hw = 2, 16, 32, 16
    conv = torch.nn.Conv2d(cin, cout, 3, padding=1, bias=False)
    bn = torch.nn.BatchNorm2d(cout)
    conv_g = torch.nn.Conv2d(cin, cout, 3, padding=1, bias=False).cuda().to(dtype)
    bn_g = torch.nn.BatchNorm2d(cout).cuda()  # BN affine/stats stay fp32
    conv_g.load_state_dict({k: v.to(dtype) for k, v in conv.state_dict().items()})
    bn_g.load_state_dict(bn.state_dict())
    bn.train(training), bn_g.train(training)

    x = torch.randn(n, cin, hw, hw)
    xg = x.cuda().to(dtype).contiguous(memory_format=CL).requires_grad_(True)
    xr = x.clone().requires_grad_(True)

    y = ops.conv_bn_act(xg, conv_g, bn_g, act=True, training=training)
    yr = F.leaky_relu(bn(conv(xr)), 0.01)
    _assert_rel(y, yr, max(tol / 2, 5e-4), "fwd")

    dy = torch.randn_like(yr)
    yr.backward(dy)
    y.backward(dy.cuda().to(dtype))
    _assert_rel(xg.grad, xr.grad, max(tol, 1e-3), "dx")
    _assert_rel(conv_g.weight.grad, conv.weight.grad, max(tol, 1e-3), "dw")
    _assert_rel(bn_g.weight.grad, bn.weight.grad, max(tol, 1e-3), "dgamma")
    _assert_rel(bn_g.bias.grad, bn.bias.grad, max(tol, 1e-3), "dbeta")
    if training:
        _assert_close(bn_g.running_mean, bn.running_mean, tol, tol, "running_mean")
        _assert_close(bn_g.running_var, bn.running_var, tol, tol, "running_var")


def test_fused_eval_conv_bn_epilogue():
    """Inference no_grad path folds BN (+residual +leaky) into the conv kernel."""
    torch.manual_seed(4)
    conv = torch.nn.Conv2d(64, 128, 3, padding=1, bias=False)
    bn = torch.nn.BatchNorm2d(128)
    bn.running_mean.normal_(0, 0.5)
    bn.running_var.uniform_(0.5, 2.0)
    conv_g = torch.nn.Conv2d(64, 128, 3, padding=1, bias=False).cuda().bfloat16()
    bn_g = torch.nn.BatchNorm2d(128).cuda()
    conv_g.load_state_dict({k: v.bfloat16() for k, v in conv.state_dict().items()})
    bn_g.load_state_dict(bn.state_dict())
    conv.eval(), bn.eval(), conv_g.eval(), bn_g.eval()
    x = torch.randn(2, 64, 32, 32)
    xg = x.cuda().bfloat16().contiguous(memory_format=CL)
    with torch.no_grad():
        y = ops.conv_bn_act(xg, conv_g, bn_g, act=True, training=False)
        yr = F.leaky_relu(bn(conv(x)), 0.01)
    _assert_rel(y, yr, 2e-2, "fused eval conv+bn+act")
    # with residual
    r = torch.randn(2, 128, 32, 32)
    rg = r.cuda().bfloat16().contiguous(memory_format=CL)
    with torch.no_grad():
        y2 = ops.conv_bn_add_act(xg, conv_g, bn_g, rg, act=True, training=False)
        yr2 = F.leaky_relu(bn(conv(x)) + r, 0.01)
    _assert_rel(y2, yr2, 2e-2, "fused eval conv+bn+res+act")


def test_conv_bn_add_act_residual():
    torch.manual_seed(1)
    conv = torch.nn.Conv2d(8, 8, 1, bias=False)
    bn = torch.nn.BatchNorm2d(8)
    conv_g = torch.nn.Conv2d(8, 8, 1, bias=False).cuda()
    bn_g = torch.nn.BatchNorm2d(8).cuda()
    conv_g.load_state_dict(conv.state_dict())
    bn_g.load_state_dict(bn.state_dict())
    x = torch.randn(2, 8, 8, 8)
    r = torch.randn(2, 8, 8, 8)
    xg = x.cuda().contiguous(memory_format=CL).requires_grad_(True)
    rg = r.cuda().contiguous(memory_format=CL).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    rr = r.clone().requires_grad_(True)
    y = ops.conv_bn_add_act(xg, conv_g, bn_g, rg, act=True, training=True)
    yr = F.leaky_relu(bn(conv(xr)) + rr, 0.01)
    _assert_close(y, yr, 1e-4, 1e-4, "fwd")
    dy = torch.randn_like(yr)
    yr.backward(dy)
    y.backward(dy.cuda())
    _assert_close(rg.grad, rr.grad, 1e-4, 1e-4, "dres")
    _assert_close(xg.grad, xr.grad, 1e-3, 1e-4, "dx")


def test_mfma_conv_shapes_vs_miopen():
    """MFMA implicit-GEMM conv vs the library conv across the model's shapes."""
    from improved_body_parts_amd.ops import conv_kernels
    ext = _backend.hip_extension()
    if not hasattr(ext, "conv_mfma_fwd"):
        pytest.skip("MFMA conv not built yet")
    cases = [
        (2, 64, 64, 64, 1, 1, 1),     # 1x1
        (2, 64, 128, 32, 3, 1, 1),    # 3x3
        (1, 128, 128, 128, 3, 1, 3),  # dilated 3
        (1, 128, 128, 64, 3, 1, 5),   # dilated 5
        (2, 384, 384, 8, 3, 1, 1),    # small spatial, wide channels
        (2, 256, 50, 32, 1, 1, 1),    # head 1x1 to 50ch
        (1, 50, 256, 32, 1, 1, 1),    # merge 1x1 from 50ch (K-tail)
        (2, 64, 64, 64, 1, 2, 1),     # 1x1 stride 2
        (1, 256, 77, 16, 1, 1, 1),    # odd Cout tail
        (3, 192, 320, 20, 3, 1, 1),   # odd M tail (3*20*20=1200 pixels)
        (2, 3, 64, 128, 7, 2, 1),     # 7x7 s2 stem (per-element tap gather)
        (2, 16, 64, 64, 4, 1, 1),     # Cin=16 multi-tap (s2d-stem shape)
        (2, 256, 256, 16, 3, 1, 1),   # BM=32 small-M tile
        (2, 320, 256, 24, 3, 1, 1),   # BM=64 tile (M=1152... 2*576=1152->32; 24^2*2=1152)
    ]
    for n, cin, cout, hw, k, s, d in cases:
        torch.manual_seed(0)
        x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16() \
            .contiguous(memory_format=CL)
        w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
        pad = (k - 1) // 2 * d
        y = conv_kernels.conv_fwd(x, w, (s, s), (pad, pad), (d, d))
        assert y is not None, f"shape not covered: {(n, cin, cout, hw, k, s, d)}"
        ref = F.conv2d(x.float(), w.float(), None, s, pad, d)
        _assert_rel(y, ref, 1e-2, f"conv {(n, cin, cout, hw, k, s, d)}")


def test_mfma_conv_dgrad_vs_reference():
    from improved_body_parts_amd.ops import conv_kernels
    ext = _backend.hip_extension()
    if not hasattr(ext, "conv_mfma_fwd"):
        pytest.skip("MFMA conv not built yet")
    for (cin, cout, hw, k, s, d) in [(64, 128, 32, 3, 1, 1),
                                     (128, 128, 16, 3, 1, 3),
                                     (64, 50, 16, 1, 1, 1),
                                     (50, 64, 16, 1, 1, 1),
                                     (64, 50, 16, 3, 1, 1),   # Cout=50 KxK tail
                                     (16, 64, 32, 3, 2, 1),   # stride-2 (zs gather)
                                     (64, 128, 32, 1, 2, 1)]: # 1x1 stride-2
        torch.manual_seed(1)
        pad = (k - 1) // 2 * d
        hw_o = (hw + 2 * pad - d * (k - 1) - 1) // s + 1
        dy = torch.randn(2, cout, hw_o, hw_o, device="cuda").bfloat16() \
            .contiguous(memory_format=CL)
        w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
        dx = conv_kernels.conv_dgrad(dy, w, (2, cin, hw, hw), (s, s),
                                     (pad, pad), (d, d))
        assert dx is not None, f"dgrad not covered: {(cin, cout, hw, k, s, d)}"
        ref = torch.nn.grad.conv2d_input((2, cin, hw, hw), w.float(),
                                         dy.float(), s, pad, d)
        _assert_rel(dx, ref, 1e-2, f"dgrad {(cin, cout, hw, k, s, d)}")


@pytest.mark.parametrize("training", [False, True])
def test_conv_post_act_residuals(training):
    """residual_post/residual_post2 join AFTER the activation — fused into the
    conv epilogue on the inference path, identity-gradient adds in training."""
    torch.manual_seed(11)
    conv = torch.nn.Conv2d(64, 64, 3, padding=1, bias=False)
    bn = torch.nn.BatchNorm2d(64)
    conv_g = torch.nn.Conv2d(64, 64, 3, padding=1, bias=False).cuda().bfloat16()
    bn_g = torch.nn.BatchNorm2d(64).cuda()
    conv_g.load_state_dict({k: v.bfloat16() for k, v in conv.state_dict().items()})
    bn_g.load_state_dict(bn.state_dict())
    if not training:
        conv.eval(), bn.eval(), conv_g.eval(), bn_g.eval()
    x = torch.randn(2, 64, 32, 32)
    r1 = torch.randn(2, 64, 32, 32)
    r2 = torch.randn(2, 64, 32, 32)
    xg = x.cuda().bfloat16().contiguous(memory_format=CL).requires_grad_(training)
    r1g = r1.cuda().bfloat16().contiguous(memory_format=CL).requires_grad_(training)
    r2g = r2.cuda().bfloat16().contiguous(memory_format=CL).requires_grad_(training)
    ctx = torch.enable_grad() if training else torch.no_grad()
    with ctx:
        y = ops.conv_bn_act(xg, conv_g, bn_g, act=True, training=training,
                            residual_post=r1g, residual_post2=r2g)
    yr = F.leaky_relu(bn(conv(x)), 0.01) + r1 + r2
    _assert_rel(y, yr, 2e-2, f"post-act residuals (training={training})")
    if training:
        y.float().pow(2).sum().backward()
        # post-residual grads are identity: dL/dr = dy
        dy_ref = (2 * y.detach().float())
        _assert_rel(r1g.grad, dy_ref, 2e-2, "dres_post identity")
        _assert_rel(r2g.grad, dy_ref, 2e-2, "dres_post2 identity")


def test_mfma_conv_fuzz_shapes():
    """Randomized shapes across the kernel envelope (tile heuristics, tap
    walk, tails) vs the library conv — guards the gather's many branches."""
    from improved_body_parts_amd.ops import conv_kernels
    rng = np.random.RandomState(20)
    for trial in range(12):
        k = int(rng.choice([1, 1, 3, 3, 5, 7]))
        cin = int(rng.choice([8, 16, 24, 48, 64, 96, 128, 50] if k == 1
                             else [8, 16, 48, 64, 128]))
        cout = int(rng.choice([16, 50, 64, 77, 128, 192]))
        hw = int(rng.choice([7, 8, 12, 16, 24, 32, 48]))
        n = int(rng.randint(1, 4))
        d = int(rng.choice([1, 1, 1, 2, 3])) if k == 3 else 1
        s = int(rng.choice([1, 1, 2]))
        pad = (k - 1) // 2 * d
        if hw + 2 * pad < d * (k - 1) + 1:
            continue
        torch.manual_seed(trial)
        x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16() \
            .contiguous(memory_format=CL)
        w = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
        y = conv_kernels.conv_fwd(x, w, (s, s), (pad, pad), (d, d))
        assert y is not None, (n, cin, cout, hw, k, s, d)
        ref = F.conv2d(x.float(), w.float(), None, s, pad, d)
        _assert_rel(y, ref, 1.5e-2, f"fuzz fwd {(n, cin, cout, hw, k, s, d)}")
        if s == 1:
            dy = (torch.randn_like(ref) * 0.1).bfloat16() \
                .contiguous(memory_format=CL)
            dw = conv_kernels.conv_wgrad(x, dy, w.shape, (s, s), (pad, pad),
                                         (d, d))
            refw = torch.nn.grad.conv2d_weight(x.float(), w.shape, dy.float(),
                                               s, pad, d)
            _assert_rel(dw, refw, 2e-2, f"fuzz wgrad {(n, cin, cout, hw, k, d)}")


def test_mfma_conv_rectangular_inputs():
    """Non-square H x W — the real inference path pads to arbitrary x64
    aspect ratios (reference predict() / padRightDownCorner)."""
    from improved_body_parts_amd.ops import conv_kernels
    for (n, cin, cout, h, w, k, s, d) in [
        (2, 64, 128, 24, 40, 3, 1, 1),
        (1, 128, 128, 16, 48, 3, 1, 3),
        (2, 256, 64, 12, 20, 1, 1, 1),
        (1, 3, 64, 96, 160, 7, 2, 1),   # stem s2d, rectangular
    ]:
        torch.manual_seed(8)
        x = torch.randn(n, cin, h, w, device="cuda").bfloat16() \
            .contiguous(memory_format=CL)
        wt = (torch.randn(cout, cin, k, k, device="cuda") * 0.05).bfloat16()
        pad = (k - 1) // 2 * d
        y = conv_kernels.conv_fwd(x, wt, (s, s), (pad, pad), (d, d))
        ref = F.conv2d(x.float(), wt.float(), None, s, pad, d)
        _assert_rel(y, ref, 1.5e-2, f"rect fwd {(n, cin, cout, h, w, k, s, d)}")
        dy = (torch.randn_like(ref) * 0.1).bfloat16().contiguous(memory_format=CL)
        dw = conv_kernels.conv_wgrad(x, dy, wt.shape, (s, s), (pad, pad), (d, d))
        refw = torch.nn.grad.conv2d_weight(x.float(), wt.shape, dy.float(),
                                           s, pad, d)
        _assert_rel(dw, refw, 2e-2, f"rect wgrad {(n, cin, cout, h, w, k, s, d)}")
        if s == 1:
            dx = conv_kernels.conv_dgrad(dy, wt, x.shape, (s, s), (pad, pad),
                                         (d, d))
            refx = torch.nn.grad.conv2d_input(x.shape, wt.float(), dy.float(),
                                              s, pad, d)
            _assert_rel(dx, refx, 1.5e-2, f"rect dgrad {(n, cin, cout, h, w, k)}")


def test_conv_kernels_bitwise_deterministic():
    """The split-K (fwd) and split-M (wgrad) designs claim determinism: no
    atomics, fixed-order slice reduction. Verify bit equality across runs."""
    from improved_body_parts_amd.ops import conv_kernels
    torch.manual_seed(6)
    # fwd with split-K active (small grid): 8^2 wide-channel layer
    x = torch.randn(4, 384, 8, 8, device="cuda").bfloat16() \
        .contiguous(memory_format=CL)
    w = (torch.randn(384, 384, 3, 3, device="cuda") * 0.05).bfloat16()
    y1 = conv_kernels.conv_fwd(x, w, (1, 1), (1, 1), (1, 1)).clone()
    y2 = conv_kernels.conv_fwd(x, w, (1, 1), (1, 1), (1, 1))
    assert torch.equal(y1, y2), "split-K conv fwd is not bitwise deterministic"
    # wgrad with many m-chunks + tree combine
    xb = torch.randn(4, 128, 64, 64, device="cuda").bfloat16() \
        .contiguous(memory_format=CL)
    dy = (torch.randn(4, 128, 64, 64, device="cuda") * 0.1).bfloat16() \
        .contiguous(memory_format=CL)
    d1 = conv_kernels.conv_wgrad(xb, dy, (128, 128, 3, 3), (1, 1), (1, 1),
                                 (1, 1)).clone()
    d2 = conv_kernels.conv_wgrad(xb, dy, (128, 128, 3, 3), (1, 1), (1, 1),
                                 (1, 1))
    assert torch.equal(d1, d2), "split-M wgrad is not bitwise deterministic"


def test_pack_conv_weight_kernel():
    """One-launch fwd+dgrad weight pack vs the eager permute/flip chains."""
    ext = _backend.hip_extension()
    torch.manual_seed(4)
    w = torch.randn(24, 50, 3, 3, device="cuda").bfloat16()
    fwd = torch.empty(24, 9 * 50, dtype=torch.bfloat16, device="cuda")
    dgr = torch.empty(50, 9 * 24, dtype=torch.bfloat16, device="cuda")
    ext.pack_conv_weight(w, fwd, dgr)
    ref_f = w.permute(0, 2, 3, 1).reshape(24, -1)
    ref_d = torch.flip(w, dims=(2, 3)).permute(1, 2, 3, 0).reshape(50, -1)
    assert torch.equal(fwd, ref_f.contiguous())
    assert torch.equal(dgr, ref_d.contiguous())


def test_tr16_probe_delivery_map():
    """Pin the ds_read_b64_tr_b16 semantics the wgrad kernel is built on:
    with per-lane address base + l*8B over a [64][16]-short LDS image, lane l
    elem j receives image element (l&15) + j*16 + (l>>4)*64."""
    ext = _backend.hip_extension()
    out = ext.tr16_probe().cpu()  # [64, 4] int16
    for lane in range(64):
        for j in range(4):
            expect = (lane & 15) + j * 16 + (lane >> 4) * 64
            assert int(out[lane, j]) == expect, \
                f"lane {lane} elem {j}: got {int(out[lane, j])}, want {expect}"


def test_mfma_conv_wgrad_vs_reference():
    """Hand-written wgrad (tr16-transposed MFMA) vs torch.nn.grad.conv2d_weight
    across the IMHN's shapes, incl. tap-crossing tails and the 7x7 s2 stem."""
    from improved_body_parts_amd.ops import conv_kernels
    ext = _backend.hip_extension()
    if not hasattr(ext, "conv_mfma_wgrad"):
        pytest.skip("wgrad kernel not built yet")
    cases = [
        # (n, cin, cout, hw, k, s, d)
        (2, 64, 64, 64, 1, 1, 1),     # 1x1
        (2, 64, 128, 32, 3, 1, 1),    # 3x3
        (1, 128, 128, 128, 3, 1, 3),  # dilated 3
        (1, 128, 128, 64, 3, 1, 5),   # dilated 5
        (2, 384, 384, 8, 3, 1, 1),    # small spatial, wide channels
        (2, 256, 50, 32, 1, 1, 1),    # head 1x1 to 50ch (cout tail)
        (1, 50, 256, 32, 1, 1, 1),    # merge 1x1 from 50ch (cin tail)
        (2, 3, 64, 64, 7, 2, 1),      # 7x7 s2 stem (elementwise gather)
        (3, 192, 320, 20, 3, 1, 1),   # odd M tail
        (2, 16, 64, 32, 4, 1, 1),     # Cin=16 multi-tap (s2d stem shape)
        (4, 256, 128, 128, 1, 1, 1),  # wide (64,128) tile (M >= 65536, 1x1)
        (4, 50, 128, 128, 1, 1, 1),   # wide tile + cin tail (ELEM)
        (4, 128, 128, 128, 3, 1, 1),  # mid (128,128)x8-wave tile (KxK, M>=65536)
        (4, 128, 128, 128, 3, 1, 5),  # mid tile, dilation 5
    ]
    for n, cin, cout, hw, k, s, d in cases:
        torch.manual_seed(3)
        x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16() \
            .contiguous(memory_format=CL)
        pad = (k - 1) // 2 * d
        ho = (hw + 2 * pad - d * (k - 1) - 1) // s + 1
        dy = (torch.randn(n, cout, ho, ho, device="cuda") * 0.1).bfloat16() \
            .contiguous(memory_format=CL)
        dw = conv_kernels.conv_wgrad(x, dy, (cout, cin, k, k), (s, s),
                                     (pad, pad), (d, d))
        assert dw is not None, f"wgrad not covered: {(n, cin, cout, hw, k, s, d)}"
        ref = torch.nn.grad.conv2d_weight(x.float(), (cout, cin, k, k),
                                          dy.float(), s, pad, d)
        _assert_rel(dw, ref, 1.5e-2, f"wgrad {(n, cin, cout, hw, k, s, d)}")


# ---------------------------------------------------------------------------
# spatial ops
# ---------------------------------------------------------------------------
def test_maxpool2x2():
    x = torch.randn(2, 32, 16, 16).cuda().contiguous(memory_format=CL)
    xg = x.requires_grad_(True)
    y = ops.maxpool2x2(xg)
    ref_in = x.detach().clone().cpu().requires_grad_(True)
    yr = F.max_pool2d(ref_in, 2, 2)
    _assert_close(y, yr, 1e-6, 1e-6)
    dy = torch.randn_like(yr)
    yr.backward(dy)
    y.backward(dy.cuda())
    _assert_close(xg.grad, ref_in.grad, 1e-6, 1e-6, "maxpool dx")


def test_upsample2x():
    x = torch.randn(2, 16, 8, 8).cuda().contiguous(memory_format=CL)
    xg = x.requires_grad_(True)
    y = ops.upsample2x_nearest(xg)
    ref_in = x.detach().clone().cpu().requires_grad_(True)
    yr = F.interpolate(ref_in, scale_factor=2, mode="nearest")
    _assert_close(y, yr, 1e-6, 1e-6)
    dy = torch.randn_like(yr)
    yr.backward(dy)
    y.backward(dy.cuda())
    _assert_close(xg.grad, ref_in.grad, 1e-6, 1e-6, "upsample dx")


def test_se_layer():
    torch.manual_seed(3)
    fc1 = torch.nn.Linear(32, 2)
    fc2 = torch.nn.Linear(2, 32)
    fc1g = torch.nn.Linear(32, 2).cuda()
    fc2g = torch.nn.Linear(2, 32).cuda()
    fc1g.load_state_dict(fc1.state_dict())
    fc2g.load_state_dict(fc2.state_dict())
    x = torch.randn(2, 32, 8, 8)
    xg = x.cuda().contiguous(memory_format=CL).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    y = ops.se_layer(xg, fc1g, fc2g)
    n, c = xr.shape[:2]
    pr = xr.mean(dim=(2, 3))
    sr = torch.sigmoid(fc2(F.leaky_relu(fc1(pr), 0.01)))
    yr = xr * sr.view(n, c, 1, 1)
    _assert_close(y, yr, 1e-4, 1e-5, "se fwd")
    dy = torch.randn_like(yr)
    yr.backward(dy)
    y.backward(dy.cuda())
    _assert_close(xg.grad, xr.grad, 1e-4, 1e-5, "se dx")
    _assert_close(fc1g.weight.grad, fc1.weight.grad, 1e-3, 1e-5, "se dfc1")


# ---------------------------------------------------------------------------
# focal loss
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("gamma", [1, 2])
def test_focal_l2_gpu_vs_cpu(gamma):
    torch.manual_seed(0)
    pred = torch.rand(2, 2, 50, 16, 16)
    gt = torch.rand(2, 50, 16, 16) * (torch.rand(2, 50, 16, 16) > 0.6)
    mask = (torch.rand(2, 1, 16, 16) > 0.2).float()
    kw = dict(heat_start=30, bkg_start=48, gamma=gamma,
              multi_task_weight=0.1, keypoint_task_weight=3.0,
              nstack_weight=(1, 2))
    pg = pred.cuda().requires_grad_(True)
    pr = pred.clone().requires_grad_(True)
    lg = ops.focal_l2_loss(pg, gt.cuda(), mask.cuda(), **kw)
    lr = ops.focal_l2_loss(pr, gt, mask, **kw)
    _assert_close(lg, lr, 1e-4, 1e-3 * float(lr.detach()), "loss")
    lg.backward()
    lr.backward()
    _assert_close(pg.grad, pr.grad, 1e-4, 1e-5, "dpred")


@pytest.mark.parametrize("r", [2, 4, 16])
def test_focal_l2_fused_pyramid(r):
    """Full-res GT/mask go straight into the kernel, which average-pools GT
    windows and bilinearly samples + thresholds mask_miss on the fly; vs the
    CPU eager path (which materialises the pyramid like the reference)."""
    torch.manual_seed(2)
    H = 64
    h = H // r
    pred = torch.rand(2, 2, 50, h, h)
    gt = torch.rand(2, 50, H, H) * (torch.rand(2, 50, H, H) > 0.6)
    mask = (torch.rand(2, 1, H, H) > 0.2).float()
    kw = dict(heat_start=30, bkg_start=48, gamma=1,
              multi_task_weight=0.1, keypoint_task_weight=3.0,
              nstack_weight=(1, 2))
    pg = pred.cuda().requires_grad_(True)
    pr = pred.clone().requires_grad_(True)
    lg = ops.focal_l2_loss(pg, gt.cuda(), mask.cuda(), **kw)
    lr = ops.focal_l2_loss(pr, gt, mask, **kw)
    _assert_close(lg, lr, 1e-4, 1e-3 * max(float(lr.detach()), 1.0), "pyr loss")
    lg.backward()
    lr.backward()
    _assert_close(pg.grad, pr.grad, 1e-4, 1e-5, "pyr dpred")


def test_focal_l2_bf16():
    torch.manual_seed(1)
    pred = torch.rand(1, 2, 50, 32, 32)
    gt = (torch.rand(2, 50, 32, 32) * (torch.rand(2, 50, 32, 32) > 0.6))
    mask = torch.ones(2, 1, 32, 32)
    kw = dict(heat_start=30, bkg_start=48, gamma=1, nstack_weight=(1,))
    pg = pred.cuda().bfloat16().requires_grad_(True)
    lr = ops.focal_l2_loss(pred.clone().requires_grad_(True), gt, mask, **kw)
    lg = ops.focal_l2_loss(pg, gt.cuda().bfloat16(), mask.cuda().bfloat16(), **kw)
    assert abs(float(lg) - float(lr)) / float(lr) < 0.05


# ---------------------------------------------------------------------------
# fused SGD
# ---------------------------------------------------------------------------
def test_fused_sgd_bf16_master_weights():
    from improved_body_parts_amd.engine import FusedSGD
    torch.manual_seed(0)
    ref = torch.nn.Linear(64, 64)
    dev = torch.nn.Linear(64, 64).cuda().bfloat16()
    dev.load_state_dict({k: v.bfloat16() for k, v in ref.state_dict().items()})
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9,
                              weight_decay=0.01)
    opt_dev = FusedSGD(dev.parameters(), lr=0.05, momentum=0.9, weight_decay=0.01)
    for _ in range(10):
        g = torch.randn(64, 64) * 0.1
        gb = torch.randn(64) * 0.1
        opt_ref.zero_grad(), opt_dev.zero_grad()
        ref.weight.grad = g.clone()
        ref.bias.grad = gb.clone()
        dev.weight.grad = g.cuda().bfloat16()
        dev.bias.grad = gb.cuda().bfloat16()
        opt_ref.step(), opt_dev.step()
    # master weights track fp32 SGD closely even after 10 bf16 steps
    master = opt_dev.state[dev.weight]["master"]
    _assert_close(master, ref.weight.detach(), 2e-2, 2e-2, "sgd master")


# ---------------------------------------------------------------------------
# post-process
# ---------------------------------------------------------------------------
def test_heatmap_nms_matches_util():
    from improved_body_parts_amd.utils import keypoint_heatmap_nms
    torch.manual_seed(0)
    heat = torch.rand(1, 18, 64, 64)
    got = ops.heatmap_nms(heat.cuda(), 0.1)
    ref = keypoint_heatmap_nms(heat, 3, 0.1)
    _assert_close(got, ref, 1e-6, 1e-6)


def test_find_peaks_device():
    from improved_body_parts_amd.ops.postproc import find_peaks_device
    heat = torch.zeros(3, 64, 64)
    heat[0, 10, 20] = 0.9
    heat[2, 40, 30] = 0.8
    # slight spread so centroid refinement has signal
    heat[0, 10, 21] = 0.5
    peaks = find_peaks_device(heat.cuda(), threshold=0.3, radius=2)
    assert peaks.shape[0] == 2
    rows = {int(r[0]): r for r in peaks}
    assert 0 in rows and 2 in rows
    assert abs(float(rows[0][1]) - 20) < 1.0 and abs(float(rows[0][2]) - 10) < 0.5
    assert abs(float(rows[2][1]) - 30) < 0.5


# ---------------------------------------------------------------------------
# end-to-end: tiny Network on GPU vs CPU oracle
# ---------------------------------------------------------------------------
def test_network_gpu_matches_cpu_fp32():
    from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
    from improved_body_parts_amd.models import Network
    torch.manual_seed(0)
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=64, increase=32, batch_size=1,
                      nstack_weight=[1, 1])
    net = Network(opt, cfg, bn=True, dist=True)
    net_g = Network(opt, cfg, bn=True, dist=True).cuda()
    net_g.load_state_dict(net.state_dict())
    img = torch.rand(1, 128, 128, 3)
    mm = torch.ones(1, 1, 32, 32)
    hm = torch.rand(1, 50, 32, 32)
    net.train(), net_g.train()
    loss = net((img, mm, hm))
    loss_g = net_g((img.cuda(), mm.cuda(), hm.cuda()))
    assert abs(float(loss_g) - float(loss)) / float(loss) < 1e-3
    loss.backward()
    loss_g.backward()
    # gradients at the very bottom of a 2-stack net pass through ~50 BN
    # backward couplings — compare in relative L2, not elementwise max
    g_cpu = net.posenet.pre.conv1.weight.grad
    g_gpu = net_g.posenet.pre.conv1.weight.grad
    # at the stem the gradient has crossed ~50 BN-backward couplings; small
    # library-vs-CPU conv differences amplify — require direction, not bits
    cos = torch.nn.functional.cosine_similarity(
        g_gpu.float().flatten().cpu(), g_cpu.float().flatten(), dim=0)
    assert cos > 0.98, f"e2e conv1 grad cosine {cos}"  # 50 BN couplings deep
    head_cpu = net.posenet.outs[0][0].conv.weight.grad
    head_gpu = net_g.posenet.outs[0][0].conv.weight.grad
    _assert_rel(head_gpu, head_cpu, 3e-3, "e2e head grad")  # deep-net fp32 accum drift


def test_network_bf16_trains():
    from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
    from improved_body_parts_amd.data import SyntheticPoseDataset
    from improved_body_parts_amd.engine import FusedSGD
    from improved_body_parts_amd.models import Network
    torch.manual_seed(0)
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=64, increase=32, batch_size=2,
                      nstack_weight=[1, 1])
    net = Network(opt, cfg, bn=True, dist=True).cuda().bfloat16()
    for m in net.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    sgd = FusedSGD(net.parameters(), lr=1e-4, momentum=0.9)
    ds = SyntheticPoseDataset(cfg, length=2)
    img, mm, hm = ds[0]
    batch = tuple(t[None].cuda().bfloat16() for t in (img, mm, hm))
    losses = []
    for _ in range(5):
        sgd.zero_grad()
        loss = net(batch)
        loss.backward()
        sgd.step()
        losses.append(float(loss))
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]


def test_se_gate_fused_inference():
    """Fused GAP->FC->leaky->FC->sigmoid->scale vs the eager module chain."""
    from improved_body_parts_amd.models import SELayer
    torch.manual_seed(3)
    for c, dtype in [(256, torch.bfloat16), (128, torch.float32),
                     (768, torch.bfloat16)]:
        se = SELayer(c).cuda().to(dtype)
        x = torch.randn(3, c, 16, 16, device="cuda").to(dtype) \
            .contiguous(memory_format=CL)
        with torch.no_grad():
            y = se(x)  # fused path (grad disabled)
        pooled = x.float().mean(dim=(2, 3))
        h = F.leaky_relu(pooled @ se.fc[0].weight.float().t()
                         + se.fc[0].bias.float(), 0.01)
        s = torch.sigmoid(h @ se.fc[2].weight.float().t() + se.fc[2].bias.float())
        ref = x.float() * s.view(3, c, 1, 1)
        _assert_rel(y, ref, 2e-2 if dtype == torch.bfloat16 else 1e-4, f"se C={c}")


def test_fused_sgd_mixed_dtypes_vs_torch():
    """HIP fused SGD on a MIXED bf16+fp32 parameter set vs torch.optim.SGD
    (one global dtype tag once corrupted the fp32 BN affine params)."""
    from improved_body_parts_amd.engine import FusedSGD
    torch.manual_seed(0)
    w_bf = torch.nn.Parameter(torch.randn(70000, device="cuda").bfloat16())
    w_fp = torch.nn.Parameter(torch.randn(333, device="cuda"))
    ref_bf = torch.nn.Parameter(w_bf.detach().float().clone())
    ref_fp = torch.nn.Parameter(w_fp.detach().clone())
    opt = FusedSGD([w_bf, w_fp], lr=0.1, momentum=0.9, weight_decay=0.01)
    ref = torch.optim.SGD([ref_bf, ref_fp], lr=0.1, momentum=0.9,
                          weight_decay=0.01)
    for step in range(4):
        g = torch.randn(70000, device="cuda")
        g2 = torch.randn(333, device="cuda")
        w_bf.grad = g.bfloat16()
        w_fp.grad = g2.clone()
        ref_bf.grad = g.bfloat16().float()  # same quantized grad
        ref_fp.grad = g2.clone()
        opt.step()
        ref.step()
    _assert_rel(w_fp, ref_fp, 1e-6, "fp32 params")
    _assert_rel(w_bf.float(), ref_bf, 1e-2, "bf16 params (fp32 master)")
    assert torch.isfinite(w_fp).all() and torch.isfinite(w_bf.float()).all()


def test_mfma_conv_ragged_splitk():
    """Shapes where splitk does not divide the K-chunk count: trailing slices
    must still be accounted (uninitialised-workspace regression,
    scripts/diag_splitk.py). Run twice so the allocator serves DIRTY memory."""
    from improved_body_parts_amd.ops import conv_kernels
    for _ in range(2):
        for (n, cin, cout, hw) in [(8, 512, 50, 32), (8, 512, 64, 32),
                                   (8, 512, 128, 32), (8, 1024, 50, 32)]:
            torch.manual_seed(1)
            x = torch.randn(n, cin, hw, hw, device="cuda").bfloat16() \
                .contiguous(memory_format=CL)
            # dirty the allocator pool so empty() workspaces are non-zero
            junk = torch.full((64 << 20,), 3.3e7, device="cuda")
            del junk
            w = (torch.randn(cout, cin, 1, 1, device="cuda") * 0.05).bfloat16()
            y = conv_kernels.conv_fwd(x, w, (1, 1), (0, 0), (1, 1))
            ref = F.conv2d(x.float(), w.float())
            _assert_rel(y, ref, 1e-2, f"ragged splitk {(n, cin, cout, hw)}")
