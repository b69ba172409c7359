"""Every model-variant family trains one bf16 step on device.

The five reference architecture forks (posenet_final / posenet2 / posenet3 /
posenet_independent / ae_pose) plus the flagship IMHN, all through the HIP
kernel path — the CPU suite covers their checkpoint/forward parity with the
reference; this covers the device compute path per family.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
from improved_body_parts_amd.models import Network


@pytest.mark.parametrize(
    "variant", ["imhn", "final", "attention", "light", "independent", "ae"])
def test_variant_trains_on_gpu_bf16(variant):
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=256, batch_size=2,
                      nstack_weight=[1, 1], model_variant=variant)
    torch.manual_seed(3)
    net = Network(opt, cfg, bn=True, dist=True).cuda().bfloat16()
    for m in net.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    net.train()
    img = torch.rand(2, 128, 128, 3, device="cuda", dtype=torch.bfloat16)
    mm = torch.ones(2, 1, 32, 32, device="cuda", dtype=torch.bfloat16)
    hm = torch.rand(2, 50, 32, 32, device="cuda", dtype=torch.bfloat16)
    loss = net((img, mm, hm))
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item(), f"{variant}: non-finite loss {loss}"
    # every RECEIVED gradient is finite; the bulk of parameters participate
    # (some variants carry reference-faithful unused modules — e.g. the
    # independent fork instantiates cross-stack merges it never applies)
    nonfinite = [n for n, p in net.named_parameters()
                 if p.grad is not None and not torch.isfinite(p.grad).all()]
    assert not nonfinite, f"{variant}: non-finite grads: {nonfinite[:5]}"
    total = sum(1 for _, p in net.named_parameters() if p.requires_grad)
    got = sum(1 for _, p in net.named_parameters() if p.grad is not None)
    assert got >= 0.7 * total, f"{variant}: only {got}/{total} params got grads"
