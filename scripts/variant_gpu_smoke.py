import sys, torch
sys.path.insert(0, '/root/repo')
from improved_body_parts_amd.config import CanonicalConfig, TrainingOpt
from improved_body_parts_amd.models import Network
for v in ["imhn", "final", "attention", "light", "independent", "ae"]:
    cfg = CanonicalConfig(128, 128, 4)
    opt = TrainingOpt(nstack=2, hourglass_inp_dim=256, batch_size=2,
                      nstack_weight=[1, 1], model_variant=v)
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
    ok = torch.isfinite(loss).item()
    print(f"{v}: loss={float(loss):.2f} finite={ok}")
    assert ok
print("ALL VARIANTS OK on GPU bf16")
