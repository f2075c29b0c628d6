"""Independent checkpoint-layout contract test: a hand-built reference-layout
checkpoint (as the ORIGINAL pretraining pipeline would emit)
must construct a model with the injected stats/affines in place."""
import torch

from dwt_amd.models import resnet50


def _ref_checkpoint(tmp_path):
    """Build the §3.4 key layout from scratch (module.-prefixed)."""
    torch.manual_seed(0)
    sd = {}

    def wh_site(prefix, c):
        sd[f"{prefix}.wh.running_mean"] = torch.randn(1, c, 1, 1)
        cov = torch.randn(c // 4, 4, 4)
        sd[f"{prefix}.wh.running_variance"] = cov @ cov.transpose(1, 2) + \
            torch.eye(4).expand(c // 4, 4, 4)
        sd[f"{prefix}.gamma"] = torch.rand(c, 1, 1) + 0.5
        sd[f"{prefix}.beta"] = torch.randn(c, 1, 1)

    def bn_site(prefix, c):
        sd[f"{prefix}.running_mean"] = torch.randn(c)
        sd[f"{prefix}.running_var"] = torch.rand(c) + 0.5
        sd[f"{prefix}.weight"] = torch.rand(c) + 0.5
        sd[f"{prefix}.bias"] = torch.randn(c)

    wh_site("bn1", 64)
    blocks = {1: (3, 64), 2: (4, 128), 3: (6, 256), 4: (3, 512)}
    for layer, (n, planes) in blocks.items():
        site = wh_site if layer == 1 else bn_site
        for i in range(n):
            site(f"layer{layer}.{i}.bn1", planes)
            site(f"layer{layer}.{i}.bn2", planes)
            site(f"layer{layer}.{i}.bn3", planes * 4)
        site(f"layer{layer}.0.downsample_bn", planes * 4)

    # conv/fc weights (subset is enough: strict=False load)
    sd["conv1.weight"] = torch.randn(64, 3, 7, 7) * 0.05
    sd["fc_out.weight"] = torch.randn(65, 2048) * 0.05
    sd["fc_out.bias"] = torch.zeros(65)

    blob = {"state_dict": {f"module.{k}": v for k, v in sd.items()}}
    path = str(tmp_path / "ref.pth.tar")
    torch.save(blob, path)
    return path, sd


def test_reference_layout_checkpoint_loads(tmp_path):
    path, sd = _ref_checkpoint(tmp_path)
    model = resnet50(path, torch.device("cpu"))

    # whitening stats injected into all three branches
    for branch in (model.bns1, model.bnt1, model.bnt1_aug):
        assert torch.allclose(branch.wh.running_mean, sd["bn1.wh.running_mean"])
        assert torch.allclose(branch.wh.running_variance,
                              sd["bn1.wh.running_variance"])
    # shared affines from the checkpoint
    assert torch.allclose(model.gamma1.data, sd["bn1.gamma"])
    assert torch.allclose(model.layer2[0].gamma1.data,
                          sd["layer2.0.bn1.weight"].view(-1, 1, 1))
    # BN branch stats
    assert torch.allclose(model.layer3[1].bnt2.running_mean,
                          sd["layer3.1.bn2.running_mean"])
    # conv/fc weights restored by load_state_dict(strict=False)
    assert torch.allclose(model.conv1.weight.data, sd["conv1.weight"])
    assert torch.allclose(model.fc_out.weight.data, sd["fc_out.weight"])

    # eval forward runs with injected stats
    model.eval()
    with torch.no_grad():
        out = model(torch.randn(2, 3, 96, 96))
    assert out.shape == (2, 65) and torch.isfinite(out).all()
