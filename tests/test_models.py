"""Model graph tests: LeNet + DWT-ResNet50 construction, stream routing,
checkpoint layout round-trip."""
import torch
import pytest

from dwt_amd.models import LeNet, ResNetDWT, Bottleneck, resnet50
from dwt_amd.models.checkpoint import (
    compute_bn_stats,
    export_reference_checkpoint,
    load_reference_state_dict,
    load_training_state,
    save_training_state,
)


def small_resnet(num_classes=7, layers=(1, 1, 1, 1)):
    return ResNetDWT(Bottleneck, list(layers), None, num_classes=num_classes)


def test_lenet_shapes_and_modes():
    model = LeNet(group_size=4)
    model.train()
    x = torch.randn(8, 1, 28, 28)  # 4 source + 4 target
    out = model(x)
    assert out.shape == (8, 10)
    model.eval()
    out = model(torch.randn(3, 1, 28, 28))
    assert out.shape == (3, 10)


def test_lenet_three_stream():
    model = LeNet(group_size=4, streams=3)
    model.train()
    out = model(torch.randn(9, 1, 28, 28))
    assert out.shape == (9, 10)


def test_lenet_domain_branches_differ():
    """Source and target streams must be normalized by different stats."""
    torch.manual_seed(0)
    model = LeNet(group_size=4)
    model.train()
    x_half = torch.randn(4, 1, 28, 28)
    x = torch.cat([x_half, x_half * 3.0 + 1.0], dim=0)
    _ = model(x)
    assert not torch.allclose(model.ws1.running_mean, model.wt1.running_mean)


def test_lenet_backward_updates_all_params():
    model = LeNet(group_size=4)
    model.train()
    out = model(torch.randn(8, 1, 28, 28))
    out.sum().backward()
    missing = [n for n, p in model.named_parameters() if p.grad is None]
    assert missing == [], missing


def test_resnet_small_fwd_bwd():
    model = small_resnet()
    model.train()
    x = torch.randn(6, 3, 64, 64)  # thirds of 2
    out = model(x)
    assert out.shape == (6, 7)
    out.sum().backward()
    missing = [n for n, p in model.named_parameters() if p.grad is None]
    assert missing == [], missing
    model.eval()
    assert model(torch.randn(2, 3, 64, 64)).shape == (2, 7)


def test_resnet_norm_site_kinds():
    model = small_resnet()
    from dwt_amd.ops.whitening import WhiteningScaleShift
    from dwt_amd.ops.batch_norm import DomainBatchNorm2d
    assert isinstance(model.layer1[0].bns1, WhiteningScaleShift)
    assert isinstance(model.layer2[0].bns1, DomainBatchNorm2d)
    # triplicated branches everywhere, downsample sites on first blocks
    assert hasattr(model.layer1[0], "downsample_bnt_aug")
    assert hasattr(model.layer3[0], "downsample_bns")


@pytest.mark.slow
def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    model = ResNetDWT(Bottleneck, [3, 4, 6, 3], None, num_classes=65)
    model.train()
    _ = model(torch.randn(6, 3, 96, 96))  # move EMA stats off their init

    path = str(tmp_path / "ref_layout.pth.tar")
    export_reference_checkpoint(model, path)

    sd = load_reference_state_dict(path)
    assert "bn1.wh.running_mean" in sd
    assert sd["bn1.wh.running_mean"].shape == (1, 64, 1, 1)
    assert sd["layer1.0.bn1.wh.running_variance"].shape == (16, 4, 4)
    assert "layer1.0.bn1.gamma" in sd
    assert "layer2.0.bn1.running_mean" in sd
    assert sd["layer2.0.bn1.weight"].shape == (128,)
    assert "layer1.0.downsample_bn.wh.running_mean" in sd
    assert "layer2.0.downsample_bn.running_var" in sd
    bn_dict = compute_bn_stats(sd)
    assert all(("bn" in k) or ("downsample" in k) for k in bn_dict)

    model2 = resnet50(path, torch.device("cpu"))
    model.eval(); model2.eval()
    x = torch.randn(2, 3, 96, 96)
    with torch.no_grad():
        y1, y2 = model(x), model2(x)
    assert torch.allclose(y1, y2, atol=1e-4), (y1 - y2).abs().max()


def test_training_state_resume(tmp_path):
    model = LeNet(group_size=4)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    sched = torch.optim.lr_scheduler.MultiStepLR(opt, milestones=[5], gamma=0.1)
    model.train()
    out = model(torch.randn(8, 1, 28, 28))
    out.sum().backward()
    opt.step(); sched.step()
    path = str(tmp_path / "state.pt")
    save_training_state(path, model, opt, sched, iteration=17)

    model2 = LeNet(group_size=4)
    opt2 = torch.optim.Adam(model2.parameters(), lr=1e-3)
    sched2 = torch.optim.lr_scheduler.MultiStepLR(opt2, milestones=[5], gamma=0.1)
    it = load_training_state(path, model2, opt2, sched2)
    assert it == 17
    for (n1, p1), (n2, p2) in zip(model.state_dict().items(), model2.state_dict().items()):
        assert n1 == n2 and torch.equal(p1, p2), n1
