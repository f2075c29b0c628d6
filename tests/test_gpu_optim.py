"""Fused optimizer kernels vs torch.optim on GPU."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    from dwt_amd.kernels import dispatch
    assert dispatch.available()
    return torch.device("cuda:0")


def _models(dev, dtype=torch.float32):
    torch.manual_seed(0)
    a = torch.nn.Sequential(torch.nn.Linear(37, 91), torch.nn.ReLU(),
                            torch.nn.Linear(91, 11)).to(dev).to(dtype)
    b = torch.nn.Sequential(torch.nn.Linear(37, 91), torch.nn.ReLU(),
                            torch.nn.Linear(91, 11)).to(dev).to(dtype)
    b.load_state_dict(a.state_dict())
    return a, b


def _run_steps(model, opt, dev, dtype, steps=5):
    torch.manual_seed(7)
    for _ in range(steps):
        x = torch.randn(16, 37, device=dev, dtype=dtype)
        loss = model(x).float().pow(2).mean()
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()


def test_fused_sgd_matches_torch_fp32(dev):
    from dwt_amd.ops.optim import FusedSGD
    a, b = _models(dev)
    oa = FusedSGD(a.parameters(), lr=0.05, momentum=0.9, weight_decay=5e-4)
    ob = torch.optim.SGD(b.parameters(), lr=0.05, momentum=0.9, weight_decay=5e-4)
    _run_steps(a, oa, dev, torch.float32)
    _run_steps(b, ob, dev, torch.float32)
    for (n1, p1), (n2, p2) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (n1, (p1 - p2).abs().max())


def test_fused_adam_matches_torch_fp32(dev):
    from dwt_amd.ops.optim import FusedAdam
    a, b = _models(dev)
    oa = FusedAdam(a.parameters(), lr=1e-3, weight_decay=5e-4)
    ob = torch.optim.Adam(b.parameters(), lr=1e-3, weight_decay=5e-4)
    _run_steps(a, oa, dev, torch.float32)
    _run_steps(b, ob, dev, torch.float32)
    for (n1, p1), (n2, p2) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (n1, (p1 - p2).abs().max())


def test_fused_sgd_bf16_master_tracks_fp32(dev):
    """bf16 params + fp32 master must track a pure-fp32 run closely — much
    closer than naive bf16 in-place updates would."""
    from dwt_amd.ops.optim import FusedSGD
    a, b = _models(dev, torch.bfloat16)
    ref, _ = _models(dev, torch.float32)
    oa = FusedSGD(a.parameters(), lr=0.05, momentum=0.9)
    oref = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    _run_steps(a, oa, dev, torch.bfloat16, steps=10)
    _run_steps(ref, oref, dev, torch.float32, steps=10)
    for (n1, p1), (_, p2) in zip(a.named_parameters(), ref.named_parameters()):
        # master weights live in optimizer state
        st = oa.state[p1]
        assert "master" in st
        assert torch.allclose(st["master"], p2, atol=5e-2, rtol=5e-2), n1
        assert torch.allclose(p1.float(), st["master"], atol=1e-2, rtol=1e-2)


def test_fused_sgd_zeroes_grads(dev):
    from dwt_amd.ops.optim import FusedSGD
    a, _ = _models(dev)
    opt = FusedSGD(a.parameters(), lr=0.01)
    _run_steps(a, opt, dev, torch.float32, steps=2)
    torch.cuda.synchronize()
    for p in a.parameters():
        assert p.grad is not None
        assert p.grad.abs().max().item() == 0.0  # kernel zeroed after update


def test_scheduler_interop(dev):
    from dwt_amd.ops.optim import FusedSGD
    a, _ = _models(dev)
    opt = FusedSGD(a.parameters(), lr=0.1, momentum=0.9)
    sched = torch.optim.lr_scheduler.MultiStepLR(opt, milestones=[2], gamma=0.1)
    for i in range(4):
        _run_steps(a, opt, dev, torch.float32, steps=1)
        sched.step()
    assert abs(opt.param_groups[0]["lr"] - 0.01) < 1e-9
