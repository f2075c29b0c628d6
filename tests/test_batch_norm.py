"""Domain BN: parity with stock F.batch_norm semantics + explicit backward."""
import pytest
import torch
import torch.nn.functional as F

from dwt_amd.ops import functional as Fdwt
from dwt_amd.ops.batch_norm import DomainBatchNorm1d, DomainBatchNorm2d


def test_train_forward_matches_stock():
    torch.manual_seed(0)
    x = torch.randn(8, 6, 5, 5, dtype=torch.float64)
    rm = torch.zeros(6, dtype=torch.float64)
    rv = torch.ones(6, dtype=torch.float64)
    rm2, rv2 = rm.clone(), rv.clone()
    w = torch.randn(6, dtype=torch.float64)
    b = torch.randn(6, dtype=torch.float64)

    y = Fdwt.BatchNormMulti.apply(
        x, w.view(6, 1, 1), b.view(6, 1, 1), [rm], [rv],
        dict(parts=1, eps=1e-5, momentum=0.1, training=True, relu=False))
    y_ref = F.batch_norm(x, rm2, rv2, w, b, True, 0.1, 1e-5)
    assert torch.allclose(y, y_ref, atol=1e-10)
    assert torch.allclose(rm, rm2, atol=1e-12)
    assert torch.allclose(rv, rv2, atol=1e-12)  # unbiased var into EMA


def test_eval_forward_matches_stock():
    torch.manual_seed(0)
    x = torch.randn(8, 6, 5, 5, dtype=torch.float64)
    rm = torch.randn(6, dtype=torch.float64)
    rv = torch.rand(6, dtype=torch.float64) + 0.5
    y = Fdwt.BatchNormMulti.apply(
        x, None, None, [rm.clone()], [rv.clone()],
        dict(parts=1, eps=1e-5, momentum=0.1, training=False, relu=False))
    y_ref = F.batch_norm(x, rm, rv, None, None, False, 0.0, 1e-5)
    assert torch.allclose(y, y_ref, atol=1e-12)


@pytest.mark.parametrize("parts", [1, 3])
@pytest.mark.parametrize("relu", [False, True])
def test_backward_matches_autograd(parts, relu):
    torch.manual_seed(1)
    b, c = 4, 6
    x = torch.randn(parts * b, c, 3, 4, dtype=torch.float64, requires_grad=True)
    gamma = torch.randn(c, 1, 1, dtype=torch.float64, requires_grad=True)
    beta = torch.randn(c, 1, 1, dtype=torch.float64, requires_grad=True)
    out = Fdwt.BatchNormMulti.apply(
        x, gamma, beta, None, None,
        dict(parts=parts, eps=1e-5, momentum=0.1, training=True, relu=relu))
    gout = torch.randn_like(out)
    gx, gg, gb = torch.autograd.grad(out, (x, gamma, beta), gout)

    x2 = x.detach().clone().requires_grad_(True)
    g2 = gamma.detach().clone().requires_grad_(True)
    b2 = beta.detach().clone().requires_grad_(True)
    ys = []
    for p in range(parts):
        xp = x2[p * b:(p + 1) * b]
        m = xp.mean(dim=(0, 2, 3), keepdim=True)
        v = xp.var(dim=(0, 2, 3), unbiased=False, keepdim=True)
        ys.append((xp - m) / torch.sqrt(v + 1e-5))
    out2 = torch.cat(ys, 0) * g2 + b2
    if relu:
        out2 = torch.relu(out2)
    gx2, gg2, gb2 = torch.autograd.grad(out2, (x2, g2, b2), gout)
    assert torch.allclose(gx, gx2, atol=1e-9)
    assert torch.allclose(gg, gg2, atol=1e-9)
    assert torch.allclose(gb, gb2, atol=1e-9)


def test_module_vs_nn_batchnorm():
    """DomainBatchNorm with fresh buffers behaves exactly like nn.BatchNorm."""
    torch.manual_seed(0)
    ours = DomainBatchNorm2d(5, affine=True)
    theirs = torch.nn.BatchNorm2d(5, affine=True)
    with torch.no_grad():
        theirs.weight.copy_(ours.weight)
        theirs.bias.copy_(ours.bias)
    for step in range(3):
        x = torch.randn(6, 5, 4, 4)
        y1, y2 = ours(x), theirs(x)
        assert torch.allclose(y1, y2, atol=1e-6), step
    assert torch.allclose(ours.running_mean, theirs.running_mean, atol=1e-6)
    assert torch.allclose(ours.running_var, theirs.running_var, atol=1e-6)
    assert ours.num_batches_tracked.item() == 3
    ours.eval(); theirs.eval()
    x = torch.randn(2, 5, 4, 4)
    assert torch.allclose(ours(x), theirs(x), atol=1e-6)


def test_bn1d_paths():
    torch.manual_seed(0)
    ours = DomainBatchNorm1d(7, affine=False)
    theirs = torch.nn.BatchNorm1d(7, affine=False)
    x = torch.randn(9, 7)
    assert torch.allclose(ours(x), theirs(x), atol=1e-6)
    with pytest.raises(ValueError):
        ours(torch.randn(2, 7, 3, 3))


def test_injected_buffers_are_used():
    rm = torch.full((4,), 2.0)
    rv = torch.full((4,), 4.0)
    mod = DomainBatchNorm2d(4, running_m=rm, running_v=rv, affine=False)
    mod.eval()
    x = torch.full((1, 4, 2, 2), 4.0)
    y = mod(x)
    assert torch.allclose(y, torch.full_like(y, (4.0 - 2.0) / 2.0), atol=1e-4)
    assert mod.running_mean.data_ptr() == rm.data_ptr()  # same storage (checkpoint aliasing)
