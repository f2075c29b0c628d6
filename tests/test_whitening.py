"""Whitening: explicit backward vs autograd-of-oracle, reference semantics."""
import pytest
import torch

from dwt_amd.ops import functional as Fdwt
from dwt_amd.ops import oracle
from dwt_amd.ops.whitening import WTransform2d, WhiteningScaleShift


def ref_whiten(x, num_groups, eps=1e-3, mode="chol"):
    """Differentiable composition of oracle pieces (training mode)."""
    m = oracle.channel_mean(x)
    xn = x - m
    cov = oracle.grouped_cov(xn, num_groups)
    cov_s = oracle.shrink_cov(cov, eps)
    if mode == "chol":
        w = oracle.whiten_matrix_chol(cov_s)
    else:
        w = oracle.whiten_matrix_ns(cov_s)
    c = x.shape[1]
    g = c // num_groups
    import torch.nn.functional as F
    return F.conv2d(xn, w.reshape(c, g, 1, 1), groups=num_groups)


@pytest.mark.parametrize("mode", ["chol", "zca"])
@pytest.mark.parametrize("shape,groups", [
    ((6, 8, 5, 7), 2),     # odd spatial
    ((4, 8, 3, 3), 8),     # group_size 1
    ((8, 16, 4, 4), 4),
    ((5, 32, 2, 2), 1),    # one big group (g=32, LeNet default config)
])
def test_forward_matches_oracle(mode, shape, groups):
    x = torch.randn(*shape, dtype=torch.float64)
    y = Fdwt.WhitenMulti.apply(
        x, None, None, None, None,
        dict(parts=1, num_groups=groups, eps=1e-3, momentum=0.1, training=True,
             mode=mode, relu=False))
    y_ref = ref_whiten(x, groups, mode=mode)
    assert torch.allclose(y, y_ref, atol=1e-10)


def test_whitening_decorrelates():
    """After whitening, the per-group covariance of y is ~identity (x(1-eps))."""
    torch.manual_seed(3)
    x = torch.randn(64, 16, 8, 8, dtype=torch.float64) * 3.0 + 1.0
    # correlate the channels
    mix = torch.randn(16, 16, dtype=torch.float64)
    x = torch.einsum("dc,nchw->ndhw", mix, x)
    for mode in ("chol", "zca"):
        y = Fdwt.WhitenMulti.apply(
            x, None, None, None, None,
            dict(parts=1, num_groups=4, eps=1e-3, momentum=0.1, training=True,
                 mode=mode, relu=False))
        cov_y = oracle.grouped_cov(y - oracle.channel_mean(y), 4)
        eye = torch.eye(4, dtype=torch.float64).expand_as(cov_y)
        # shrinkage means y covariance isn't exactly I: W Sigma W^T = I - eps(...)
        assert (cov_y - eye).abs().max() < 0.05, mode


@pytest.mark.parametrize("mode", ["chol", "zca"])
@pytest.mark.parametrize("parts", [1, 2, 3])
def test_backward_matches_autograd(mode, parts):
    """The hand-derived backward == autograd through the oracle composition."""
    torch.manual_seed(1)
    b, c, h, w = 4, 8, 3, 5
    x = torch.randn(parts * b, c, h, w, dtype=torch.float64, requires_grad=True)
    gamma = torch.randn(c, 1, 1, dtype=torch.float64, requires_grad=True)
    beta = torch.randn(c, 1, 1, dtype=torch.float64, requires_grad=True)

    out = Fdwt.WhitenMulti.apply(
        x, gamma, beta, None, None,
        dict(parts=parts, num_groups=2, eps=1e-3, momentum=0.1, training=True,
             mode=mode, relu=True))
    gout = torch.randn_like(out)
    gx, ggamma, gbeta = torch.autograd.grad(out, (x, gamma, beta), gout)

    # autograd reference
    x2 = x.detach().clone().requires_grad_(True)
    gamma2 = gamma.detach().clone().requires_grad_(True)
    beta2 = beta.detach().clone().requires_grad_(True)
    ys = []
    for p in range(parts):
        ys.append(ref_whiten(x2[p * b:(p + 1) * b], 2, mode=("chol" if mode == "chol" else "zca")))
    out2 = torch.relu(torch.cat(ys, 0) * gamma2 + beta2)
    gx2, ggamma2, gbeta2 = torch.autograd.grad(out2, (x2, gamma2, beta2), gout)

    assert torch.allclose(gx, gx2, atol=1e-8), (gx - gx2).abs().max()
    assert torch.allclose(ggamma, ggamma2, atol=1e-8)
    assert torch.allclose(gbeta, gbeta2, atol=1e-8)


def test_gradcheck_small():
    torch.manual_seed(2)
    x = torch.randn(4, 4, 2, 3, dtype=torch.float64, requires_grad=True)
    for mode in ("chol", "zca"):
        assert torch.autograd.gradcheck(
            lambda t: Fdwt.WhitenMulti.apply(
                t, None, None, None, None,
                dict(parts=2, num_groups=2, eps=1e-3, momentum=0.1,
                     training=True, mode=mode, relu=False)),
            (x,), eps=1e-6, atol=1e-6)


def test_ema_update_semantics():
    """running = (1-m)*running + m*batch; variance stores the UNSHRUNK cov
    (whitening.py:57-59 — SURVEY quirk #2)."""
    torch.manual_seed(0)
    mod = WTransform2d(8, 4)
    mod.train()
    x = torch.randn(6, 8, 4, 4)
    rm0 = mod.running_mean.clone()
    rv0 = mod.running_variance.clone()
    _ = mod(x)
    m = oracle.channel_mean(x)
    cov = oracle.grouped_cov(x - m, 2)
    assert torch.allclose(mod.running_mean, 0.9 * rm0 + 0.1 * m, atol=1e-6)
    assert torch.allclose(mod.running_variance, 0.9 * rv0 + 0.1 * cov, atol=1e-5)


def test_eval_uses_running_stats():
    torch.manual_seed(0)
    mod = WTransform2d(8, 4)
    mod.train()
    for _ in range(5):
        _ = mod(torch.randn(16, 8, 4, 4) * 2 + 0.5)
    mod.eval()
    x = torch.randn(3, 8, 4, 4)
    y = mod(x)
    # manual eval-path computation (whitening.py:42-43, 50-51)
    m = mod.running_mean
    cov_s = oracle.shrink_cov(mod.running_variance, mod.eps)
    w = oracle.whiten_matrix_chol(cov_s)
    import torch.nn.functional as F
    y_ref = F.conv2d(x - m, w.reshape(8, 4, 1, 1), groups=2)
    assert torch.allclose(y, y_ref, atol=1e-5)
    # eval must not touch the buffers
    rm = mod.running_mean.clone()
    _ = mod(torch.randn(3, 8, 4, 4))
    assert torch.equal(rm, mod.running_mean)


def test_group_size_validation():
    mod = WTransform2d(48, 32)  # min(48,32)=32 doesn't divide 48
    with pytest.raises(ValueError):
        mod(torch.randn(2, 48, 3, 3))
    with pytest.raises(ValueError):
        WTransform2d(8, 4)(torch.randn(2, 8, 3))  # 3D input


def test_scale_shift_module():
    mod = WhiteningScaleShift(8, 4, affine=True)
    mod.train()
    x = torch.randn(4, 8, 3, 3)
    y = mod(x)
    assert y.shape == x.shape
    names = dict(mod.named_buffers())
    assert "wh.running_mean" in names and "wh.running_variance" in names
    assert names["wh.running_mean"].shape == (1, 8, 1, 1)
    assert names["wh.running_variance"].shape == (2, 4, 4)
