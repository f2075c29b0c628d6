"""Property-based coverage (hypothesis): the whitening Function across random
shapes/groups/modes always (a) matches the oracle composition and (b)
decorrelates, and the hand-derived backward always matches autograd."""
import pytest
import torch

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from dwt_amd.ops import functional as Fdwt
from dwt_amd.ops import oracle


@st.composite
def whiten_case(draw):
    g = draw(st.sampled_from([1, 2, 3, 4, 5, 8]))
    groups = draw(st.integers(1, 4))
    c = g * groups
    b = draw(st.integers(2, 5))
    parts = draw(st.sampled_from([1, 2, 3]))
    h = draw(st.integers(1, 5))
    w = draw(st.integers(1, 5))
    if b * h * w < 2:
        b = 2
    mode = draw(st.sampled_from(["chol", "zca"]))
    seed = draw(st.integers(0, 2 ** 16))
    return parts, b, c, groups, h, w, mode, seed


@settings(max_examples=25, deadline=None, derandomize=True)
@given(whiten_case())
def test_whiten_forward_backward_property(case):
    parts, b, c, groups, h, w, mode, seed = case
    torch.manual_seed(seed)
    x = torch.randn(parts * b, c, h, w, dtype=torch.float64, requires_grad=True)
    cfg = dict(parts=parts, num_groups=groups, eps=1e-3, momentum=0.1,
               training=True, mode=mode, relu=False)
    out = Fdwt.WhitenMulti.apply(x, None, None, None, None, cfg)

    # (a) forward == oracle per part
    import torch.nn.functional as F
    g = c // groups
    for p in range(parts):
        xp = x[p * b:(p + 1) * b].detach()
        m = oracle.channel_mean(xp)
        cov_s = oracle.shrink_cov(oracle.grouped_cov(xp - m, groups), 1e-3)
        wm = (oracle.whiten_matrix_chol(cov_s) if mode == "chol"
              else oracle.whiten_matrix_ns(cov_s))
        ref = F.conv2d(xp - m, wm.reshape(c, g, 1, 1), groups=groups)
        assert torch.allclose(out[p * b:(p + 1) * b], ref, atol=1e-8)

    # (b) hand-derived backward == autograd through the oracle composition
    gout = torch.randn_like(out)
    (gx,) = torch.autograd.grad(out, (x,), gout)
    x2 = x.detach().clone().requires_grad_(True)
    ys = []
    for p in range(parts):
        xp = x2[p * b:(p + 1) * b]
        m = oracle.channel_mean(xp)
        cov_s = oracle.shrink_cov(oracle.grouped_cov(xp - m, groups), 1e-3)
        wm = (oracle.whiten_matrix_chol(cov_s) if mode == "chol"
              else oracle.whiten_matrix_ns(cov_s))
        ys.append(F.conv2d(xp - m, wm.reshape(c, g, 1, 1), groups=groups))
    (gx2,) = torch.autograd.grad(torch.cat(ys, 0), (x2,), gout)
    assert torch.allclose(gx, gx2, atol=1e-7), (gx - gx2).abs().max()


@settings(max_examples=15, deadline=None, derandomize=True)
@given(st.integers(0, 2 ** 16), st.sampled_from([2, 4, 8]),
       st.sampled_from(["chol", "zca"]))
def test_whitening_decorrelates_property(seed, g, mode):
    torch.manual_seed(seed)
    c = g * 3
    x = torch.randn(32, c, 6, 6, dtype=torch.float64)
    mix = torch.randn(c, c, dtype=torch.float64) + torch.eye(c)
    x = torch.einsum("dc,nchw->ndhw", mix, x)
    cfg = dict(parts=1, num_groups=3, eps=1e-3, momentum=0.1, training=True,
               mode=mode, relu=False)
    y = Fdwt.WhitenMulti.apply(x, None, None, None, None, cfg)
    cov_y = oracle.grouped_cov(y - oracle.channel_mean(y), 3)
    eye = torch.eye(g, dtype=torch.float64).expand_as(cov_y)
    # shrinkage leaves O(eps * conditioning) residual
    assert (cov_y - eye).abs().max() < 0.2
