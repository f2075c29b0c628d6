"""GPU numerics: every HIP kernel vs the (CPU-validated) torch reference path.

All tests @pytest.mark.gpu — run on an MI355X via gpurun / the driver.
The torch reference Functions here were themselves gradchecked against
autograd of the fp64 oracle on CPU (tests/test_whitening.py), so agreement
HIP<->torch-fp32 closes the chain HIP <-> oracle.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu



@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available(), "gpu tests need a GPU"
    from dwt_amd.kernels import dispatch
    assert dispatch.available(), "HIP extension must be built+loaded on a GPU box"
    return torch.device("cuda:0")


def _whiten_cfg(parts, groups, training=True, mode="chol", relu=False, affine=True):
    return dict(parts=parts, num_groups=groups, eps=1e-3, momentum=0.1,
                training=training, mode=mode, relu=relu)


def run_whiten(x, gamma, beta, cfg, use_hip, rms=None, rvs=None):
    if use_hip:
        from dwt_amd.kernels.hip_ops import _HipWhitenMulti
        return _HipWhitenMulti.apply(x, gamma, beta, rms, rvs, cfg)
    from dwt_amd.ops.functional import WhitenMulti
    return WhitenMulti.apply(x, gamma, beta, rms, rvs, cfg)


@pytest.mark.parametrize("mode", ["chol", "zca"])
@pytest.mark.parametrize("g,groups_of", [
    (4, 16), (2, 4), (8, 4),
    # generic LDS-tiled path (8 < g <= 32): the reference digits configs
    # beyond the register-blocked sizes (g=16 covers C=32/48; g=32 only
    # C%32==0 sites — the reference's own default raises on C=48)
    (16, 3), (32, 1),
])
@pytest.mark.parametrize("relu", [True, False])
def test_whiten_fwd_bwd_parity_fp32(dev, mode, g, groups_of, relu):
    torch.manual_seed(0)
    c = g * groups_of
    parts = 3
    x = torch.randn(parts * 4, c, 7, 9, device=dev)  # odd spatial -> scalar path
    gamma = torch.randn(c, 1, 1, device=dev)
    beta = torch.randn(c, 1, 1, device=dev)
    cfg = _whiten_cfg(parts, groups_of, mode=mode, relu=relu)

    xa = x.clone().requires_grad_(True)
    ga = gamma.clone().requires_grad_(True)
    ba = beta.clone().requires_grad_(True)
    out_h = run_whiten(xa, ga, ba, cfg, True)
    xb = x.clone().requires_grad_(True)
    gb = gamma.clone().requires_grad_(True)
    bb = beta.clone().requires_grad_(True)
    out_t = run_whiten(xb, gb, bb, cfg, False)
    assert torch.allclose(out_h, out_t, atol=2e-4), (out_h - out_t).abs().max()

    gout = torch.randn_like(out_h)
    out_h.backward(gout)
    out_t.backward(gout)
    assert torch.allclose(xa.grad, xb.grad, atol=2e-3), (xa.grad - xb.grad).abs().max()
    assert torch.allclose(ga.grad, gb.grad, atol=2e-2), (ga.grad - gb.grad).abs().max()
    assert torch.allclose(ba.grad, bb.grad, atol=2e-2)


def test_whiten_vectorized_path(dev):
    """HW divisible by 8 + aligned -> vector kernels; same numbers."""
    torch.manual_seed(1)
    c, groups = 64, 16
    x = torch.randn(6, c, 16, 16, device=dev)
    cfg = _whiten_cfg(3, groups)
    out_h = run_whiten(x.clone().requires_grad_(False), None, None, cfg, True)
    out_t = run_whiten(x.clone(), None, None, cfg, False)
    assert torch.allclose(out_h, out_t, atol=2e-4)


def test_whiten_bf16(dev):
    torch.manual_seed(2)
    c, groups = 64, 16
    x32 = torch.randn(6, c, 16, 16, device=dev)
    x = x32.to(torch.bfloat16).requires_grad_(True)
    gamma = torch.randn(c, 1, 1, device=dev, dtype=torch.bfloat16, requires_grad=True)
    beta = torch.zeros(c, 1, 1, device=dev, dtype=torch.bfloat16, requires_grad=True)
    cfg = _whiten_cfg(3, groups, relu=True)
    out = run_whiten(x, gamma, beta, cfg, True)
    # fp32 torch reference on the bf16-quantized input
    x2 = x.detach().clone().requires_grad_(True)
    g2 = gamma.detach().clone().requires_grad_(True)
    b2 = beta.detach().clone().requires_grad_(True)
    ref = run_whiten(x2, g2, b2, cfg, False)
    assert torch.allclose(out.float(), ref.float(), atol=0.1, rtol=0.05)
    gout = torch.randn_like(out)
    out.backward(gout)
    ref.backward(gout)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=0.2, rtol=0.1)


def test_whiten_ema_and_eval(dev):
    torch.manual_seed(3)
    c, groups, g = 16, 4, 4
    rm_h = [torch.zeros(1, c, 1, 1, device=dev)]
    rv_h = [torch.eye(g, device=dev).repeat(groups, 1, 1)]
    rm_t = [t.clone() for t in rm_h]
    rv_t = [t.clone() for t in rv_h]
    for _ in range(3):
        x = torch.randn(8, c, 6, 6, device=dev) * 2 + 0.3
        cfg = _whiten_cfg(1, groups)
        run_whiten(x, None, None, cfg, True, rm_h, rv_h)
        run_whiten(x, None, None, cfg, False, rm_t, rv_t)
    assert torch.allclose(rm_h[0], rm_t[0], atol=1e-4)
    assert torch.allclose(rv_h[0], rv_t[0], atol=1e-4)
    # eval parity
    x = torch.randn(5, c, 6, 6, device=dev)
    cfg = _whiten_cfg(1, groups, training=False)
    out_h = run_whiten(x, None, None, cfg, True, rm_h, rv_h)
    out_t = run_whiten(x, None, None, cfg, False, rm_t, rv_t)
    assert torch.allclose(out_h, out_t, atol=2e-4)


@pytest.mark.parametrize("spatial", [True, False])
@pytest.mark.parametrize("relu", [True, False])
def test_bn_parity(dev, spatial, relu):
    torch.manual_seed(4)
    from dwt_amd.kernels.hip_ops import _HipBatchNormMulti
    from dwt_amd.ops.functional import BatchNormMulti
    c, parts = 32, 3
    shape = (parts * 6, c, 8, 8) if spatial else (parts * 6, c)
    gshape = (c, 1, 1) if spatial else (1, c)
    x = torch.randn(*shape, device=dev)
    gamma = torch.randn(*gshape, device=dev)
    beta = torch.randn(*gshape, device=dev)
    rms_h = [torch.zeros(c, device=dev) for _ in range(parts)]
    rvs_h = [torch.ones(c, device=dev) for _ in range(parts)]
    rms_t = [t.clone() for t in rms_h]
    rvs_t = [t.clone() for t in rvs_h]
    cfg = dict(parts=parts, eps=1e-5, momentum=0.1, training=True, relu=relu)

    xa = x.clone().requires_grad_(True); ga = gamma.clone().requires_grad_(True)
    ba = beta.clone().requires_grad_(True)
    out_h = _HipBatchNormMulti.apply(xa, ga, ba, rms_h, rvs_h, cfg)
    xb = x.clone().requires_grad_(True); gb = gamma.clone().requires_grad_(True)
    bb = beta.clone().requires_grad_(True)
    out_t = BatchNormMulti.apply(xb, gb, bb, rms_t, rvs_t, cfg)
    assert torch.allclose(out_h, out_t, atol=1e-4)
    for h, t in zip(rms_h + rvs_h, rms_t + rvs_t):
        assert torch.allclose(h, t, atol=1e-4)
    gout = torch.randn_like(out_h)
    out_h.backward(gout); out_t.backward(gout)
    assert torch.allclose(xa.grad, xb.grad, atol=1e-3)
    assert torch.allclose(ga.grad, gb.grad, atol=1e-2)
    assert torch.allclose(ba.grad, bb.grad, atol=1e-2)
    # eval parity
    cfg_e = dict(cfg, training=False)
    oh = _HipBatchNormMulti.apply(x, gamma, beta, rms_h[:1], rvs_h[:1], dict(cfg_e, parts=1))
    ot = BatchNormMulti.apply(x, gamma, beta, rms_t[:1], rvs_t[:1], dict(cfg_e, parts=1))
    assert torch.allclose(oh, ot, atol=1e-4)


def test_losses_parity(dev):
    torch.manual_seed(5)
    from dwt_amd.kernels import hip_ops
    from dwt_amd.ops import functional as Fdwt
    x = torch.randn(18, 65, device=dev, requires_grad=True)
    y = torch.randn(18, 65, device=dev, requires_grad=True)
    l_h = hip_ops.mec_loss(x, y)
    l_t = Fdwt.MecLossFn.apply(x.detach().clone(), y.detach().clone())
    assert torch.allclose(l_h, l_t, atol=1e-5)
    l_h.backward()
    x2 = x.detach().clone().requires_grad_(True)
    y2 = y.detach().clone().requires_grad_(True)
    Fdwt.MecLossFn.apply(x2, y2).backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(y.grad, y2.grad, atol=1e-6)

    z = torch.randn(32, 10, device=dev, requires_grad=True)
    e_h = hip_ops.entropy_loss(z)
    e_t = Fdwt.EntropyLossFn.apply(z.detach().clone())
    assert torch.allclose(e_h, e_t, atol=1e-5)
    e_h.backward()
    z2 = z.detach().clone().requires_grad_(True)
    Fdwt.EntropyLossFn.apply(z2).backward()
    assert torch.allclose(z.grad, z2.grad, atol=1e-6)


def test_model_step_uses_hip_and_trains(dev):
    """Full 3-stream R50-small step on GPU: finite loss, grads flow, and the
    dispatch layer actually routes through the extension."""
    import dwt_amd.kernels.dispatch as dispatch
    assert dispatch.available()
    import torch.nn.functional as F
    from dwt_amd.models import Bottleneck, ResNetDWT
    from dwt_amd.ops import functional as Fdwt

    torch.manual_seed(0)
    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=7)
    model = model.to(dev).to(torch.bfloat16).train()
    opt = torch.optim.SGD(model.parameters(), lr=1e-3, momentum=0.9)
    losses = []
    for _ in range(4):
        data = torch.randn(6, 3, 64, 64, device=dev, dtype=torch.bfloat16)
        labels = torch.randint(0, 7, (2,), device=dev)
        opt.zero_grad(set_to_none=True)
        out = model(data)
        s, t, td = torch.split(out, 2, dim=0)
        loss = F.nll_loss(F.log_softmax(s.float(), 1), labels) + 0.1 * Fdwt.mec_loss(t, td)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    # stats buffers must be fp32 despite bf16 model
    assert model.bnt1.wh.running_mean.dtype == torch.float32


def test_lenet_gpu_step(dev):
    import torch.nn.functional as F
    from dwt_amd.models import LeNet
    from dwt_amd.ops import functional as Fdwt
    model = LeNet(group_size=4).to(dev).train()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    data = torch.randn(16, 1, 28, 28, device=dev)
    labels = torch.randint(0, 10, (8,), device=dev)
    out = model(data)
    s, t = torch.split(out, 8, dim=0)
    loss = F.nll_loss(F.log_softmax(s, 1), labels) + 0.1 * Fdwt.entropy_loss(t)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)


# ---------------------------------------------------------------------------
# channels_last (NHWC) kernel variants
# ---------------------------------------------------------------------------


@pytest.mark.parametrize("mode", ["chol", "zca"])
@pytest.mark.parametrize("c,groups_of", [(64, 16), (256, 64), (512, 128)])
def test_whiten_channels_last_parity(dev, mode, c, groups_of):
    torch.manual_seed(10)
    parts = 3
    x = torch.randn(parts * 2, c, 6, 10, device=dev)
    x_cl = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x_t = x.clone().requires_grad_(True)
    gamma = torch.randn(c, 1, 1, device=dev)
    beta = torch.randn(c, 1, 1, device=dev)
    ga, gb = gamma.clone().requires_grad_(True), gamma.clone().requires_grad_(True)
    ba, bb = beta.clone().requires_grad_(True), beta.clone().requires_grad_(True)
    cfg = _whiten_cfg(parts, groups_of, mode=mode, relu=True)

    out_cl = run_whiten(x_cl, ga, ba, cfg, True)
    out_t = run_whiten(x_t, gb, bb, cfg, False)
    assert out_cl.is_contiguous(memory_format=torch.channels_last)
    assert torch.allclose(out_cl, out_t, atol=3e-4), (out_cl - out_t).abs().max()

    gout = torch.randn_like(out_t)
    out_cl.backward(gout)
    out_t.backward(gout)
    assert torch.allclose(x_cl.grad, x_t.grad, atol=2e-3), (x_cl.grad - x_t.grad).abs().max()
    assert torch.allclose(ga.grad, gb.grad, atol=3e-2)
    assert torch.allclose(ba.grad, bb.grad, atol=3e-2)


def test_whiten_channels_last_bf16(dev):
    torch.manual_seed(11)
    c, groups_of = 64, 16
    x = torch.randn(6, c, 8, 8, device=dev).to(torch.bfloat16)
    x_cl = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x_t = x.clone().requires_grad_(True)
    cfg = _whiten_cfg(3, groups_of)
    out_cl = run_whiten(x_cl, None, None, cfg, True)
    out_t = run_whiten(x_t, None, None, cfg, False)
    assert torch.allclose(out_cl.float(), out_t.float(), atol=0.1, rtol=0.05)
    gout = torch.randn_like(out_t)
    out_cl.backward(gout)
    out_t.backward(gout)
    assert torch.allclose(x_cl.grad.float(), x_t.grad.float(), atol=0.2, rtol=0.1)


def test_bn_channels_last_parity(dev):
    torch.manual_seed(12)
    from dwt_amd.kernels.hip_ops import _HipBatchNormMulti
    from dwt_amd.ops.functional import BatchNormMulti
    c, parts = 128, 3
    x = torch.randn(parts * 4, c, 7, 5, device=dev)
    x_cl = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x_t = x.clone().requires_grad_(True)
    gamma = torch.randn(c, 1, 1, device=dev)
    beta = torch.randn(c, 1, 1, device=dev)
    ga, gb = gamma.clone().requires_grad_(True), gamma.clone().requires_grad_(True)
    ba, bb = beta.clone().requires_grad_(True), beta.clone().requires_grad_(True)
    rms_h = [torch.zeros(c, device=dev) for _ in range(parts)]
    rvs_h = [torch.ones(c, device=dev) for _ in range(parts)]
    rms_t = [t.clone() for t in rms_h]
    rvs_t = [t.clone() for t in rvs_h]
    cfg = dict(parts=parts, eps=1e-5, momentum=0.1, training=True, relu=True)
    out_cl = _HipBatchNormMulti.apply(x_cl, ga, ba, rms_h, rvs_h, cfg)
    out_t = BatchNormMulti.apply(x_t, gb, bb, rms_t, rvs_t, cfg)
    assert torch.allclose(out_cl, out_t, atol=1e-4)
    for h, t in zip(rms_h + rvs_h, rms_t + rvs_t):
        assert torch.allclose(h, t, atol=1e-4)
    gout = torch.randn_like(out_t)
    out_cl.backward(gout)
    out_t.backward(gout)
    assert torch.allclose(x_cl.grad, x_t.grad, atol=1e-3)
    assert torch.allclose(ga.grad, gb.grad, atol=1e-2)


def test_model_channels_last_step(dev):
    import torch.nn.functional as F
    from dwt_amd.models import Bottleneck, ResNetDWT
    from dwt_amd.ops import functional as Fdwt
    torch.manual_seed(1)
    model = ResNetDWT(Bottleneck, [1, 1, 1, 1], None, num_classes=7)
    model = model.to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last).train()
    data = torch.randn(6, 3, 64, 64, device=dev, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    labels = torch.randint(0, 7, (2,), device=dev)
    out = model(data)
    s, t, td = torch.split(out, 2, dim=0)
    loss = F.nll_loss(F.log_softmax(s.float(), 1), labels) + 0.1 * Fdwt.mec_loss(t, td)
    loss.backward()
    assert torch.isfinite(loss)


def test_add_relu_parity(dev):
    from dwt_amd.ops.functional import AddReluFn
    torch.manual_seed(13)
    for mf in (torch.contiguous_format, torch.channels_last):
        a = torch.randn(4, 64, 8, 8, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        b = torch.randn(4, 64, 8, 8, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        a2 = a.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        av = a.contiguous(memory_format=mf)
        out = AddReluFn.apply(av, b.contiguous(memory_format=mf))
        ref = torch.relu(a2 + b2)
        assert torch.allclose(out.float(), ref.float(), atol=1e-2)
        g = torch.randn_like(ref)
        out.backward(g.contiguous(memory_format=mf))
        ref.backward(g)
        assert torch.allclose(a.grad.float(), a2.grad.float(), atol=1e-2)
        assert torch.allclose(b.grad.float(), b2.grad.float(), atol=1e-2)


def test_ce_loss_parity(dev):
    import torch.nn.functional as F
    from dwt_amd.kernels import hip_ops
    torch.manual_seed(14)
    x = torch.randn(18, 65, device=dev, requires_grad=True)
    t = torch.randint(0, 65, (18,), device=dev)
    l_h = hip_ops.ce_loss(x, t)
    ref = F.nll_loss(F.log_softmax(x.detach().float(), dim=1), t)
    assert torch.allclose(l_h, ref, atol=1e-5)
    l_h.backward()
    x2 = x.detach().clone().requires_grad_(True)
    F.nll_loss(F.log_softmax(x2, dim=1), t).backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    # bf16 logits path
    xb = torch.randn(6, 10, device=dev, dtype=torch.bfloat16, requires_grad=True)
    tb = torch.randint(0, 10, (6,), device=dev)
    hip_ops.ce_loss(xb, tb).backward()
    assert xb.grad is not None


def test_pooling_parity(dev):
    import torch.nn.functional as F
    from dwt_amd.ops.pooling import MaxPool2dFn, GlobalAvgPoolFn
    torch.manual_seed(15)
    for ks, st, pad, shape in [(3, 2, 1, (4, 64, 15, 15)),
                               (2, 2, 0, (4, 32, 28, 28))]:
        x = torch.randn(*shape, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last).requires_grad_(True)
        x2 = x.detach().clone().requires_grad_(True)
        out = MaxPool2dFn.apply(x, ks, st, pad)
        ref = F.max_pool2d(x2, kernel_size=ks, stride=st, padding=pad)
        assert torch.allclose(out.float(), ref.float(), atol=1e-2), (ks, st)
        g = torch.randn_like(ref)
        out.backward(g.contiguous(memory_format=torch.channels_last))
        ref.backward(g)
        assert torch.allclose(x.grad.float(), x2.grad.float(), atol=1e-2)

    x = torch.randn(6, 256, 7, 7, device=dev, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    out = GlobalAvgPoolFn.apply(x)
    ref = F.adaptive_avg_pool2d(x2, (1, 1)).reshape(6, 256)
    assert torch.allclose(out.float(), ref.float(), atol=1e-2)
    g = torch.randn_like(ref)
    out.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=1e-2)


def test_sync_stats_split_parity(dev):
    """stats_sync=True at world_size=1 exercises the HIP partial-sums ->
    (all-reduce) -> finalize kernel split; outputs and grads must match the
    fused single-launch path (the gloo ws=2 CPU test pins the cross-rank
    semantics; this pins the GPU kernel split)."""
    from dwt_amd.kernels.hip_ops import _HipWhitenMulti, _HipBatchNormMulti
    torch.manual_seed(0)
    x = torch.randn(12, 16, 6, 6, device=dev)
    gamma = torch.randn(16, 1, 1, device=dev)
    beta = torch.randn(16, 1, 1, device=dev)
    for cl in (False, True):
        xx = x.contiguous(memory_format=torch.channels_last) if cl else x
        for sync in (False, True):
            cfg = dict(parts=3, num_groups=4, eps=1e-3, momentum=0.1,
                       training=True, mode="chol", relu=True,
                       stats_sync=sync)
            xa = xx.clone().requires_grad_(True)
            out = _HipWhitenMulti.apply(xa, gamma.clone().requires_grad_(True),
                                        beta.clone().requires_grad_(True),
                                        None, None, cfg)
            out.sum().backward()
            if sync:
                assert torch.allclose(out, ref_out, atol=1e-6)
                assert torch.allclose(xa.grad, ref_grad, atol=1e-6)
            else:
                ref_out, ref_grad = out.detach(), xa.grad.detach()

        for sync in (False, True):
            bcfg = dict(parts=3, eps=1e-5, momentum=0.1, training=True,
                        relu=True, stats_sync=sync)
            xa = xx.clone().requires_grad_(True)
            out = _HipBatchNormMulti.apply(xa, gamma.clone().requires_grad_(True),
                                           beta.clone().requires_grad_(True),
                                           None, None, bcfg)
            out.sum().backward()
            if sync:
                assert torch.allclose(out, bref_out, atol=1e-6)
                assert torch.allclose(xa.grad, bref_grad, atol=1e-6)
            else:
                bref_out, bref_grad = out.detach(), xa.grad.detach()
