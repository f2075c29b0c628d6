"""Schedule-stability pass over the LDS-reduction kernels (SURVEY §5 race
row; docs/SANITIZER.md describes the full methodology).

A missing barrier or a racy LDS protocol shows up as run-to-run output
divergence far above rounding (lost updates drop whole per-block partials;
torn tiles produce garbage).  The reductions here accumulate per-block fp32
partials with atomicAdd, whose ARRIVAL ORDER legitimately varies between
launches, so exact bitwise equality is not the right oracle — instead each
op is launched many times on identical inputs and the spread across runs
must stay within a few ulps-worth of fp32 reassociation noise (races
measured during development produced relative errors of 1e-2..inf, seven+
orders of magnitude above this bound)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

REPEATS = 12


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    from dwt_amd.kernels import dispatch
    assert dispatch.available()
    return torch.device("cuda:0")


def _spread(fn, n=REPEATS):
    """max over runs/outputs of rel. deviation from run 0."""
    outs = [fn() for _ in range(n)]
    ref = outs[0]
    worst = 0.0
    for o in outs[1:]:
        for a, b in zip(ref, o):
            assert torch.isfinite(b.float()).all()
            d = (a.float() - b.float()).abs().max().item()
            s = a.float().abs().max().clamp_min(1e-6).item()
            worst = max(worst, d / s)
    return worst


def test_whiten_reduction_schedule_stability(dev):
    from dwt_amd.kernels.hip_ops import _HipWhitenMulti
    torch.manual_seed(0)
    for cl in (False, True):
        x0 = torch.randn(12, 64, 28, 28, device=dev)
        if cl:
            x0 = x0.contiguous(memory_format=torch.channels_last)
        gamma = torch.randn(64, 1, 1, device=dev)
        beta = torch.randn(64, 1, 1, device=dev)
        cfg = dict(parts=3, num_groups=16, eps=1e-3, momentum=0.1,
                   training=True, mode="zca", relu=True)
        gout = torch.randn(12, 64, 28, 28, device=dev)

        def step():
            x = x0.clone().requires_grad_(True)
            g = gamma.clone().requires_grad_(True)
            b = beta.clone().requires_grad_(True)
            out = _HipWhitenMulti.apply(x, g, b, None, None, cfg)
            out.backward(gout)
            return out.detach().clone(), x.grad.clone(), g.grad.clone(), b.grad.clone()

        assert _spread(step) < 1e-4


def test_bn_reduction_schedule_stability(dev):
    from dwt_amd.kernels.hip_ops import _HipBatchNormMulti
    torch.manual_seed(1)
    x0 = torch.randn(12, 128, 14, 14, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    gamma = torch.randn(128, 1, 1, device=dev)
    beta = torch.randn(128, 1, 1, device=dev)
    cfg = dict(parts=3, eps=1e-5, momentum=0.1, training=True, relu=True)
    gout = torch.randn_like(x0)

    def step():
        x = x0.clone().requires_grad_(True)
        g = gamma.clone().requires_grad_(True)
        b = beta.clone().requires_grad_(True)
        out = _HipBatchNormMulti.apply(x, g, b, None, None, cfg)
        out.backward(gout)
        return out.detach().clone(), x.grad.clone(), g.grad.clone(), b.grad.clone()

    assert _spread(step) < 1e-4


def test_generic_group_schedule_stability(dev):
    """The g=16/32 LDS-tiled kernels (pair-owned accumulators + atomics)."""
    from dwt_amd.kernels.hip_ops import _HipWhitenMulti
    torch.manual_seed(2)
    for g, c in ((16, 48), (32, 32)):
        x0 = torch.randn(9, c, 9, 9, device=dev)
        cfg = dict(parts=3, num_groups=c // g, eps=1e-3, momentum=0.1,
                   training=True, mode="chol", relu=False)
        gout = torch.randn_like(x0)

        def step():
            x = x0.clone().requires_grad_(True)
            out = _HipWhitenMulti.apply(x, None, None, None, None, cfg)
            out.backward(gout)
            return out.detach().clone(), x.grad.clone()

        assert _spread(step) < 1e-4


def test_wgrad_splitk_schedule_stability(dev):
    """Split-K wgrad: many k-slab blocks atomicAdd into the fp32 workspace;
    arrival-order reassociation is bounded rounding, a tile/barrier race is
    not."""
    from dwt_amd.ops.mfma import conv2d_wgrad
    torch.manual_seed(3)
    x = torch.randn(4, 128, 28, 28, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = torch.randn(4, 128, 28, 28, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)

    def step():
        return (conv2d_wgrad(dy, x, (128, 128, 3, 3), stride=1, padding=1).float(),)

    # bf16 output quantization: order effects either vanish in the rounding
    # or flip one ulp
    assert _spread(step) < 2e-2
