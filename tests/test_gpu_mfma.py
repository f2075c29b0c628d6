"""MFMA GEMM / implicit-GEMM conv kernels vs torch (rocBLAS/MIOpen)."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    from dwt_amd.kernels import dispatch
    assert dispatch.available()
    return torch.device("cuda:0")


def _rel_err(a, b):
    return ((a.float() - b.float()).abs().max() /
            b.float().abs().max().clamp_min(1e-6)).item()


@pytest.mark.parametrize("m,k,n", [
    (256, 128, 128),       # exact tiles
    (300, 72, 65),         # every dim ragged (K padded on host)
    (1024, 2048, 65),      # fc_out shape
    (192, 2048, 512),
])
def test_mfma_gemm_vs_matmul(dev, m, k, n):
    torch.manual_seed(0)
    from dwt_amd.ops.mfma import mfma_gemm
    a = torch.randn(m, k, device=dev).to(torch.bfloat16)
    bt = torch.randn(n, k, device=dev).to(torch.bfloat16)
    ours = mfma_gemm(a, bt)
    ref = (a.float() @ bt.float().t()).to(torch.bfloat16)
    assert _rel_err(ours, ref) < 0.02, _rel_err(ours, ref)


def test_mfma_gemm_asymmetric_transpose_check(dev):
    """Transpose-detecting inputs (guide §5.4 rule 16): A=I with asymmetric B."""
    from dwt_amd.ops.mfma import mfma_gemm
    k = 64
    a = torch.eye(k, device=dev).to(torch.bfloat16)
    bt = (torch.arange(k * k, device=dev).reshape(k, k).float() % 37).to(torch.bfloat16)
    ours = mfma_gemm(a, bt).float()
    assert torch.allclose(ours, bt.float().t(), atol=1e-2)


def test_mfma_gemm_bias_relu(dev):
    from dwt_amd.ops.mfma import mfma_gemm
    a = torch.randn(100, 64, device=dev).to(torch.bfloat16)
    bt = torch.randn(32, 64, device=dev).to(torch.bfloat16)
    bias = torch.randn(32, device=dev)
    ours = mfma_gemm(a, bt, bias=bias, relu=True)
    ref = torch.relu(a.float() @ bt.float().t() + bias)
    assert _rel_err(ours, ref.to(torch.bfloat16)) < 0.03


def test_mfma_linear_fwd_bwd(dev):
    from dwt_amd.ops.mfma import MFMALinear
    torch.manual_seed(1)
    lin = MFMALinear(2048, 65).to(dev).to(torch.bfloat16)
    ref = torch.nn.Linear(2048, 65).to(dev).to(torch.bfloat16)
    with torch.no_grad():
        ref.weight.copy_(lin.weight)
        ref.bias.copy_(lin.bias)
    x1 = torch.randn(192, 2048, device=dev, dtype=torch.bfloat16, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1, y2 = lin(x1), ref(x2)
    assert _rel_err(y1, y2) < 0.05
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert _rel_err(x1.grad, x2.grad) < 0.05
    assert _rel_err(lin.weight.grad, ref.weight.grad) < 0.05
    assert _rel_err(lin.bias.grad, ref.bias.grad) < 0.05


@pytest.mark.parametrize("shape", [
    # (N, Cin, H, W, Cout, K, stride, pad) — R50 conv shapes
    (6, 64, 56, 56, 64, 1, 1, 0),       # bottleneck conv1
    (6, 64, 56, 56, 64, 3, 1, 1),       # bottleneck conv2
    (6, 64, 56, 56, 256, 1, 1, 0),      # bottleneck conv3
    (6, 256, 56, 56, 128, 1, 1, 0),
    (6, 128, 56, 56, 128, 3, 2, 1),     # strided 3x3
    (6, 3, 64, 64, 64, 7, 2, 3),        # stem (Cin=3 slow-gather path)
])
def test_mfma_conv_fwd_vs_miopen(dev, shape):
    torch.manual_seed(2)
    from dwt_amd.ops.mfma import conv2d_fwd
    n, cin, h, w, cout, k, stride, pad = shape
    x = torch.randn(n, cin, h, w, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(cout, cin, k, k, device=dev) / (k * cin ** 0.5)) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    ours = conv2d_fwd(x, wt, stride=stride, padding=pad)
    ref = F.conv2d(x, wt, stride=stride, padding=pad)
    err = _rel_err(ours, ref)
    assert err < 0.05, (shape, err)
    assert ours.is_contiguous(memory_format=torch.channels_last)


@pytest.mark.parametrize("shape", [
    (4, 64, 28, 28, 64, 1, 1, 0),
    (4, 64, 28, 28, 64, 3, 1, 1),
    (4, 128, 28, 28, 128, 3, 2, 1),
])
def test_mfma_conv_dgrad_vs_miopen(dev, shape):
    torch.manual_seed(3)
    from dwt_amd.ops.mfma import conv2d_dgrad
    n, cin, h, w, cout, k, stride, pad = shape
    x = torch.randn(n, cin, h, w, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    wt = (torch.randn(cout, cin, k, k, device=dev) / (k * cin ** 0.5)) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    out = F.conv2d(x, wt, stride=stride, padding=pad)
    g = torch.randn_like(out).contiguous(memory_format=torch.channels_last)
    out.backward(g)
    dx_ref = x.grad
    dx = conv2d_dgrad(g, wt, x.shape, stride=stride, padding=pad)
    assert _rel_err(dx, dx_ref) < 0.05, (shape, _rel_err(dx, dx_ref))


def test_mfma_conv2d_module_trains(dev):
    from dwt_amd.ops.mfma import MFMAConv2d
    torch.manual_seed(4)
    conv = MFMAConv2d(64, 128, kernel_size=3, stride=1, padding=1, bias=False)
    conv = conv.to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
    ref = torch.nn.Conv2d(64, 128, 3, 1, 1, bias=False)
    ref = ref.to(dev).to(torch.bfloat16).to(memory_format=torch.channels_last)
    with torch.no_grad():
        ref.weight.copy_(conv.weight)
    x1 = torch.randn(4, 64, 16, 16, device=dev, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1, y2 = conv(x1), ref(x2)
    assert _rel_err(y1, y2) < 0.05
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert _rel_err(x1.grad, x2.grad) < 0.05
    assert _rel_err(conv.weight.grad, ref.weight.grad) < 0.05


@pytest.mark.parametrize("shape", [
    (4, 64, 28, 28, 64, 1, 1, 0),
    (4, 64, 28, 28, 64, 3, 1, 1),
    (4, 128, 28, 28, 128, 3, 2, 1),
    (4, 3, 64, 64, 64, 7, 2, 3),     # stem: Cin=3 scalar-gather B path
    (2, 512, 14, 14, 512, 3, 2, 1),  # deep layer, Cin%64==0 aligned path
    (4, 256, 14, 14, 1024, 1, 1, 0),
])
def test_mfma_conv_wgrad_vs_miopen(dev, shape):
    torch.manual_seed(5)
    from dwt_amd.ops.mfma import conv2d_wgrad
    n, cin, h, w, cout, k, stride, pad = shape
    x = torch.randn(n, cin, h, w, device=dev).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(cout, cin, k, k, device=dev) / (k * cin ** 0.5)) \
        .to(torch.bfloat16).contiguous(memory_format=torch.channels_last) \
        .requires_grad_(True)
    out = F.conv2d(x, wt, stride=stride, padding=pad)
    g = torch.randn_like(out).contiguous(memory_format=torch.channels_last)
    out.backward(g)
    dw = conv2d_wgrad(g, x, tuple(wt.shape), stride=stride, padding=pad)
    assert _rel_err(dw, wt.grad) < 0.05, (shape, _rel_err(dw, wt.grad))
