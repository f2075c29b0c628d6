"""CPU validation of the wgrad formulation + index decode: the exact gather
the HIP kernel performs (replicated in _wgrad_patches_reference) times dy^T
must equal aten's conv weight gradient."""
import pytest
import torch
import torch.nn.functional as F

from dwt_amd.ops.mfma import _wgrad_patches_reference


@pytest.mark.parametrize("shape", [
    # (N, Cin, H, W, Cout, K, stride, pad)
    (2, 3, 8, 8, 4, 3, 1, 1),
    (2, 4, 9, 7, 5, 3, 2, 1),
    (3, 2, 6, 6, 4, 1, 1, 0),
    (1, 3, 10, 10, 2, 7, 2, 3),
])
def test_wgrad_formulation_matches_autograd(shape):
    torch.manual_seed(0)
    n, cin, h, w, cout, k, stride, pad = shape
    x = torch.randn(n, cin, h, w, dtype=torch.float64)
    wt = torch.randn(cout, cin, k, k, dtype=torch.float64, requires_grad=True)
    out = F.conv2d(x, wt, stride=stride, padding=pad)
    g = torch.randn_like(out)
    out.backward(g)
    dw_ref = wt.grad  # (Cout, Cin, K, K)

    # kernel formulation: dw[co][(r,s,ci)] = dy^T @ patches^T
    p, q = out.shape[2], out.shape[3]
    dyt = g.permute(0, 2, 3, 1).reshape(n * p * q, cout).t()  # (Cout, NPQ)
    x_nhwc = x.permute(0, 2, 3, 1).contiguous()
    # pad K to %8 like the GPU wrapper does — padded columns must be zeros
    npq = n * p * q
    npq_pad = (npq + 7) // 8 * 8
    cols = _wgrad_patches_reference(x_nhwc, (cout, cin, k, k), stride, pad,
                                    npq_pad=npq_pad)
    dyt_pad = torch.zeros(cout, npq_pad, dtype=torch.float64)
    dyt_pad[:, :npq] = dyt
    dw_flat = dyt_pad @ cols.t()  # (Cout, K*K*Cin), columns (r,s,ci)
    dw_ours = dw_flat.reshape(cout, k, k, cin).permute(0, 3, 1, 2)
    assert torch.allclose(dw_ours, dw_ref, atol=1e-9), \
        (dw_ours - dw_ref).abs().max()
