"""fpn_fuse op tests: CPU fallback correctness + GPU kernel numerics."""

import pytest
import torch
import torch.nn.functional as F

from maskrcnn_amd.ops.fpn_fuse import fpn_fuse


def _ref(x, w, b, up):
    out = F.conv2d(x.float(), w.float().view(w.size(0), -1, 1, 1),
                   b.float())
    if up is not None:
        out = out + F.interpolate(up.float(), scale_factor=2,
                                  mode="nearest")
    return out


def test_cpu_fallback_matches_conv():
    torch.manual_seed(0)
    x = torch.randn(2, 32, 8, 8)
    w = torch.randn(16, 32, 1, 1)
    b = torch.randn(16)
    up = torch.randn(2, 16, 4, 4)
    out = fpn_fuse(x, w, b, up)
    torch.testing.assert_close(out, _ref(x, w, b, up), atol=1e-5,
                               rtol=1e-5)


def test_cpu_backward_matches_autograd():
    torch.manual_seed(1)
    x = torch.randn(1, 32, 4, 4, requires_grad=True)
    w = torch.randn(8, 32, 1, 1, requires_grad=True)
    b = torch.randn(8, requires_grad=True)
    up = torch.randn(1, 8, 2, 2, requires_grad=True)

    out = fpn_fuse(x, w, b, up)
    g = torch.randn_like(out)
    out.backward(g)
    gx, gw, gb, gu = x.grad, w.grad, b.grad, up.grad

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    u2 = up.detach().clone().requires_grad_(True)
    _ref(x2, w2, b2, u2).backward(g)
    torch.testing.assert_close(gx, x2.grad, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(gw, w2.grad, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(gb, b2.grad, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(gu, u2.grad, atol=1e-4, rtol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize("cin,cout,h", [(256, 256, 32), (2048, 256, 32),
                                        (512, 256, 64), (256, 64, 16)])
def test_gpu_kernel_matches_reference(cin, cout, h):
    from maskrcnn_amd.ops._ext import has_ext
    assert has_ext()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    x = torch.randn(2, cin, h, h, device=dev).bfloat16() \
        .to(memory_format=torch.channels_last)
    w = torch.randn(cout, cin, 1, 1, device=dev) * 0.05
    b = torch.randn(cout, device=dev)
    up = torch.randn(2, cout, h // 2, h // 2, device=dev).bfloat16() \
        .to(memory_format=torch.channels_last)
    out = fpn_fuse(x, w, b, up)
    assert out.dtype == torch.bfloat16
    ref = _ref(x.float(), w, b, up.float())
    # bf16 inputs + fp32 accum: tolerance scaled to K=cin dot length.
    torch.testing.assert_close(out.float(), ref, atol=0.15, rtol=0.05)


@pytest.mark.gpu
def test_gpu_kernel_no_up():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    x = torch.randn(1, 256, 16, 16, device=dev).bfloat16() \
        .to(memory_format=torch.channels_last)
    w = torch.randn(256, 256, 1, 1, device=dev) * 0.05
    b = torch.zeros(256, device=dev)
    out = fpn_fuse(x, w, b)
    ref = _ref(x.float(), w, b, None)
    torch.testing.assert_close(out.float(), ref, atol=0.1, rtol=0.05)


@pytest.mark.gpu
def test_gpu_kernel_asymmetric_catches_transpose():
    """Asymmetric weight (G9 rule): detects operand/output transposes."""
    dev = torch.device("cuda:0")
    x = torch.zeros(1, 32, 16, 16, device=dev).bfloat16() \
        .to(memory_format=torch.channels_last)
    x[0, 3, 5, 7] = 2.0
    w = torch.zeros(64, 32, 1, 1, device=dev)
    w[11, 3] = 1.5
    b = torch.zeros(64, device=dev)
    out = fpn_fuse(x, w, b).float()
    assert out[0, 11, 5, 7].item() == pytest.approx(3.0, rel=0.01)
    assert out.abs().sum().item() == pytest.approx(3.0, rel=0.01)
