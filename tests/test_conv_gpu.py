"""Direct NHWC conv kernel numerics vs fp32 torch conv (tolerance-tiered
bf16, same style as test_kernels_gpu.py)."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
@pytest.mark.parametrize("cin,h,cout,k,s,p", [
    (64, 56, 64, 3, 1, 1),
    (128, 28, 128, 3, 1, 1),
    (256, 14, 256, 3, 1, 1),
    (512, 7, 512, 3, 1, 1),     # split-K path (starved grid)
    (64, 56, 128, 3, 2, 1),
    (256, 14, 512, 1, 2, 0),
])
def test_conv_fwd_matches_fp32(cin, h, cout, k, s, p):
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from vantage6_amd.ops import hip_ops

    torch.manual_seed(0)
    x = torch.randn(8, cin, h, h, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    w = (torch.randn(cout, cin, k, k, device="cuda", dtype=torch.bfloat16)
         * (2.0 / (cin * k * k)) ** 0.5).to(memory_format=torch.channels_last)
    ref = F.conv2d(x.float(), w.float(), stride=s, padding=p)
    got = hip_ops().conv_nhwc_fwd(x, w, s, p).float()
    scale = ref.abs().max().item()
    assert (got - ref).abs().max().item() <= 0.03 * scale + 0.05


@pytest.mark.timeout(300)
def test_conv_module_backward_matches_library():
    """Conv2d wrapper: gradients equal the all-library path."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    import torch.nn as nn

    from vantage6_amd.ops.conv import Conv2d as HipConv2d

    torch.manual_seed(1)
    x0 = torch.randn(4, 64, 28, 28, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    ours = HipConv2d(64, 128, 3, padding=1, bias=False).cuda().bfloat16().to(
        memory_format=torch.channels_last)
    lib = nn.Conv2d(64, 128, 3, padding=1, bias=False).cuda().bfloat16().to(
        memory_format=torch.channels_last)
    with torch.no_grad():
        lib.weight.copy_(ours.weight)
    xa = x0.clone().requires_grad_(True)
    xb = x0.clone().requires_grad_(True)
    ya = ours(xa)
    yb = lib(xb)
    scale = yb.float().abs().max().item()
    assert (ya.float() - yb.float()).abs().max().item() <= 0.03 * scale + 0.05
    ya.float().pow(2).mean().backward()
    yb.float().pow(2).mean().backward()
    gs = xb.grad.float().abs().max().item()
    assert (xa.grad.float() - xb.grad.float()).abs().max().item() \
        <= 0.05 * gs + 0.05
    ws = lib.weight.grad.float().abs().max().item()
    assert (ours.weight.grad.float() - lib.weight.grad.float()).abs().max() \
        .item() <= 0.05 * ws + 0.05


@pytest.mark.timeout(300)
@pytest.mark.parametrize("cin,h,cout,k,s,p", [
    (64, 56, 64, 3, 1, 1),
    (128, 28, 128, 3, 1, 1),
    (512, 7, 512, 3, 1, 1),
    (64, 56, 128, 3, 2, 1),
    (256, 14, 512, 1, 2, 0),
])
def test_conv_wgrad_matches_fp32(cin, h, cout, k, s, p):
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from vantage6_amd.ops import hip_ops

    torch.manual_seed(2)
    x = torch.randn(8, cin, h, h, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)
    oh = (h + 2 * p - k) // s + 1
    dy = torch.randn(8, cout, oh, oh, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last) * 0.1
    # fp32 reference weight grad
    xr = x.float().requires_grad_(True)
    wr = torch.zeros(cout, cin, k, k, device="cuda", requires_grad=True)
    F.conv2d(xr, wr, stride=s, padding=p).backward(dy.float())
    ref = wr.grad
    got = hip_ops().conv_nhwc_wgrad(dy, x, k, k, s, p).float()
    scale = ref.abs().max().item()
    err = (got - ref).abs().max().item()
    assert err <= 0.05 * scale + 0.05, (err, scale)


@pytest.mark.timeout(300)
def test_conv_bn_fused_stats_matches_plain():
    """Conv-epilogue BN-stat accumulation must reproduce the separate
    partials pass (same bf16-rounded inputs; only atomic add order
    differs)."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from vantage6_amd.ops.batchnorm import FusedBatchNorm2d
    from vantage6_amd.ops.conv import Conv2d as HipConv2d

    torch.manual_seed(0)
    x = torch.randn(8, 64, 28, 28, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last)

    def build(fused):
        torch.manual_seed(1)
        conv = HipConv2d(64, 128, 3, padding=1, bias=False).cuda().bfloat16() \
            .to(memory_format=torch.channels_last)
        bn = FusedBatchNorm2d(128).cuda()
        if fused:
            object.__setattr__(conv, "stats_bn", bn)
        conv.train(), bn.train()
        y = bn(conv(x), relu=True)
        return y, bn

    y_plain, bn_plain = build(False)
    y_fused, bn_fused = build(True)
    assert bn_fused._pending_stats is False  # consumed
    scale = y_plain.float().abs().max().item()
    err = (y_fused.float() - y_plain.float()).abs().max().item()
    assert err <= 2e-2 * scale + 2e-2, err
    rm = (bn_fused.running_mean - bn_plain.running_mean).abs().max().item()
    rv = (bn_fused.running_var - bn_plain.running_var).abs().max().item()
    assert rm < 1e-3 and rv < 1e-3, (rm, rv)


@pytest.mark.timeout(300)
def test_maxpool_nhwc_matches_torch():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from vantage6_amd.ops.conv import MaxPool2d as HipMaxPool2d

    torch.manual_seed(4)
    for h in (112, 56, 17):  # odd size exercises the tail windows
        x0 = torch.randn(4, 64, h, h, device="cuda", dtype=torch.bfloat16).to(
            memory_format=torch.channels_last)
        xa = x0.clone().requires_grad_(True)
        xb = x0.clone().requires_grad_(True)
        ya = HipMaxPool2d()(xa)
        yb = torch.nn.functional.max_pool2d(xb, 3, stride=2, padding=1)
        assert torch.equal(ya.float(), yb.float()), h
        g = torch.randn_like(ya)
        ya.backward(g)
        yb.backward(g)
        # argmax TIES may be broken differently — compare where unique;
        # bf16 values tie often, so allow a tolerance via sums instead
        assert torch.allclose(xa.grad.float().sum(), xb.grad.float().sum(),
                              rtol=1e-3), h
        d = (xa.grad.float() - xb.grad.float()).abs()
        assert (d > 0).float().mean().item() < 0.02, h  # only tie positions
