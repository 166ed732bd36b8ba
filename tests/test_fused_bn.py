"""Fused BN(+add)+ReLU: CPU fallback equivalence (always) and GPU kernel
numerics vs the eager composition (gpu-marked)."""
import pytest
import torch
import torch.nn.functional as F

from cpd_amd.models.fused_bn import FusedBNReLU
from cpd_amd.models import res_cifar


def test_cpu_fallback_matches_eager_bn():
    torch.manual_seed(0)
    m = FusedBNReLU(8, relu=True)
    ref = torch.nn.BatchNorm2d(8)
    ref.load_state_dict({k: v for k, v in m.state_dict().items()})
    x = torch.randn(4, 8, 6, 6)
    y = m(x)
    want = F.relu(ref(x))
    torch.testing.assert_close(y, want)
    torch.testing.assert_close(m.running_mean, ref.running_mean)
    torch.testing.assert_close(m.running_var, ref.running_var)


def test_fused_model_statedict_compatible():
    a = res_cifar(fused_bn=True)
    b = res_cifar(fused_bn=False)
    ka = set(a.state_dict().keys())
    kb = set(b.state_dict().keys())
    assert ka == kb
    # cross-load works
    b.load_state_dict(a.state_dict())
    a.load_state_dict(b.state_dict())


@pytest.mark.gpu
@pytest.mark.parametrize("relu,with_res", [(True, False), (False, False),
                                           (True, True)])
def test_fused_bn_gpu_matches_eager(relu, with_res):
    torch.manual_seed(1)
    N, C, H, W = 16, 32, 8, 8
    x = torch.randn(N, C, H, W, device="cuda", requires_grad=True)
    res = torch.randn(N, C, H, W, device="cuda", requires_grad=True) \
        if with_res else None

    m = FusedBNReLU(C, relu=relu).cuda().train()
    with torch.no_grad():
        m.weight.mul_(0).add_(torch.rand(C, device="cuda") + 0.5)
        m.bias.add_(torch.randn(C, device="cuda") * 0.1)

    ref_bn = torch.nn.BatchNorm2d(C).cuda().train()
    ref_bn.load_state_dict({k: v for k, v in m.state_dict().items()})

    y = m(x, residual=res)
    xe = x.detach().clone().requires_grad_(True)
    rese = res.detach().clone().requires_grad_(True) if with_res else None
    ye = ref_bn(xe)
    if with_res:
        ye = ye + rese
    if relu:
        ye = F.relu(ye)
    torch.testing.assert_close(y, ye, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(m.running_mean, ref_bn.running_mean,
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(m.running_var, ref_bn.running_var,
                               rtol=1e-5, atol=1e-6)

    g = torch.randn_like(y)
    y.backward(g)
    ye.backward(g)
    torch.testing.assert_close(x.grad, xe.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.weight.grad, ref_bn.weight.grad,
                               rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(m.bias.grad, ref_bn.bias.grad,
                               rtol=1e-4, atol=1e-4)
    if with_res:
        torch.testing.assert_close(res.grad, rese.grad, rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_fused_resnet18_matches_eager_model():
    torch.manual_seed(2)
    fused = res_cifar(fused_bn=True).cuda().train()
    eager = res_cifar(fused_bn=False).cuda().train()
    eager.load_state_dict(fused.state_dict())
    x = torch.randn(8, 3, 32, 32, device="cuda")
    y1 = fused(x)
    y2 = eager(x)
    # fp32 with a different per-channel reduction order than MIOpen BN —
    # differences compound through 18 layers, so compare relative error
    rel = (y1 - y2).norm() / y2.norm().clamp_min(1e-12)
    assert rel < 1e-3, float(rel)
    (y1.square().sum()).backward()
    (y2.square().sum()).backward()
    for (n1, p1), (n2, p2) in zip(fused.named_parameters(),
                                  eager.named_parameters()):
        relg = (p1.grad - p2.grad).norm() / p2.grad.norm().clamp_min(1e-8)
        assert relg < 5e-2, (n1, float(relg))


@pytest.mark.gpu
@pytest.mark.parametrize("relu,with_res,C,HW", [
    (True, False, 64, (8, 8)), (False, False, 32, (7, 9)),
    (True, True, 128, (4, 4)), (True, False, 68, (6, 6)),  # C%8!=0: no mask
])
def test_fused_bn_nhwc_matches_eager(relu, with_res, C, HW):
    """channels_last path: native NHWC kernels (no layout transposes) vs
    the eager NCHW composition."""
    torch.manual_seed(3)
    N, (H, W) = 16, HW
    fmt = torch.channels_last
    x = torch.randn(N, C, H, W, device="cuda").to(memory_format=fmt)
    x.requires_grad_(True)
    res = None
    if with_res:
        res = torch.randn(N, C, H, W, device="cuda").to(memory_format=fmt)
        res.requires_grad_(True)

    m = FusedBNReLU(C, relu=relu).cuda().train()
    with torch.no_grad():
        m.weight.mul_(0).add_(torch.rand(C, device="cuda") + 0.5)
        m.bias.add_(torch.randn(C, device="cuda") * 0.1)
    ref_bn = torch.nn.BatchNorm2d(C).cuda().train()
    ref_bn.load_state_dict({k: v for k, v in m.state_dict().items()})

    y = m(x, residual=res)
    assert y.is_contiguous(memory_format=fmt)
    xe = x.detach().clone().contiguous().requires_grad_(True)
    rese = res.detach().clone().contiguous().requires_grad_(True) \
        if with_res else None
    ye = ref_bn(xe)
    if with_res:
        ye = ye + rese
    if relu:
        ye = F.relu(ye)
    torch.testing.assert_close(y.contiguous(), ye, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(m.running_mean, ref_bn.running_mean,
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(m.running_var, ref_bn.running_var,
                               rtol=1e-5, atol=1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    ye.backward(g.contiguous())
    torch.testing.assert_close(x.grad.contiguous(), xe.grad,
                               rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.weight.grad, ref_bn.weight.grad,
                               rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(m.bias.grad, ref_bn.bias.grad,
                               rtol=1e-4, atol=1e-4)
    if with_res:
        torch.testing.assert_close(res.grad.contiguous(), rese.grad,
                                   rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
def test_fused_bn_nhwc_nchw_agree():
    """Same data through both layout paths: channel sums use the same
    fixed-shape two-level reduction, so results must agree to fp32
    round-off; both must be deterministic."""
    torch.manual_seed(4)
    N, C, H, W = 32, 64, 16, 16
    xc = torch.randn(N, C, H, W, device="cuda")
    outs = {}
    for fmt in (torch.contiguous_format, torch.channels_last):
        m = FusedBNReLU(C, relu=True).cuda().train()
        # clone: .to(memory_format=contiguous_format) would alias xc
        x = xc.clone(memory_format=fmt).requires_grad_(True)
        y = m(x)
        y.square().sum().backward()
        outs[fmt] = (y.detach().contiguous(), x.grad.contiguous(),
                     m.weight.grad.clone())
    a, b = outs[torch.contiguous_format], outs[torch.channels_last]
    torch.testing.assert_close(a[0], b[0], rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(a[1], b[1], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(a[2], b[2], rtol=1e-4, atol=1e-4)
