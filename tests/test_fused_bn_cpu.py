"""FusedBatchNorm2d CPU reference path vs torch BatchNorm2d."""

import torch

from chainermn_amd.links import FusedBatchNorm2d


def test_fused_bn_cpu_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(4, 6, 5, 5)
    xr = x.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    bn = FusedBatchNorm2d(6)
    ref = torch.nn.BatchNorm2d(6)
    y = bn(xf)
    yr = ref(xr)
    gy = torch.randn_like(y)
    y.backward(gy)
    yr.backward(gy)
    assert torch.allclose(y, yr, atol=1e-5)
    assert torch.allclose(xf.grad, xr.grad, atol=1e-5)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=1e-4)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-5)


def test_fused_bn_relu_and_add_relu_cpu():
    from chainermn_amd.links import FusedBNReLU2d
    torch.manual_seed(1)
    x = torch.randn(4, 8, 5, 5)
    res = torch.randn(4, 8, 5, 5)
    # bn+relu vs composed
    xf = x.clone().requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    bn = FusedBNReLU2d(8)
    ref = torch.nn.BatchNorm2d(8)
    y = bn(xf)
    yr = torch.relu(ref(xr))
    gy = torch.randn_like(y)
    y.backward(gy)
    yr.backward(gy)
    assert torch.allclose(y, yr, atol=1e-5)
    assert torch.allclose(xf.grad, xr.grad, atol=1e-5)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=1e-4)
    # bn+add+relu vs composed
    from chainermn_amd.links import FusedBatchNorm2d
    xf2 = x.clone().requires_grad_(True)
    rf2 = res.clone().requires_grad_(True)
    xr2 = x.clone().requires_grad_(True)
    rr2 = res.clone().requires_grad_(True)
    bn2 = FusedBatchNorm2d(8)
    ref2 = torch.nn.BatchNorm2d(8)
    y2 = bn2.forward_add_relu(xf2, rf2)
    yr2 = torch.relu(ref2(xr2) + rr2)
    gy2 = torch.randn_like(y2)
    y2.backward(gy2)
    yr2.backward(gy2)
    assert torch.allclose(y2, yr2, atol=1e-5)
    assert torch.allclose(xf2.grad, xr2.grad, atol=1e-5)
    assert torch.allclose(rf2.grad, rr2.grad, atol=1e-5)


def test_resnet_fused_ops_cpu_matches_plain():
    from chainermn_amd.models import ResNet50
    torch.manual_seed(2)
    m1 = ResNet50(num_classes=10, fused_ops=True)
    torch.manual_seed(2)
    m2 = ResNet50(num_classes=10)
    x = torch.randn(2, 3, 32, 32)
    y1 = m1(x)
    y2 = m2(x)
    assert torch.allclose(y1, y2, atol=1e-3), (y1 - y2).abs().max()
