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
