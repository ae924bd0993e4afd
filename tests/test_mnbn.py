"""MultiNodeBatchNormalization gold test (reference:
links_tests/test_batch_normalization.py): MNBN over 2 ranks with half
batches must match plain BatchNorm run on the concatenated global batch —
forward, input grads, and parameter grads."""

import numpy
import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.links import MultiNodeBatchNormalization
from chainermn_amd.links import create_mnbn_model
from chainermn_amd.testing import run_distributed

B, C, H, W = 4, 5, 3, 3


def _mnbn_worker(rank, size, x_all_np, gy_all_np):
    comm = chainermn_amd.create_communicator('naive')
    x_all = torch.from_numpy(x_all_np)
    gy_all = torch.from_numpy(gy_all_np)
    x = x_all[rank * B:(rank + 1) * B].clone().requires_grad_(True)
    gy = gy_all[rank * B:(rank + 1) * B]
    bn = MultiNodeBatchNormalization(C, comm, decay=0.9, eps=2e-5)
    y = bn(x)
    y.backward(gy)
    return {
        'y': y.detach(),
        'gx': x.grad,
        'ggamma': bn.weight.grad,
        'gbeta': bn.bias.grad,
        'running_mean': bn.running_mean,
        'running_var': bn.running_var,
    }


def test_mnbn_matches_global_batch_bn():
    size = 2
    rng = numpy.random.RandomState(0)
    x_all = rng.randn(size * B, C, H, W).astype(numpy.float32)
    gy_all = rng.randn(size * B, C, H, W).astype(numpy.float32)
    results = run_distributed(_mnbn_worker, size, args=(x_all, gy_all))

    # serial reference: plain BN over the whole batch (biased var, like BN)
    x = torch.from_numpy(x_all).clone().requires_grad_(True)
    bn = nn.BatchNorm2d(C, eps=2e-5, momentum=0.1)
    y = bn(x)
    y.backward(torch.from_numpy(gy_all))

    for rank in range(size):
        r = results[rank]
        got_y = torch.from_numpy(r['y'])
        ref_y = y.detach()[rank * B:(rank + 1) * B]
        assert torch.allclose(got_y, ref_y, atol=1e-5), \
            (got_y - ref_y).abs().max()
        got_gx = torch.from_numpy(r['gx'])
        ref_gx = x.grad[rank * B:(rank + 1) * B]
        assert torch.allclose(got_gx, ref_gx, atol=1e-5), \
            (got_gx - ref_gx).abs().max()
        # parameter grads: every rank returns the GLOBAL sums == serial grads
        assert torch.allclose(torch.from_numpy(r['ggamma']),
                              bn.weight.grad, atol=1e-4)
        assert torch.allclose(torch.from_numpy(r['gbeta']),
                              bn.bias.grad, atol=1e-4)
        # running stats: upstream Chainer BN folds the UNBIASED batch var
        # (m/(m-1), m = global element count) into running_var, like torch
        n = size * B * H * W
        unbias = n / (n - 1)
        batch_var = x.detach().var(dim=(0, 2, 3), unbiased=False)
        expect_var = 0.9 * 1.0 + 0.1 * batch_var * unbias  # decay=0.9
        got_var = torch.from_numpy(r['running_var'])
        assert torch.allclose(got_var, expect_var, atol=1e-5), \
            (got_var, expect_var, unbias)


def _mnbn_2d_worker(rank, size):
    """MNBN on 2-D (N, C) inputs — the Linear/BN1d case."""
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(7)
    x_all = torch.randn(6, C)
    x = x_all[rank * 3:(rank + 1) * 3].clone().requires_grad_(True)
    bn = MultiNodeBatchNormalization(C, comm)
    y = bn(x)
    y.sum().backward()
    return {'y': y.detach(), 'gx': x.grad}


def test_mnbn_2d():
    results = run_distributed(_mnbn_2d_worker, 2)
    torch.manual_seed(7)
    x_all = torch.randn(6, C).requires_grad_(True)
    bn = nn.BatchNorm1d(C, eps=2e-5)
    y = bn(x_all)
    y.sum().backward()
    for rank in range(2):
        got = torch.from_numpy(results[rank]['y'])
        assert torch.allclose(got, y.detach()[rank * 3:(rank + 1) * 3],
                              atol=1e-5)
        gotg = torch.from_numpy(results[rank]['gx'])
        assert torch.allclose(gotg, x_all.grad[rank * 3:(rank + 1) * 3],
                              atol=1e-5)


def _create_mnbn_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.Sequential(nn.Conv2d(3, 8, 3), nn.BatchNorm2d(8), nn.ReLU())
    mnbn = create_mnbn_model(model, comm)
    assert isinstance(mnbn[1], MultiNodeBatchNormalization)
    assert mnbn[1].num_features == 8
    # swapped module still runs
    x = torch.randn(2, 3, 8, 8)
    y = mnbn(x)
    assert y.shape == (2, 8, 6, 6)
    return True


def test_create_mnbn_model():
    run_distributed(_create_mnbn_worker, 2)


def _fused_mnbn_worker(rank, size, x_all_np, gy_all_np):
    """Fused-path MNBN relu variant across 2 ranks vs composed serial."""
    from chainermn_amd.links.batch_normalization import MultiNodeBNReLU2d
    comm = chainermn_amd.create_communicator('naive')
    x_all = torch.from_numpy(x_all_np)
    gy_all = torch.from_numpy(gy_all_np)
    x = x_all[rank * B:(rank + 1) * B].clone().requires_grad_(True)
    gy = gy_all[rank * B:(rank + 1) * B]
    bn = MultiNodeBNReLU2d(C, comm)
    y = bn(x)
    y.backward(gy)
    return {'y': y.detach(), 'gx': x.grad, 'ggamma': bn.weight.grad}


def test_fused_mnbn_relu_matches_global_batch():
    size = 2
    rng = numpy.random.RandomState(3)
    x_all = rng.randn(size * B, C, H, W).astype(numpy.float32)
    gy_all = rng.randn(size * B, C, H, W).astype(numpy.float32)
    results = run_distributed(_fused_mnbn_worker, size, args=(x_all, gy_all))
    x = torch.from_numpy(x_all).clone().requires_grad_(True)
    bn = nn.BatchNorm2d(C, eps=2e-5)
    y = torch.relu(bn(x))
    y.backward(torch.from_numpy(gy_all))
    for rank in range(size):
        r = results[rank]
        assert torch.allclose(torch.from_numpy(r['y']),
                              y.detach()[rank * B:(rank + 1) * B],
                              atol=1e-5)
        assert torch.allclose(torch.from_numpy(r['gx']),
                              x.grad[rank * B:(rank + 1) * B], atol=1e-5)
        assert torch.allclose(torch.from_numpy(r['ggamma']),
                              bn.weight.grad, atol=1e-4)


def test_create_mnbn_model_converts_fused_bn():
    """A model built with the FUSED local BN classes converts too (the
    +ReLU variant keeps its fusion via MultiNodeBNReLU2d)."""
    import torch.nn as nn
    from chainermn_amd.links.fused_batch_norm import (FusedBatchNorm2d,
                                                      FusedBNReLU2d)
    from chainermn_amd.links.batch_normalization import (
        MultiNodeBatchNormalization, MultiNodeBNReLU2d)
    comm = chainermn_amd.create_communicator('dummy')
    m = nn.Sequential(FusedBNReLU2d(8), FusedBatchNorm2d(4),
                      nn.BatchNorm2d(6))
    with torch.no_grad():
        m[0].weight.mul_(2.0)
        m[1].running_var.fill_(3.0)
    mn = create_mnbn_model(m, comm)
    assert isinstance(mn[0], MultiNodeBNReLU2d)
    assert isinstance(mn[1], MultiNodeBatchNormalization)
    assert not isinstance(mn[1], MultiNodeBNReLU2d)
    assert isinstance(mn[2], MultiNodeBatchNormalization)
    assert torch.allclose(mn[0].weight, m[0].weight)
    assert torch.allclose(mn[1].running_var, m[1].running_var)
    # double conversion is a no-op
    mn2 = create_mnbn_model(mn, comm)
    assert isinstance(mn2[0], MultiNodeBNReLU2d)
