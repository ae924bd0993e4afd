"""Multi-node optimizer: params stay identical across ranks and equal the
serially-computed mean-grad update (reference:
optimizer_tests/test_multi_node_optimizer.py)."""

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.testing import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 2))


def _serial_reference(steps, world):
    model = _make_model()
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    for step in range(steps):
        opt.zero_grad()
        losses = []
        for rank in range(world):
            torch.manual_seed(1000 + step * world + rank)
            x = torch.randn(6, 4)
            losses.append(model(x).pow(2).mean())
        # mean over ranks == mean-grad of per-rank losses
        (sum(losses) / world).backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def _dp_worker(rank, size, steps):
    comm = chainermn_amd.create_communicator('naive')
    model = _make_model(seed=rank * 31 + 5)  # intentionally different init
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    opt = chainermn_amd.create_multi_node_optimizer(opt, comm)
    opt.setup(model)  # broadcasts rank-0 params — but ranks differ...
    for step in range(steps):
        opt.zero_grad(set_to_none=False)
        torch.manual_seed(1000 + step * size + rank)
        x = torch.randn(6, 4)
        loss = model(x).pow(2).mean()
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_multi_node_sgd_matches_serial():
    steps = 3
    world = 2
    results = run_distributed(_dp_worker, world, args=(steps,))
    results = [[torch.from_numpy(a) for a in r] for r in results]
    # all ranks identical
    for p0, p1 in zip(results[0], results[1]):
        assert torch.allclose(p0, p1, atol=1e-6)
    # broadcast made everyone start from rank 0's init (seed 5)
    ref_model = _make_model(seed=5)
    ref_opt = torch.optim.SGD(ref_model.parameters(), lr=0.1, momentum=0.9)
    for step in range(steps):
        ref_opt.zero_grad()
        losses = []
        for rank in range(world):
            torch.manual_seed(1000 + step * world + rank)
            x = torch.randn(6, 4)
            losses.append(ref_model(x).pow(2).mean())
        (sum(losses) / world).backward()
        ref_opt.step()
    for p, r in zip(results[0], ref_model.parameters()):
        assert torch.allclose(p, r.detach(), atol=1e-5), (p, r)


def _update_api_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = _make_model()
    opt = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.05), comm).setup(model)

    def lossfun():
        torch.manual_seed(42 + rank)
        return model(torch.randn(3, 4)).sum()

    loss = opt.update(lossfun)
    assert loss is not None
    # attribute proxying reaches the wrapped optimizer
    assert opt.param_groups[0]['lr'] == 0.05
    opt.lr_dummy = 1  # setattr proxies without error
    return True


def test_update_api_and_proxy():
    run_distributed(_update_api_worker, 2)
