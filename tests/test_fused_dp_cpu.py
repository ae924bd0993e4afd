"""FusedDataParallel logic on a 2-process gloo world (CPU, fp32 arenas,
overlap=False): ranks stay identical and match the serial mean-grad
update — insurance for the 8-GPU run the driver performs."""

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.parallel import FusedDataParallelSGD
from chainermn_amd.testing import run_distributed


def _model(seed):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(6, 16), nn.ReLU(), nn.Linear(16, 4))


def _worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = _model(0)
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               weight_decay=1e-4, overlap=False,
                               grad_dtype=torch.float32)
    for step in range(3):
        opt.zero_grad()
        torch.manual_seed(50 + step * size + rank)
        x = torch.randn(5, 6)
        model(x).pow(2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_fused_dp_matches_serial():
    results = run_distributed(_worker, 2)
    results = [[torch.from_numpy(a) for a in r] for r in results]
    for a, b in zip(results[0], results[1]):
        assert torch.allclose(a, b, atol=1e-6)
    # serial reference
    ref = _model(0)
    opt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9,
                          weight_decay=1e-4)
    for step in range(3):
        opt.zero_grad()
        losses = []
        for rank in range(2):
            torch.manual_seed(50 + step * 2 + rank)
            x = torch.randn(5, 6)
            losses.append(ref(x).pow(2).mean())
        (sum(losses) / 2).backward()
        opt.step()
    for p, r in zip(results[0], ref.parameters()):
        assert torch.allclose(p, r.detach(), atol=1e-5), \
            (p - r.detach()).abs().max()


def test_bench_cpu_world2():
    """bench.py runs end-to-end on a 2-proc CPU world (mlp model)."""
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29617', 'bench.py', '--model', 'mlp',
         '--steps', '3', '--warmup', '1', '--batch-size', '8'],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    assert '"n_gpus": 2' in out.stdout, out.stdout


def _bucketed_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = _model(0)
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               weight_decay=1e-4, overlap='bucketed',
                               grad_dtype=torch.float32, bucket_mb=1)
    for step in range(3):
        opt.zero_grad()
        torch.manual_seed(50 + step * size + rank)
        x = torch.randn(5, 6)
        model(x).pow(2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_bucketed_overlap_matches_serial():
    results = run_distributed(_bucketed_worker, 2)
    results = [[torch.from_numpy(a) for a in r] for r in results]
    for a, b in zip(results[0], results[1]):
        assert torch.allclose(a, b, atol=1e-6)
    ref = _model(0)
    opt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9,
                          weight_decay=1e-4)
    for step in range(3):
        opt.zero_grad()
        losses = []
        for rank in range(2):
            torch.manual_seed(50 + step * 2 + rank)
            x = torch.randn(5, 6)
            losses.append(ref(x).pow(2).mean())
        (sum(losses) / 2).backward()
        opt.step()
    for p, r in zip(results[0], ref.parameters()):
        assert torch.allclose(p, r.detach(), atol=1e-5)


def test_fused_dp_state_roundtrip():
    comm = chainermn_amd.create_communicator('dummy')
    model = _model(3)
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               overlap=False, grad_dtype=torch.float32)
    for _ in range(2):
        opt.zero_grad()
        model(torch.randn(4, 6)).pow(2).mean().backward()
        opt.step()
    sd = {k: (v.clone() if torch.is_tensor(v) else v)
          for k, v in opt.state_dict().items()}
    w_after = [p.detach().clone() for p in model.parameters()]
    # perturb then restore
    opt.state.master.add_(1.0)
    opt.load_state_dict(sd)
    assert torch.allclose(opt.state.master,
                          sd['master'])
    for p, w in zip(model.parameters(), w_after):
        assert torch.allclose(p.data, w, atol=1e-6)


def test_fused_dp_preserves_channels_last():
    """Wrapping a channels_last model must keep conv-weight memory format
    (kernel selection on MI355X depends on it) AND train identically to
    the same model in contiguous format (ADVICE.md round-1 finding)."""
    comm = chainermn_amd.create_communicator('dummy')

    def conv_model(seed):
        torch.manual_seed(seed)
        return nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                             nn.Conv2d(8, 4, 1))

    model = conv_model(11).to(memory_format=torch.channels_last)
    opt = FusedDataParallelSGD(model, comm, lr=0.05, momentum=0.9,
                               overlap=False, grad_dtype=torch.float32)
    for p in model.parameters():
        if p.dim() == 4:
            assert p.data.is_contiguous(
                memory_format=torch.channels_last), p.shape
            if p.shape[2] * p.shape[3] > 1:  # 1x1: CL ≡ contiguous
                assert not p.data.is_contiguous(), p.shape
            assert p.grad.is_contiguous(
                memory_format=torch.channels_last), p.shape

    ref = conv_model(11)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    for step in range(3):
        torch.manual_seed(90 + step)
        x = torch.randn(2, 3, 8, 8)
        opt.zero_grad()
        model(x.to(memory_format=torch.channels_last)
              ).pow(2).mean().backward()
        opt.step()
        ropt.zero_grad()
        ref(x).pow(2).mean().backward()
        ropt.step()
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, r.detach(), atol=1e-5), \
            (p.data - r.detach()).abs().max()
    # state roundtrip keeps the format too
    sd = {k: (v.clone() if torch.is_tensor(v) else v)
          for k, v in opt.state_dict().items()}
    opt.load_state_dict(sd)
    for p in model.parameters():
        if p.dim() == 4:
            assert p.data.is_contiguous(memory_format=torch.channels_last)


def _world4_bucketed_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = _model(0)
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               overlap='bucketed', bucket_mb=0.0001,
                               grad_dtype=torch.float32)
    for step in range(2):
        opt.zero_grad()
        torch.manual_seed(70 + step * size + rank)
        model(torch.randn(5, 6)).pow(2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_bucketed_overlap_world4():
    """Bucketed engine at world 4 with tiny buckets (many buckets, all
    ranks must issue the same fixed collective order)."""
    results = run_distributed(_world4_bucketed_worker, 4)
    results = [[torch.from_numpy(a) for a in r] for r in results]
    for other in results[1:]:
        for a, b in zip(results[0], other):
            assert torch.allclose(a, b, atol=1e-6)
