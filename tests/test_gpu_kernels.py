"""GPU kernel numerics: every CDNA4 kernel vs a plain PyTorch fp32
reference of the same op.  Run on an MI355X box:
  python -m pytest tests -m gpu -x -q
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from chainermn_amd import ops
    DEV = torch.device('cuda:0')
else:
    pytest.skip('needs ROCm GPU', allow_module_level=True)


def _ext():
    return ops.require()


class P(object):
    """Minimal param carrier for PackPlan."""

    def __init__(self, t):
        self.data = t
        self.grad = torch.randn_like(t)

    def numel(self):
        return self.data.numel()


def _mk_params(shapes, dtype=torch.float32):
    torch.manual_seed(0)
    return [P(torch.randn(s, device=DEV, dtype=dtype)) for s in shapes]


SHAPES = [(64, 3, 7, 7), (64,), (1000, 512), (3,), (17, 5), (2048,)]


@pytest.mark.parametrize('wire', [torch.float32, torch.bfloat16,
                                  torch.float16])
def test_pack_unpack_roundtrip(wire):
    params = _mk_params(SHAPES)
    plan = ops.PackPlan(params, DEV)
    flat = torch.zeros(plan.total, device=DEV, dtype=wire)
    ops.pack_grads(plan, flat)
    torch.cuda.synchronize()
    # reference: each grad lands at its offset, cast to wire dtype
    for p, off, n in zip(params, plan.offsets_cpu, plan.sizes_cpu):
        ref = p.grad.reshape(-1).to(wire)
        assert torch.equal(flat[off:off + n], ref), (off, n)
        # padding is zero-filled
        slot_end = (off + n + ops.ALIGN - 1) // ops.ALIGN * ops.ALIGN
        if slot_end > off + n and slot_end <= plan.total:
            assert (flat[off + n:slot_end] == 0).all()
    # unpack with scale
    scale = 0.25
    originals = [p.grad.clone() for p in params]
    ops.unpack_grads(plan, flat, scale=scale)
    torch.cuda.synchronize()
    for p, orig in zip(params, originals):
        ref = (orig.reshape(-1).to(wire).float() * scale).to(
            p.grad.dtype).view_as(p.grad)
        assert torch.allclose(p.grad, ref, atol=1e-6), \
            (p.grad - ref).abs().max()


def test_pack_bf16_grads():
    params = _mk_params(SHAPES, dtype=torch.bfloat16)
    plan = ops.PackPlan(params, DEV)
    flat = torch.zeros(plan.total, device=DEV, dtype=torch.bfloat16)
    ops.pack_grads(plan, flat)
    torch.cuda.synchronize()
    for p, off, n in zip(params, plan.offsets_cpu, plan.sizes_cpu):
        assert torch.equal(flat[off:off + n], p.grad.reshape(-1))


@pytest.mark.parametrize('src,dst', [(torch.float32, torch.bfloat16),
                                     (torch.bfloat16, torch.float32),
                                     (torch.float32, torch.float16),
                                     (torch.float32, torch.float32)])
def test_cast_flat(src, dst):
    torch.manual_seed(1)
    n = 1 << 20 | 3  # odd tail
    x = torch.randn(n, device=DEV, dtype=src)
    out = torch.empty(n, device=DEV, dtype=dst)
    ops.cast_flat(x, out, scale=0.5)
    torch.cuda.synchronize()
    ref = (x.float() * 0.5).to(dst)
    assert torch.equal(out, ref)


@pytest.mark.parametrize('gdtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('nesterov', [False, True])
def test_fused_sgd_matches_torch(gdtype, nesterov):
    torch.manual_seed(2)
    n = 100003
    p_ref = torch.randn(n, device=DEV)
    p = p_ref.clone()
    v = torch.zeros(n, device=DEV)
    ref_param = p_ref.clone().requires_grad_(True)
    opt = torch.optim.SGD([ref_param], lr=0.1, momentum=0.9,
                          weight_decay=1e-4, nesterov=nesterov)
    for step in range(3):
        g32 = torch.randn(n, device=DEV)
        g = g32.to(gdtype)
        ref_param.grad = g.float()
        opt.step()
        ops.fused_sgd(p, g, v, lr=0.1, momentum=0.9, weight_decay=1e-4,
                      nesterov=nesterov, grad_scale=1.0)
        torch.cuda.synchronize()
        assert torch.allclose(p, ref_param.detach(), atol=1e-5), \
            (step, (p - ref_param.detach()).abs().max())


def test_fused_sgd_lowp_mirror():
    torch.manual_seed(3)
    n = 4096 + 5
    p = torch.randn(n, device=DEV)
    lowp = p.to(torch.bfloat16)
    v = torch.zeros(n, device=DEV)
    g = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    ops.fused_sgd(p, g, v, lr=0.1, momentum=0.9, p_lowp=lowp,
                  grad_scale=0.5)
    torch.cuda.synchronize()
    assert torch.equal(lowp, p.to(torch.bfloat16))


@pytest.mark.parametrize('adamw', [False, True])
def test_fused_adam_matches_torch(adamw):
    torch.manual_seed(4)
    n = 50001
    p = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    ref_param = p.clone().requires_grad_(True)
    cls = torch.optim.AdamW if adamw else torch.optim.Adam
    opt = cls([ref_param], lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
              weight_decay=1e-2)
    for step in range(1, 4):
        g = torch.randn(n, device=DEV)
        ref_param.grad = g.clone()
        opt.step()
        ops.fused_adam(p, g, m, v, lr=1e-3, beta1=0.9, beta2=0.999,
                       eps=1e-8, weight_decay=1e-2, step=step, adamw=adamw)
        torch.cuda.synchronize()
        assert torch.allclose(p, ref_param.detach(), atol=1e-6), \
            (step, (p - ref_param.detach()).abs().max())


def test_segment_sq_norms():
    params = _mk_params(SHAPES)
    plan = ops.PackPlan(params, DEV)
    flat = torch.randn(plan.total, device=DEV)
    out = torch.zeros(len(SHAPES), device=DEV)
    ops.segment_sq_norms(flat, plan.offsets, plan.sizes, out)
    torch.cuda.synchronize()
    for i, (off, n) in enumerate(zip(plan.offsets_cpu, plan.sizes_cpu)):
        ref = flat[off:off + n].pow(2).sum()
        assert torch.allclose(out[i], ref, rtol=1e-4), (i, out[i], ref)


def test_lars_update_matches_cpu():
    params = _mk_params(SHAPES)
    plan = ops.PackPlan(params, DEV)
    n = plan.total
    torch.manual_seed(5)
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    p_cpu, g_cpu, v_cpu = p.cpu(), g.cpu(), v.cpu()
    p_sqn = torch.zeros(len(SHAPES), device=DEV)
    g_sqn = torch.zeros(len(SHAPES), device=DEV)
    ops.segment_sq_norms(p, plan.offsets, plan.sizes, p_sqn)
    ops.segment_sq_norms(g, plan.offsets, plan.sizes, g_sqn)
    kw = dict(lr=0.1, momentum=0.9, weight_decay=5e-5, eta=0.001, eps=1e-9,
              grad_scale=0.5)
    ops.lars_update(p, g, v, plan.offsets, plan.sizes, p_sqn, g_sqn, **kw)
    torch.cuda.synchronize()
    off_cpu = plan.offsets.cpu()
    sz_cpu = plan.sizes.cpu()
    ops.lars_update(p_cpu, g_cpu, v_cpu, off_cpu, sz_cpu,
                    p_sqn.cpu(), g_sqn.cpu(), **kw)
    # padding regions differ (kernel skips them); compare segment data only
    for i, (off, n_el) in enumerate(zip(plan.offsets_cpu, plan.sizes_cpu)):
        assert torch.allclose(p[off:off + n_el].cpu(),
                              p_cpu[off:off + n_el], atol=1e-5)
        assert torch.allclose(v[off:off + n_el].cpu(),
                              v_cpu[off:off + n_el], atol=1e-5)


# ---------------- MNBN kernels ----------------
@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('cl', [False, True])
def test_bn_stats(dtype, cl):
    torch.manual_seed(6)
    x = torch.randn(8, 32, 14, 14, device=DEV, dtype=dtype)
    if cl:
        x = x.to(memory_format=torch.channels_last)
    s, sq, count = _ext().bn_stats(x)
    torch.cuda.synchronize()
    xf = x.float()
    ref_s = xf.sum(dim=(0, 2, 3))
    ref_sq = (xf * xf).sum(dim=(0, 2, 3))
    assert count == 8 * 14 * 14
    tol = 1e-2 if dtype == torch.bfloat16 else 1e-3
    assert torch.allclose(s, ref_s, atol=tol, rtol=1e-4), \
        (s - ref_s).abs().max()
    assert torch.allclose(sq, ref_sq, atol=tol, rtol=1e-4)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('cl', [False, True])
def test_bn_fwd_bwd_vs_torch(dtype, cl):
    """Full MNBN autograd function on GPU (dummy comm) vs torch BN fp32."""
    import chainermn_amd
    from chainermn_amd.functions.batch_normalization import \
        multi_node_batch_norm
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(7)
    N, C, H, W = 6, 16, 8, 8
    x32 = torch.randn(N, C, H, W, device=DEV)
    x = x32.to(dtype).clone()
    if cl:
        x = x.to(memory_format=torch.channels_last)
    x.requires_grad_(True)
    gamma = torch.randn(C, device=DEV).abs() + 0.5
    beta = torch.randn(C, device=DEV)
    gamma_p = gamma.to(dtype).clone().requires_grad_(True)
    beta_p = beta.to(dtype).clone().requires_grad_(True)
    y, mean, var = multi_node_batch_norm(x, gamma_p, beta_p, comm)
    gy = torch.randn(N, C, H, W, device=DEV).to(dtype)
    if cl:
        gy = gy.to(memory_format=torch.channels_last)
    y.backward(gy)

    # torch fp32 reference
    xr = x32.clone().requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C, eps=2e-5, affine=True).to(DEV)
    with torch.no_grad():
        bn.weight.copy_(gamma)
        bn.bias.copy_(beta)
    yr = bn(xr)
    yr.backward(gy.float())
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(y.float(), yr.detach(), atol=tol), \
        (y.float() - yr.detach()).abs().max()
    assert torch.allclose(x.grad.float(), xr.grad, atol=tol), \
        (x.grad.float() - xr.grad).abs().max()
    assert torch.allclose(gamma_p.grad.float(), bn.weight.grad,
                          atol=tol * 10, rtol=1e-2)
    assert torch.allclose(beta_p.grad.float(), bn.bias.grad,
                          atol=tol * 10, rtol=1e-2)


def test_fused_dp_sgd_single_gpu_matches_eager():
    """FusedDataParallelSGD (arena path, world=1) vs eager torch SGD."""
    import chainermn_amd
    import torch.nn as nn
    from chainermn_amd.parallel import FusedDataParallelSGD
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(8)
    model = nn.Sequential(nn.Linear(32, 64), nn.ReLU(),
                          nn.Linear(64, 10)).to(DEV)
    ref = nn.Sequential(nn.Linear(32, 64), nn.ReLU(),
                        nn.Linear(64, 10)).to(DEV)
    ref.load_state_dict(model.state_dict())
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               weight_decay=1e-4, overlap=False)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9,
                           weight_decay=1e-4)
    for step in range(3):
        torch.manual_seed(100 + step)
        x = torch.randn(16, 32, device=DEV)
        opt.zero_grad(set_to_none=False)
        loss = model(x).pow(2).mean()
        loss.backward()
        opt.step()
        ropt.zero_grad()
        rloss = ref(x).pow(2).mean()
        rloss.backward()
        ropt.step()
        torch.cuda.synchronize()
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, r.data, atol=1e-5), \
            (p.data - r.data).abs().max()


def test_fused_dp_overlap_staleness_semantics():
    """overlap=True applies grads one step late; after a final extra step
    with zero loss, params match the non-overlapped run one step back."""
    import chainermn_amd
    import torch.nn as nn
    from chainermn_amd.parallel import FusedDataParallelSGD
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(9)
    model = nn.Linear(8, 8).to(DEV)
    ref = nn.Linear(8, 8).to(DEV)
    ref.load_state_dict(model.state_dict())
    init_sd = {k: v.clone() for k, v in model.state_dict().items()}
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.0,
                               overlap=True)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.1)
    xs = [torch.randn(4, 8, device=DEV) for _ in range(4)]
    for x in xs:
        opt.zero_grad()
        model(x).pow(2).mean().backward()
        opt.step()
    # one flush step with zero grads applies the last pending gradient
    opt.zero_grad()
    opt.step()
    torch.cuda.synchronize()
    # exact 1-step-stale reference: step k computes g_k at the CURRENT
    # params, THEN applies g_{k-1} (fp32 arenas, momentum 0 → bitwise-
    # comparable update order)
    pending = None
    for x in xs:
        ropt.zero_grad()
        ref(x).pow(2).mean().backward()
        g = [p.grad.clone() for p in ref.parameters()]
        if pending is not None:
            with torch.no_grad():
                for p, gp in zip(ref.parameters(), pending):
                    p.sub_(gp, alpha=0.1)
        pending = g
    with torch.no_grad():
        for p, gp in zip(ref.parameters(), pending):
            p.sub_(gp, alpha=0.1)
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, r.data, atol=1e-6), \
            (p.data - r.data).abs().max()
    # and the stale trajectory must differ from the synchronous one —
    # otherwise this test would pass vacuously on a non-overlapped
    # implementation
    sync = nn.Linear(8, 8).to(DEV)
    sync.load_state_dict(init_sd)
    sopt = torch.optim.SGD(sync.parameters(), lr=0.1)
    for x in xs:
        sopt.zero_grad()
        sync(x).pow(2).mean().backward()
        sopt.step()
    diff = max((p.data - s.data).abs().max().item()
               for p, s in zip(model.parameters(), sync.parameters()))
    assert diff > 1e-5, diff


def test_bench_smoke():
    """__graft_entry__.smoke() runs end to end on this box."""
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        '__graft_entry__', os.path.join(os.path.dirname(__file__), '..',
                                        '__graft_entry__.py'))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.smoke()


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('cl', [False, True])
def test_fused_batch_norm_vs_torch(dtype, cl):
    from chainermn_amd.links import FusedBatchNorm2d
    torch.manual_seed(10)
    N, C, H, W = 8, 64, 14, 14
    x32 = torch.randn(N, C, H, W, device=DEV)
    bn = FusedBatchNorm2d(C, eps=1e-5, momentum=0.1).to(DEV)
    ref = torch.nn.BatchNorm2d(C, eps=1e-5, momentum=0.1).to(DEV)
    x = x32.to(dtype).clone()
    if cl:
        x = x.to(memory_format=torch.channels_last)
    x.requires_grad_(True)
    xr = x32.clone().requires_grad_(True)
    y = bn(x)
    yr = ref(xr)
    gy = torch.randn_like(yr)
    y.backward(gy.to(dtype) if not cl
               else gy.to(dtype).to(memory_format=torch.channels_last))
    yr.backward(gy)
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(y.float(), yr.detach(), atol=tol)
    assert torch.allclose(x.grad.float(), xr.grad, atol=tol)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=tol * 20,
                          rtol=1e-2)
    assert torch.allclose(bn.bias.grad, ref.bias.grad, atol=tol * 20,
                          rtol=1e-2)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-3)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-3)
    # eval mode path
    bn.eval(); ref.eval()
    ye = bn(x32.to(dtype))
    yre = ref(x32)
    assert torch.allclose(ye.float(), yre, atol=tol)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_fused_bn_relu_add_gpu(dtype):
    from chainermn_amd.links import FusedBatchNorm2d, FusedBNReLU2d
    torch.manual_seed(11)
    N, C, H, W = 8, 32, 14, 14
    x32 = torch.randn(N, C, H, W, device=DEV)
    res32 = torch.randn(N, C, H, W, device=DEV)
    gy32 = torch.randn(N, C, H, W, device=DEV)
    to = lambda t: t.to(dtype).to(memory_format=torch.channels_last)
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    # bn+relu
    x = to(x32).requires_grad_(True)
    xr = x32.clone().requires_grad_(True)
    bn = FusedBNReLU2d(C).to(DEV)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    y = bn(x)
    yr = torch.relu(ref(xr))
    y.backward(to(gy32))
    yr.backward(gy32)

    def close_enough(a, b):
        # bf16 rounds y near the relu boundary differently from the fp32
        # reference, flipping the mask on isolated elements: require the
        # bulk to match and only a tiny fraction of boundary flips.
        bad = (~torch.isclose(a, b, atol=tol)).float().mean()
        assert bad < 5e-3, bad

    close_enough(y.float(), yr.detach())
    close_enough(x.grad.float(), xr.grad)
    assert torch.allclose(bn.weight.grad.float(), ref.weight.grad,
                          atol=tol * 20, rtol=5e-2)
    # bn+add+relu
    x2 = to(x32).requires_grad_(True)
    r2 = to(res32).requires_grad_(True)
    xr2 = x32.clone().requires_grad_(True)
    rr2 = res32.clone().requires_grad_(True)
    bn2 = FusedBatchNorm2d(C).to(DEV)
    ref2 = torch.nn.BatchNorm2d(C).to(DEV)
    y2 = bn2.forward_add_relu(x2, r2)
    yr2 = torch.relu(ref2(xr2) + rr2)
    y2.backward(to(gy32))
    yr2.backward(gy32)
    close_enough(y2.float(), yr2.detach())
    close_enough(x2.grad.float(), xr2.grad)
    close_enough(r2.grad.float(), rr2.grad)


def test_resnet_fused_ops_gpu_loss_decreases():
    import chainermn_amd
    from chainermn_amd.models import ResNet50
    from chainermn_amd.parallel import FusedDataParallelSGD
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(3)
    model = ResNet50(num_classes=100, fused_ops=True).to(
        device=DEV, dtype=torch.bfloat16,
        memory_format=torch.channels_last)
    opt = FusedDataParallelSGD(model, comm, lr=0.05, momentum=0.9,
                               overlap=False)
    x = torch.randn(16, 3, 64, 64, device=DEV, dtype=torch.bfloat16
                    ).to(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (16,), device=DEV)
    losses = []
    for _ in range(12):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.7, losses


def test_double_buffering_optimizer_gpu():
    """Generic DoubleBufferingOptimizer (arena + side-stream cast path) on
    one GPU: finite losses, params update after step 1, wait() drains."""
    import chainermn_amd
    import torch.nn as nn
    comm = chainermn_amd.create_communicator('pure_rccl')
    torch.manual_seed(5)
    model = nn.Linear(16, 8).to(DEV)
    opt = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.1), comm,
        double_buffering=True).setup(model)
    w0 = model.weight.detach().clone()
    for i in range(3):
        opt.zero_grad()
        loss = model(torch.randn(4, 16, device=DEV)).pow(2).mean()
        loss.backward()
        opt.step()
    opt.wait()
    assert torch.isfinite(model.weight).all()
    assert not torch.allclose(model.weight, w0)


def test_pure_rccl_mean_grad_single_rank():
    """PureRcclCommunicator pack->allreduce->unpack on world_size=1."""
    import chainermn_amd
    import torch.nn as nn
    comm = chainermn_amd.create_communicator(
        'pure_rccl', allreduce_grad_dtype='bf16')
    model = nn.Linear(33, 7).to(DEV)
    for p in model.parameters():
        p.grad = torch.randn_like(p)
    before = [p.grad.clone() for p in model.parameters()]
    comm.multi_node_mean_grad(model)
    for p, b in zip(model.parameters(), before):
        # world=1: value preserved through the bf16 wire round-trip
        assert torch.allclose(p.grad, b.to(torch.bfloat16).float(),
                              atol=1e-2)


def test_fused_dp_adam_matches_eager():
    import chainermn_amd
    import torch.nn as nn
    from chainermn_amd.parallel import FusedDataParallelAdam
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(21)
    model = nn.Sequential(nn.Linear(12, 24), nn.ReLU(),
                          nn.Linear(24, 6)).to(DEV)
    ref = nn.Sequential(nn.Linear(12, 24), nn.ReLU(),
                        nn.Linear(24, 6)).to(DEV)
    ref.load_state_dict(model.state_dict())
    opt = FusedDataParallelAdam(model, comm, lr=1e-2, weight_decay=1e-2,
                                adamw=True, overlap=False)
    ropt = torch.optim.AdamW(ref.parameters(), lr=1e-2, weight_decay=1e-2)
    for step in range(3):
        torch.manual_seed(200 + step)
        x = torch.randn(8, 12, device=DEV)
        opt.zero_grad()
        model(x).pow(2).mean().backward()
        opt.step()
        ropt.zero_grad()
        ref(x).pow(2).mean().backward()
        ropt.step()
    torch.cuda.synchronize()
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, r.data, atol=1e-5), \
            (p.data - r.data).abs().max()


def test_bucketed_overlap_gpu_single_rank():
    import chainermn_amd
    import torch.nn as nn
    from chainermn_amd.parallel import FusedDataParallelSGD
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(31)
    model = nn.Sequential(nn.Linear(64, 128), nn.ReLU(),
                          nn.Linear(128, 10)).to(DEV)
    ref = nn.Sequential(nn.Linear(64, 128), nn.ReLU(),
                        nn.Linear(128, 10)).to(DEV)
    ref.load_state_dict(model.state_dict())
    opt = FusedDataParallelSGD(model, comm, lr=0.1, momentum=0.9,
                               overlap='bucketed', bucket_mb=1)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9)
    for step in range(3):
        torch.manual_seed(300 + step)
        x = torch.randn(8, 64, device=DEV)
        opt.zero_grad()
        model(x).pow(2).mean().backward()
        opt.step()
        ropt.zero_grad()
        ref(x).pow(2).mean().backward()
        ropt.step()
    torch.cuda.synchronize()
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.data, r.data, atol=1e-5)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_fused_maxpool_vs_torch(dtype):
    from chainermn_amd.links import FusedMaxPool2d
    torch.manual_seed(41)
    for N, C, H, W in [(4, 64, 112, 112), (2, 8, 9, 9), (1, 16, 7, 10)]:
        x32 = torch.randn(N, C, H, W, device=DEV)
        x = x32.to(dtype).to(memory_format=torch.channels_last)
        x.requires_grad_(True)
        xr = x32.to(dtype).clone().requires_grad_(True)
        pool = FusedMaxPool2d(3, 2, 1)
        y = pool(x)
        yr = torch.nn.functional.max_pool2d(xr, 3, 2, 1)
        assert y.shape == yr.shape
        assert torch.equal(y.float(),
                           yr.float().to(memory_format=torch.channels_last)
                           .float()), (N, C, H, W)
        gy = torch.randn_like(yr)
        y.backward(gy.to(memory_format=torch.channels_last))
        yr.backward(gy)
        assert torch.allclose(x.grad.float(), xr.grad.float(), atol=1e-3), \
            (N, C, H, W, (x.grad.float() - xr.grad.float()).abs().max())


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_fused_conv1x1_vs_conv2d(dtype):
    from chainermn_amd.links import FusedConv1x1
    torch.manual_seed(51)
    N, C, H, W, Cout = 8, 64, 14, 14, 128
    x32 = torch.randn(N, C, H, W, device=DEV)
    gy32 = torch.randn(N, Cout, H, W, device=DEV)
    conv = FusedConv1x1(C, Cout).to(DEV, dtype)
    x = x32.to(dtype).to(memory_format=torch.channels_last)
    x.requires_grad_(True)
    xr = x32.to(dtype).clone().requires_grad_(True)
    y = conv(x)
    yr = torch.nn.functional.conv2d(xr, conv.weight)
    tol = 5e-2 if dtype == torch.bfloat16 else 1e-4
    assert torch.allclose(y.float(), yr.float(), atol=tol, rtol=1e-2)
    w_grad_ref = torch.autograd.grad(
        yr, [xr, conv.weight], gy32.to(dtype), retain_graph=False)
    y.backward(gy32.to(dtype).to(memory_format=torch.channels_last))
    assert torch.allclose(x.grad.float(), w_grad_ref[0].float(), atol=tol,
                          rtol=1e-2)
    assert torch.allclose(conv.weight.grad.float(),
                          w_grad_ref[1].float(), atol=tol * 10, rtol=5e-2)


def test_flat_and_staged_communicators_world1():
    """flat/single_node/hierarchical/two_dimensional mean_grad on a
    1-rank GPU world (exercises the pack->reduce->unpack machinery and
    sub-group construction with RCCL available)."""
    import chainermn_amd
    import torch.nn as nn
    for name in ['flat', 'single_node', 'hierarchical', 'two_dimensional',
                 'non_cuda_aware', 'pure_rccl']:
        comm = chainermn_amd.create_communicator(name)
        model = nn.Linear(17, 5).to(DEV)
        for p in model.parameters():
            p.grad = torch.full_like(p, 2.0)
        comm.multi_node_mean_grad(model)
        for p in model.parameters():
            assert torch.allclose(p.grad, torch.full_like(p.grad, 2.0),
                                  atol=1e-2), name


def test_mfma_gemm_bt_vs_matmul():
    """Hand-written MFMA GEMM vs hipBLASLt, asymmetric operands
    (transpose-detecting — guide §5.4 rule 16)."""
    torch.manual_seed(61)
    for M, N, K in [(128, 128, 64), (256, 384, 192), (512, 1024, 512)]:
        a = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        # asymmetric structure so a row/col swap cannot pass
        a[0, :] += torch.arange(K, device=DEV, dtype=torch.bfloat16) * 0.01
        d = ops.gemm_bt(a, b)
        ref = a @ b.t()
        err = (d.float() - ref.float()).abs().max()
        ref_mag = ref.float().abs().max()
        assert err <= 0.02 * ref_mag + 0.1, (M, N, K, err, ref_mag)


def test_mfma_gemm_bt256_vs_matmul():
    """256x256-tile 8-wave GEMM numerics vs hipBLASLt/fp32 reference."""
    torch.manual_seed(55)
    for M, N, K in [(256, 256, 64), (512, 256, 128), (256, 512, 192),
                    (1024, 512, 320), (768, 768, 64)]:
        a = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        # asymmetric structure so a row/col swap cannot pass
        a[0, :] += torch.arange(K, device=DEV, dtype=torch.bfloat16) * 0.01
        b[1, :] -= torch.arange(K, device=DEV, dtype=torch.bfloat16) * 0.02
        ref = a.float() @ b.float().t()
        ref_mag = ref.abs().max()
        runs = [('simple', lambda: _ext().gemm_bt256_bf16(a, b))]
        runs += [('sched%d' % s,
                  lambda s=s: _ext().gemm_bt256p_bf16(a, b, s))
                 for s in range(4)]
        for name, fn in runs:
            d = fn()
            err = (d.float() - ref).abs().max()
            assert err <= 0.02 * ref_mag + 0.1, \
                (name, M, N, K, err, ref_mag)


@pytest.mark.parametrize('C', [12, 24, 40, 1028, 2560])
def test_bn_odd_channel_counts(C):
    """Channels-last reduce paths for C%4==0 but not power-of-two lane
    mappings (non-pow2 cross-row trees, CPT=4/8 splits).

    C=1028 (CPT=4, C/CPT=257>256) and C=2560 (CPT=8, C/CPT=320>256)
    force the channel-tiled path of bn_reduce_cl_kernel, where the last
    tile has threads past C — regression test for the OOB read/write in
    that tile (ADVICE.md round-1 high finding)."""
    from chainermn_amd.links import FusedBatchNorm2d
    torch.manual_seed(71)
    x32 = torch.randn(6, C, 9, 9, device=DEV)
    x = x32.to(torch.bfloat16).to(memory_format=torch.channels_last)
    x.requires_grad_(True)
    xr = x32.clone().requires_grad_(True)
    bn = FusedBatchNorm2d(C).to(DEV)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    y = bn(x)
    yr = ref(xr)
    gy = torch.randn_like(yr)
    y.backward(gy.to(torch.bfloat16).to(memory_format=torch.channels_last))
    yr.backward(gy)
    assert torch.allclose(y.float(), yr.detach(), atol=5e-2), C
    assert torch.allclose(x.grad.float(), xr.grad, atol=5e-2), C
    assert torch.allclose(bn.weight.grad.float(), ref.weight.grad,
                          atol=0.5, rtol=5e-2), C


def test_pure_rccl_mean_grad_async_world1():
    """Side-stream async mean-grad API (pack+all-reduce on a caller
    stream, unpack after event wait)."""
    import chainermn_amd
    import torch.nn as nn
    comm = chainermn_amd.create_communicator(
        'pure_rccl', allreduce_grad_dtype='bf16')
    model = nn.Linear(40, 9).to(DEV)
    for p in model.parameters():
        p.grad = torch.randn_like(p)
    before = [p.grad.clone() for p in model.parameters()]
    stream = torch.cuda.Stream()
    handle = comm.mean_grad_async(model, stream)
    handle.unpack()
    torch.cuda.synchronize()
    for p, b in zip(model.parameters(), before):
        assert torch.allclose(p.grad, b.to(torch.bfloat16).float(),
                              atol=1e-2)


def test_conv3x3_implicit_gemm_fwd():
    """Implicit-GEMM 3x3/s1/p1 conv forward vs F.conv2d, all three wave
    geometries (Cout 64 / 128 / 256) incl. border-dominated shapes."""
    torch.manual_seed(81)
    for N, C, H, W, Cout in [(2, 64, 16, 16, 128), (1, 128, 32, 32, 128),
                             (2, 64, 32, 16, 256), (16, 64, 4, 4, 64),
                             (4, 128, 8, 8, 64), (1, 192, 16, 16, 320),
                             (16, 128, 28, 28, 128)]:
        x = torch.randn(N, C, H, W, device=DEV, dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last)
        w = torch.randn(Cout, C, 3, 3, device=DEV, dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last) * 0.05
        y = _ext().conv3x3s1_fwd_bf16(x, w)
        ref = torch.nn.functional.conv2d(x, w, padding=1)
        err = (y.float() - ref.float()).abs().max()
        mag = ref.float().abs().max()
        assert err <= 0.03 * mag + 0.05, (N, C, H, W, Cout, err, mag)


def test_fused_conv3x3_fwd_bwd_vs_conv2d():
    """FusedConv3x3 autograd (custom fwd + dgrad-via-flipped-transpose +
    MIOpen wgrad) vs F.conv2d, bf16 channels_last."""
    from chainermn_amd.links import FusedConv3x3
    torch.manual_seed(52)
    for N, C, H, W, Cout in [(4, 64, 8, 8, 64), (2, 128, 16, 16, 128),
                             (1, 64, 16, 16, 256)]:
        x32 = torch.randn(N, C, H, W, device=DEV)
        gy32 = torch.randn(N, Cout, H, W, device=DEV) * 0.1
        conv = FusedConv3x3(C, Cout).to(DEV, torch.bfloat16)
        with torch.no_grad():
            conv.weight.mul_(0.05)
        conv = conv.to(memory_format=torch.channels_last)
        x = x32.to(torch.bfloat16).to(memory_format=torch.channels_last)
        x.requires_grad_(True)
        xr = x32.to(torch.bfloat16).clone().requires_grad_(True)
        wr = conv.weight.detach().clone().requires_grad_(True)
        y = conv(x)
        yr = torch.nn.functional.conv2d(xr, wr, padding=1)
        mag = yr.float().abs().max()
        assert (y.float() - yr.float()).abs().max() <= 0.03 * mag + 0.05, \
            (N, C, H, W, Cout)
        gy = gy32.to(torch.bfloat16)
        y.backward(gy.to(memory_format=torch.channels_last))
        yr.backward(gy)
        gmag = xr.grad.float().abs().max()
        assert (x.grad.float() - xr.grad.float()).abs().max() \
            <= 0.05 * gmag + 0.05, (N, C, H, W, Cout, 'dgrad')
        wmag = wr.grad.float().abs().max()
        assert (conv.weight.grad.float() - wr.grad.float()).abs().max() \
            <= 0.05 * wmag + 0.05, (N, C, H, W, Cout, 'wgrad')


def test_fused_conv3x3_stride2_fallback():
    """stride-2 FusedConv3x3 must match F.conv2d exactly (fallback)."""
    from chainermn_amd.links import FusedConv3x3
    torch.manual_seed(53)
    conv = FusedConv3x3(64, 128, stride=2).to(DEV, torch.bfloat16)
    x = torch.randn(2, 64, 16, 16, device=DEV, dtype=torch.bfloat16
                    ).to(memory_format=torch.channels_last)
    y = conv(x)
    yr = torch.nn.functional.conv2d(x, conv.weight, stride=2, padding=1)
    assert torch.equal(y.float(), yr.float())


def test_producer_fused_bn_stats_match_bn_reduce():
    """Conv/GEMM epilogue stats partials == the standalone bn_stats pass."""
    torch.manual_seed(57)
    # conv3x3 epilogue (all three wave geometries)
    for N, C, H, W, Cout in [(4, 64, 8, 8, 64), (2, 128, 16, 16, 128),
                             (1, 64, 16, 16, 256)]:
        x = torch.randn(N, C, H, W, device=DEV, dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last)
        w = (torch.randn(Cout, C, 3, 3, device=DEV, dtype=torch.bfloat16)
             * 0.05).to(memory_format=torch.channels_last)
        stats = torch.zeros(16, 2 * Cout, device=DEV)
        y = _ext().conv3x3s1_fwd_bf16(x, w, stats)
        ref = torch.empty(2 * Cout, device=DEV)
        _ext().bn_stats_into(y, ref)
        got = stats.sum(0)
        assert torch.allclose(got, ref, rtol=1e-3, atol=2e-2), \
            (N, C, H, W, Cout, (got - ref).abs().max())
    # 256-tile phased GEMM epilogue
    a = torch.randn(512, 128, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(256, 128, device=DEV, dtype=torch.bfloat16) * 0.05
    stats = torch.zeros(16, 512, device=DEV)
    d = _ext().gemm_bt256p_stats_bf16(a, b, stats, 3)
    ref = torch.empty(512, device=DEV)
    _ext().bn_stats_into(d, ref)
    got = stats.sum(0)
    assert torch.allclose(got, ref, rtol=1e-3, atol=2e-2), \
        (got - ref).abs().max()


def test_bn_pre_stats_path_matches_plain():
    """FusedBatchNorm2d(pre_stats=...) == the standalone-stats path,
    forward and backward."""
    from chainermn_amd.links import FusedConv3x3
    from chainermn_amd.links.fused_batch_norm import FusedBNReLU2d
    torch.manual_seed(58)
    conv = FusedConv3x3(64, 128).to(DEV, torch.bfloat16)
    with torch.no_grad():
        conv.weight.mul_(0.05)
    conv = conv.to(memory_format=torch.channels_last)
    bn = FusedBNReLU2d(128).to(DEV)
    bn2 = FusedBNReLU2d(128).to(DEV)
    bn2.load_state_dict(bn.state_dict())
    x32 = torch.randn(4, 64, 16, 16, device=DEV)
    x = x32.to(torch.bfloat16).to(memory_format=torch.channels_last
                                  ).requires_grad_(True)
    xr = x32.to(torch.bfloat16).to(memory_format=torch.channels_last
                                   ).requires_grad_(True)
    y, st = conv.forward_stats(x)
    assert st is not None  # the custom kernel must have been dispatched
    out = bn(y, pre_stats=st)
    ref = bn2(conv(xr))
    assert torch.allclose(out.float(), ref.float(), atol=3e-2), \
        (out.float() - ref.float()).abs().max()
    g = torch.randn_like(ref)
    out.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad.float(), xr.grad.float(), atol=3e-2)
    assert torch.allclose(bn.weight.grad, bn2.weight.grad,
                          rtol=1e-2, atol=1e-2)
    assert torch.allclose(bn.running_var, bn2.running_var,
                          rtol=1e-3, atol=1e-4)


def test_conv3x3_wgrad_vs_reference():
    """Hand-written wgrad kernel vs aten.convolution_backward (MIOpen),
    both against an fp32 conv_backward reference."""
    torch.manual_seed(61)
    for N, C, H, W, Co in [(4, 64, 8, 8, 64), (2, 128, 16, 16, 128),
                           (1, 64, 16, 16, 256), (3, 64, 7, 7, 64),
                           (2, 192, 9, 9, 64)]:
        x32 = torch.randn(N, C, H, W, device=DEV)
        gy32 = torch.randn(N, Co, H, W, device=DEV) * 0.1
        x = x32.to(torch.bfloat16).to(memory_format=torch.channels_last)
        gy = gy32.to(torch.bfloat16).to(
            memory_format=torch.channels_last)
        gw = _ext().conv3x3_wgrad_bf16(x, gy)
        # fp32 reference on the same bf16-rounded inputs
        w_ref = torch.zeros(Co, C, 3, 3, device=DEV, requires_grad=True)
        yr = torch.nn.functional.conv2d(x.float(), w_ref, padding=1)
        yr.backward(gy.float())
        ref = w_ref.grad
        err = (gw.float() - ref).abs().max()
        mag = ref.abs().max()
        assert err <= 0.02 * mag + 0.05, (N, C, H, W, Co, err, mag)
        assert gw.is_contiguous(memory_format=torch.channels_last)
        assert gw.dtype == torch.bfloat16


def test_mfma_gemm_btw_vs_matmul():
    """Narrow-N phased GEMM (BN 128/64) numerics + fused stats."""
    torch.manual_seed(63)
    for M, N, K in [(256, 128, 64), (512, 64, 128), (768, 192, 320),
                    (256, 128, 1024), (512, 320, 64)]:
        a = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        a[0, :] += torch.arange(K, device=DEV, dtype=torch.bfloat16) * 0.01
        b[1, :] -= torch.arange(K, device=DEV, dtype=torch.bfloat16) * 0.02
        ref = a.float() @ b.float().t()
        mag = ref.abs().max()
        d = _ext().gemm_btw_bf16(a, b)
        assert (d.float() - ref).abs().max() <= 0.02 * mag + 0.1, (M, N, K)
        stats = torch.zeros(16, 2 * N, device=DEV)
        d2 = _ext().gemm_btw_bf16(a, b, stats)
        assert torch.equal(d, d2)
        sref = torch.empty(2 * N, device=DEV)
        _ext().bn_stats_into(d, sref)
        assert torch.allclose(stats.sum(0), sref, rtol=1e-3, atol=2e-2), \
            (M, N, K, (stats.sum(0) - sref).abs().max())


def test_residual_junction_fold_matches_unfused():
    """The _ResidualEntry + conv1-dgrad accumulate fold must give the
    same grads as the SAME fused stack with the fold disabled (second
    step, when the per-shape GEMM choices are decided and the fold
    engages).  Identical kernel stacks, so tolerances are tight."""
    from chainermn_amd.models.resnet import Bottleneck
    torch.manual_seed(77)
    fused = torch.nn.Sequential(
        Bottleneck(256, 64, fused_ops=True),
        Bottleneck(256, 64, fused_ops=True),
    ).to(DEV, torch.bfloat16).to(memory_format=torch.channels_last)
    ref = torch.nn.Sequential(
        Bottleneck(256, 64, fused_ops=True),
        Bottleneck(256, 64, fused_ops=True),
    ).to(DEV, torch.bfloat16).to(memory_format=torch.channels_last)
    ref.load_state_dict(fused.state_dict())
    try:
        for step in range(2):
            torch.manual_seed(300 + step)
            x32 = torch.randn(4, 256, 16, 16, device=DEV) * 0.5
            x = x32.to(torch.bfloat16).to(
                memory_format=torch.channels_last).requires_grad_(True)
            xr = x32.to(torch.bfloat16).to(
                memory_format=torch.channels_last).requires_grad_(True)
            Bottleneck._use_junction_fold = True
            y = fused(x)
            Bottleneck._use_junction_fold = False
            yr = ref(xr)
            # forward differs only by the atomic-order rounding of the
            # producer-fused BN stats (+-1 ulp on mean/var)
            fmag = yr.float().abs().max()
            assert (y.float() - yr.float()).abs().max() \
                <= 0.01 * fmag + 0.01, step
            g = torch.randn_like(yr) * 0.1
            Bottleneck._use_junction_fold = True
            y.backward(g)
            Bottleneck._use_junction_fold = False
            yr.backward(g)
            # the fold changes only the junction-add rounding order
            # (fp32 accumulate + one bf16 round vs bf16 + bf16 add)
            gmag = xr.grad.float().abs().max()
            assert (x.grad.float() - xr.grad.float()).abs().max() \
                <= 0.02 * gmag + 0.02, (step, 'input grad')
            for (n1, p1), (n2, p2) in zip(fused.named_parameters(),
                                          ref.named_parameters()):
                if p1.grad is None or p2.grad is None:
                    continue
                pm = p2.grad.float().abs().max()
                assert (p1.grad.float() - p2.grad.float()).abs().max() \
                    <= 0.02 * pm + 0.02, (step, n1)
            fused.zero_grad()
            ref.zero_grad()
    finally:
        Bottleneck._use_junction_fold = True


def test_bnbwd_epilogue_matches_standalone_reduce():
    """The dgrad-fused BN-backward reduce == bn_bwd_reduce_act_into
    (relu_mode 1) computed on the produced gy."""
    torch.manual_seed(91)
    # GEMM form (1x1 dgrad): d = a @ b^T is the BN's gy
    for M, N, K in [(512, 256, 128), (256, 128, 64), (512, 64, 192)]:
        a = torch.randn(M, K, device=DEV, dtype=torch.bfloat16) * 0.1
        b = torch.randn(N, K, device=DEV, dtype=torch.bfloat16)
        bn_x = torch.randn(M, N, device=DEV, dtype=torch.bfloat16)
        mean = torch.randn(N, device=DEV) * 0.1
        rstd = torch.rand(N, device=DEV) + 0.5
        gamma = torch.randn(N, device=DEV, dtype=torch.bfloat16)
        beta = torch.randn(N, device=DEV, dtype=torch.bfloat16) * 0.1
        partials = torch.zeros(16, 2 * N, device=DEV)
        if N % 256 == 0:
            d = _ext().gemm_bt256p_bnbwd_bf16(a, b, bn_x, mean, rstd,
                                              gamma, beta, partials)
        else:
            d = _ext().gemm_btw_bnbwd_bf16(a, b, bn_x, mean, rstd,
                                           gamma, beta, partials)
        ref = torch.empty(2 * N, device=DEV)
        _ext().bn_bwd_reduce_act_into(bn_x, d, mean, rstd, gamma, beta,
                                      bn_x.new_empty(0), 1, ref)
        got = partials.sum(0)
        assert torch.allclose(got, ref, rtol=1e-3, atol=5e-2), \
            (M, N, K, (got - ref).abs().max())
    # conv form (3x3 dgrad)
    x = (torch.randn(2, 64, 16, 16, device=DEV, dtype=torch.bfloat16)
         * 0.1).to(memory_format=torch.channels_last)
    w = (torch.randn(128, 64, 3, 3, device=DEV, dtype=torch.bfloat16)
         * 0.05).to(memory_format=torch.channels_last)
    Cout = 128
    Mt = 2 * 16 * 16
    bn_x = torch.randn(2, Cout, 16, 16, device=DEV, dtype=torch.bfloat16
                       ).to(memory_format=torch.channels_last)
    mean = torch.randn(Cout, device=DEV) * 0.1
    rstd = torch.rand(Cout, device=DEV) + 0.5
    gamma = torch.randn(Cout, device=DEV, dtype=torch.bfloat16)
    beta = torch.randn(Cout, device=DEV, dtype=torch.bfloat16) * 0.1
    partials = torch.zeros(16, 2 * Cout, device=DEV)
    y = _ext().conv3x3s1_fwd_bnbwd_bf16(
        x, w, x.new_empty(0, dtype=torch.float32), bn_x, mean, rstd,
        gamma, beta, partials)
    ref = torch.empty(2 * Cout, device=DEV)
    _ext().bn_bwd_reduce_act_into(bn_x, y, mean, rstd, gamma, beta,
                                  bn_x.new_empty(0), 1, ref)
    got = partials.sum(0)
    assert torch.allclose(got, ref, rtol=1e-3, atol=5e-2), \
        (got - ref).abs().max()


def test_resnet_fused_ops_fp32_composition():
    """fused_ops model at fp32: convs fall back to MIOpen while the BN
    kernels run their fp32 paths — the mixed composition must train."""
    import chainermn_amd
    from chainermn_amd.models import ResNet50
    from chainermn_amd.parallel import FusedDataParallelSGD
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(21)
    model = ResNet50(num_classes=10, fused_ops=True).to(
        device=DEV, memory_format=torch.channels_last)  # fp32
    opt = FusedDataParallelSGD(model, comm, lr=0.05, momentum=0.9,
                               overlap=False, grad_dtype=torch.float32)
    x = torch.randn(8, 3, 64, 64, device=DEV
                    ).to(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (8,), device=DEV)
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert all(l == l for l in losses)  # finite
    assert losses[-1] < losses[0], losses
