"""utils: trace ranges, StepTimer, DeviceBuffer growth, PackPlan layout."""

import torch

from chainermn_amd import ops
from chainermn_amd.utils import StepTimer, trace_range
from chainermn_amd.utils.memory import DeviceBuffer


def test_trace_range_noop_cpu():
    with trace_range('phase'):
        pass  # must be silent without a GPU


def test_step_timer():
    t = StepTimer()
    for _ in range(3):
        with t.phase('fwd'):
            sum(range(1000))
    s = t.summary()
    assert s['fwd']['total_s'] > 0
    assert abs(s['fwd']['mean_ms'] - s['fwd']['total_s'] / 3 * 1000) < 1e-9


def test_device_buffer_grows_and_reuses():
    b = DeviceBuffer()
    t1 = b.array(100, torch.float32)
    p1 = t1.data_ptr()
    t2 = b.array(50, torch.float32)  # smaller: reuse
    assert t2.data_ptr() == p1
    t3 = b.array(200, torch.float32)  # grow: new backing
    assert t3.numel() == 200


def test_pack_plan_alignment():
    class P:
        def __init__(self, n):
            self.data = torch.zeros(n)
            self.grad = torch.zeros(n)

        def numel(self):
            return self.data.numel()

    params = [P(5), P(64), P(129)]
    plan = ops.PackPlan(params, torch.device('cpu'))
    assert plan.offsets_cpu == [0, 64, 128]
    assert plan.total == 128 + 192  # 129 rounded up to 192
    for off in plan.offsets_cpu:
        assert off % ops.ALIGN == 0


def test_gemm_wrappers_cpu_fallback():
    import torch
    a = torch.randn(32, 16)
    b = torch.randn(24, 16)
    assert torch.allclose(ops.gemm_bt(a, b), a @ b.t(), atol=1e-5)
    c = torch.randn(16, 24)
    assert torch.allclose(ops.gemm_nn(a, c), a @ c, atol=1e-5)


def test_conv_ops_cpu_fallbacks():
    """ops.conv3x3 / conv3x3_with_stats / conv3x3_wgrad fall back to
    torch ops for CPU tensors (GPU kernels dispatch only when eligible)."""
    import torch
    import torch.nn.functional as F
    from chainermn_amd import ops
    torch.manual_seed(5)
    x = torch.randn(2, 8, 6, 6)
    w = torch.randn(8, 8, 3, 3) * 0.1
    y = ops.conv3x3(x, w)
    assert torch.allclose(y, F.conv2d(x, w, padding=1))
    y2, st = ops.conv3x3_with_stats(x, w)
    assert st is None and torch.equal(y, y2)
    gy = torch.randn_like(y)
    gw = ops.conv3x3_wgrad(x, gy, w)
    ref = torch.ops.aten.convolution_backward(
        gy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
        [False, True, False])[1]
    assert torch.allclose(gw, ref)
    # gemm stats fallback: CPU / undecided shapes return stats=None
    a = torch.randn(128, 64)
    b = torch.randn(128, 64)
    d, st = ops.gemm_bt_with_stats(a, b)
    assert st is None and torch.allclose(d, a @ b.t(), atol=1e-5)
    # accumulate fold: CPU tensors never fold (caller owns the add)
    bmat = torch.randn(64, 32)
    out = torch.zeros(128, 32)
    r, folded = ops.gemm_nn_accum(a, bmat, out)
    assert not folded
    assert torch.allclose(r, a @ bmat, atol=1e-5)
    # bnbwd fallback returns no partials off-GPU
    r2, parts = ops.gemm_nn_bnbwd(a, bmat, {})
    assert parts is None and torch.allclose(r2, a @ bmat, atol=1e-5)


def test_residual_entry_function_semantics():
    """_ResidualEntry: explicit fork whose backward adds the two path
    grads, unless the junction was marked 'fused' (then the identity
    grad is already folded into the main grad and must not double)."""
    import torch
    from chainermn_amd.links.fused_conv import _ResidualEntry
    x = torch.randn(4, requires_grad=True)
    j = {}
    a, b = _ResidualEntry.apply(x, j)
    (a * 2 + b * 3).sum().backward()
    assert torch.allclose(x.grad, torch.full((4,), 5.0))
    # fused flag short-circuits the add (g_main already contains g_id)
    x2 = torch.randn(4, requires_grad=True)
    j2 = {}
    a2, b2 = _ResidualEntry.apply(x2, j2)
    loss = (a2 * 2 + b2 * 3).sum()
    j2['fused'] = True
    loss.backward()
    assert torch.allclose(x2.grad, torch.full((4,), 2.0))
    assert 'fused' not in j2  # consumed


def test_wgrad_magic_divisor_formula():
    """Python mirror of wg_magic() in _hip/csrc/wgrad3x3.hip: the
    round-up magic (mul, shift) pair must give exact floor division for
    every n <= n_max.  The kernel-side correctness is covered end to end
    by the GPU wgrad test; this guards the FORMULA (shift search +
    exactness bound) against future edits."""
    import random
    rng = random.Random(13)

    def magic(d, n_max):
        if d & (d - 1) == 0:  # power of two incl. 1
            k = d.bit_length() - 1
            return (1 << 31, k - 1)
        for s in range(30, -1, -1):
            m = ((1 << (32 + s)) + d - 1) // d
            if m >= (1 << 32):
                continue
            e = m * d - (1 << (32 + s))
            if e == 0 or (1 << (32 + s)) // e > n_max:
                return (m, s)
        raise AssertionError(d)

    n_max = 2 ** 31 - 1
    for d in [1, 2, 3, 5, 6, 7, 9, 14, 16, 28, 49, 56, 64, 100, 196,
              784, 3136, 12544, 50176]:
        m, s = magic(d, n_max)
        samples = list(range(0, 3000)) + \
            [rng.randrange(0, n_max) for _ in range(20000)] + \
            [n_max, n_max - 1, d - 1, d, d + 1, 2 * d - 1]
        for n in samples:
            assert (n * m) >> (32 + s) == n // d, (d, n, m, s)
