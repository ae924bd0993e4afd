"""Within-probe interleaved A/B of the 256-tile GEMM schedule variants
(guide rule 24: N variants x M rounds in ONE process, report median/min)."""
import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops

ext = ops.require()

def timeit(fn, n=10):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

for MNK in (4096, 8192):
    M = N = K = MNK
    a = (torch.rand(M, K, device='cuda', dtype=torch.bfloat16)*2-1)
    b = (torch.rand(N, K, device='cuda', dtype=torch.bfloat16)*2-1)
    ref = ext.gemm_bt256_bf16(a, b)
    variants = {
        'simple': lambda: ext.gemm_bt256_bf16(a, b),
        'p0(read,2bar)': lambda: ext.gemm_bt256p_bf16(a, b, 0),
        'p1(stage,2bar)': lambda: ext.gemm_bt256p_bf16(a, b, 1),
        'p2(read,1bar)': lambda: ext.gemm_bt256p_bf16(a, b, 2),
        'p3(stage,1bar)': lambda: ext.gemm_bt256p_bf16(a, b, 3),
        'blas': lambda: a @ b.t(),
    }
    # numerics check for every variant first
    for name, fn in variants.items():
        d = fn()
        err = (d.float() - ref.float()).abs().max().item()
        assert err < 1.0, (name, err)
    fl = 2.0 * M * N * K
    rounds = {k: [] for k in variants}
    for r in range(6):
        for name, fn in variants.items():
            rounds[name].append(timeit(fn, n=8))
    print(f"== {MNK}^3 ==")
    for name, ts in rounds.items():
        med = sorted(ts)[len(ts)//2]; mn = min(ts)
        print(f"  {name:16s} median {fl/med/1e12:7.1f} TF  best {fl/mn/1e12:7.1f} TF")
