"""Profiling target: the three bf16 GEMM kernels at one shape, N reps
each, for rocprofv3 PMC / kernel-stats runs."""
import os, sys, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops

M = N = K = int(os.environ.get("GEMM_N", 4096))
reps = int(os.environ.get("GEMM_REPS", 20))
a = (torch.rand(M, K, device='cuda', dtype=torch.bfloat16)*2-1)
b = (torch.rand(N, K, device='cuda', dtype=torch.bfloat16)*2-1)
ext = ops.require()
for fn in (ext.gemm_bt_bf16, ext.gemm_bt256_bf16, ext.gemm_bt256p_bf16):
    for _ in range(3): fn(a, b)
    torch.cuda.synchronize()
    for _ in range(reps): fn(a, b)
    torch.cuda.synchronize()
print("done")
