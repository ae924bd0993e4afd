"""Profiling target: conv3x3 fwd (3 geometries) for rocprofv3 PMC."""
import os, sys, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops
ext = ops.require()
for N, C, H, W, Co in [(512, 64, 56, 56, 64), (512, 128, 28, 28, 128),
                       (512, 256, 14, 14, 256)]:
    x = (torch.rand(N, C, H, W, device='cuda', dtype=torch.bfloat16)*2-1
         ).to(memory_format=torch.channels_last)
    w = ((torch.rand(Co, C, 3, 3, device='cuda', dtype=torch.bfloat16)*2-1)
         * 0.05).to(memory_format=torch.channels_last)
    for _ in range(3): ext.conv3x3s1_fwd_bf16(x, w)
    torch.cuda.synchronize()
    for _ in range(15): ext.conv3x3s1_fwd_bf16(x, w)
    torch.cuda.synchronize()
print("done")
