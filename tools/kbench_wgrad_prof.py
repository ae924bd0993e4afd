import os, sys, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops
N, C, H, W, Co = 512, 256, 14, 14, 256
x = (torch.rand(N, C, H, W, device='cuda', dtype=torch.bfloat16)*2-1).to(memory_format=torch.channels_last)
gy = ((torch.rand(N, Co, H, W, device='cuda', dtype=torch.bfloat16)*2-1)*0.1).to(memory_format=torch.channels_last)
ext = ops.require()
for _ in range(3): ext.conv3x3_wgrad_bf16(x, gy)
torch.cuda.synchronize()
for _ in range(15): ext.conv3x3_wgrad_bf16(x, gy)
torch.cuda.synchronize()
print("done")
