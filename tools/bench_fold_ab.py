"""Within-box A/B: residual-junction fold on vs off (full ResNet step)."""
import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import chainermn_amd
from chainermn_amd.models import ResNet50
from chainermn_amd.models.resnet import Bottleneck
from chainermn_amd.parallel import FusedDataParallelSGD

comm = chainermn_amd.create_communicator('dummy')
torch.manual_seed(1)
m = ResNet50(fused_ops=True).to('cuda', torch.bfloat16,
                                memory_format=torch.channels_last)
opt = FusedDataParallelSGD(m, comm, lr=0.05, momentum=0.9, overlap=True)
x = torch.randn(512, 3, 224, 224, device='cuda', dtype=torch.bfloat16
                ).to(memory_format=torch.channels_last)
yt = torch.randint(0, 1000, (512,), device='cuda')
lf = torch.nn.CrossEntropyLoss()

def step():
    opt.zero_grad()
    loss = lf(m(x).float(), yt)
    loss.backward()
    opt.step()

for _ in range(8): step()
torch.cuda.synchronize()

import collections
res = collections.defaultdict(list)
for rnd in range(4):
    for fold in (True, False):
        Bottleneck._use_junction_fold = fold
        for _ in range(2): step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(12): step()
        torch.cuda.synchronize()
        res[fold].append((time.perf_counter() - t0) / 12 * 1000)
for fold, ts in res.items():
    print(f"fold={fold}: median {sorted(ts)[len(ts)//2]:.3f} ms  best {min(ts):.3f} ms  all {[round(t,2) for t in ts]}")

# diagnostic: how many junctions actually folded in one step?
from chainermn_amd import ops as _ops
import chainermn_amd.links.fused_conv as fc
orig = _ops.gemm_nn_accum
cnt = {'called': 0, 'folded': 0}
def wrapped(a, b, out):
    cnt['called'] += 1
    r, f = orig(a, b, out)
    cnt['folded'] += int(f)
    return r, f
_ops.gemm_nn_accum = wrapped
fc.ops.gemm_nn_accum = wrapped
Bottleneck._use_junction_fold = True
step()
torch.cuda.synchronize()
print("junction fold calls:", cnt)
