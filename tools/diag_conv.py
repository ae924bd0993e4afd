import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import chainermn_amd
from chainermn_amd import ops
from chainermn_amd.models import ResNet50
comm = chainermn_amd.create_communicator('dummy')
m = ResNet50(fused_ops=True).to('cuda', torch.bfloat16,
                                memory_format=torch.channels_last)
x = torch.randn(512, 3, 224, 224, device='cuda', dtype=torch.bfloat16
                ).to(memory_format=torch.channels_last)
y = torch.randint(0, 1000, (512,), device='cuda')
lf = torch.nn.CrossEntropyLoss()
for i in range(3):
    loss = lf(m(x).float(), y)
    loss.backward()
    for p in m.parameters(): p.grad = None
torch.cuda.synchronize()
c3 = {k: v for k, v in ops._GEMM_CHOICE.items() if isinstance(k, tuple) and k[0] == 'c3'}
print("conv3 choices:", c3)
print("n gemm choices:", len(ops._GEMM_CHOICE))
# time fwd+bwd
t0 = time.perf_counter()
for i in range(5):
    loss = lf(m(x).float(), y)
    loss.backward()
    for p in m.parameters(): p.grad = None
torch.cuda.synchronize()
print("fwd+bwd ms:", (time.perf_counter()-t0)/5*1000)
