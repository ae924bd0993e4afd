"""Fixed-batch memorization probe (round-1 protocol): ResNet-50 on one
synthetic batch, 300 steps, lr 0.02 — the full fused stack must drive
the loss to ~0, like the torch-native stack does.  Round-2 re-run with
the phased conv kernels, producer-fused BN stats and the residual
junction fold all active."""
import os, sys, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import chainermn_amd
from chainermn_amd.models import ResNet50
from chainermn_amd.parallel import FusedDataParallelSGD

def run(fused):
    comm = chainermn_amd.create_communicator('dummy')
    torch.manual_seed(5)
    m = ResNet50(num_classes=100, fused_ops=fused).to(
        'cuda', torch.bfloat16, memory_format=torch.channels_last)
    opt = FusedDataParallelSGD(m, comm, lr=0.02, momentum=0.9,
                               overlap=False)
    torch.manual_seed(9)
    x = torch.randn(64, 3, 64, 64, device='cuda', dtype=torch.bfloat16
                    ).to(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (64,), device='cuda')
    hist = []
    for step in range(300):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(m(x).float(), y)
        loss.backward()
        opt.step()
        if step % 50 == 0 or step == 299:
            hist.append((step, round(float(loss), 4)))
    return hist

for fused in (True, False):
    h = run(fused)
    print(("fused " if fused else "torch ") + "stack:", h, flush=True)
