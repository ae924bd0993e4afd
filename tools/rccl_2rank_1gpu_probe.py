"""can RCCL run 2 ranks on ONE MI355X (shared device)?

If yes, this executes the real GPU collective/p2p code paths
cross-process (pure_rccl mean-grad, MNBN stats, MultiNodeChainList
backward) without an 8-GPU node.  If RCCL refuses shared-device ranks,
we record that and keep gloo coverage (the reference position).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
       --master-addr 127.0.0.1 tools/rccl_2rank_1gpu_probe.py
"""
import os, sys
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import torch
import torch.distributed as dist

rank = int(os.environ['RANK'])
os.environ.setdefault('HSA_ENABLE_IPC_MODE_LEGACY', '0')
torch.cuda.set_device(0)  # BOTH ranks on device 0
try:
    dist.init_process_group('nccl', timeout=__import__('datetime').timedelta(seconds=60))
    x = torch.full((1024,), float(rank + 1), device='cuda')
    dist.all_reduce(x)
    torch.cuda.synchronize()
    ok = bool((x == 3.0).all().item())
    print(f"rank {rank}: allreduce ok={ok}", flush=True)
    # p2p send/recv pair
    if rank == 0:
        t = torch.arange(64, device='cuda', dtype=torch.float32)
        dist.send(t, dst=1)
    else:
        t = torch.empty(64, device='cuda')
        dist.recv(t, src=0)
        print(f"rank 1: recv ok={bool((t == torch.arange(64, device='cuda', dtype=torch.float32)).all().item())}", flush=True)
    dist.barrier()
    dist.destroy_process_group()
    print(f"rank {rank}: SHARED-DEVICE RCCL WORKS", flush=True)
except Exception as e:
    print(f"rank {rank}: RCCL shared-device REFUSED/FAILED: {type(e).__name__}: {str(e)[:300]}", flush=True)
    sys.exit(1)
