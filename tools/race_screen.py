"""Race screen for the phased-schedule kernels (guide two-lane
discipline: sync-structure edits need multi-run screens, not one
refcheck).  20 rounds x fresh random data x all schedule variants and
geometries; any nondeterministic wait/barrier bug shows as a sporadic
mismatch."""
import os, sys, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import torch.nn.functional as F
from chainermn_amd import ops
ext = ops.require()

fails = 0
for rnd in range(20):
    torch.manual_seed(1000 + rnd)
    # GEMM variants
    for M, N, K in [(512, 256, 192), (256, 512, 64), (768, 768, 320)]:
        a = torch.randn(M, K, device='cuda', dtype=torch.bfloat16)
        b = torch.randn(N, K, device='cuda', dtype=torch.bfloat16)
        ref = a.float() @ b.float().t()
        mag = float(ref.abs().max())
        for name, fn in [('256', lambda: ext.gemm_bt256_bf16(a, b))] + [
                (f'p{s}', lambda s=s: ext.gemm_bt256p_bf16(a, b, s))
                for s in range(4)] + [('w', lambda: ext.gemm_btw_bf16(a, b))]:
            d = fn()
            err = float((d.float() - ref).abs().max())
            if err > 0.02 * mag + 0.1:
                fails += 1
                print(f"FAIL rnd{rnd} gemm{name} {M}x{N}x{K} err={err}")
    # conv geometries (all three + 2-phase c64)
    for Nb, C, H, W, Co in [(2, 64, 16, 16, 256), (4, 128, 8, 8, 128),
                            (16, 64, 4, 4, 64), (1, 192, 16, 16, 320)]:
        x = torch.randn(Nb, C, H, W, device='cuda', dtype=torch.bfloat16
                        ).to(memory_format=torch.channels_last)
        w = (torch.randn(Co, C, 3, 3, device='cuda',
                         dtype=torch.bfloat16) * 0.05
             ).to(memory_format=torch.channels_last)
        y = ext.conv3x3s1_fwd_bf16(x, w)
        ref = F.conv2d(x, w, padding=1)
        err = float((y.float() - ref.float()).abs().max())
        mag = float(ref.float().abs().max())
        if err > 0.03 * mag + 0.05:
            fails += 1
            print(f"FAIL rnd{rnd} conv {Nb},{C},{H},{W}->{Co} err={err}")
    # wgrad
    x = (torch.randn(3, 64, 9, 9, device='cuda', dtype=torch.bfloat16)
         ).to(memory_format=torch.channels_last)
    gy = (torch.randn(3, 64, 9, 9, device='cuda', dtype=torch.bfloat16)
          * 0.1).to(memory_format=torch.channels_last)
    gw = ext.conv3x3_wgrad_bf16(x, gy)
    wr = torch.zeros(64, 64, 3, 3, device='cuda', requires_grad=True)
    F.conv2d(x.float(), wr, padding=1).backward(gy.float())
    err = float((gw.float() - wr.grad).abs().max())
    mag = float(wr.grad.abs().max())
    if err > 0.02 * mag + 0.05:
        fails += 1
        print(f"FAIL rnd{rnd} wgrad err={err}")
print("race screen:", "CLEAN (20 rounds)" if fails == 0 else f"{fails} FAILURES")
