"""wgrad residue probe: time MIOpen wrw for the ResNet 3x3 shapes under
the shipped DB vs exhaustive find, plus hipBLASLt GEMM formulations."""
import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
import torch.nn.functional as F

def t(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n

shapes = [(512, 64, 56, 56, 64), (512, 128, 28, 28, 128),
          (512, 256, 14, 14, 256), (512, 512, 7, 7, 512)]
for N, C, H, W, Co in shapes:
    x = torch.randn(N, C, H, W, device='cuda', dtype=torch.bfloat16
                    ).to(memory_format=torch.channels_last)
    gy = torch.randn(N, Co, H, W, device='cuda', dtype=torch.bfloat16
                     ).to(memory_format=torch.channels_last)
    w = torch.randn(Co, C, 3, 3, device='cuda', dtype=torch.bfloat16
                    ).to(memory_format=torch.channels_last)
    def wrw():
        return torch.ops.aten.convolution_backward(
            gy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
    dt = t(wrw)
    fl = 2.0 * N * H * W * Co * 9 * C
    # GEMM formulation: gw[co, 9ci] via unfold is too big; try per-tap
    # bmm-free: hipBLASLt on [Co, M] x [M, C] per tap (9 gemms)
    M = N * H * W
    gy2 = gy.permute(0, 2, 3, 1).reshape(M, Co)
    x2 = x.permute(0, 2, 3, 1).reshape(M, C)
    def tap_gemm():  # center tap only, x9 estimate
        return gy2.t() @ x2
    dt2 = t(tap_gemm)
    print(f"N{N} C{C} {H}x{W} Co{Co}: wrw {dt*1e6:7.1f}us ({fl/dt/1e12:6.1f} TF) | 9x center-tap-gemm est {9*dt2*1e6:7.1f}us")
