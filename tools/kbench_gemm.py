"""MFMA GEMM microbench: hand-written kernel vs hipBLASLt."""
import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops

def t(fn, n=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n

print("bf16 GEMM D = A @ B^T, uniform random [-1,1) operands:")
for M, N, K in [(1024,1024,1024),(4096,4096,4096),(8192,8192,8192),
                (401408,128,512),(100352,256,1024),(401408,256,512),
                (100352,512,1024),(25088,1024,2048),(6272,2048,1024)]:
    a = (torch.rand(M, K, device='cuda', dtype=torch.bfloat16)*2-1)
    b = (torch.rand(N, K, device='cuda', dtype=torch.bfloat16)*2-1)
    fl = 2.0*M*N*K
    dt1 = t(lambda: ops.require().gemm_bt_bf16(a, b))
    dt2 = t(lambda: a @ b.t())
    line = f"  M{M} N{N} K{K}: 128t {fl/dt1/1e12:7.1f} TF | hipBLASLt {fl/dt2/1e12:7.1f} TF"
    if M % 256 == 0 and N % 256 == 0:
        dt3 = t(lambda: ops.require().gemm_bt256_bf16(a, b))
        dt4 = t(lambda: ops.require().gemm_bt256p_bf16(a, b))
        line += f" | 256t {fl/dt3/1e12:7.1f} TF | 256t8p {fl/dt4/1e12:7.1f} TF"
    print(line)

print("\nimplicit-GEMM 3x3/s1/p1 conv fwd vs torch/MIOpen:")
import torch.nn.functional as F
for N, C, H, W, Cout in [(512, 64, 56, 56, 64), (512, 128, 28, 28, 128),
                         (512, 256, 14, 14, 256), (512, 512, 7, 7, 512)]:
    if (N*H*W) % 256: continue
    x = (torch.rand(N, C, H, W, device='cuda', dtype=torch.bfloat16)*2-1).to(memory_format=torch.channels_last)
    w = ((torch.rand(Cout, C, 3, 3, device='cuda', dtype=torch.bfloat16)*2-1)*0.05).to(memory_format=torch.channels_last)
    fl = 2.0*N*H*W*Cout*9*C
    d1 = t(lambda: ops.require().conv3x3s1_fwd_bf16(x, w))
    d2 = t(lambda: F.conv2d(x, w, padding=1))
    print(f"  N{N} C{C} {H}x{W} -> {Cout}: ours {fl/d1/1e12:6.1f} TF ({d1*1e6:6.1f}us) | miopen {fl/d2/1e12:6.1f} TF ({d2*1e6:6.1f}us)")

print("\n3x3 conv WGRAD: ours vs MIOpen wrw (aten.convolution_backward):")
for N, C, H, W, Co in [(512, 64, 56, 56, 64), (512, 128, 28, 28, 128),
                       (512, 256, 14, 14, 256), (512, 512, 7, 7, 512)]:
    x = (torch.rand(N, C, H, W, device='cuda', dtype=torch.bfloat16)*2-1).to(memory_format=torch.channels_last)
    gy = ((torch.rand(N, Co, H, W, device='cuda', dtype=torch.bfloat16)*2-1)*0.1).to(memory_format=torch.channels_last)
    w = torch.zeros(Co, C, 3, 3, device='cuda', dtype=torch.bfloat16).to(memory_format=torch.channels_last)
    fl = 2.0*N*H*W*Co*9*C
    d1 = t(lambda: ops.require().conv3x3_wgrad_bf16(x, gy))
    d2 = t(lambda: torch.ops.aten.convolution_backward(
        gy, x, w, None, [1,1], [1,1], [1,1], False, [0,0], 1,
        [False, True, False])[1])
    print(f"  N{N} C{C} {H}x{W} Co{Co}: ours {fl/d1/1e12:6.1f} TF ({d1*1e6:7.1f}us) | miopen {fl/d2/1e12:6.1f} TF ({d2*1e6:7.1f}us)")

print("\nnarrow-N phased GEMM (BN 128/64) vs 128-tile vs hipBLASLt:")
for M, N, K in [(401408, 128, 512), (401408, 64, 256), (100352, 128, 512),
                (1605632, 64, 576), (100352, 192, 1024)]:
    a = (torch.rand(M, K, device='cuda', dtype=torch.bfloat16)*2-1)
    b = (torch.rand(N, K, device='cuda', dtype=torch.bfloat16)*2-1)
    fl = 2.0*M*N*K
    dw = t(lambda: ops.require().gemm_btw_bf16(a, b))
    db_ = t(lambda: a @ b.t())
    line = f"  M{M} N{N} K{K}: wide {fl/dw/1e12:6.1f} TF | blas {fl/db_/1e12:6.1f} TF"
    if N % 128 == 0:
        d1 = t(lambda: ops.require().gemm_bt_bf16(a, b))
        line += f" | 128t {fl/d1/1e12:6.1f} TF"
    print(line)
