"""Gradient-path kernel bandwidth microbench (written to profiles/)."""
import os, sys, time, torch
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "."))
from chainermn_amd import ops

def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n

N = 256 * (1<<20) // 4  # 64M elements (fp32 = 256 MB, past L3)
dev = 'cuda'
res = []
# cast fp32 -> bf16 (pack path): read 4B + write 2B per elem
src = torch.randn(N, device=dev); dst = torch.empty(N, device=dev, dtype=torch.bfloat16)
dt = t(lambda: ops.cast_flat(src, dst))
res.append(("cast_flat fp32->bf16 (K3/pack)", N*6/dt/1e12))
# cast bf16 -> fp32 with scale (unpack path)
dt = t(lambda: ops.cast_flat(dst, src, scale=0.5))
res.append(("cast_flat bf16->fp32 x scale (unpack)", N*6/dt/1e12))
# fused SGD momentum, bf16 grads: read p4+g2+v4, write p4+v4+lowp2 = 20 B/elem
p = torch.randn(N, device=dev); v = torch.zeros(N, device=dev)
g = torch.randn(N, device=dev, dtype=torch.bfloat16); lp = p.to(torch.bfloat16)
dt = t(lambda: ops.fused_sgd(p, g, v, 0.0, momentum=0.9, p_lowp=lp))
res.append(("fused_sgd mom + bf16 mirror (K5)", N*20/dt/1e12))
# fused Adam: read p,g2,m,v write p,m,v,lowp2 = 24 B/elem
m = torch.zeros(N, device=dev); vv = torch.zeros(N, device=dev)
dt = t(lambda: ops.fused_adam(p, g, m, vv, 0.0, 0.9, 0.999, 1e-8, p_lowp=lp))
res.append(("fused_adam + bf16 mirror (K5)", N*24/dt/1e12))
# batched pack bf16 grads (K1): ResNet-50-like param list scaled up
class P:
    def __init__(s, t_): s.data = t_; s.grad = torch.randn_like(t_)
    def numel(s): return s.data.numel()
import numpy.random as npr
npr.seed(0)
sizes = [int(x) for x in (npr.lognormal(11, 2, 160).clip(64, 4<<20))]
params = [P(torch.randn(sz, device=dev, dtype=torch.bfloat16)) for sz in sizes]
plan = ops.PackPlan(params, torch.device(dev))
flat = torch.empty(plan.total, device=dev, dtype=torch.bfloat16)
nbytes = sum(sizes)*2
dt = t(lambda: ops.pack_grads(plan, flat))
res.append((f"batched_pack bf16 {nbytes>>20} MB, 160 params (K1)", nbytes*2/dt/1e12))
dt = t(lambda: ops.unpack_grads(plan, flat, 0.5))
res.append(("batched_unpack + scale (K2)", nbytes*2/dt/1e12))
# segment_sq_norms (LARS)
dt = t(lambda: ops.segment_sq_norms(src, plan.offsets[:1], torch.tensor([N], device=dev), torch.zeros(1, device=dev)))
res.append(("segment_sq_norms 256MB single segment", N*4/dt/1e12))
print("kernel bandwidth microbench, 1xMI355X (achievable HBM ~6.3 TB/s):")
for name, tb in res:
    print(f"  {name:45s} {tb:5.2f} TB/s  ({100*tb/6.3:.0f}% of achievable)")
