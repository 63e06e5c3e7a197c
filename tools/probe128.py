"""gemm256 vs the 128-tile dispatch at K=512 256-block shapes
(the round-2 viability-rule relax: +4-24% measured — gemm256.hip).

    python tools/probe128.py   (GPU box)
"""
import sys, torch
sys.path.insert(0, ".")
from transformer_amd.ops import ext
E = ext()
def bench(fns, iters=30):
    for f in fns:
        for _ in range(3): f()
    torch.cuda.synchronize()
    best = [1e9]*len(fns)
    for _ in range(4):
        for i, f in enumerate(fns):
            s = torch.cuda.Event(True); e = torch.cuda.Event(True)
            s.record()
            for _ in range(iters): f()
            e.record(); torch.cuda.synchronize()
            best[i] = min(best[i], s.elapsed_time(e)/iters)
    return best
for M, N, K in [(16384,512,512),(16384,1024,512),(16320,512,512),(16384,512,1024)]:
    a = torch.randn(M,K,device="cuda",dtype=torch.bfloat16)*0.05
    w = torch.randn(N,K,device="cuda",dtype=torch.bfloat16)*0.05
    b = torch.randn(N,device="cuda",dtype=torch.bfloat16)
    r = bench([lambda: E.gemm_nt(a,w,b,0), lambda: E.gemm256_nt(a,w,b,0)])
    tf = lambda ms: 2.0*M*N*K/(ms*1e-3)/1e12
    c1 = E.gemm_nt(a,w,b,0).float(); c2 = E.gemm256_nt(a,w,b,0).float()
    ref = a.float()@w.float().T + b.float()
    e1 = (c1-ref).abs().max()/ref.abs().max(); e2 = (c2-ref).abs().max()/ref.abs().max()
    print(f"{M}x{N}x{K}: dispatch(128) {r[0]*1000:6.1f}us {tf(r[0]):4.0f}TF | g256 {r[1]*1000:6.1f}us {tf(r[1]):4.0f}TF | err {e1:.3f}/{e2:.3f}")
