import math, random, sys, torch
sys.path.insert(0, "/root/repo")
from transformer_amd.ops import ext
E = ext()
bad = 0
def chk(got, ref, tol, name):
    global bad
    err = ((got.float()-ref.float()).abs().max() / ref.float().abs().max().clamp(min=1.0)).item()
    if err > tol:
        bad += 1
        print("FAIL", name, err)
for case in range(6):
    rng = random.Random(900+case)
    m, n = rng.randrange(500, 3000), rng.randrange(500, 3000)
    k = 64*rng.randrange(2, 40)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    chk(E.gemm256_nt(a, w, torch.Tensor(), 0), a.float()@w.float().T, .04, f"g256 {m}x{n}x{k}")
    del a, w
for case in range(4):
    rng = random.Random(950+case)
    B, H, dh = rng.randrange(1,3), rng.choice([2,4]), 64
    S = rng.randrange(700, 1700)
    q = torch.randn(B,S,H,dh,device="cuda",dtype=torch.bfloat16)
    kk = torch.randn(B,S,H,dh,device="cuda",dtype=torch.bfloat16)
    v = torch.randn(B,S,H,dh,device="cuda",dtype=torch.bfloat16)
    sc = 1/math.sqrt(dh)
    o, lse = E.attn_fwd(q,kk,v,torch.Tensor(),True,sc)
    qt,kt,vt = (t.float().permute(0,2,1,3) for t in (q,kk,v))
    s = qt@kt.transpose(-1,-2)*sc + torch.triu(torch.ones(S,S,device="cuda"),1)*-1e9
    ref = (torch.softmax(s,-1)@vt).permute(0,2,1,3)
    chk(o, ref, .05, f"attn S{S}")
    del q,kk,v,o,lse,qt,kt,vt,s,ref
for case in range(4):
    rng = random.Random(980+case)
    mt = rng.randrange(5000, 40000)
    n, k = rng.randrange(100, 2100), rng.randrange(100, 2100)
    dy = torch.randn(mt, n, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(mt, k, device="cuda", dtype=torch.bfloat16)
    chk(E.gemm_dw(dy, x), dy.float().t()@x.float(), .04, f"dw {mt}x{n}x{k}")
    del dy, x
print("wide fuzz done, failures:", bad)
