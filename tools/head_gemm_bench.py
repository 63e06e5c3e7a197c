"""Logits-head forward GEMM: our gemm256 (bias epilogue) vs hipBLASLt via
F.linear (bias epilogue) and torch.matmul+add, at model fwd shapes."""
import os, sys, time
import torch
import torch.nn.functional as F
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from transformer_amd import ops
E = ops.ext()
torch.manual_seed(0)

def t(fn, iters=30):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6

M = 16320
for (N, K, tag) in [(32770, 512, "logits"), (2048, 512, "ffn1(no relu)"),
                    (512, 2048, "ffn2"), (1536, 512, "qkv")]:
    a = torch.randn(M, K, device="cuda").bfloat16().contiguous()
    w = torch.randn(N, K, device="cuda").bfloat16().contiguous()
    b = torch.randn(N, device="cuda").bfloat16().contiguous()
    fl = 2*M*N*K/1e12
    u1 = t(lambda: E.gemm_nt(a, w, b, 0))
    u2 = t(lambda: F.linear(a, w, b))
    u3 = t(lambda: torch.matmul(a, w.t()))
    c1 = E.gemm_nt(a, w, b, 0).float(); c2 = F.linear(a, w, b).float()
    err = (c1-c2).abs().max().item()/c2.abs().max().item()
    print(f"{tag:14s} ours {u1:7.1f}us ({fl/u1*1e6:5.0f}TF) | F.linear "
          f"{u2:7.1f} ({fl/u2*1e6:5.0f}) | matmul(nobias) {u3:7.1f} "
          f"({fl/u3*1e6:5.0f}) | relerr {err:.1e}")

# relu epilogue: ours vs torch._addmm_activation (hipBLASLt RELU epilogue?)
a = torch.randn(M, 512, device="cuda").bfloat16().contiguous()
w = torch.randn(2048, 512, device="cuda").bfloat16().contiguous()
b = torch.randn(2048, device="cuda").bfloat16().contiguous()
fl = 2*M*2048*512/1e12
u1 = t(lambda: E.gemm_nt(a, w, b, 1))
u2 = t(lambda: torch._addmm_activation(b, a, w.t()))
c1 = E.gemm_nt(a, w, b, 1).float()
c2 = torch._addmm_activation(b, a, w.t()).float()
err = (c1-c2).abs().max().item()/c2.abs().max().item()
print(f"ffn1+relu      ours {u1:7.1f}us ({fl/u1*1e6:5.0f}TF) | addmm_act "
      f"{u2:7.1f} ({fl/u2*1e6:5.0f}) | relerr {err:.1e}")
# small o-proj shape
w5 = torch.randn(512, 512, device="cuda").bfloat16().contiguous()
b5 = torch.randn(512, device="cuda").bfloat16().contiguous()
fl = 2*M*512*512/1e12
u1 = t(lambda: E.gemm_nt(a, w5, b5, 0))
u2 = t(lambda: F.linear(a, w5, b5))
print(f"o-proj 512x512 ours {u1:7.1f}us ({fl/u1*1e6:5.0f}TF) | F.linear "
      f"{u2:7.1f} ({fl/u2*1e6:5.0f})")
