"""Minimal kernel driver for PMC counter collection (run under rocprofv3
--pmc, which must not be combined with tracing on this pool)."""
import sys
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from transformer_amd.ops import ext

E = ext()
torch.manual_seed(0)
which = sys.argv[1] if len(sys.argv) > 1 else "all"
nop = torch.Tensor()
if which in ("all", "dw"):
    dy = torch.randn(16384, 2048, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(16384, 512, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        E.gemm_dw(dy, x)
if which in ("all", "g256"):
    a = torch.randn(16384, 1024, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(3072, 1024, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        E.gemm256_nt(a, w, nop, 0)
if which in ("all", "attn"):
    q = torch.randn(64, 256, 8, 64, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        E.attn_fwd(q, q, q, nop, False, 0.125, 1)
torch.cuda.synchronize()
print("done")
