"""Per-shape timing of the reduction/elementwise kernel fleet vs the HBM
bound (~6.3 TB/s achievable), to find which of the 1.4 ms/step colsum pool,
0.77 ms CE pair and 0.35 ms relu_bwd is worth optimizing.  Also times
gemm_nn vs hipBLASLt at the dh = dY2 @ W2 shape (candidate relu-mask
epilogue fusion site)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from transformer_amd import ops

E = ops.ext()
torch.manual_seed(0)


def t(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


M = 16320
print("== colsum (db): read bound = bytes / 6.3 TB/s ==")
for N in (512, 2048, 32770):
    dy = torch.randn(M, N, device="cuda").bfloat16().contiguous()
    us = t(lambda: E.colsum(dy))
    gb = M * N * 2 / 1e9
    ref = dy.float().sum(0)
    got = E.colsum(dy).float()
    err = (ref - got).abs().max().item() / ref.abs().max().item()
    print(f"  colsum {M}x{N}: {us:8.1f} us  ({gb/us*1e3:7.0f} TB/s, "
          f"bound {gb/6.3*1e3:6.1f} us)  relerr {err:.1e}")

print("== ce fwd/bwd: logits 16320x32770 ==")
V = 32770
logits = torch.randn(M, V, device="cuda").bfloat16().contiguous()
tgt = torch.randint(0, V, (M,), device="cuda")
tgt[::7] = 0
loss_lse = E.ce_fwd(logits, tgt, 64.0, 0.1)
lf = logits.float()
ref_lse = torch.logsumexp(lf, -1)
real = tgt != 0
per = ref_lse - 0.9 * lf.gather(1, tgt[:, None]).squeeze(1) - 0.1 * lf.mean(1)
ref_loss = per[real].sum() / 64.0
print(f"  ce loss relerr {abs(loss_lse[0].item()-ref_loss.item())/abs(ref_loss.item()):.1e}")
us = t(lambda: E.ce_fwd(logits, tgt, 64.0, 0.1))
gb = M * V * 2 / 1e9
print(f"  ce_fwd : {us:8.1f} us  ({gb/us*1e3:7.0f} TB/s, bound {gb/6.3*1e3:6.1f} us)")
dloss = torch.ones((), device="cuda")
us = t(lambda: E.ce_bwd(logits, tgt, loss_lse[1], dloss, 64.0, 0.1))
print(f"  ce_bwd : {us:8.1f} us  ({2*gb/us*1e3:7.0f} TB/s r+w, bound {2*gb/6.3*1e3:6.1f} us)")

print("== relu_bwd 16320x2048 ==")
h = torch.randn(M, 2048, device="cuda").bfloat16().contiguous()
dyh = torch.randn(M, 2048, device="cuda").bfloat16().contiguous()
us = t(lambda: E.relu_bwd(dyh, h))
gb = 3 * M * 2048 * 2 / 1e9
print(f"  relu_bwd: {us:7.1f} us  ({gb/us*1e3:7.0f} TB/s rrw, bound {gb/6.3*1e3:6.1f} us)")

print("== ln_gb shapes (dgamma/dbeta over rows, D=512) ==")
s = torch.randn(M, 512, device="cuda").bfloat16().contiguous()
dy5 = torch.randn(M, 512, device="cuda").bfloat16().contiguous()
g = torch.ones(512, device="cuda").bfloat16()
mean = torch.zeros(M, device="cuda")
rstd = torch.ones(M, device="cuda")
us = t(lambda: E.ln_bwd(dy5, s, g, mean, rstd))
gb = 3 * M * 512 * 2 / 1e9
_, dgm, dbt = E.ln_bwd(dy5, s, g, mean, rstd)
xh = (s.float() - mean[:, None]) * rstd[:, None]
rg = (dy5.float() * xh).sum(0)
rb = dy5.float().sum(0)
eg = (rg - dgm.float()).abs().max().item() / rg.abs().max().item()
eb = (rb - dbt.float()).abs().max().item() / rb.abs().max().item()
print(f"  ln_bwd+gb: {us:6.1f} us  ({gb/us*1e3:7.0f} TB/s, bound {gb/6.3*1e3:6.1f} us)  relerr g {eg:.1e} b {eb:.1e}")

print("== dh GEMM: dY2[16320,512] @ W2[512,2048] (NN) ==")
a = torch.randn(M, 512, device="cuda").bfloat16().contiguous()
b = torch.randn(512, 2048, device="cuda").bfloat16().contiguous()
fl = 2 * M * 512 * 2048 / 1e12
us = t(lambda: torch.matmul(a, b))
print(f"  blaslt NN : {us:7.1f} us ({fl/us*1e6:6.0f} TF/s)")
us = t(lambda: E.gemm_nn(a, b))
print(f"  gemm_nn   : {us:7.1f} us ({fl/us*1e6:6.0f} TF/s)")
c1 = torch.matmul(a, b).float()
c2 = E.gemm_nn(a, b).float()
print(f"  rel err   : {(c1-c2).abs().max().item()/c1.abs().max().item():.3e}")
print("eltwise bench done")
