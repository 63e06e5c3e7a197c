"""A/B the fused attention kernels at model shapes (GPU box).

Forward: trv=1 (tr16 reads of natural V) vs trv=0 (transposed V image).
Backward: timed as-is.  python tools/attn_bench.py [iters]
"""
import sys
import time

import torch

sys.path.insert(0, ".")
from transformer_amd.ops import ext  # noqa: E402


def timeit(fn, iters):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# (B, H, Sq, Sk, dh, causal, tag)
SHAPES = [
    (64, 8, 256, 256, 64, False, "base enc self"),
    (64, 8, 255, 255, 64, True, "base dec self"),
    (64, 8, 255, 256, 64, False, "base cross"),
    (64, 16, 256, 256, 64, False, "big enc self"),
    (8, 16, 4096, 4096, 64, False, "big4k enc"),
    (8, 16, 4095, 4095, 64, True, "big4k dec"),
]


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    E = ext()
    torch.manual_seed(0)
    nop = torch.Tensor()
    for (B, H, Sq, Sk, dh, causal, tag) in SHAPES:
        q = torch.randn(B, Sq, H, dh, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Sk, H, dh, device="cuda", dtype=torch.bfloat16)
        sc = dh ** -0.5
        t1 = timeit(lambda: E.attn_fwd(q, k, v, nop, causal, sc, 1), iters)
        t0 = timeit(lambda: E.attn_fwd(q, k, v, nop, causal, sc, 0), iters)
        o, lse = E.attn_fwd(q, k, v, nop, causal, sc, 1)
        do = torch.randn_like(o)
        tb = timeit(lambda: E.attn_bwd(q, k, v, o, do, lse, nop, causal,
                                       sc, 0), iters)
        fl = 4.0 * B * H * Sq * Sk * dh * (0.5 if causal else 1.0)
        print(f"{tag:14s} fwd trv {t1*1e3:7.3f} ms ({fl/t1/1e12:6.1f} TF) | "
              f"fwd img {t0*1e3:7.3f} ({fl/t0/1e12:6.1f}) | "
              f"bwd {tb*1e3:7.3f} ({2.5*fl/tb/1e12:6.1f})")


if __name__ == "__main__":
    main()
