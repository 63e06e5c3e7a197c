"""Microbenchmark gemm_nt at the model's actual shapes (run on GPU box).

    python tools/gemm_bench.py [iters]
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from transformer_amd.ops import ext  # noqa: E402

# (M, N, K, tag) — transformer-base, B=64, S=256 training shapes
SHAPES = [
    (16384, 1536, 512, "enc QKV fwd"),
    (16384, 512, 512, "attn O fwd"),
    (16384, 2048, 512, "FFN1 fwd"),
    (16384, 512, 2048, "FFN2 fwd"),
    (16320, 32770, 512, "logits fwd"),
    (16320, 512, 32770, "logits dx (odd K)"),
    (32770, 512, 16320, "logits dW"),
    (512, 512, 16384, "attn O dW"),
    (2048, 512, 16384, "FFN1 dW"),
    (16384, 512, 1536, "QKV dx"),
    (1536, 512, 16384, "QKV dW"),
]


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    E = ext()
    torch.manual_seed(0)
    print(f"{'tag':18s} {'M':>6s} {'N':>6s} {'K':>6s} {'ms':>8s} {'TF/s':>7s}")
    total_t, total_f = 0.0, 0.0
    for (m, n, k, tag) in SHAPES:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):
            E.gemm_nt(a, w, b, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            E.gemm_nt(a, w, b, 0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        fl = 2.0 * m * n * k
        print(f"{tag:18s} {m:6d} {n:6d} {k:6d} {dt*1e3:8.3f} {fl/dt/1e12:7.1f}")
        total_t += dt
        total_f += fl
    print(f"{'TOTAL':18s} {'':21s} {total_t*1e3:8.3f} {total_f/total_t/1e12:7.1f}")


if __name__ == "__main__":
    main()
