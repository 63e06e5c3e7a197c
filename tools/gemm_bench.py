"""Microbenchmark the GEMM paths at the model's actual shapes (GPU box).

    python tools/gemm_bench.py [iters]

For each backward shape it times BOTH implementations:
  dX:  gemm_nn(dy, w)            vs  transpose2d(w) + gemm_nt(dy, wT)
  dW:  gemm_tn(dy, x)            vs  transpose2d(dy)+transpose2d(x)+gemm_nt
so the functional layer's dispatch choices are measurement-driven.
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


# forward shapes: (M, N, K, tag) — transformer-base B=64 S=256 + big-model
FWD = [
    (16384, 1536, 512, "QKV fwd"),
    (16384, 512, 512, "attn O fwd"),
    (16384, 2048, 512, "FFN1 fwd"),
    (16384, 512, 2048, "FFN2 fwd"),
    (16320, 32770, 512, "logits fwd"),
    (16384, 3072, 1024, "big QKV fwd"),
    (16384, 4096, 1024, "big FFN1 fwd"),
]

# dX shapes: dy (M, N_out) @ w (N_out, K_in): (M, N_out, K_in, tag)
DX = [
    (16384, 1536, 512, "QKV dx"),
    (16384, 512, 512, "attn O dx"),
    (16384, 512, 2048, "FFN1 dx"),
    (16384, 2048, 512, "FFN2 dx"),
    (16320, 32770, 512, "logits dx"),
]

# dW shapes: dy (Mtok, N_out), x (Mtok, K_in): (Mtok, N_out, K_in, tag)
DW = [
    (16384, 1536, 512, "QKV dW"),
    (16384, 512, 512, "attn O dW"),
    (16384, 2048, 512, "FFN1 dW"),
    (16384, 512, 2048, "FFN2 dW"),
    (16320, 32770, 512, "logits dW"),
]


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    E = ext()
    torch.manual_seed(0)

    print("== 256-template vs 128-kernel vs torch.matmul (hipBLASLt) ==")
    for (m, n, k, tag) in FWD + [(16384, 512, 2048, "FFN2 fwd"),
                                 (1536, 512, 16384, "QKV dW"),
                                 (32770, 512, 16320, "logits dW"),
                                 (4096, 4096, 4096, "4096^3 cal"),
                                 (8192, 8192, 8192, "8192^3 cal")]:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        nob = torch.Tensor()
        ref = (a[:512].float() @ w.t().float()[:, :512])
        got = E.gemm256_nt(a, w, nob, 0)[:512, :512].float()
        err = (got - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        t256 = timeit(lambda: E.gemm256_nt(a, w, nob, 0), iters)
        t128 = timeit(lambda: E.gemm128_nt(a, w, nob, 0), iters)
        tbl = timeit(lambda: torch.matmul(a, w.t()), iters)
        fl = 2.0 * m * n * k
        print(f"{tag:14s} 256 {t256*1e3:8.3f} ms ({fl/t256/1e12:6.1f} TF) | "
              f"128 {t128*1e3:8.3f} ({fl/t128/1e12:6.1f}) | "
              f"blaslt {tbl*1e3:8.3f} ({fl/tbl/1e12:6.1f})  relerr {err:.2e}")

    print("== forward: gemm_nt ==")
    for (m, n, k, tag) in FWD:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: E.gemm_nt(a, w, b, 0), iters)
        fl = 2.0 * m * n * k
        print(f"{tag:14s} M{m:6d} N{n:6d} K{k:6d} {dt*1e3:8.3f} ms {fl/dt/1e12:7.1f} TF/s")

    print("== dX: gemm_nn vs transpose(w)+gemm_nt ==")
    for (m, n, k, tag) in DX:
        dy = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        t_nn = timeit(lambda: E.gemm_nn(dy, w), iters)
        t_tr = timeit(lambda: E.gemm_nt(dy, E.transpose2d(w), torch.Tensor(), 0), iters)
        t_bl = timeit(lambda: torch.matmul(dy, w), iters)
        fl = 2.0 * m * n * k
        print(f"{tag:14s} nn {t_nn*1e3:8.3f} ms ({fl/t_nn/1e12:6.1f} TF) | "
              f"tr+nt {t_tr*1e3:8.3f} ms ({fl/t_tr/1e12:6.1f} TF) | "
              f"blaslt {t_bl*1e3:8.3f} ({fl/t_bl/1e12:6.1f})")

    print("== dW: gemm_tn vs transpose(dy)+transpose(x)+gemm_nt ==")
    for (mt, n, k, tag) in DW:
        dy = torch.randn(mt, n, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(mt, k, device="cuda", dtype=torch.bfloat16)
        t_tn = timeit(lambda: E.gemm_tn(dy, x), iters)
        t_tr = timeit(lambda: E.gemm_nt(E.transpose2d(dy), E.transpose2d(x),
                                        torch.Tensor(), 0), iters)
        t_bl = timeit(lambda: torch.matmul(dy.t(), x), iters)
        t_dw = timeit(lambda: E.gemm_dw(dy, x), iters)
        fl = 2.0 * mt * n * k
        print(f"{tag:14s} tn {t_tn*1e3:8.3f} ms ({fl/t_tn/1e12:6.1f} TF) | "
              f"tr+nt {t_tr*1e3:8.3f} ms ({fl/t_tr/1e12:6.1f} TF) | "
              f"blaslt {t_bl*1e3:8.3f} ({fl/t_bl/1e12:6.1f}) | "
              f"dw {t_dw*1e3:8.3f} ({fl/t_dw/1e12:6.1f})")


if __name__ == "__main__":
    main()
