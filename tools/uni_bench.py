"""A/B the round-2 gemm_uni kernels against the round-1 dispatch targets
at the model's actual training shapes (GPU box).

    python tools/uni_bench.py [iters]

Per shape, interleaved rounds (guide §5.4 rule 24): report min ms and TF.
  fwd : gemm_uni_nt (counted-vmcnt + setprio)  vs  gemm_nt (r1 dispatch)
        vs hipBLASLt fused epilogue (F.linear/_addmm_activation)
  dX  : gemm_uni_nn (NTxTR)                    vs  torch.matmul (r1)
  dW  : gemm_uni_tn (TRxTR, splitr sweep)      vs  gemm_dw / torch.matmul
"""

import sys

import torch

sys.path.insert(0, ".")
from transformer_amd.ops import ext  # noqa: E402

E = ext()


def bench_pair(fns, iters=30, rounds=5):
    """Interleaved rounds; returns per-fn min-of-round-means (ms)."""
    for fn in fns:
        for _ in range(3):
            fn()
    torch.cuda.synchronize()
    best = [float("inf")] * len(fns)
    for _ in range(rounds):
        for i, fn in enumerate(fns):
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            for _ in range(iters):
                fn()
            e.record()
            torch.cuda.synchronize()
            best[i] = min(best[i], s.elapsed_time(e) / iters)
    return best


def tf(m, n, k, ms):
    return 2.0 * m * n * k / (ms * 1e-3) / 1e12


FWD = [
    (16384, 1536, 512, 0, "QKV fwd"),
    (16384, 512, 512, 0, "attn O fwd"),
    (16384, 2048, 512, 1, "FFN1 fwd (relu)"),
    (16384, 512, 2048, 0, "FFN2 fwd"),
    (16320, 32770, 512, 0, "logits fwd"),
    (16384, 3072, 1024, 0, "big QKV fwd"),
    (16384, 4096, 1024, 1, "big FFN1 fwd"),
    (16384, 1024, 4096, 0, "big FFN2 fwd"),
]

DX = [
    (16384, 1536, 512, "QKV dx"),
    (16384, 512, 512, "attn O dx"),
    (16384, 2048, 512, "FFN1 dx"),    # dy (M,2048) @ W1 (2048,512)
    (16384, 512, 2048, "FFN2 dx"),    # dy (M,512)  @ W2 (512,2048)
    (16384, 3072, 1024, "big QKV dx"),
    (16384, 4096, 1024, "big FFN1 dx"),
]

DW = [
    (16384, 1536, 512, "QKV dW"),
    (16384, 512, 512, "attn O dW"),
    (16384, 2048, 512, "FFN1 dW"),
    (16384, 512, 2048, "FFN2 dW"),
    (16320, 32768, 512, "logits dW (aligned probe)"),
    (16384, 3072, 1024, "big QKV dW"),
    (16384, 1024, 1024, "big O dW"),
    (16384, 4096, 1024, "big FFN1 dW"),
    (16384, 1024, 4096, "big FFN2 dW"),
]


SCHED_NAMES = {0: "cnt+prio", 1: "cnt", 2: "r1-style", 3: "cnt+stat",
               4: "cnt+Bq1", 5: "ring", 6: "nobar", 7: "2bar", 8: "ring128x2"}


def sched_ab(iters):
    print("== NT schedule A/B (gemm_uni_nt_ab) ==")
    dt = torch.bfloat16
    for M, N, K, _, tag in [FWD[0], FWD[3], FWD[4], FWD[6]]:
        a = torch.randn(M, K, device="cuda", dtype=dt)
        w = torch.randn(N, K, device="cuda", dtype=dt) * 0.05
        scheds = [2, 5, 6, 8]
        fns = [lambda s=s: E.gemm_uni_nt_ab(a, w, s) for s in scheds]
        fns.append(lambda: E.gemm_nt(a, w, torch.Tensor(), 0))
        r = bench_pair(fns, iters)
        parts = " | ".join(f"{SCHED_NAMES[s]} {tf(M,N,K,r[i]):5.0f}TF"
                           for i, s in enumerate(scheds))
        print(f"{tag:18s} {parts} | r1-nt {tf(M,N,K,r[-1]):5.0f}TF")
        # numerics sanity on one round
        ref = (a.float() @ w.float().T)
        for s in scheds:
            c = E.gemm_uni_nt_ab(a, w, s).float()
            err = (c - ref).abs().max() / ref.abs().max().clamp(min=1)
            assert err < 0.03, (tag, s, err)


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    dev = "cuda"
    dt = torch.bfloat16
    torch.manual_seed(0)

    if len(sys.argv) > 2 and sys.argv[2] == "sched":
        sched_ab(iters)
        return

    print("== forward: uni vs r1 gemm_nt vs hipBLASLt ==")
    for M, N, K, epi, tag in FWD:
        a = torch.randn(M, K, device=dev, dtype=dt)
        w = torch.randn(N, K, device=dev, dtype=dt) * 0.05
        b = torch.randn(N, device=dev, dtype=dt)
        wt = w.t().contiguous().t()  # for F.linear (same layout)
        fns = [lambda: E.gemm_uni_nt(a, w, b, epi),
               lambda: E.gemm_nt(a, w, b, epi)]
        if epi == 1 and hasattr(torch, "_addmm_activation"):
            fns.append(lambda: torch._addmm_activation(b, a, w.t()))
        else:
            fns.append(lambda: torch.nn.functional.linear(a, w, b))
        r = bench_pair(fns, iters)
        print(f"{tag:22s} uni {r[0]*1000:7.1f}us {tf(M,N,K,r[0]):6.0f}TF | "
              f"r1 {r[1]*1000:7.1f}us {tf(M,N,K,r[1]):6.0f}TF | "
              f"blaslt {r[2]*1000:7.1f}us {tf(M,N,K,r[2]):6.0f}TF")

    print("== dX: uni_nn (NTxTR) vs wt+gemm_nt vs torch.matmul ==")
    for M, N, K, tag in DX:
        dy = torch.randn(M, N, device=dev, dtype=dt) * 0.05
        w = torch.randn(N, K, device=dev, dtype=dt) * 0.05
        wt = w.t().contiguous()
        r = bench_pair([lambda: E.gemm_uni_nn(dy, w),
                        lambda: E.gemm_nt(dy, wt, torch.Tensor(), 0),
                        lambda: torch.matmul(dy, w)], iters)
        print(f"{tag:22s} uni {r[0]*1000:7.1f}us {tf(M,N,K,r[0]):6.0f}TF | "
              f"wt+nt {r[1]*1000:7.1f}us {tf(M,N,K,r[1]):6.0f}TF | "
              f"blaslt {r[2]*1000:7.1f}us {tf(M,N,K,r[2]):6.0f}TF")

    print("== dW: uni_tn (TRxTR) vs gemm_dw vs torch.matmul ==")
    for Mt, N, K, tag in DW:
        dy = torch.randn(Mt, N, device=dev, dtype=dt) * 0.05
        x = torch.randn(Mt, K, device=dev, dtype=dt) * 0.05
        fns = [lambda: E.gemm_dw(dy, x, None, None, None),
               lambda: torch.matmul(dy.t(), x)]
        splits = [1, 2, 4, 8, 16, 32]
        for s in splits:
            fns.append(lambda s=s: E.gemm_uni_tn(dy, x, None, s))
        r = bench_pair(fns, max(10, iters // 2))
        base = " ".join(f"uni@s{s} {r[2+i]*1000:6.1f}us"
                        f" {tf(Mt,N,K,r[2+i]):5.0f}TF"
                        for i, s in enumerate(splits))
        print(f"{tag:22s} dw {r[0]*1000:7.1f}us {tf(Mt,N,K,r[0]):6.0f}TF | "
              f"blaslt {r[1]*1000:7.1f}us {tf(Mt,N,K,r[1]):6.0f}TF")
        print(f"{'':22s} {base}")


if __name__ == "__main__":
    main()
