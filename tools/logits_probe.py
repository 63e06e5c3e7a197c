"""A/B tile-walk order x XCD remap at the L3-resident logits-head shapes
(the biggest single GEMM pool: ~2.4 ms/step).

    TFMX_G256_ORDER=nm TFMX_G256_XCD=0 python tools/logits_probe.py ...
is env-driven; this script re-execs itself over the matrix.
"""
import os
import subprocess
import sys

import torch

sys.path.insert(0, ".")


def run_one():
    from transformer_amd.ops import ext
    E = ext()
    dt = torch.bfloat16
    torch.manual_seed(0)
    shapes = [
        (16320, 32770, 512, "logits fwd"),
        (16320, 512, 33024, "logits dX (padded K)"),
        (16384, 1536, 512, "QKV fwd"),
        (16384, 512, 2048, "FFN2 fwd"),
    ]
    for M, N, K, tag in shapes:
        a = torch.randn(M, K, device="cuda", dtype=dt) * 0.05
        w = torch.randn(N, K, device="cuda", dtype=dt) * 0.05
        for _ in range(3):
            E.gemm_nt(a, w, torch.Tensor(), 0)
        torch.cuda.synchronize()
        best = float("inf")
        for _ in range(4):
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            for _ in range(10):
                E.gemm_nt(a, w, torch.Tensor(), 0)
            e.record()
            torch.cuda.synchronize()
            best = min(best, s.elapsed_time(e) / 10)
        tf = 2.0 * M * N * K / (best * 1e-3) / 1e12
        print(f"  {tag:22s} {best*1000:7.1f}us {tf:6.0f}TF", flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "one":
        run_one()
    else:
        for order in ("m", "n"):
            for xcd in ("1", "0"):
                print(f"== order={'mn' if order=='m' else 'nm'} xcd={xcd} ==",
                      flush=True)
                env = dict(os.environ, TFMX_G256_ORDER=order,
                           TFMX_G256_XCD=xcd)
                subprocess.run([sys.executable, __file__, "one"], env=env)
