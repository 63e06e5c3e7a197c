"""A/B the gemm_dw staging ring depth (TFMX_DW_RING=2|3) at the dW
training shapes — re-execs itself per arm."""
import json
import os
import subprocess
import sys

import torch

sys.path.insert(0, ".")

SHAPES = [(16384, 1536, 512, "qkv"), (16384, 2048, 512, "ffn1"),
          (16384, 512, 2048, "ffn2"), (16384, 512, 512, "o"),
          (16384, 3072, 1024, "big qkv"), (16384, 4096, 1024, "big ffn1")]


def one():
    from transformer_amd.ops import ext
    E = ext()
    torch.manual_seed(0)
    out = {}
    for M, N, K, tag in SHAPES:
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.05
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.05
        ref = dy.float().T @ x.float()
        got = E.gemm_dw(dy, x, None, None, None).float()
        err = ((got - ref).abs().max() / ref.abs().max().clamp(min=1)).item()
        assert err < 0.03, (tag, err)
        for _ in range(3):
            E.gemm_dw(dy, x, None, None, None)
        torch.cuda.synchronize()
        best = 1e9
        for _ in range(4):
            s = torch.cuda.Event(True)
            e = torch.cuda.Event(True)
            s.record()
            for _ in range(20):
                E.gemm_dw(dy, x, None, None, None)
            e.record()
            torch.cuda.synchronize()
            best = min(best, s.elapsed_time(e) / 20)
        out[tag] = round(best * 1000, 1)
    print(json.dumps(out))


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "one":
        one()
    else:
        for ring in ("2", "3"):
            env = dict(os.environ, TFMX_DW_RING=ring)
            r = subprocess.run([sys.executable, __file__, "one"], env=env,
                               capture_output=True, text=True)
            tail = (r.stdout.strip().splitlines() or [r.stderr[-300:]])[-1]
            print(f"RING{ring}: {tail}", flush=True)
