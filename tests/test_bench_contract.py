"""The driver depends on bench.py's exact contract: default flags finish
quickly and rank 0 prints ONE JSON line with the documented fields."""

import json
import subprocess
import sys


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--model", "tiny", "--batch", "2", "--seq_len", "16"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 1 and d["warmup"] == 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["unit"] == "tokens/s" and d["value"] > 0
    assert d["dtype"] in ("bf16", "fp32")  # bf16 on GPU, fp32 CPU fallback
    assert "synthetic" in d["data"]
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg, key


def test_bench_torchrun_world2_gloo():
    """The driver launches bench.py under torchrun for N>1.  Run the exact
    launch shape with 2 CPU ranks (gloo): rendezvous, DDP broadcast +
    bucketed all-reduce, barriers, the cross-rank MAX reduction, and the
    single rank-0 JSON line must all work."""
    import json
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--model", "tiny", "--batch", "2",
         "--seq_len", "16"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected ONE JSON line, got {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4  # 2 per rank, weak scaling
