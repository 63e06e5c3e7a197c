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
