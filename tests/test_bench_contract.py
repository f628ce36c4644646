"""Pin the driver-facing bench.py contract: one JSON line on stdout with
the schema the round-end harness parses (metric/config per BASELINE.json).
Runs the CPU smoke path (seq clamps small off-GPU) — the schema is the
same object the MI355X run prints."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout[-2000:]
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in out, key
    assert out["metric"] == "attention tokens/sec"
    assert out["unit"] == "tokens/s"
    assert out["n_gpus"] == 1 and out["steps"] == 3 and out["warmup"] == 1
    assert out["higher_is_better"] is True and out["scaling"] == "weak"
    assert out["dtype"] == "bf16" and out["data"] == "synthetic"
    assert out["value"] > 0 and out["ms_per_step"] > 0
    cfg = out["config"]
    for key in ("model", "global_batch", "seq_len", "num_heads",
                "head_dim", "parallelism"):
        assert key in cfg, key
    assert cfg["num_heads"] == 32 and cfg["head_dim"] == 128
    assert cfg["parallelism"] == "sp1"
    # whole-job aggregate: tokens attended / s == B*seq*steps / elapsed
    expect = (cfg["global_batch"] * cfg["seq_len"] * out["steps"]
              / (out["ms_per_step"] * out["steps"] / 1e3))
    assert abs(expect - out["value"]) / out["value"] < 1e-6
