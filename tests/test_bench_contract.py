"""bench.py driver-contract test: one JSON line on stdout with the agreed
schema, runnable on CPU with the tiny model."""
import json
import os
import subprocess
import sys


def run_bench(*extra):
    env = dict(os.environ)
    env["SHTENS_PORT_BASE"] = "53611"
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "tiny", "--batch", "2", "--seq", "32", *extra],
        capture_output=True, text=True, timeout=240, env=env)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [l for l in out.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"stdout must be ONE JSON line, got: {lines}"
    return json.loads(lines[0])


def test_train_contract():
    d = run_bench()
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["metric"] == "async-DP tokens/sec GPT-2-small"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism", "codec",
                "staleness_p50", "paramsync_wire_gbps"):
        assert key in cfg, key


def test_paramsync_contract():
    d = run_bench("--mode", "paramsync", "--numel", "4096",
                  "--interval", "0.001")
    assert d["unit"] == "GB/s logical"
    assert d["n_gpus"] == 1
    assert "staleness_p50" in d["config"]
