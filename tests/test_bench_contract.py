"""bench.py driver-contract test: one JSON line on stdout with the agreed
schema, runnable on CPU with the tiny model."""
import json
import os
import subprocess
import sys


def run_bench(*extra, port_base=None, timeout=240, span=1):
    from sharedtensor_amd.utils import free_port
    if port_base is None:
        port_base = str(free_port(span=span))
    env = dict(os.environ)
    env["SHTENS_PORT_BASE"] = str(port_base)
    env.pop("WORLD_SIZE", None)  # exercise the driver's plain invocation
    env.pop("RANK", None)
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "tiny", "--batch", "2", "--seq", "32", *extra],
        capture_output=True, text=True, timeout=timeout, env=env)
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


def test_n1_paramsync_self_describing():
    d = run_bench()
    cfg = d["config"]
    assert cfg["paramsync"] == "n/a at n_gpus=1 (no links)"
    assert cfg["paramsync_wire_gbps"] is None
    assert cfg["sync_rounds_per_s"] is None


def test_self_launch_gpus2():
    """The driver's plain `python bench.py --gpus 2` must complete unaided:
    bench forks both ranks itself (no torchrun, no preset WORLD_SIZE)."""
    d = run_bench("--gpus", "2", timeout=420, span=2)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "async-dp2"
    # with links up, the param-sync half of the metric is measured
    assert d["config"]["paramsync_wire_gbps"] is not None
    assert d["config"]["sync_rounds_per_s"] >= 0
    links = d["config"]["links_rank0"]  # root's per-link view
    assert len(links) >= 1
    for l in links:
        for key in ("peer", "rccl", "rounds_sent_per_s", "rounds_recv_per_s"):
            assert key in l, key


def test_self_launch_gpus8_dry():
    """`--gpus 8` self-launch dry-run on CPU: the full 8-rank tree forms,
    steps, and reports without hanging (pre-stages the driver's SCALE run)."""
    d = run_bench("--gpus", "8", "--steps", "1", timeout=600, span=8)
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "async-dp8"


def test_paramsync_contract():
    d = run_bench("--mode", "paramsync", "--numel", "4096",
                  "--interval", "0.001")
    assert d["unit"] == "GB/s logical"
    assert d["n_gpus"] == 1
    assert "staleness_p50" in d["config"]
